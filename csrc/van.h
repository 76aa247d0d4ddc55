// Van: the single transport. TCP side channel for bootstrap / barriers /
// heartbeats / control and as the correctness data path; a pluggable
// DataPlane (shm rings + hipIpc/xGMI, gpu_plane.h) carries same-host
// device payloads zero-copy.
//
// Reference parity: ps-lite include/ps/internal/van.h + src/van.cc
// (bootstrap, rank assignment, barrier counting, heartbeat, recv loop)
// with the backend zoo (zmq/rdma/fabric/ucx/multivan) collapsed into one
// MI355X-native design per SURVEY.md §5.8.
#pragma once

#include <atomic>
#include <memory>
#include <random>
#include <thread>
#include <chrono>
#include <map>
#include <unordered_map>
#include <vector>
#include <functional>

#include "message.h"
#include "tcp.h"

namespace xps {

class Postoffice;
class Resender;

// Fast-path transport for data messages (installed by gpu_plane.cc).
class DataPlane {
 public:
  virtual ~DataPlane() = default;
  // may this message ride the fast path to `peer`?
  virtual bool CanSend(const Message& msg, const Node& peer) = 0;
  virtual int64_t Send(Message& msg, const Node& peer) = 0;
  virtual void OnPeer(const Node& peer) {}  // called when a peer becomes known
  // fill my Node's pool/device/shm fields before ADD_NODE is sent
  virtual void FillSelf(Node* self) {}
  // import every known peer's pool (called from the Start thread inside
  // the staged bootstrap sequence — exactly one instance imports at a
  // time cluster-wide; see Postoffice::Start)
  virtual void ImportPeers() {}
  virtual void Stop() {}
};

class Van {
 public:
  explicit Van(Postoffice* po);
  ~Van();

  void Start(int customer_id);
  void Stop();

  // route and send; expands group-mask recvers. Returns total bytes sent.
  int64_t Send(Message& msg);

  const Node& my_node() const { return my_node_; }
  Node GetNode(int id);
  bool IsReady() const { return ready_.load(); }

  void SetDataPlane(std::shared_ptr<DataPlane> p) { plane_ = std::move(p); }
  DataPlane* plane() const { return plane_.get(); }

  // Deliver a message produced locally (loopback or data-plane receive).
  void Deliver(Message msg);
  // Data-plane fast delivery: run the customer handler inline on the
  // caller (the single ring-poll thread) instead of the customer queue.
  void DeliverInline(Message& msg);

  Postoffice* postoffice() const { return po_; }

  // Slow-path send used by the data plane when a deferred fast-path
  // delivery fails (ring gone / full): the response must reach the
  // waiting requester over TCP rather than be dropped (liveness).
  int64_t SendOverTcp(Message& msg, int id);

  std::atomic<int64_t> send_bytes_{0};
  std::atomic<int64_t> recv_bytes_{0};

 private:
  int64_t SendToNode(Message& msg, int id);
  std::shared_ptr<TcpConn> GetOrDial(int id);
  void OnNewConnection(int fd);
  void RecvLoop(std::shared_ptr<TcpConn> conn);
  void ProcessControl(Message& msg, const std::shared_ptr<TcpConn>& conn);
  void ProcessAddNodeAtScheduler(Message& msg, const std::shared_ptr<TcpConn>& conn);
  void ProcessRecoveryAtScheduler(Message& msg, const std::shared_ptr<TcpConn>& conn);
  void ProcessNodeListAssigned(Message& msg);
  void ProcessBarrierAtScheduler(Message& msg);
  void ProcessHeartbeat(Message& msg);
  void DeliverData(Message& msg);
  void DeliverDataNow(Message& msg);
  void InlineData(Message& msg);
  // per-sender FIFO across transports: a sequenced data message is
  // delivered (inline, in seq order, under the flow's lock) or held
  void GatedDeliver(Message& msg);
  void ResetFlow(int peer_id);  // recovery: peer restarted with fresh counters
  void HeartbeatLoop();

  Postoffice* po_;
  Node my_node_;
  Node scheduler_;
  uint64_t my_uid_ = 0;  // nonce to recognize myself in the assigned node list
  TcpListener listener_;

  std::mutex conn_mu_;
  // cross-transport ordering state (see Meta::seq)
  std::mutex order_mu_;  // guards send_seq_ + the flow map (not delivery)
  std::unordered_map<int, uint64_t> send_seq_;
  struct Flow {
    std::mutex mu;  // held ACROSS delivery so seq order = execution order
    uint64_t expected = 1;
    std::map<uint64_t, Message> held;
    std::chrono::steady_clock::time_point hold_since;
  };
  std::unordered_map<int, std::unique_ptr<Flow>> recv_flows_;
  std::unordered_map<int, std::shared_ptr<TcpConn>> conns_;  // node id -> conn
  std::vector<std::thread> recv_threads_;
  std::atomic<bool> stopping_{false};

  std::mutex nodes_mu_;
  std::unordered_map<int, Node> nodes_;  // node id -> info (all roles + scheduler)

  // scheduler-only bootstrap/barrier state
  std::mutex sched_mu_;
  std::vector<std::pair<Node, std::shared_ptr<TcpConn>>> pending_nodes_;
  // group -> (requester id, its per-call token) pairs
  std::unordered_map<int, std::vector<std::pair<int, int>>> barrier_waiters_;
  std::unordered_map<int, std::unordered_map<uint64_t, time_t>> dead_ignore_;  // unused yet

  std::shared_ptr<DataPlane> plane_;
  std::unique_ptr<Resender> resender_;
  std::atomic<bool> ready_{false};
  int drop_rate_ = 0;
  std::mt19937 drop_rng_{12345};
  std::thread heartbeat_thread_;
  int heartbeat_interval_ = 0;

  // message tracing (ENABLE_PROFILING parity, van.cc:38-77)
  void MaybeTrace(const Message& msg, bool recv);
  FILE* trace_file_ = nullptr;
  std::mutex trace_mu_;
};

// Rank-assignment ordering for a full ADD_NODE batch (reference
// src/van.cc:126-177 placement policies):
//   BYTEPS_ORDERED_HOSTS="h1,h2,..."  — nodes sort by the host's position
//     in the list (unlisted hosts come after, by name);
//   BYTEPS_ENABLE_MIXED_MODE=1        — servers on hosts with NO worker
//     (non-colocated) sort before colocated servers, so they take the
//     low server ranks;
// then deterministic (hostname, port). Exposed for unit tests.
std::function<bool(const Node&, const Node&)> NodeRankOrder(const std::vector<Node>& batch);

}  // namespace xps
