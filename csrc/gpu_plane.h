// The MI355X data plane: same-host metadata over lock-free shm rings,
// payloads moved GPU-to-GPU over xGMI (hipIpc-mapped pools +
// hipMemcpyAsync / HIP kernels on per-peer side streams), completions
// via a hipEvent poller thread.
//
// Reference parity: replaces ps-lite's RDMA data path — rendezvous +
// one-sided RDMA_WRITE (src/rdma_van.h, rdma_transport.h) and the POSIX
// shm IPCTransport (rdma_transport.h:591-617) — per SURVEY.md §5.8:
//  * per-(key,peer) buffer rendezvous -> a single pool hipIpc handle
//    exchanged once at ADD_NODE (steady state ships {offset, len});
//  * unsignaled RDMA_WRITE + write-with-imm -> async copy/kernel on the
//    peer's stream + ring metadata sent after the completion event;
//  * PollCQ busy-spin thread -> the event-poller (completion) thread;
//  * IPCTransport shm segments -> the worker pool itself (zero copy).
#pragma once

#include <hip/hip_runtime.h>

#include <atomic>
#include <deque>
#include <memory>
#include <shared_mutex>
#include <thread>
#include <tuple>
#include <unordered_map>
#include <vector>

#include "shm_ring.h"
#include "van.h"

namespace xps {

// zero-copy reception counter (by-ref blobs resolved straight into a
// mapped peer pool; see gpu_plane.cc)
extern std::atomic<uint64_t> g_zero_copy_recv;

class Postoffice;

class GpuPlane : public DataPlane {
 public:
  GpuPlane(Postoffice* po, int device);
  ~GpuPlane() override;

  void FillSelf(Node* self) override;
  void ImportPeers() override;
  bool CanSend(const Message& msg, const Node& peer) override;
  int64_t Send(Message& msg, const Node& peer) override;
  void OnPeer(const Node& peer) override;
  void Stop() override;

  // ---- used by the GPU server handlers -------------------------------
  // Per-peer HIP stream LANES (XPS_STREAMS_PER_PEER, default 2):
  // lane 0 carries push/accumulate kernels, the pull lane carries pull-
  // response copies, so a key's response copy overlaps the next key's
  // push kernel (same xGMI link, different queues). Same-key cross-lane
  // ordering is the handlers' job via their last_ev event chains — the
  // dense handler enables chaining whenever lanes > 1. With 1 lane both
  // calls return the same stream (the round-1 behavior).
  hipStream_t StreamForPeer(int node_id);      // lane 0: push/compute
  hipStream_t PullStreamForPeer(int node_id);  // pull-response lane
  int lanes() const { return lanes_; }
  // per-peer traffic counters: (node id, tx bytes, rx bytes) — the
  // per-link utilization report of SURVEY §5.8 (each peer pair rides
  // its own xGMI link)
  std::vector<std::tuple<int, int64_t, int64_t>> PeerBytes();
  int device() const { return device_; }

  // One-sided pull (the RDMA_READ analog): copy `len` bytes from the
  // peer's advertised store entry straight into `dst` with OUR kernel
  // on the peer's lane-0 stream (ordered after our one-sided pushes of
  // the same key), then deliver the synthetic in-place response to
  // OURSELVES once the copy completes. Returns false when the peer's
  // pool is not mapped (caller sends a normal pull request).
  bool LocalPullRead(int peer_id, void* dst, uint64_t entry_off, size_t len, Message resp);
  // pooled events (shared with the server handlers)
  hipEvent_t GetEvent();
  void PutEvent(hipEvent_t ev);
  // resolve a peer-pool global offset to a locally mapped pointer
  // (handlers use this to write pull responses in place themselves)
  char* PeerDst(int peer_id, uint64_t global_off, uint64_t len) {
    return ResolvePeer(GetPeer(peer_id), global_off, len);
  }

 private:
  struct Peer {
    Node node;
    // set when this peer's Postoffice lives in OUR process (joint
    // mode): sends bypass serialize/ring/poll entirely (direct call)
    std::atomic<Postoffice*> local_po{nullptr};
    // producer handle on the peer's inbound ring. shared_ptr so a
    // RECOVERY (same id, new process, new shm_uid) can swap in a fresh
    // ring while in-flight pushes keep the old mapping alive (swapping
    // a by-value ring under a concurrent Push would munmap live memory)
    std::shared_ptr<ShmRing> ring;
    bool ring_tried = false;
    // fast-path flag: senders read this without taking mu
    std::atomic<bool> ring_ok{false};
    std::vector<void*> slab_bases;  // peer pool slabs mapped into our space
    bool pool_tried = false;
    // lane 0 = push, 1 = pull; created once, then read lock-free
    std::atomic<hipStream_t> streams[2] = {{nullptr}, {nullptr}};
    std::atomic<int64_t> tx_bytes{0};
    std::atomic<int64_t> rx_bytes{0};
    std::mutex mu;
  };

  struct Pending {
    hipEvent_t ev;
    int peer_id;
    std::string payload;
    // TCP-resendable form of this message: if the ring push fails after
    // the event fires, the van sends this instead (never drop a response
    // — the requester is blocked in Wait). For in-place writes this is
    // the meta-only kOptInPlace message (the data already landed).
    // For same-process peers it is the message delivered directly.
    Message resend;
    Message keepalive;  // holds pool temporaries until the event fires
    int64_t bytes;
    Postoffice* local_po = nullptr;  // same-process: deliver, don't ring-push
  };

  Peer* GetPeer(int id);
  // the peer's Postoffice when it lives in THIS process, else nullptr
  Postoffice* LocalPeer(Peer* p);
  // same-process send: no serialize, no ring — direct delivery (deferred
  // on the lane event when GPU kernels must complete first)
  int64_t SendLocal(Message& msg, Peer* p, Postoffice* lpo);
  void DeliverLocal(Postoffice* lpo, Message& msg, int64_t bytes);
  // hand a message to MY poll thread (I am the receiving side's plane):
  // keeps the sender's app thread free (pipelining) and runs the seq
  // gate on one thread. Falls back to inline delivery when the poll
  // thread is disabled.
  void EnqueueLocal(Message msg, int64_t bytes);
  bool EnsureRing(Peer* p);
  // the peer's current ring (nullptr when none could be opened)
  std::shared_ptr<ShmRing> RingOf(Peer* p);
  // import every slab of the peer's pool (idempotent)
  bool ImportPeerSlabs(Peer* p);
  // translate a peer-pool GLOBAL offset range to a mapped pointer
  // (nullptr if unmapped or the range is out of bounds / spans slabs)
  char* ResolvePeer(Peer* p, uint64_t global_off, uint64_t len);
  // serialize msg (meta + blobs, by-ref where flagged) into `out`;
  // by_ref[i] true => blob i encoded as pool offset
  bool Serialize(const Message& msg, const std::vector<char>& by_ref, std::string* out);
  void RingPollLoop();
  void CompletionLoop();
  hipStream_t StreamLane(int node_id, int lane);
  void DeferSendInternal(Peer* p, int peer_id, hipStream_t stream, Message resend,
                         Message keepalive, std::string payload, int64_t bytes);

  Postoffice* po_;
  int device_;
  int lanes_ = 2;  // XPS_STREAMS_PER_PEER
  uint64_t my_host_hash_;
  ShmRing in_ring_;
  bool started_ = false;
  bool inline_deliver_ = true;  // XPS_INLINE_HANDLER (default on)

  // read-mostly after bootstrap: shared lock on the hot send/recv paths
  std::shared_timed_mutex peers_mu_;
  std::unordered_map<int, std::unique_ptr<Peer>> peers_;

  std::thread poll_thread_;
  std::thread comp_thread_;
  std::atomic<bool> stop_{false};

  std::mutex pend_mu_;
  std::unordered_map<int, std::deque<Pending>> pending_;  // per peer, FIFO
  std::atomic<int> pending_count_{0};

  // same-process deliveries queued for MY poll thread
  std::mutex local_mu_;
  std::deque<std::pair<Message, int64_t>> local_q_;
  std::atomic<int> local_count_{0};
  bool poll_running_ = false;  // poll thread exists (set in FillSelf)

  std::mutex ev_mu_;
  std::vector<hipEvent_t> event_pool_;
};

// Returns the plane for this van, or nullptr when no GPU is attached.
std::shared_ptr<DataPlane> CreateGpuPlane(Postoffice* po, int device);

}  // namespace xps
