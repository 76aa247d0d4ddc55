// Postoffice: per-role runtime instance — node registry, key ranges,
// barriers, customer registry. A process may host several instances
// (scheduler / server / worker / joint), each with its own Van and port.
//
// Reference parity: ps-lite include/ps/internal/postoffice.h,
// src/postoffice.cc. Re-designed: instances of all three roles can
// coexist in one process (the reference needs one process per role,
// except JOINT); MI355X deployments run one JOINT instance pair per GPU.
#pragma once

#include <atomic>
#include <condition_variable>
#include <memory>
#include <mutex>
#include <unordered_map>
#include <vector>

#include "customer.h"
#include "message.h"

namespace xps {

class Van;

struct Range {
  uint64_t begin = 0;
  uint64_t end = 0;
  uint64_t size() const { return end - begin; }
};

class Postoffice {
 public:
  Postoffice(int role, int instance_idx = 0);
  ~Postoffice();

  // Create the Van for the next Start cycle (idempotent). Exists so a
  // DataPlane can be installed on the van before Start registers us.
  void EnsureVan();
  // Bring up the van, register with the scheduler, optional global barrier.
  void Start(int customer_id, bool do_barrier);
  void Finalize(int customer_id, bool do_barrier);

  Van* van() const { return van_.get(); }
  int role() const { return role_; }
  bool is_worker() const { return role_ == Node::WORKER; }
  bool is_server() const { return role_ == Node::SERVER; }
  bool is_scheduler() const { return role_ == Node::SCHEDULER; }
  int num_workers() const { return num_workers_; }
  int num_servers() const { return num_servers_; }
  int instance_idx() const { return instance_idx_; }

  int my_rank() const;
  int node_id() const { return node_id_; }
  void set_node_id(int id) { node_id_ = id; }
  // preferred instance rank (DMLC_RANK expanded by DMLC_GROUP_SIZE)
  int preferred_rank() const { return preferred_rank_; }
  void set_preferred_rank(int r) { preferred_rank_ = r; }

  // ids of every node in a group mask (kScheduler|kServerGroup|kWorkerGroup)
  std::vector<int> GetNodeIDs(int group) const;
  // key-space partition over servers: server i owns [i*kMaxKey/n, (i+1)*...)
  const std::vector<Range>& GetServerKeyRanges();

  // customers
  void AddCustomer(Customer* c);
  void RemoveCustomer(Customer* c);
  Customer* GetCustomer(int app_id, int customer_id, int timeout_sec = 15) const;

  // blocking barrier over `group` via the scheduler
  void Barrier(int customer_id, int group);
  // handle a barrier-release / control notification from the van
  void Manage(const Message& msg);

  // dead-node bookkeeping (heartbeats recorded by the van)
  void UpdateHeartbeat(int node_id, time_t t);
  std::vector<int> GetDeadNodes(int timeout_sec = 60);

  void RegisterExitCallback(std::function<void()> cb) { exit_cb_ = std::move(cb); }

  // ---- process-wide registry -------------------------------------------
  // Creates (if needed) and returns instances. "joint" creates one server
  // and one worker instance.
  static Postoffice* GetScheduler();
  static Postoffice* GetServer(int idx = 0);
  static Postoffice* GetWorker(int idx = 0);
  // the STARTED instance in THIS process whose assigned node id is
  // `id`, or nullptr — the same-process direct-delivery fast path asks
  // this for every data send (joint mode: worker+server in one process)
  static Postoffice* FindByNodeId(int id);
  static void ClearRegistry();  // after finalize of everything

 private:
  friend class Van;

  int role_;
  int instance_idx_;
  // atomic: FindByNodeId reads it from other threads during bootstrap
  std::atomic<int> node_id_{kEmptyNodeID};
  int preferred_rank_ = -1;
  int num_workers_ = 0;
  int num_servers_ = 0;
  std::unique_ptr<Van> van_;
  bool started_ = false;

  mutable std::mutex mu_;
  std::unordered_map<int, Customer*> customers_;  // (app_id<<16|customer_id) -> Customer
  mutable std::condition_variable customer_cv_;

  std::mutex barrier_mu_;
  std::condition_variable barrier_cv_;
  bool barrier_done_ = false;
  int barrier_seq_ = 0;    // per-call token generator
  int barrier_token_ = 0;  // token of the outstanding Barrier() call
  int barrier_group_ = 0;  // group of the outstanding Barrier() call

  std::mutex heartbeat_mu_;
  std::unordered_map<int, time_t> heartbeats_;
  time_t start_time_ = time(nullptr);

  std::vector<Range> server_key_ranges_;
  std::function<void()> exit_cb_;
};

}  // namespace xps
