#include "postoffice.h"

#include <chrono>

#include "van.h"

namespace xps {

Postoffice::Postoffice(int role, int instance_idx) : role_(role), instance_idx_(instance_idx) {}

Postoffice::~Postoffice() = default;

void Postoffice::EnsureVan() {
  if (!van_) van_.reset(new Van(this));
}

void Postoffice::Start(int customer_id, bool do_barrier) {
  if (started_) return;
  auto* env = Environment::Get();
  num_workers_ = env->GetInt("DMLC_NUM_WORKER", 0);
  num_servers_ = env->GetInt("DMLC_NUM_SERVER", 0);
  node_id_ = kEmptyNodeID;
  {
    std::lock_guard<std::mutex> lk(mu_);
    server_key_ranges_.clear();
  }
  EnsureVan();
  started_ = true;
  van_->Start(customer_id);
  // a recovered node must not wait on the start barrier: the cluster is
  // already past it (ps-lite is_recovery behavior)
  if (van_->my_node().is_recovery) do_barrier = false;
  if (do_barrier) {
    Barrier(customer_id, kScheduler | kServerGroup | kWorkerGroup);
  }
  // Staged hipIpc bootstrap: exactly ONE instance cluster-wide imports
  // its peers' pools at a time, from this (app) thread. Concurrent
  // cross-process hipIpcOpenMemHandle — and imports issued from the
  // data-plane recv threads — deadlock inside the ROCm runtime
  // (measured; see gpu_plane.cc). Every server/worker instance walks the
  // same barrier ladder so the sequencing is global; CPU-only instances
  // just pass through the barriers.
  if (do_barrier && !is_scheduler()) {
    for (int id : GetNodeIDs(kServerGroup | kWorkerGroup)) {
      if (id == node_id_ && van_->plane()) van_->plane()->ImportPeers();
      Barrier(customer_id, kServerGroup | kWorkerGroup);
    }
  }
}

void Postoffice::Finalize(int customer_id, bool do_barrier) {
  if (!started_) return;
  if (do_barrier) {
    Barrier(customer_id, kScheduler | kServerGroup | kWorkerGroup);
  }
  van_->Stop();
  van_.reset();  // next Start builds a fresh Van
  started_ = false;
  if (exit_cb_) exit_cb_();
}

int Postoffice::my_rank() const { return IDtoRank(node_id_); }

std::vector<int> Postoffice::GetNodeIDs(int group) const {
  std::vector<int> ids;
  if (group & kScheduler) ids.push_back(kScheduler);
  if (group & kServerGroup) {
    for (int r = 0; r < num_servers_; ++r) ids.push_back(ServerRankToID(r));
  }
  if (group & kWorkerGroup) {
    for (int r = 0; r < num_workers_; ++r) ids.push_back(WorkerRankToID(r));
  }
  return ids;
}

const std::vector<Range>& Postoffice::GetServerKeyRanges() {
  std::lock_guard<std::mutex> lk(mu_);
  if (server_key_ranges_.empty()) {
    int n = std::max(num_servers_, 1);
    uint64_t step = kMaxKey / n;
    for (int i = 0; i < n; ++i) {
      Range r;
      r.begin = step * i;
      r.end = (i == n - 1) ? kMaxKey : step * (i + 1);
      server_key_ranges_.push_back(r);
    }
  }
  return server_key_ranges_;
}

void Postoffice::AddCustomer(Customer* c) {
  std::lock_guard<std::mutex> lk(mu_);
  int key = (c->app_id() << 16) | (c->customer_id() & 0xFFFF);
  XPS_CHECK_EQ(customers_.count(key), 0u)
      << "duplicate customer app=" << c->app_id() << " id=" << c->customer_id();
  customers_[key] = c;
  customer_cv_.notify_all();
}

void Postoffice::RemoveCustomer(Customer* c) {
  std::lock_guard<std::mutex> lk(mu_);
  customers_.erase((c->app_id() << 16) | (c->customer_id() & 0xFFFF));
}

Customer* Postoffice::GetCustomer(int app_id, int customer_id, int timeout_sec) const {
  int key = (app_id << 16) | (customer_id & 0xFFFF);
  std::unique_lock<std::mutex> lk(mu_);
  bool ok = customer_cv_.wait_for(lk, std::chrono::seconds(timeout_sec),
                                  [&] { return customers_.count(key) > 0; });
  return ok ? customers_.at(key) : nullptr;
}

void Postoffice::Barrier(int customer_id, int group) {
  XPS_CHECK(van_->IsReady());
  int token;
  {
    std::lock_guard<std::mutex> lk(barrier_mu_);
    barrier_done_ = false;
    token = ++barrier_seq_;
    barrier_token_ = token;
    barrier_group_ = group;
  }
  Message req;
  req.meta.control.cmd = Control::BARRIER;
  req.meta.control.barrier_group = group;
  req.meta.request = true;
  req.meta.recver = kScheduler;
  req.meta.app_id = 0;
  req.meta.customer_id = customer_id;
  req.meta.timestamp = token;  // echoed by the scheduler; matched in Manage
  // NOTE: the lock must NOT be held across Send — on the scheduler the
  // loopback delivery can complete the barrier inline on this thread and
  // re-enter Manage (which takes barrier_mu_)
  van_->Send(req);
  std::unique_lock<std::mutex> lk(barrier_mu_);
  barrier_cv_.wait(lk, [this] { return barrier_done_; });
}

void Postoffice::Manage(const Message& msg) {
  if (msg.meta.control.cmd == Control::BARRIER && !msg.meta.request) {
    std::lock_guard<std::mutex> lk(barrier_mu_);
    // a stale or duplicated response (e.g. a resend whose original was
    // slow) must not release a LATER Barrier() call: the scheduler
    // echoes the group and this node's per-call token; both must match
    // the outstanding call
    if (msg.meta.control.barrier_group != barrier_group_ ||
        msg.meta.timestamp != barrier_token_) {
      XPS_VLOG(2) << "ignoring stale barrier response (group "
                  << msg.meta.control.barrier_group << " token " << msg.meta.timestamp << ")";
      return;
    }
    barrier_done_ = true;
    barrier_cv_.notify_all();
  }
}

void Postoffice::UpdateHeartbeat(int node_id, time_t t) {
  std::lock_guard<std::mutex> lk(heartbeat_mu_);
  heartbeats_[node_id] = t;
}

std::vector<int> Postoffice::GetDeadNodes(int timeout_sec) {
  std::vector<int> dead;
  if (!van_->IsReady() || timeout_sec == 0) return dead;
  time_t now = time(nullptr);
  std::lock_guard<std::mutex> lk(heartbeat_mu_);
  for (int id : GetNodeIDs(kWorkerGroup | kServerGroup)) {
    auto it = heartbeats_.find(id);
    if ((it == heartbeats_.end() || it->second + timeout_sec < now) &&
        start_time_ + timeout_sec < now) {
      dead.push_back(id);
    }
  }
  return dead;
}

// ------------------------------------------------------------- registry
namespace {
std::mutex reg_mu;
std::unique_ptr<Postoffice> reg_scheduler;
std::vector<std::unique_ptr<Postoffice>> reg_servers;
std::vector<std::unique_ptr<Postoffice>> reg_workers;
}  // namespace

Postoffice* Postoffice::GetScheduler() {
  std::lock_guard<std::mutex> lk(reg_mu);
  if (!reg_scheduler) reg_scheduler.reset(new Postoffice(Node::SCHEDULER));
  return reg_scheduler.get();
}

Postoffice* Postoffice::GetServer(int idx) {
  std::lock_guard<std::mutex> lk(reg_mu);
  while (static_cast<int>(reg_servers.size()) <= idx) {
    reg_servers.emplace_back(new Postoffice(Node::SERVER, reg_servers.size()));
  }
  return reg_servers[idx].get();
}

Postoffice* Postoffice::GetWorker(int idx) {
  std::lock_guard<std::mutex> lk(reg_mu);
  while (static_cast<int>(reg_workers.size()) <= idx) {
    reg_workers.emplace_back(new Postoffice(Node::WORKER, reg_workers.size()));
  }
  return reg_workers[idx].get();
}

Postoffice* Postoffice::FindByNodeId(int id) {
  std::lock_guard<std::mutex> lk(reg_mu);
  auto match = [id](Postoffice* po) {
    return po && po->started_ && po->node_id_ == id && po->van_ && po->van_->IsReady();
  };
  if (match(reg_scheduler.get())) return reg_scheduler.get();
  for (auto& p : reg_servers) {
    if (match(p.get())) return p.get();
  }
  for (auto& p : reg_workers) {
    if (match(p.get())) return p.get();
  }
  return nullptr;
}

void Postoffice::ClearRegistry() {
  // Postoffice instances are process-permanent: live app objects
  // (Customers) may still point at them, and instances are restartable
  // (Start after Finalize builds a fresh Van). Deliberately leak instead
  // of destroying.
  std::lock_guard<std::mutex> lk(reg_mu);
  if (reg_scheduler) reg_scheduler.release();
  for (auto& p : reg_servers) p.release();
  for (auto& p : reg_workers) p.release();
  reg_servers.clear();
  reg_workers.clear();
}

}  // namespace xps
