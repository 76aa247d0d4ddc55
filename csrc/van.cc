#include <tuple>

#include "van.h"

#include <unistd.h>

#include <algorithm>
#include <chrono>
#include <random>
#include <set>

#include "hip_util.h"
#include "postoffice.h"
#include "resender.h"
#include "wire.h"

namespace xps {

Van::Van(Postoffice* po) : po_(po) {}

Van::~Van() { Stop(); }

void Van::Start(int customer_id) {
  auto* env = Environment::Get();
  scheduler_.role = Node::SCHEDULER;
  scheduler_.id = kScheduler;
  scheduler_.hostname = env->GetStr("DMLC_PS_ROOT_URI", "127.0.0.1");
  scheduler_.port = env->GetInt("DMLC_PS_ROOT_PORT", 9000);

  my_node_.role = po_->role();
  my_node_.hostname = LocalIP();
  my_node_.host_hash = HostHash();
  my_node_.aux_id = po_->preferred_rank() >= 0 ? po_->preferred_rank()
                                               : env->GetInt("DMLC_RANK", -1);
  std::random_device rd;
  my_uid_ = (static_cast<uint64_t>(rd()) << 32) ^ rd() ^ (static_cast<uint64_t>(getpid()) << 16);
  my_node_.shm_uid = my_uid_;
  if (plane_) plane_->FillSelf(&my_node_);

  drop_rate_ = env->GetInt("PS_DROP_MSG", 0);
  heartbeat_interval_ = env->GetInt("PS_HEARTBEAT_INTERVAL", 0);
  if (env->GetInt("PS_RESEND", 0)) {
    resender_.reset(new Resender(env->GetInt("PS_RESEND_TIMEOUT", 1000), 10, this));
  }
  if (env->GetInt("ENABLE_PROFILING", 0)) {
    std::string path = env->GetStr("PROFILE_PATH", "");
    if (path.empty()) {
      path = std::string("xps_profile_van_") +
             (po_->is_worker() ? "worker" : po_->is_server() ? "server" : "sched") + "_" +
             std::to_string(getpid());
    } else {
      path += "_" + std::to_string(getpid());
    }
    trace_file_ = fopen(path.c_str(), "w");
  }

  int want_port = 0;
  if (po_->is_scheduler()) {
    want_port = scheduler_.port;
    my_node_ = scheduler_;
    my_node_.host_hash = HostHash();
    my_node_.shm_uid = my_uid_;
    if (plane_) plane_->FillSelf(&my_node_);
  } else if (env->GetInt("DMLC_PORT", 0) && po_->instance_idx() == 0) {
    want_port = env->GetInt("DMLC_PORT", 0);
  }
  int bound = listener_.Bind(want_port, po_->is_scheduler() ? 1 : 40);
  XPS_CHECK_GE(bound, 0) << "failed to bind port " << want_port << " role "
                         << Node::RoleStr(my_node_.role);
  my_node_.port = bound;
  listener_.StartAccepting([this](int fd) { OnNewConnection(fd); });

  XPS_VLOG(1) << "van up: " << my_node_.DebugString();

  if (po_->is_scheduler()) {
    po_->set_node_id(kScheduler);
    {
      std::lock_guard<std::mutex> lk(nodes_mu_);
      nodes_[kScheduler] = my_node_;
    }
    if (po_->num_workers() + po_->num_servers() == 0) ready_ = true;
  } else {
    // dial the scheduler and register
    int fd = TcpConnect(scheduler_.hostname, scheduler_.port);
    XPS_CHECK_GE(fd, 0) << "cannot reach scheduler at " << scheduler_.hostname << ":"
                        << scheduler_.port;
    auto conn = std::make_shared<TcpConn>(fd);
    {
      std::lock_guard<std::mutex> lk(conn_mu_);
      conns_[kScheduler] = conn;
      recv_threads_.emplace_back([this, conn] { RecvLoop(conn); });
    }
    {
      std::lock_guard<std::mutex> lk(nodes_mu_);
      nodes_[kScheduler] = scheduler_;
    }
    Message req;
    req.meta.control.cmd = Control::ADD_NODE;
    req.meta.request = true;
    req.meta.sender = kEmptyNodeID;
    req.meta.recver = kScheduler;
    req.meta.control.node.push_back(my_node_);
    std::string meta;
    PackMeta(req.meta, &meta);
    conn->SendFrame(meta, {});
  }

  // wait until the node list is assigned / all nodes joined
  auto deadline = std::chrono::steady_clock::now() + std::chrono::seconds(
      Environment::Get()->GetInt("XPS_BOOTSTRAP_TIMEOUT", 300));
  while (!ready_.load()) {
    XPS_CHECK(std::chrono::steady_clock::now() < deadline)
        << "bootstrap timeout: " << my_node_.DebugString();
    usleep(2000);
  }

  if (heartbeat_interval_ > 0 && !po_->is_scheduler()) {
    heartbeat_thread_ = std::thread([this] { HeartbeatLoop(); });
  }
}

void Van::Stop() {
  if (stopping_.exchange(true)) return;
  if (plane_) plane_->Stop();
  listener_.Stop();
  // swap the thread list out under conn_mu_: GetOrDial/OnNewConnection
  // check stopping_ under the same lock before spawning, so no thread
  // can be added after the swap (joining while a concurrent dialer
  // emplaces into the same vector would be a data race)
  std::vector<std::thread> threads;
  {
    std::lock_guard<std::mutex> lk(conn_mu_);
    for (auto& kv : conns_) kv.second->Close();
    threads.swap(recv_threads_);
  }
  for (auto& t : threads) {
    if (t.joinable()) t.join();
  }
  if (heartbeat_thread_.joinable()) heartbeat_thread_.join();
  resender_.reset();
  if (trace_file_) {
    fclose(trace_file_);
    trace_file_ = nullptr;
  }
}

Node Van::GetNode(int id) {
  std::lock_guard<std::mutex> lk(nodes_mu_);
  auto it = nodes_.find(id);
  XPS_CHECK(it != nodes_.end()) << "unknown node id " << id;
  return it->second;
}

int64_t Van::Send(Message& msg) {
  if (msg.meta.sender == kEmptyNodeID) msg.meta.sender = my_node_.id;
  int recver = msg.meta.recver;
  XPS_CHECK_NE(recver, kEmptyNodeID) << "message has no recver";
  if (recver >= 8 || recver == kScheduler) return SendToNode(msg, recver);
  // group mask
  int64_t total = 0;
  for (int id : po_->GetNodeIDs(recver)) {
    Message copy = msg;  // shallow: SArray payloads shared
    copy.meta.recver = id;
    int64_t n = SendToNode(copy, id);
    if (n >= 0) total += n;
  }
  return total;
}

int64_t Van::SendToNode(Message& msg, int id) {
  MaybeTrace(msg, /*recv=*/false);
  if (id != my_node_.id && msg.meta.control.empty() && msg.meta.seq == 0) {
    std::lock_guard<std::mutex> lk(order_mu_);
    msg.meta.seq = ++send_seq_[id];
  }
  if (id == my_node_.id) {
    // loopback without touching the wire
    Message copy = msg;
    int64_t approx = static_cast<int64_t>(msg.meta.val_len);
    send_bytes_ += approx;
    Deliver(std::move(copy));
    return approx;
  }
  if (plane_ && msg.meta.control.empty()) {
    Node peer;
    {
      std::lock_guard<std::mutex> lk(nodes_mu_);
      auto it = nodes_.find(id);
      if (it != nodes_.end()) peer = it->second;
    }
    if (peer.id != kEmptyNodeID && plane_->CanSend(msg, peer)) {
      int64_t n = plane_->Send(msg, peer);
      if (n >= 0) {
        send_bytes_ += n;
        return n;
      }
    }
  }
  return SendOverTcp(msg, id);
}

int64_t Van::SendOverTcp(Message& msg, int id) {
  // retransmission covers the TCP path only: shm/xGMI plane messages are
  // lossless by construction and their recv path sends no ACKs
  if (resender_ && msg.meta.control.empty()) resender_->AddOutgoing(msg);
  auto conn = GetOrDial(id);
  if (!conn) {
    XPS_LOG(Warning) << "no route to node " << id << " from " << my_node_.id;
    return -1;
  }
  std::string meta;
  PackMeta(msg.meta, &meta);
  std::vector<SArray<char>> host_data;
  host_data.reserve(msg.data.size());
  for (auto& d : msg.data) {
    host_data.push_back(d.on_device() ? gpu::StageToHost(d) : d);
  }
  int64_t n = conn->SendFrame(meta, host_data);
  if (n > 0) send_bytes_ += n;
  return n;
}

std::shared_ptr<TcpConn> Van::GetOrDial(int id) {
  {
    std::lock_guard<std::mutex> lk(conn_mu_);
    auto it = conns_.find(id);
    if (it != conns_.end()) return it->second;
  }
  Node n;
  {
    std::lock_guard<std::mutex> lk(nodes_mu_);
    auto it = nodes_.find(id);
    if (it == nodes_.end()) return nullptr;
    n = it->second;
  }
  int fd = TcpConnect(n.hostname, n.port, 20);
  if (fd < 0) return nullptr;
  auto conn = std::make_shared<TcpConn>(fd);
  {
    std::lock_guard<std::mutex> lk(conn_mu_);
    if (stopping_.load()) {  // Stop already joined the recv threads
      conn->Close();
      return nullptr;
    }
    auto it = conns_.find(id);
    if (it != conns_.end()) {
      conn->Close();
      return it->second;  // raced with another dialer
    }
    conns_[id] = conn;
    recv_threads_.emplace_back([this, conn] { RecvLoop(conn); });
  }
  // identify ourselves so the peer can reply on this connection
  Message hs;
  hs.meta.control.cmd = Control::HANDSHAKE;
  hs.meta.sender = my_node_.id;
  hs.meta.recver = id;
  std::string meta;
  PackMeta(hs.meta, &meta);
  conn->SendFrame(meta, {});
  return conn;
}

void Van::OnNewConnection(int fd) {
  auto conn = std::make_shared<TcpConn>(fd);
  std::lock_guard<std::mutex> lk(conn_mu_);
  if (stopping_.load()) {
    conn->Close();
    return;
  }
  recv_threads_.emplace_back([this, conn] { RecvLoop(conn); });
}

void Van::RecvLoop(std::shared_ptr<TcpConn> conn) {
  while (true) {
    std::string meta;
    std::vector<SArray<char>> data;
    int64_t n = conn->RecvFrame(&meta, &data);
    if (n < 0) {
      if (!stopping_.load()) XPS_VLOG(2) << "connection closed on " << my_node_.id;
      // forget any route that used this connection
      std::lock_guard<std::mutex> lk(conn_mu_);
      for (auto it = conns_.begin(); it != conns_.end();) {
        if (it->second == conn) {
          it = conns_.erase(it);
        } else {
          ++it;
        }
      }
      return;
    }
    recv_bytes_ += n;
    Message msg;
    UnpackMeta(meta.data(), meta.size(), &msg.meta);
    msg.data = std::move(data);
    // learn (or refresh) the reply route for this connection
    if (msg.meta.sender != kEmptyNodeID) {
      std::lock_guard<std::mutex> lk(conn_mu_);
      auto it = conns_.find(msg.meta.sender);
      if (it == conns_.end() || it->second->closed()) conns_[msg.meta.sender] = conn;
    }
    if (!msg.meta.control.empty()) {
      ProcessControl(msg, conn);
    } else {
      MaybeTrace(msg, /*recv=*/true);
      if (drop_rate_ > 0 && msg.meta.request) {
        // drop-test BEFORE the resender ACK so dropped messages get resent
        std::lock_guard<std::mutex> lk(sched_mu_);
        if (static_cast<int>(drop_rng_() % 100) < drop_rate_) {
          XPS_VLOG(2) << "drop-test: dropping " << msg.DebugString();
          continue;
        }
      }
      if (resender_ && resender_->AddIncoming(msg)) continue;  // duplicate
      DeliverData(msg);
    }
  }
}

void Van::Deliver(Message msg) {
  if (!msg.meta.control.empty()) {
    ProcessControl(msg, nullptr);
  } else {
    DeliverData(msg);
  }
}

// single execution lane for sequenced flows: handler runs on the
// DELIVERING thread (plane poll thread or TCP recv thread), serialized
// per customer by handle_mu_ — a queued lane for one transport and an
// inline lane for the other would reorder right after the gate
void Van::InlineData(Message& msg) {
  int app_id = msg.meta.app_id;
  int customer_id = po_->is_worker() ? msg.meta.customer_id : app_id;
  Customer* c = po_->GetCustomer(app_id, customer_id, 15);
  XPS_CHECK(c) << "no customer (app=" << app_id << ", customer=" << customer_id
               << ") on node " << my_node_.id << " for " << msg.DebugString();
  c->ProcessInline(msg);
}

void Van::GatedDeliver(Message& msg) {
  uint64_t seq = msg.meta.seq;
  Flow* f;
  {
    std::lock_guard<std::mutex> lk(order_mu_);
    auto& slot = recv_flows_[msg.meta.sender];
    if (!slot) slot.reset(new Flow());
    f = slot.get();
  }
  // f->mu is held ACROSS handler execution: release order and execution
  // order are then the same thing, even when one message arrived over
  // the shm ring (inline poll thread) and its predecessor over the TCP
  // fallback (recv thread)
  std::unique_lock<std::mutex> lk(f->mu);
  if (seq < f->expected) {
    // late duplicate (resend whose original arrived): deliver — the
    // resender's signature dedup upstream already filtered true dups
    InlineData(msg);
    return;
  }
  if (f->held.empty()) f->hold_since = std::chrono::steady_clock::now();
  f->held.emplace(seq, std::move(msg));
  while (!f->held.empty() && f->held.begin()->first == f->expected) {
    Message m = std::move(f->held.begin()->second);
    f->held.erase(f->held.begin());
    f->expected++;
    InlineData(m);
    f->hold_since = std::chrono::steady_clock::now();
  }
  if (f->held.empty()) return;
  // a later message overtook its predecessor on the other transport and
  // the gap has not filled. Liveness guard: without a resender a lost
  // message would stall the flow forever — flush in order after a
  // timeout (ordering is then best-effort, matching a lossy fault).
  int timeout_ms = Environment::Get()->GetInt("XPS_ORDER_TIMEOUT_MS", 10000);
  if (std::chrono::steady_clock::now() - f->hold_since >
      std::chrono::milliseconds(timeout_ms)) {
    XPS_LOG(Warning) << "sequence gap from node "
                     << f->held.begin()->second.meta.sender << " (expected " << f->expected
                     << ") unfilled for " << timeout_ms << " ms; flushing out of order";
    while (!f->held.empty()) {
      Message m = std::move(f->held.begin()->second);
      uint64_t s2 = f->held.begin()->first;
      f->held.erase(f->held.begin());
      f->expected = s2 + 1;
      InlineData(m);
    }
  }
}

void Van::ResetFlow(int peer_id) {
  Flow* f = nullptr;
  {
    std::lock_guard<std::mutex> lk(order_mu_);
    send_seq_.erase(peer_id);
    auto it = recv_flows_.find(peer_id);
    if (it != recv_flows_.end()) f = it->second.get();
  }
  if (f) {
    // reset in place: erasing would destroy a mutex another thread may
    // be holding mid-delivery
    std::lock_guard<std::mutex> lk(f->mu);
    f->expected = 1;
    f->held.clear();
  }
}

void Van::DeliverData(Message& msg) {
  if (msg.meta.seq != 0) {
    GatedDeliver(msg);
    return;
  }
  DeliverDataNow(msg);
}

void Van::DeliverDataNow(Message& msg) {
  int app_id = msg.meta.app_id;
  int customer_id = po_->is_worker() ? msg.meta.customer_id : app_id;
  Customer* c = po_->GetCustomer(app_id, customer_id, 15);
  XPS_CHECK(c) << "no customer (app=" << app_id << ", customer=" << customer_id << ") on node "
               << my_node_.id << " for " << msg.DebugString();
  c->Accept(std::move(msg));
}

void Van::DeliverInline(Message& msg) {
  if (!msg.meta.control.empty()) {
    ProcessControl(msg, nullptr);
    return;
  }
  if (msg.meta.seq != 0) {
    GatedDeliver(msg);
    return;
  }
  int app_id = msg.meta.app_id;
  int customer_id = po_->is_worker() ? msg.meta.customer_id : app_id;
  Customer* c = po_->GetCustomer(app_id, customer_id, 15);
  XPS_CHECK(c) << "no customer (app=" << app_id << ", customer=" << customer_id
               << ") on node " << my_node_.id << " for " << msg.DebugString();
  c->ProcessInline(msg);
}

void Van::ProcessControl(Message& msg, const std::shared_ptr<TcpConn>& conn) {
  auto& ctrl = msg.meta.control;
  switch (ctrl.cmd) {
    case Control::HANDSHAKE:
      break;  // route registration already happened in RecvLoop
    case Control::ADD_NODE:
      if (po_->is_scheduler() && msg.meta.request) {
        ProcessAddNodeAtScheduler(msg, conn);
      } else {
        ProcessNodeListAssigned(msg);
      }
      break;
    case Control::BARRIER:
      if (msg.meta.request) {
        ProcessBarrierAtScheduler(msg);
      } else {
        po_->Manage(msg);
      }
      break;
    case Control::HEARTBEAT:
      ProcessHeartbeat(msg);
      break;
    case Control::ACK:
      if (resender_) resender_->HandleAck(ctrl.msg_sig);
      break;
    case Control::TERMINATE:
      stopping_ = true;
      break;
    default:
      XPS_LOG(Warning) << "unknown control " << ctrl.cmd;
  }
}

std::function<bool(const Node&, const Node&)> NodeRankOrder(const std::vector<Node>& batch) {
  auto* env = Environment::Get();
  std::map<std::string, int> host_idx;  // BYTEPS_ORDERED_HOSTS position
  {
    std::string hosts = env->GetStr("BYTEPS_ORDERED_HOSTS", "");
    int idx = 0;
    size_t start = 0;
    while (start <= hosts.size() && !hosts.empty()) {
      size_t comma = hosts.find(',', start);
      std::string h = hosts.substr(start, comma == std::string::npos ? comma : comma - start);
      if (!h.empty()) host_idx[h] = idx++;
      if (comma == std::string::npos) break;
      start = comma + 1;
    }
  }
  bool mixed = env->GetInt("BYTEPS_ENABLE_MIXED_MODE", 0) != 0;
  std::set<std::string> worker_hosts;
  if (mixed) {
    for (auto& n : batch)
      if (n.role == Node::WORKER) worker_hosts.insert(n.hostname);
  }
  auto key = [host_idx, mixed, worker_hosts](const Node& n) {
    auto it = host_idx.find(n.hostname);
    int order = it == host_idx.end() ? (1 << 30) : it->second;
    int coloc = 0;
    if (mixed && n.role == Node::SERVER) coloc = worker_hosts.count(n.hostname) ? 1 : 0;
    return std::make_tuple(order, coloc, n.hostname, n.port);
  };
  return [key](const Node& a, const Node& b) { return key(a) < key(b); };
}

void Van::ProcessAddNodeAtScheduler(Message& msg, const std::shared_ptr<TcpConn>& conn) {
  if (ready_.load()) {
    // late registration = recovery: a restarted process re-joins and
    // inherits a dead node's id (ps-lite van.cc:266-320 behavior)
    ProcessRecoveryAtScheduler(msg, conn);
    return;
  }
  std::vector<std::pair<Node, std::shared_ptr<TcpConn>>> batch;
  {
    std::lock_guard<std::mutex> lk(sched_mu_);
    XPS_CHECK_EQ(msg.meta.control.node.size(), 1u);
    pending_nodes_.emplace_back(msg.meta.control.node[0], conn);
    int expected = po_->num_workers() + po_->num_servers();
    XPS_VLOG(1) << "scheduler: " << pending_nodes_.size() << "/" << expected << " nodes joined ("
                << msg.meta.control.node[0].DebugString() << ")";
    if (static_cast<int>(pending_nodes_.size()) < expected) return;
    batch.swap(pending_nodes_);
  }
  // deterministic order with the BytePS placement policies (ordered
  // hosts / mixed mode — see NodeRankOrder); recovery/aux pinning first
  {
    std::vector<Node> nodes;
    nodes.reserve(batch.size());
    for (auto& p : batch) nodes.push_back(p.first);
    auto less = NodeRankOrder(nodes);
    std::sort(batch.begin(), batch.end(),
              [&less](const auto& a, const auto& b) { return less(a.first, b.first); });
  }
  std::set<int> taken;
  auto assign = [&](Node& n) {
    int rank = n.aux_id;
    if (rank >= 0) {
      int id = n.role == Node::WORKER ? WorkerRankToID(rank) : ServerRankToID(rank);
      XPS_CHECK(!taken.count(id)) << "duplicate pinned rank " << rank;
      n.id = id;
      taken.insert(id);
      return;
    }
    for (int r = 0;; ++r) {
      int id = n.role == Node::WORKER ? WorkerRankToID(r) : ServerRankToID(r);
      if (!taken.count(id)) {
        n.id = id;
        taken.insert(id);
        return;
      }
    }
  };
  // pinned ranks first so they cannot collide with auto-assignment
  for (auto& p : batch)
    if (p.first.aux_id >= 0) assign(p.first);
  for (auto& p : batch)
    if (p.first.aux_id < 0) assign(p.first);

  Message reply;
  reply.meta.control.cmd = Control::ADD_NODE;
  reply.meta.request = false;
  reply.meta.sender = kScheduler;
  reply.meta.control.node.push_back(my_node_);
  {
    std::lock_guard<std::mutex> lk(nodes_mu_);
    for (auto& p : batch) {
      nodes_[p.first.id] = p.first;
      reply.meta.control.node.push_back(p.first);
    }
  }
  {
    std::lock_guard<std::mutex> lk(conn_mu_);
    for (auto& p : batch) conns_[p.first.id] = p.second;
  }
  if (plane_) {
    for (auto& p : batch) plane_->OnPeer(p.first);
  }
  std::string meta;
  for (auto& p : batch) {
    reply.meta.recver = p.first.id;
    PackMeta(reply.meta, &meta);
    p.second->SendFrame(meta, {});
  }
  XPS_VLOG(1) << "scheduler: assigned ids, cluster up (" << batch.size() << " nodes)";
  ready_ = true;
}

void Van::ProcessRecoveryAtScheduler(Message& msg, const std::shared_ptr<TcpConn>& conn) {
  XPS_CHECK_EQ(msg.meta.control.node.size(), 1u);
  Node node = msg.meta.control.node[0];
  int timeout = Environment::Get()->GetInt("PS_HEARTBEAT_TIMEOUT", 10);
  std::vector<int> dead = po_->GetDeadNodes(timeout);
  int assigned = kEmptyNodeID;
  for (int id : dead) {
    bool want_worker = node.role == Node::WORKER;
    if (IsWorkerID(id) == want_worker) {
      assigned = id;
      break;
    }
  }
  if (assigned == kEmptyNodeID) {
    XPS_LOG(Warning) << "late ADD_NODE but no dead " << Node::RoleStr(node.role)
                     << " to recover (" << node.DebugString() << ")";
    return;
  }
  node.id = assigned;
  node.is_recovery = 1;
  XPS_VLOG(1) << "scheduler: recovering node " << assigned << " as " << node.DebugString();
  {
    std::lock_guard<std::mutex> lk(nodes_mu_);
    nodes_[assigned] = node;
  }
  {
    std::lock_guard<std::mutex> lk(conn_mu_);
    conns_[assigned] = conn;
  }
  po_->UpdateHeartbeat(assigned, time(nullptr));
  if (plane_) plane_->OnPeer(node);
  // full node list to the recovered node; its updated record to survivors
  Message reply;
  reply.meta.control.cmd = Control::ADD_NODE;
  reply.meta.request = false;
  reply.meta.sender = kScheduler;
  reply.meta.recver = assigned;
  {
    std::lock_guard<std::mutex> lk(nodes_mu_);
    for (auto& kv : nodes_) reply.meta.control.node.push_back(kv.second);
  }
  std::string meta;
  PackMeta(reply.meta, &meta);
  conn->SendFrame(meta, {});
  Message update;
  update.meta.control.cmd = Control::ADD_NODE;
  update.meta.request = false;
  update.meta.sender = kScheduler;
  update.meta.control.node.push_back(node);
  for (int id : po_->GetNodeIDs(kServerGroup | kWorkerGroup)) {
    if (id == assigned) continue;
    update.meta.recver = id;
    Message copy = update;
    SendToNode(copy, id);
  }
}

void Van::ProcessNodeListAssigned(Message& msg) {
  bool was_ready = ready_.load();
  {
    std::lock_guard<std::mutex> lk(nodes_mu_);
    for (auto& n : msg.meta.control.node) {
      auto it = nodes_.find(n.id);
      if (it != nodes_.end() && it->second.shm_uid != 0 && n.shm_uid != 0 &&
          it->second.shm_uid != n.shm_uid) {
        // the id was re-assigned (recovery): drop the stale connection so
        // the next send redials the new address. (shm_uid==0 means a
        // pre-bootstrap stub record, NOT a re-assignment.)
        {
          std::lock_guard<std::mutex> lk2(conn_mu_);
          auto cit = conns_.find(n.id);
          if (cit != conns_.end()) {
            cit->second->Close();
            conns_.erase(cit);
          }
        }
        ResetFlow(n.id);  // the replacement node starts a fresh sequence
      }
      nodes_[n.id] = n;
      if (n.shm_uid == my_uid_ && n.role == my_node_.role) {
        my_node_ = n;
        po_->set_node_id(n.id);
      }
    }
  }
  if (plane_) {
    std::lock_guard<std::mutex> lk(nodes_mu_);
    for (auto& kv : nodes_) {
      if (kv.second.id != my_node_.id) plane_->OnPeer(kv.second);
    }
  }
  if (was_ready) {
    XPS_VLOG(1) << "node list update processed (" << msg.meta.control.node.size() << " nodes)";
    return;  // survivor receiving a recovery update
  }
  XPS_CHECK_NE(my_node_.id, kEmptyNodeID) << "node list did not contain me";
  if (my_node_.is_recovery) {
    XPS_VLOG(1) << "recovered as " << my_node_.DebugString();
    if (plane_) plane_->ImportPeers();
  } else {
    XPS_VLOG(1) << "joined as " << my_node_.DebugString();
  }
  ready_ = true;
}

void Van::ProcessBarrierAtScheduler(Message& msg) {
  XPS_CHECK(po_->is_scheduler());
  int group = msg.meta.control.barrier_group;
  std::vector<std::pair<int, int>> release;  // (node id, its call token)
  {
    std::lock_guard<std::mutex> lk(sched_mu_);
    auto& waiters = barrier_waiters_[group];
    waiters.emplace_back(msg.meta.sender, msg.meta.timestamp);
    size_t expected = po_->GetNodeIDs(group).size();
    XPS_VLOG(2) << "barrier group " << group << ": " << waiters.size() << "/" << expected;
    if (waiters.size() < expected) return;
    release.swap(waiters);
  }
  // release self LAST: waking our own Finalize first would let Van::Stop
  // close the connections before the other waiters get their responses
  std::stable_partition(release.begin(), release.end(),
                        [this](const auto& w) { return w.first != my_node_.id; });
  Message res;
  res.meta.control.cmd = Control::BARRIER;
  res.meta.request = false;
  res.meta.control.barrier_group = group;
  res.meta.sender = my_node_.id;
  for (auto& w : release) {
    res.meta.recver = w.first;
    res.meta.timestamp = w.second;  // echo the waiter's token (Manage matches it)
    Message copy = res;
    SendToNode(copy, w.first);
  }
}

void Van::ProcessHeartbeat(Message& msg) {
  if (msg.meta.request) {
    po_->UpdateHeartbeat(msg.meta.sender, time(nullptr));
    Message ack;
    ack.meta.control.cmd = Control::HEARTBEAT;
    ack.meta.request = false;
    ack.meta.recver = msg.meta.sender;
    Send(ack);
  } else {
    po_->UpdateHeartbeat(kScheduler, time(nullptr));
  }
}

void Van::HeartbeatLoop() {
  while (!stopping_.load()) {
    for (int i = 0; i < heartbeat_interval_ * 10 && !stopping_.load(); ++i) usleep(100 * 1000);
    if (stopping_.load()) break;
    Message hb;
    hb.meta.control.cmd = Control::HEARTBEAT;
    hb.meta.request = true;
    hb.meta.recver = kScheduler;
    Send(hb);
  }
}

void Van::MaybeTrace(const Message& msg, bool recv) {
  if (!trace_file_ || !msg.meta.control.empty()) return;
  // format mirrors ps-lite: key \t {role}_van_{recv,send}_{push,pull} \t usec
  auto now = std::chrono::duration_cast<std::chrono::microseconds>(
                 std::chrono::system_clock::now().time_since_epoch())
                 .count();
  std::lock_guard<std::mutex> lk(trace_mu_);
  fprintf(trace_file_, "%llu\t%s_van_%s_%s\t%lld\n",
          static_cast<unsigned long long>(msg.meta.key),
          po_->is_worker() ? "worker" : "server", recv ? "recv" : "send",
          msg.meta.push ? "push" : "pull", static_cast<long long>(now));
}

}  // namespace xps
