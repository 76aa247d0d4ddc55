#include "gpu_plane.h"

#include <fcntl.h>
#include <hip/hip_runtime.h>
#include <semaphore.h>
#include <sys/prctl.h>
#include <time.h>
#include <unistd.h>

#include <cstring>

#include "hip_pool.h"
#include "host_par.h"
#include "host_pool.h"
#include "kernels.h"
#include "postoffice.h"
#include "wire.h"

namespace xps {

static const size_t kInlineMax = 4096;  // host blobs above this ride the shm pool

#define XPS_HIP_CHECK(cmd)                                                            \
  do {                                                                                \
    hipError_t e_ = (cmd);                                                            \
    XPS_CHECK(e_ == hipSuccess) << "HIP error: " << hipGetErrorString(e_) << " in " #cmd; \
  } while (0)

// count of blobs received BY REFERENCE (pool offset, no copy) — the
// zero-copy assertion hook (reference test_benchmark.cc:169-181 checks
// pointer equality against registered buffers; here the transport
// itself reports it)
std::atomic<uint64_t> g_zero_copy_recv{0};

namespace {
// process-wide dedup of hipIpcOpenMemHandle (two vans of a joint process
// and multiple planes share peer pool mappings; never closed — pools are
// process-lifetime)
std::mutex g_map_mu;
struct HandleKey {
  char h[kIpcHandleBytes];
  bool operator==(const HandleKey& o) const { return memcmp(h, o.h, kIpcHandleBytes) == 0; }
};
struct HandleHash {
  size_t operator()(const HandleKey& k) const {
    size_t v = 1469598103934665603ull;
    for (char c : k.h) {
      v ^= static_cast<unsigned char>(c);
      v *= 1099511628211ull;
    }
    return v;
  }
};
std::unordered_map<HandleKey, void*, HandleHash> g_mapped;

// Host-wide lock around hipIpcOpenMemHandle. Two processes importing
// each other's pools CONCURRENTLY deadlock inside the HIP runtime
// (measured on ROCm 7.x: the import handshake needs the exporter's
// runtime lock, held by its own in-flight import — cross-process AB-BA).
// Serializing imports host-wide breaks the cycle.
class IpcOpenLock {
 public:
  IpcOpenLock() {
    sem_ = sem_open("/xps_ipc_open_lock", O_CREAT, 0600, 1);
  }
  void Lock() {
    if (sem_ == SEM_FAILED) return;
    struct timespec ts;
    clock_gettime(CLOCK_REALTIME, &ts);
    ts.tv_sec += 60;
    if (sem_timedwait(sem_, &ts) != 0) {
      XPS_LOG(Warning) << "ipc-open lock timeout; proceeding unlocked";
      timed_out_ = true;
    }
  }
  void Unlock() {
    if (sem_ != SEM_FAILED && !timed_out_) sem_post(sem_);
    timed_out_ = false;
  }

 private:
  sem_t* sem_ = SEM_FAILED;
  bool timed_out_ = false;
};
IpcOpenLock g_ipc_lock;
}  // namespace

GpuPlane::GpuPlane(Postoffice* po, int device) : po_(po), device_(device) {
  my_host_hash_ = HostHash();
  inline_deliver_ = Environment::Get()->GetInt("XPS_INLINE_HANDLER", 1) != 0;
  lanes_ = Environment::Get()->GetInt("XPS_STREAMS_PER_PEER", 2);
  if (lanes_ < 1) lanes_ = 1;
  if (lanes_ > 2) lanes_ = 2;
}

GpuPlane::~GpuPlane() { Stop(); }

void GpuPlane::FillSelf(Node* self) {
  // reap segments left by crashed/killed runs before creating ours
  static std::once_flag gc_once;
  std::call_once(gc_once, [] { HostShmPool::GcStaleSegments(); });
  self->dev_id = device_;
  auto* pool = HbmPool::Get();
  if (device_ >= 0 && pool->initialized()) {
    self->pool_capacity = pool->capacity();
    self->pool_slab_bytes = pool->slab_bytes();
    self->pool_handles.resize(pool->slab_count());
    for (size_t i = 0; i < pool->slab_count(); ++i) {
      memcpy(self->pool_handles[i].data(), pool->slab_handle(i), kIpcHandleBytes);
    }
  }
  // host-shm arena: zero-copy HOST payloads for same-host peers (one per
  // process; both nodes of a joint process advertise the same uid)
  auto* hpool = HostShmPool::Get();
  hpool->Init(self->shm_uid);
  self->host_pool_uid = hpool->uid();
  self->host_pool_capacity = hpool->capacity();
  if (!started_) {
    auto* env = Environment::Get();
    if (!env->GetInt("XPS_PROBE_NO_RING", 0)) {
      XPS_CHECK(in_ring_.Create(self->shm_uid)) << "cannot create shm ring";
    }
    started_ = true;
    if (!env->GetInt("XPS_PROBE_NO_POLL", 0)) {
      poll_running_ = true;
      poll_thread_ = std::thread([this] { RingPollLoop(); });
    }
    if (device_ >= 0 && !env->GetInt("XPS_PROBE_NO_COMP", 0)) {
      comp_thread_ = std::thread([this] { CompletionLoop(); });
    }
  }
}

void GpuPlane::Stop() {
  if (stop_.exchange(true)) return;
  PrintStageStats(device_ >= 0 ? "gpu plane" : "host plane");
  if (poll_thread_.joinable()) poll_thread_.join();
  if (comp_thread_.joinable()) comp_thread_.join();
  // deliver any same-process messages the poll thread didn't get to
  // (customers are still alive: the van stops after its plane)
  {
    std::lock_guard<std::mutex> lk(local_mu_);
    for (auto& lm : local_q_) po_->van()->Deliver(std::move(lm.first));
    local_q_.clear();
  }
  // drain pending sends synchronously so responses are not lost on stop
  {
    std::lock_guard<std::mutex> lk(pend_mu_);
    for (auto& kv : pending_) {
      for (auto& p : kv.second) {
        (void)hipEventSynchronize(p.ev);
        if (p.local_po) {
          DeliverLocal(p.local_po, p.resend, p.bytes);
          (void)hipEventDestroy(p.ev);
          continue;
        }
        Peer* peer = GetPeer(p.peer_id);
        auto ring = peer ? RingOf(peer) : nullptr;
        bool sent = ring && ring->Push(p.payload.data(),
                                       static_cast<uint32_t>(p.payload.size()));
        if (!sent) {
          // van TCP conns are still open (plane stops first): best-effort
          (void)po_->van()->SendOverTcp(p.resend, p.peer_id);
        }
        (void)hipEventDestroy(p.ev);
      }
    }
    pending_.clear();
  }
  in_ring_.CloseAndUnlink();
  HostShmPool::Get()->Unlink();  // idempotent; mappings stay valid
  std::lock_guard<std::mutex> lk(ev_mu_);
  for (auto ev : event_pool_) (void)hipEventDestroy(ev);
  event_pool_.clear();
}

void GpuPlane::OnPeer(const Node& peer) {
  if (peer.host_hash != my_host_hash_) return;
  std::unique_lock<std::shared_timed_mutex> lk(peers_mu_);
  auto& p = peers_[peer.id];
  if (!p) p.reset(new Peer());
  if (p->node.shm_uid != 0 && peer.shm_uid != 0 && p->node.shm_uid != peer.shm_uid) {
    // RECOVERY: the id now names a NEW process. Every cached resource
    // of the old one is stale — the ring segment has no consumer, the
    // slab mappings point at a dead process's pool (writing through
    // them would corrupt whatever reused that memory), and local_po may
    // name the wrong instance. Reset; the next send re-opens/re-imports
    // lazily. The old ring object stays alive via shared_ptr for any
    // in-flight push.
    std::lock_guard<std::mutex> lk2(p->mu);
    XPS_LOG(Warning) << "peer " << peer.id << " was recovered (new process); resetting "
                        "plane caches for it";
    p->ring_ok.store(false, std::memory_order_release);
    p->ring_tried = false;
    std::atomic_store_explicit(&p->ring, std::shared_ptr<ShmRing>(),
                               std::memory_order_release);
    p->pool_tried = false;
    p->slab_bases.clear();
    p->local_po.store(nullptr, std::memory_order_release);
  }
  p->node = peer;
}

void GpuPlane::ImportPeers() {
  // collect ids first (don't hold peers_mu_ across the import)
  std::vector<int> ids;
  {
    std::shared_lock<std::shared_timed_mutex> lk(peers_mu_);
    for (auto& kv : peers_) ids.push_back(kv.first);
  }
  for (int id : ids) {
    Peer* p = GetPeer(id);
    if (p->node.pool_capacity) {
      bool ok = ImportPeerSlabs(p);
      XPS_VLOG(1) << "imported pool of peer " << id << " (" << p->node.pool_handles.size()
                  << " slabs): " << (ok ? "ok" : "FAILED");
    }
  }
}

GpuPlane::Peer* GpuPlane::GetPeer(int id) {
  {
    std::shared_lock<std::shared_timed_mutex> lk(peers_mu_);
    auto it = peers_.find(id);
    if (it != peers_.end()) return it->second.get();
  }
  std::unique_lock<std::shared_timed_mutex> lk(peers_mu_);
  auto& p = peers_[id];
  if (!p) {
    p.reset(new Peer());
    p->node = po_->van()->GetNode(id);
  }
  return p.get();
}

bool GpuPlane::EnsureRing(Peer* p) {
  if (p->ring_ok.load(std::memory_order_acquire)) return true;
  std::lock_guard<std::mutex> lk(p->mu);
  {
    auto cur = std::atomic_load_explicit(&p->ring, std::memory_order_acquire);
    if (cur && cur->ok()) return true;
  }
  if (p->ring_tried) return false;
  p->ring_tried = true;
  if (p->node.shm_uid == 0) return false;
  auto ring = std::make_shared<ShmRing>();
  if (!ring->Open(p->node.shm_uid)) return false;
  std::atomic_store_explicit(&p->ring, std::shared_ptr<ShmRing>(std::move(ring)),
                             std::memory_order_release);
  p->ring_ok.store(true, std::memory_order_release);
  return true;
}

std::shared_ptr<ShmRing> GpuPlane::RingOf(Peer* p) {
  if (!EnsureRing(p)) return nullptr;
  // lock-free snapshot: the recovery path swaps the pointer with
  // atomic_store, so the hot send path never takes p->mu
  return std::atomic_load_explicit(&p->ring, std::memory_order_acquire);
}

bool GpuPlane::ImportPeerSlabs(Peer* p) {
  {
    std::lock_guard<std::mutex> lk(p->mu);
    if (p->pool_tried) return !p->slab_bases.empty();
    p->pool_tried = true;
  }
  std::vector<void*> bases;
  auto* pool = HbmPool::Get();
  size_t nslabs = p->node.pool_handles.size();
  bool same_process =
      pool->initialized() && nslabs == pool->slab_count() &&
      memcmp(p->node.pool_handles[0].data(), pool->slab_handle(0), kIpcHandleBytes) == 0;
  // Cross-DEVICE import: our kernels will load/store the mapping over
  // xGMI, which needs peer access from our device to the exporter's.
  // Enable it eagerly at bootstrap — a lazy failure would surface as a
  // memory fault inside a steady-state kernel. peer dev_id is the peer
  // PROCESS's device index; it translates directly when all ranks see
  // the same HIP_VISIBLE_DEVICES ordering (the one-proc-per-GPU layout).
  int peer_dev = p->node.dev_id;
  if (!same_process && peer_dev >= 0 && peer_dev != device_ &&
      Environment::Get()->GetInt("XPS_PEER_ACCESS_CHECK", 1)) {
    int ndev = 0;
    (void)hipGetDeviceCount(&ndev);
    if (peer_dev < ndev) {
      int can = 0;
      (void)hipDeviceCanAccessPeer(&can, device_, peer_dev);
      if (!can) {
        XPS_LOG(Warning) << "device " << device_ << " cannot peer-access device " << peer_dev
                         << " (peer node " << p->node.id
                         << "): keeping this peer on the TCP path";
        std::lock_guard<std::mutex> lk(p->mu);
        p->slab_bases.clear();
        return false;
      }
      XPS_HIP_CHECK(hipSetDevice(device_));
      hipError_t pe = hipDeviceEnablePeerAccess(peer_dev, 0);
      if (pe != hipSuccess && pe != hipErrorPeerAccessAlreadyEnabled) {
        XPS_LOG(Warning) << "hipDeviceEnablePeerAccess(" << peer_dev
                         << ") failed: " << hipGetErrorString(pe)
                         << " (continuing; hipIpc lazy enable may still cover it)";
      }
    }
  }
  for (size_t i = 0; i < nslabs; ++i) {
    void* base = nullptr;
    if (same_process) {
      base = pool->slab_base(i);  // joint process: use the local mapping
    } else {
      HandleKey key;
      memcpy(key.h, p->node.pool_handles[i].data(), kIpcHandleBytes);
      std::lock_guard<std::mutex> lk(g_map_mu);
      auto it = g_mapped.find(key);
      if (it != g_mapped.end()) {
        base = it->second;
      } else {
        XPS_HIP_CHECK(hipSetDevice(device_));
        hipIpcMemHandle_t h;
        memcpy(&h, p->node.pool_handles[i].data(), sizeof(h));
        XPS_VLOG(3) << "ipc-open slab " << i << " of peer " << p->node.id;
        g_ipc_lock.Lock();
        hipError_t e = hipIpcOpenMemHandle(&base, h, hipIpcMemLazyEnablePeerAccess);
        g_ipc_lock.Unlock();
        XPS_VLOG(3) << "ipc-open slab " << i << " of peer " << p->node.id << " -> " << base;
        if (e != hipSuccess) {
          XPS_LOG(Warning) << "hipIpcOpenMemHandle(peer " << p->node.id << " slab " << i
                           << ") failed: " << hipGetErrorString(e)
                           << " — peer stays on the TCP path";
          base = nullptr;
        } else {
          // probe the mapping NOW (one 8-byte D2H read): a dead or
          // access-less mapping must fail loudly at bootstrap, not as a
          // memory fault inside a steady-state kernel
          char probe[8];
          hipError_t e2 = hipMemcpy(probe, base, sizeof(probe), hipMemcpyDeviceToHost);
          if (e2 != hipSuccess) {
            XPS_LOG(Warning) << "peer " << p->node.id << " slab " << i
                             << " mapped but unreadable (" << hipGetErrorString(e2)
                             << ") — peer stays on the TCP path";
            base = nullptr;
          }
        }
        g_mapped[key] = base;
      }
    }
    if (!base) {
      std::lock_guard<std::mutex> lk(p->mu);
      p->slab_bases.clear();
      return false;
    }
    bases.push_back(base);
  }
  std::lock_guard<std::mutex> lk(p->mu);
  p->slab_bases = std::move(bases);
  return !p->slab_bases.empty();
}

char* GpuPlane::ResolvePeer(Peer* p, uint64_t global_off, uint64_t len) {
  if (!ImportPeerSlabs(p)) return nullptr;
  uint64_t slab_bytes = p->node.pool_slab_bytes;
  if (slab_bytes == 0) return nullptr;
  size_t idx = static_cast<size_t>(global_off / slab_bytes);
  uint64_t local = global_off % slab_bytes;
  std::lock_guard<std::mutex> lk(p->mu);
  // compare against the remaining room, never local + len (which can
  // wrap for a hostile/corrupt length)
  if (idx >= p->slab_bases.size() || len > slab_bytes - local) {
    XPS_LOG(Warning) << "peer offset out of bounds: off=" << global_off << " len=" << len
                     << " slab=" << idx;
    return nullptr;
  }
  return static_cast<char*>(p->slab_bases[idx]) + local;
}

hipStream_t GpuPlane::StreamLane(int node_id, int lane) {
  if (lane >= lanes_) lane = lanes_ - 1;
  Peer* p = GetPeer(node_id);
  hipStream_t s = p->streams[lane].load(std::memory_order_acquire);
  if (s) return s;
  std::lock_guard<std::mutex> lk(p->mu);
  s = p->streams[lane].load(std::memory_order_relaxed);
  if (!s) {
    XPS_HIP_CHECK(hipSetDevice(device_));
    XPS_HIP_CHECK(hipStreamCreateWithFlags(&s, hipStreamNonBlocking));
    p->streams[lane].store(s, std::memory_order_release);
  }
  return s;
}

hipStream_t GpuPlane::StreamForPeer(int node_id) { return StreamLane(node_id, 0); }

hipStream_t GpuPlane::PullStreamForPeer(int node_id) {
  // The pull lane pays only CROSS-device: push kernels (reads from the
  // peer over xGMI) and pull copies (writes to the peer) then use the
  // link full duplex. Same-device peers share HBM, so a second lane
  // adds chain events without adding bandwidth (measured -11% on the
  // 64 MB same-GPU config) — collapse to lane 0.
  Peer* p = GetPeer(node_id);
  if (lanes_ < 2 || p->node.dev_id == device_) return StreamLane(node_id, 0);
  return StreamLane(node_id, 1);
}

std::vector<std::tuple<int, int64_t, int64_t>> GpuPlane::PeerBytes() {
  std::vector<std::tuple<int, int64_t, int64_t>> out;
  std::shared_lock<std::shared_timed_mutex> lk(peers_mu_);
  for (auto& kv : peers_) {
    int64_t tx = kv.second->tx_bytes.load(std::memory_order_relaxed);
    int64_t rx = kv.second->rx_bytes.load(std::memory_order_relaxed);
    if (tx || rx) out.emplace_back(kv.first, tx, rx);
  }
  return out;
}

hipEvent_t GpuPlane::GetEvent() {
  {
    std::lock_guard<std::mutex> lk(ev_mu_);
    if (!event_pool_.empty()) {
      hipEvent_t ev = event_pool_.back();
      event_pool_.pop_back();
      return ev;
    }
  }
  hipEvent_t ev;
  XPS_HIP_CHECK(hipSetDevice(device_));
  XPS_HIP_CHECK(hipEventCreateWithFlags(&ev, hipEventDisableTiming));
  return ev;
}

void GpuPlane::PutEvent(hipEvent_t ev) {
  std::lock_guard<std::mutex> lk(ev_mu_);
  event_pool_.push_back(ev);
}

Postoffice* GpuPlane::LocalPeer(Peer* p) {
  Postoffice* po = p->local_po.load(std::memory_order_acquire);
  if (po) return po;
  if (p->node.id == kEmptyNodeID) return nullptr;
  po = Postoffice::FindByNodeId(p->node.id);
  if (po) p->local_po.store(po, std::memory_order_release);
  return po;
}

// does this (device-vals) pull response REQUIRE the staging path? The
// worker-side merge expects host bytes unless an in-place HBM
// destination was advertised.
static bool NeedsStaging(const Message& msg) {
  return !msg.meta.request && msg.meta.pull && msg.data.size() > 1 &&
         msg.data[1].on_device() && msg.data[1].size() > 0 &&
         (!(msg.meta.option & kOptPullAddr) || (msg.meta.option & kOptHostAddr));
}

bool GpuPlane::CanSend(const Message& msg, const Node& peer) {
  if (!started_ || stop_.load()) return false;
  if (peer.host_hash != my_host_hash_ || peer.shm_uid == 0) return false;
  // same-process peer (joint mode): everything rides the direct path —
  // no ring-size limits, no pool-membership requirement (one address
  // space) — except responses that need the TCP host-staging contract
  if (LocalPeer(GetPeer(peer.id))) return !NeedsStaging(msg);
  // If the bootstrap import of this peer's pool failed, assume the
  // reverse import failed too (same mechanism, same host) and keep
  // everything on the TCP path — slow but never dropped.
  if (peer.pool_capacity) {
    Peer* p = GetPeer(peer.id);
    std::lock_guard<std::mutex> lk(p->mu);
    if (p->pool_tried && p->slab_bases.empty()) return false;
  }
  size_t est = 160 + msg.meta.body.size();
  for (size_t i = 0; i < msg.data.size(); ++i) {
    const auto& d = msg.data[i];
    if (d.on_device()) {
      uint64_t off;
      if (!HbmPool::Get()->OffsetOf(d.data(), &off)) return false;  // not our pool -> TCP
      est += 24;
      // device vals in a pull response need an HBM in-place destination
      if (!msg.meta.request && msg.meta.pull && i == 1) {
        if (!(msg.meta.option & kOptPullAddr) || (msg.meta.option & kOptHostAddr) ||
            peer.pool_capacity == 0) {
          return false;
        }
      }
    } else if (d.size() > kInlineMax) {
      uint64_t off;
      if (!HostShmPool::Get()->OffsetOf(d.data(), &off)) return false;  // big + unpooled -> TCP
      est += 24;
    } else {
      est += 24 + d.size();
    }
  }
  return est <= ShmRing::MaxPayload();
}

bool GpuPlane::Serialize(const Message& msg, const std::vector<char>& by_ref, std::string* out) {
  XPS_STAGE(serialize);
  std::string meta;
  PackMeta(msg.meta, &meta);
  out->clear();
  out->reserve(meta.size() + 64);
  ByteWriter w(out);
  w.U64(meta.size());
  w.Raw(meta.data(), meta.size());
  w.U8(static_cast<uint8_t>(msg.data.size()));
  for (size_t i = 0; i < msg.data.size(); ++i) {
    const auto& d = msg.data[i];
    if (by_ref[i] == 1) {  // HBM pool ref
      uint64_t off = 0;
      XPS_CHECK(HbmPool::Get()->OffsetOf(d.data(), &off));
      w.U8(1);
      w.U64(off);
      w.U64(d.size());
    } else if (by_ref[i] == 2) {  // host shm pool ref
      uint64_t off = 0;
      XPS_CHECK(HostShmPool::Get()->OffsetOf(d.data(), &off));
      w.U8(2);
      w.U64(off);
      w.U64(d.size());
    } else {
      w.U8(0);
      w.U64(d.size());
      w.Raw(d.data(), d.size());
    }
  }
  return out->size() <= ShmRing::MaxPayload();
}

void GpuPlane::DeliverLocal(Postoffice* lpo, Message& msg, int64_t bytes) {
  Van* van = lpo->van();
  if (!van) {
    XPS_LOG(Warning) << "local peer " << msg.meta.recver << " already finalized; dropping";
    return;
  }
  auto* rplane = dynamic_cast<GpuPlane*>(van->plane());
  // In-handler sends must be queued to the receiver's poll thread:
  // delivering them inline would nest two customers' handle_mu_ in the
  // opposite order of the worker-callback -> request chain (see
  // customer.cc). Requests deliver inline by default (lowest latency —
  // measured better than queue-pipelining for single-flow traffic);
  // XPS_LOCAL_QUEUE_REQ=1 opts into the queued/pipelined variant.
  static const bool queue_req = Environment::Get()->GetInt("XPS_LOCAL_QUEUE_REQ", 0) != 0;
  if (InCustomerHandler() || (queue_req && msg.meta.request)) {
    if (rplane && rplane->poll_running_) {
      rplane->EnqueueLocal(std::move(msg), bytes);
      return;
    }
    XPS_CHECK(!InCustomerHandler())
        << "in-handler local send needs the receiver's poll thread";
  }
  van->recv_bytes_ += bytes;
  // blobs handed over by reference in one address space = zero-copy
  // reception (the transport-level analog of the reference's
  // registered-buffer pointer-equality assertion)
  for (auto& d : msg.data) {
    if (d.size() && (d.on_device() || d.size() > kInlineMax)) {
      g_zero_copy_recv.fetch_add(1, std::memory_order_relaxed);
    }
  }
  if (inline_deliver_) {
    van->DeliverInline(msg);
  } else {
    van->Deliver(std::move(msg));
  }
}

bool GpuPlane::LocalPullRead(int peer_id, void* dst, uint64_t entry_off, size_t len,
                             Message resp) {
  if (device_ < 0 || !started_ || stop_.load()) return false;
  Peer* p = GetPeer(peer_id);
  char* src = ResolvePeer(p, entry_off, len);
  if (!src) return false;
  XPS_STAGE(local_pull_read);
  // lane 0 of the stream our entry pushes used: for a same-process peer
  // that is the RECEIVER plane's stream for us (also the stream its
  // handler kernels run on), else our own per-peer stream
  hipStream_t stream;
  if (Postoffice* elpo = LocalPeer(p)) {
    auto* rplane = dynamic_cast<GpuPlane*>(elpo->van() ? elpo->van()->plane() : nullptr);
    if (!rplane) return false;
    // ordering relies on the preceding LOCAL push's kernel being
    // enqueued on this same stream BEFORE we enqueue the read — true
    // only with synchronous inline request delivery (the default). With
    // queued delivery the push launch races us: fall back to a normal
    // pull request, which the seq gate orders.
    static const bool queue_req = Environment::Get()->GetInt("XPS_LOCAL_QUEUE_REQ", 0) != 0;
    if (!inline_deliver_ || queue_req) return false;
    stream = rplane->StreamForPeer(po_->node_id());
  } else {
    stream = StreamForPeer(peer_id);
  }
  XPS_HIP_CHECK(hipSetDevice(device_));
  kern::DenseAssign(dst, src, len, stream);
  g_zero_copy_recv.fetch_add(1, std::memory_order_relaxed);
  hipEvent_t ev = GetEvent();
  XPS_HIP_CHECK(hipEventRecord(ev, stream));
  int64_t bytes = static_cast<int64_t>(len) + 64;
  {
    std::lock_guard<std::mutex> lk(pend_mu_);
    pending_[peer_id].push_back(Pending{ev, peer_id, std::string(), std::move(resp), Message(),
                                        bytes, po_});
  }
  pending_count_.fetch_add(1);
  p->rx_bytes.fetch_add(bytes, std::memory_order_relaxed);
  return true;
}

void GpuPlane::EnqueueLocal(Message msg, int64_t bytes) {
  {
    std::lock_guard<std::mutex> lk(local_mu_);
    local_q_.emplace_back(std::move(msg), bytes);
  }
  local_count_.fetch_add(1, std::memory_order_release);
}

// Same-process fast path (joint mode: worker + co-located server share
// the process). This is the reference's IPCTransport/shared_node_mapping_
// locality taken to its limit: no serialization, no ring hop, no poll
// thread — a direct call, with GPU completion still gated by the lane
// event for responses (the requester's buffers must not be released
// before the kernels that read/write them finish).
int64_t GpuPlane::SendLocal(Message& msg, Peer* p, Postoffice* lpo) {
  XPS_STAGE(plane_send_local);
  bool response = !msg.meta.request;
  int64_t bytes = 64;
  for (auto& d : msg.data) bytes += static_cast<int64_t>(d.size());

  // ---- pull response, HOST vals + advertised HOST destination: copy
  // into the requester's buffer NOW (the synchronous-handler contract —
  // the store may mutate right after Response returns), meta-only after
  if (response && msg.meta.pull && msg.data.size() > 1 && !msg.data[1].on_device() &&
      msg.data[1].size() > 0 && (msg.meta.option & kOptPullAddr) &&
      (msg.meta.option & kOptHostAddr)) {
    void* base = HostShmPool::MapPeer(p->node.host_pool_uid, p->node.host_pool_capacity);
    SArray<char> vals = msg.data[1];
    if (base && msg.meta.addr <= p->node.host_pool_capacity &&
        vals.size() <= p->node.host_pool_capacity - msg.meta.addr) {
      HostPar::CopyBytes(static_cast<char*>(base) + msg.meta.addr, vals.data(), vals.size());
      g_zero_copy_recv.fetch_add(1, std::memory_order_relaxed);
      Message meta_msg;
      meta_msg.meta = msg.meta;
      meta_msg.meta.option |= kOptInPlace;
      meta_msg.meta.val_len = static_cast<int64_t>(vals.size());
      meta_msg.meta.data_type.clear();
      for (size_t i = 0; i < msg.data.size(); ++i) {
        if (i == 1 || msg.data[i].on_device()) continue;
        SArray<char> copy(msg.data[i].size());  // snapshot small keys/lens too
        memcpy(copy.data(), msg.data[i].data(), msg.data[i].size());
        meta_msg.data.push_back(copy);
        meta_msg.meta.data_type.push_back(msg.meta.data_type[i]);
      }
      DeliverLocal(lpo, meta_msg, bytes);
      p->tx_bytes.fetch_add(bytes, std::memory_order_relaxed);
      return bytes;
    }
  }

  // device-vals pull response with an advertised HBM destination: run
  // the in-place write now, deliver the meta once the copy completes
  if (response && msg.meta.pull && msg.data.size() > 1 && msg.data[1].on_device() &&
      (msg.meta.option & kOptPullAddr) && !(msg.meta.option & kOptHostAddr)) {
    SArray<char> vals = msg.data[1];
    char* dst = ResolvePeer(p, msg.meta.addr, vals.size());
    if (!dst) {
      for (int l = 0; l < lanes_; ++l) (void)hipStreamSynchronize(StreamLane(p->node.id, l));
      return -1;  // TCP fallback (streams drained first)
    }
    hipStream_t stream = (msg.meta.option & kOptPullLane) ? PullStreamForPeer(p->node.id)
                                                          : StreamForPeer(p->node.id);
    XPS_HIP_CHECK(hipSetDevice(device_));
    kern::DenseAssign(dst, vals.data(), vals.size(), stream);
    g_zero_copy_recv.fetch_add(1, std::memory_order_relaxed);
    Message meta_msg;
    meta_msg.meta = msg.meta;
    meta_msg.meta.option |= kOptInPlace;
    meta_msg.meta.val_len = static_cast<int64_t>(vals.size());
    meta_msg.meta.data_type.clear();
    for (size_t i = 0; i < msg.data.size(); ++i) {
      if (i == 1 || msg.data[i].on_device()) continue;
      meta_msg.data.push_back(msg.data[i]);
      meta_msg.meta.data_type.push_back(msg.meta.data_type[i]);
    }
    msg.data.clear();
    Message keepalive;
    keepalive.data.push_back(vals);
    hipEvent_t ev = GetEvent();
    XPS_HIP_CHECK(hipEventRecord(ev, stream));
    {
      std::lock_guard<std::mutex> lk(pend_mu_);
      pending_[p->node.id].push_back(Pending{ev, p->node.id, std::string(),
                                             std::move(meta_msg), std::move(keepalive), bytes,
                                             lpo});
    }
    pending_count_.fetch_add(1);
    p->tx_bytes.fetch_add(bytes, std::memory_order_relaxed);
    return bytes;
  }

  // ack of a one-sided push: the handler launched nothing for it — no
  // event to wait, deliver right away
  bool kernel_less_ack = response && msg.meta.push && !msg.meta.pull &&
                         (msg.meta.option & kOptInPlace);
  if (response && device_ >= 0 && !kernel_less_ack) {
    // GPU handler output: deliver once this peer's lane drained
    XPS_STAGE(local_defer);
    hipStream_t stream = (msg.meta.option & kOptPullLane) ? PullStreamForPeer(p->node.id)
                                                          : StreamForPeer(p->node.id);
    hipEvent_t ev = GetEvent();
    XPS_HIP_CHECK(hipSetDevice(device_));
    XPS_HIP_CHECK(hipEventRecord(ev, stream));
    Message resend = msg;
    {
      std::lock_guard<std::mutex> lk(pend_mu_);
      pending_[p->node.id].push_back(Pending{ev, p->node.id, std::string(), std::move(resend),
                                             Message(), bytes, lpo});
    }
    pending_count_.fetch_add(1);
  } else {
    Message copy = msg;
    if (response) {
      // synchronous (host) handler responses may be VIEWS of mutable
      // server state — the old wire paths serialized them before the
      // handler could mutate again; snapshot host blobs to keep that
      // contract (requests need no copy: the sender's buffers are
      // pinned until the response arrives)
      for (auto& d : copy.data) {
        if (!d.on_device() && d.size()) {
          SArray<char> snap(d.size());
          memcpy(snap.data(), d.data(), d.size());
          d = snap;
        }
      }
    }
    DeliverLocal(lpo, copy, bytes);
  }
  p->tx_bytes.fetch_add(bytes, std::memory_order_relaxed);
  return bytes;
}

int64_t GpuPlane::Send(Message& msg, const Node& peer_node) {
  XPS_STAGE(plane_send);
  XPS_VLOG(3) << "plane send -> " << peer_node.id << ": " << msg.DebugString();
  Peer* p = GetPeer(peer_node.id);
  {
    std::lock_guard<std::mutex> lk(p->mu);
    if (p->node.shm_uid == 0) p->node = peer_node;
  }

  // ---- one-sided assign push (reference rdma steady state: unsignaled
  // RDMA_WRITE of vals + write-with-imm of meta, rdma_transport.h:341-
  // 356): the worker writes the server's advertised store entry with
  // ITS OWN kernel and sends the meta-only notification once the write
  // completes. The server's push handling shrinks to an ack.
  static const bool local_one_sided =
      Environment::Get()->GetInt("XPS_LOCAL_ONE_SIDED", 0) != 0;
  if (msg.meta.request && msg.meta.push && !msg.meta.pull &&
      (msg.meta.option & kOptEntryPush) && device_ >= 0 && msg.data.size() > 1 &&
      msg.data[1].on_device()) {
    // Cross-process this removes the server-side launch entirely
    // (2012 -> 2603 GB/s on the 2-joint-procs-1-GPU config). For a
    // SAME-process peer it is measured SLOWER (61 vs 84 GB/s per-key
    // 1 MB, RTT 50 vs 42 us, same-box A/B: the app thread becomes the
    // single launch lane) — default off locally; XPS_LOCAL_ONE_SIDED=1
    // re-enables it (the write then runs on the RECEIVER plane's stream
    // for us, so ordering stays stream-based).
    Postoffice* elpo = LocalPeer(p);
    GpuPlane* rplane =
        elpo ? dynamic_cast<GpuPlane*>(elpo->van() ? elpo->van()->plane() : nullptr) : nullptr;
    bool can = elpo ? (local_one_sided && rplane != nullptr) : EnsureRing(p);
    SArray<char> vals = msg.data[1];
    char* dst = can ? ResolvePeer(p, msg.meta.addr, vals.size()) : nullptr;
    if (dst) {
      XPS_STAGE(entry_push);
      hipStream_t stream = elpo ? rplane->StreamForPeer(po_->node_id())
                                : StreamForPeer(p->node.id);
      XPS_HIP_CHECK(hipSetDevice(device_));
      kern::DenseAssign(dst, vals.data(), vals.size(), stream);
      g_zero_copy_recv.fetch_add(1, std::memory_order_relaxed);
      Message meta_msg;
      meta_msg.meta = msg.meta;
      meta_msg.meta.option |= kOptInPlace;
      meta_msg.meta.val_len = static_cast<int64_t>(vals.size());
      meta_msg.meta.data_type.clear();
      for (size_t i = 0; i < msg.data.size(); ++i) {
        if (i == 1 || msg.data[i].on_device()) continue;
        meta_msg.data.push_back(msg.data[i]);
        meta_msg.meta.data_type.push_back(msg.meta.data_type[i]);
      }
      int64_t bytes = 64 + static_cast<int64_t>(vals.size());
      Message keepalive;
      keepalive.data.push_back(vals);
      std::string payload;
      if (!elpo) {
        std::vector<char> br(meta_msg.data.size(), 0);
        XPS_CHECK(Serialize(meta_msg, br, &payload));
      }
      hipEvent_t ev = GetEvent();
      XPS_HIP_CHECK(hipEventRecord(ev, stream));
      {
        std::lock_guard<std::mutex> lk(pend_mu_);
        pending_[p->node.id].push_back(Pending{ev, p->node.id, std::move(payload),
                                               std::move(meta_msg), std::move(keepalive),
                                               bytes, elpo});
      }
      pending_count_.fetch_add(1);
      p->tx_bytes.fetch_add(bytes, std::memory_order_relaxed);
      return bytes;
    }
    msg.meta.option &= ~kOptEntryPush;  // entry unmapped: normal path
  }

  if (Postoffice* lpo = LocalPeer(p)) return SendLocal(msg, p, lpo);
  bool response = !msg.meta.request;
  auto sync_peer_lanes = [this, &peer_node] {
    for (int l = 0; l < lanes_; ++l) (void)hipStreamSynchronize(StreamLane(peer_node.id, l));
  };
  if (!EnsureRing(p)) {
    // TCP fallback for a response must not outrun handler kernels still
    // running on this peer's streams (the worker may reuse buffers on ack)
    if (response && device_ >= 0) sync_peer_lanes();
    return -1;
  }

  // ---- pull response with HOST vals + host destination: memcpy now --
  if (response && msg.meta.pull && msg.data.size() > 1 && !msg.data[1].on_device() &&
      msg.data[1].size() > 0 && (msg.meta.option & kOptPullAddr) &&
      (msg.meta.option & kOptHostAddr)) {
    void* base = HostShmPool::MapPeer(p->node.host_pool_uid, p->node.host_pool_capacity);
    SArray<char> vals = msg.data[1];
    if (base && msg.meta.addr <= p->node.host_pool_capacity &&
        vals.size() <= p->node.host_pool_capacity - msg.meta.addr) {
      HostPar::CopyBytes(static_cast<char*>(base) + msg.meta.addr, vals.data(), vals.size());
      Message meta_msg;
      meta_msg.meta = msg.meta;
      meta_msg.meta.option |= kOptInPlace;
      meta_msg.meta.val_len = static_cast<int64_t>(vals.size());
      for (size_t i = 0; i < msg.data.size(); ++i) {
        if (i == 1 || msg.data[i].on_device() || msg.data[i].size() > kInlineMax) continue;
        meta_msg.data.push_back(msg.data[i]);
        meta_msg.meta.data_type.push_back(msg.meta.data_type[i]);
      }
      std::vector<char> br(meta_msg.data.size(), 0);
      std::string payload;
      if (!Serialize(meta_msg, br, &payload)) {
        meta_msg.data.clear();
        meta_msg.meta.data_type.clear();
        XPS_CHECK(Serialize(meta_msg, {}, &payload));
      }
      int64_t bytes = static_cast<int64_t>(vals.size() + payload.size());
      auto ring = RingOf(p);
      if (!ring || !ring->Push(payload.data(), static_cast<uint32_t>(payload.size()))) return -1;
      p->tx_bytes.fetch_add(bytes, std::memory_order_relaxed);
      return bytes;
    }
    return -1;  // cannot map the destination: TCP fallback
  }

  // ---- pull response with device vals: one-sided xGMI write ----------
  if (response && msg.meta.pull && msg.data.size() > 1 && msg.data[1].on_device() &&
      (msg.meta.option & kOptPullAddr)) {
    SArray<char> vals = msg.data[1];
    char* dst = ResolvePeer(p, msg.meta.addr, vals.size());
    if (!dst) {
      sync_peer_lanes();
      return -1;  // TCP fallback (streams drained first)
    }
    // pull lane (only when the handler prepared its ordering there): the
    // response copy overlaps the next push kernel on lane 0
    hipStream_t stream = (msg.meta.option & kOptPullLane) ? PullStreamForPeer(peer_node.id)
                                                          : StreamForPeer(peer_node.id);
    XPS_HIP_CHECK(hipSetDevice(device_));
    // copy KERNEL instead of hipMemcpyAsync: measured 5.2 vs 4.85 TB/s
    // same-device, and it reads/writes hipIpc-mapped peer memory alike
    kern::DenseAssign(dst, vals.data(), vals.size(), stream);
    Message meta_msg;
    meta_msg.meta = msg.meta;
    meta_msg.meta.option |= kOptInPlace;
    meta_msg.meta.val_len = static_cast<int64_t>(vals.size());
    meta_msg.meta.data_type.clear();
    // keep host-side keys/lens blobs for the merge bookkeeping
    for (size_t i = 0; i < msg.data.size(); ++i) {
      if (i == 1 || msg.data[i].on_device()) continue;
      meta_msg.data.push_back(msg.data[i]);
      meta_msg.meta.data_type.push_back(msg.meta.data_type[i]);
    }
    std::vector<char> by_ref(meta_msg.data.size(), 0);
    std::string payload;
    if (!Serialize(meta_msg, by_ref, &payload)) {
      meta_msg.data.clear();
      meta_msg.meta.data_type.clear();
      XPS_CHECK(Serialize(meta_msg, {}, &payload));
    }
    int64_t bytes = static_cast<int64_t>(vals.size() + payload.size());
    msg.data.clear();
    // the original msg owns vals; keep it alive until the event fires.
    // meta_msg (kOptInPlace, host blobs only) is the TCP-resendable form:
    // by the time a fallback happens the event has fired, i.e. the
    // in-place write already landed in the peer's pool.
    Message keepalive;
    keepalive.data.push_back(vals);
    DeferSendInternal(p, peer_node.id, stream, std::move(meta_msg), std::move(keepalive),
                      std::move(payload), bytes);
    p->tx_bytes.fetch_add(bytes, std::memory_order_relaxed);
    return bytes;
  }

  // ---- general path: inline blobs + by-ref device/host-pool blobs ----
  std::vector<char> by_ref(msg.data.size(), 0);
  int64_t ref_bytes = 0;
  for (size_t i = 0; i < msg.data.size(); ++i) {
    if (msg.data[i].on_device()) {
      by_ref[i] = 1;
      ref_bytes += msg.data[i].size();
      if (i == 1) msg.meta.option |= kOptValsByRef;
    } else if (msg.data[i].size() > kInlineMax) {
      by_ref[i] = 2;  // CanSend verified host-pool membership
      ref_bytes += msg.data[i].size();
    }
  }
  std::string payload;
  if (!Serialize(msg, by_ref, &payload)) {
    if (response && device_ >= 0) (void)hipStreamSynchronize(StreamForPeer(peer_node.id));
    return -1;
  }
  int64_t bytes = static_cast<int64_t>(payload.size()) + ref_bytes;
  // ack of a one-sided push: the handler launched nothing — no event
  bool kernel_less_ack = response && msg.meta.push && !msg.meta.pull &&
                         (msg.meta.option & kOptInPlace);
  if (response && device_ >= 0 && !kernel_less_ack) {
    // order behind any handler kernels on this peer's lane (the handler
    // flags pull-lane work via kOptPullLane; everything else is lane 0).
    // The resend copy doubles as the keepalive (shallow: SArrays shared).
    hipStream_t stream = (msg.meta.option & kOptPullLane) ? PullStreamForPeer(peer_node.id)
                                                          : StreamForPeer(peer_node.id);
    Message resend = msg;
    DeferSendInternal(p, peer_node.id, stream, std::move(resend), Message(), std::move(payload),
                      bytes);
  } else {
    // requests — and every send of a host-only plane — go out now (host
    // responses were produced synchronously; nothing to wait for)
    auto ring = RingOf(p);
    if (!ring || !ring->Push(payload.data(), static_cast<uint32_t>(payload.size()))) return -1;
    XPS_VLOG(3) << "plane send done -> " << peer_node.id;
  }
  p->tx_bytes.fetch_add(bytes, std::memory_order_relaxed);
  return bytes;
}

void GpuPlane::DeferSendInternal(Peer* p, int peer_id, hipStream_t stream, Message resend,
                                 Message keepalive, std::string payload, int64_t bytes) {
  XPS_STAGE(defer_queue);
  XPS_CHECK_GE(device_, 0) << "deferred sends are a GPU-plane feature";
  hipEvent_t ev = GetEvent();
  XPS_HIP_CHECK(hipSetDevice(device_));
  XPS_HIP_CHECK(hipEventRecord(ev, stream));
  {
    std::lock_guard<std::mutex> lk(pend_mu_);
    pending_[peer_id].push_back(Pending{ev, peer_id, std::move(payload), std::move(resend),
                                        std::move(keepalive), bytes});
  }
  pending_count_.fetch_add(1);
  XPS_VLOG(3) << "deferred send queued -> " << peer_id;
}

void GpuPlane::CompletionLoop() {
  XPS_HIP_CHECK(hipSetDevice(device_));
  // 1 µs timer slack: the default 50 µs slack turns the idle usleep into
  // a ~70 µs latency floor per hop (dominates small-message RTT)
  prctl(PR_SET_TIMERSLACK, 1000);
  const int kSpin = Environment::Get()->GetInt("XPS_POLL_SPIN", 200000);
  int idle = 0;
  // NOTE on locking: pend_mu_ is held only for deque push/pop/front
  // snapshots, never across hipEventQuery or ring pushes — holding it
  // through the poll loop put ~6 µs of lock contention into EVERY
  // deferred response (measured via XPS_TIMING, round 2). This thread is
  // the only popper, so a front observed ready stays front until we pop.
  std::vector<int> ids;
  while (!stop_.load()) {
    bool did = false;
    if (pending_count_.load(std::memory_order_acquire) > 0) {
      ids.clear();
      {
        std::lock_guard<std::mutex> lk(pend_mu_);
        for (auto& kv : pending_) {
          if (!kv.second.empty()) ids.push_back(kv.first);
        }
      }
      for (int id : ids) {
        while (!stop_.load()) {
          hipEvent_t front_ev = nullptr;
          {
            std::lock_guard<std::mutex> lk(pend_mu_);
            auto& dq = pending_[id];
            if (dq.empty()) break;
            front_ev = dq.front().ev;
          }
          hipError_t e = hipEventQuery(front_ev);
          if (e == hipErrorNotReady) break;
          XPS_STAGE(defer_release);
          XPS_CHECK(e == hipSuccess) << "hipEventQuery: " << hipGetErrorString(e);
          Pending front;
          {
            std::lock_guard<std::mutex> lk(pend_mu_);
            auto& dq = pending_[id];
            front = std::move(dq.front());
            dq.pop_front();
          }
          pending_count_.fetch_sub(1);
          if (front.local_po) {  // same-process: direct delivery
            DeliverLocal(front.local_po, front.resend, front.bytes);
            PutEvent(front.ev);
            did = true;
            continue;
          }
          Peer* peer = GetPeer(front.peer_id);
          auto ring = peer ? RingOf(peer) : nullptr;
          bool sent = ring && ring->Push(front.payload.data(),
                                         static_cast<uint32_t>(front.payload.size()));
          if (sent) {
            XPS_VLOG(3) << "deferred send done -> " << front.peer_id;
          } else {
            // never drop: the requester is blocked in Wait on this
            // response — deliver it over the TCP path instead
            XPS_LOG(Warning) << "deferred plane send failed -> " << front.peer_id
                             << "; falling back to TCP";
            if (po_->van()->SendOverTcp(front.resend, front.peer_id) < 0) {
              XPS_LOG(Warning) << "TCP fallback to " << front.peer_id
                               << " failed too (peer dead?)";
            }
          }
          PutEvent(front.ev);
          did = true;
        }
      }
    }
    if (did) {
      idle = 0;
    } else if (++idle > kSpin) {
      usleep(20);
    }
  }
}

void GpuPlane::RingPollLoop() {
  if (device_ >= 0) XPS_HIP_CHECK(hipSetDevice(device_));
  prctl(PR_SET_TIMERSLACK, 1000);
  const int kSpin = Environment::Get()->GetInt("XPS_POLL_SPIN", 200000);
  std::vector<char> buf(ShmRing::MaxPayload());
  int idle = 0;
  std::deque<std::pair<Message, int64_t>> local;
  while (!stop_.load()) {
    // same-process deliveries first (they need no parsing)
    if (local_count_.load(std::memory_order_acquire) > 0) {
      {
        std::lock_guard<std::mutex> lk(local_mu_);
        local.swap(local_q_);
      }
      local_count_.fetch_sub(static_cast<int>(local.size()));
      for (auto& lm : local) {
        XPS_STAGE(local_handle);
        po_->van()->recv_bytes_ += lm.second;
        for (auto& d : lm.first.data) {
          if (d.size() && (d.on_device() || d.size() > kInlineMax)) {
            g_zero_copy_recv.fetch_add(1, std::memory_order_relaxed);
          }
        }
        if (inline_deliver_) {
          po_->van()->DeliverInline(lm.first);
        } else {
          po_->van()->Deliver(std::move(lm.first));
        }
      }
      idle = 0;
      local.clear();
    }
    uint32_t n = in_ring_.Pop(buf.data());
    if (n == 0) {
      if (++idle > kSpin) usleep(20);
      continue;
    }
    idle = 0;
    XPS_STAGE(ring_handle);
    ByteReader r(buf.data(), n);
    uint64_t meta_len = r.U64();
    Message msg;
    {
      std::vector<char> meta(meta_len);
      r.Raw(meta.data(), meta_len);
      UnpackMeta(meta.data(), meta_len, &msg.meta);
    }
    int nblobs = r.U8();
    int64_t ref_bytes = 0;
    bool ok = true;
    for (int i = 0; i < nblobs; ++i) {
      int kind = r.U8();
      if (kind == 0) {
        uint64_t len = r.U64();
        SArray<char> d(len);
        r.Raw(d.data(), len);
        msg.data.push_back(d);
      } else if (kind == 1) {
        uint64_t off = r.U64();
        uint64_t len = r.U64();
        Peer* sender = GetPeer(msg.meta.sender);
        char* ptr = sender ? ResolvePeer(sender, off, len) : nullptr;
        if (!ptr) {
          XPS_LOG(Warning) << "dropping by-ref blob: sender pool not mapped (from "
                           << msg.meta.sender << ")";
          ok = false;
          break;
        }
        msg.data.push_back(SArray<char>(ptr, len, device_));
        ref_bytes += len;
        g_zero_copy_recv.fetch_add(1, std::memory_order_relaxed);
      } else {  // kind 2: host shm pool
        uint64_t off = r.U64();
        uint64_t len = r.U64();
        Peer* sender = GetPeer(msg.meta.sender);
        void* base = sender ? HostShmPool::MapPeer(sender->node.host_pool_uid,
                                                   sender->node.host_pool_capacity)
                            : nullptr;
        if (!base || off > sender->node.host_pool_capacity ||
            len > sender->node.host_pool_capacity - off) {
          XPS_LOG(Warning) << "dropping host-ref blob: sender host pool not mapped (from "
                           << msg.meta.sender << ")";
          ok = false;
          break;
        }
        msg.data.push_back(SArray<char>(static_cast<char*>(base) + off, len, kCPU));
        ref_bytes += len;
        g_zero_copy_recv.fetch_add(1, std::memory_order_relaxed);
      }
    }
    if (!ok) continue;
    XPS_VLOG(3) << "ring recv: " << msg.DebugString();
    po_->van()->recv_bytes_ += n + ref_bytes;
    if (msg.meta.sender != kEmptyNodeID) {
      GetPeer(msg.meta.sender)->rx_bytes.fetch_add(n + ref_bytes, std::memory_order_relaxed);
    }
    if (inline_deliver_) {
      po_->van()->DeliverInline(msg);
    } else {
      po_->van()->Deliver(std::move(msg));
    }
  }
}

std::shared_ptr<DataPlane> CreateGpuPlane(Postoffice* po, int device) {
  // device < 0 builds the host-only variant (shm rings + host pool)
  return std::make_shared<GpuPlane>(po, device);
}

}  // namespace xps
