#include "server_handlers.h"

#include "gpu_plane.h"
#include "hip_pool.h"
#include "kernels.h"

namespace xps {

#define XPS_HIP_CHECK(cmd)                                                            \
  do {                                                                                \
    hipError_t e_ = (cmd);                                                            \
    XPS_CHECK(e_ == hipSuccess) << "HIP error: " << hipGetErrorString(e_) << " in " #cmd; \
  } while (0)

static GpuPlane* ThePlane(Postoffice* po) {
  return dynamic_cast<GpuPlane*>(po->van() ? po->van()->plane() : nullptr);
}

static hipStream_t PeerStream(Postoffice* po, int sender, hipStream_t* fallback) {
  auto* plane = ThePlane(po);
  if (plane) return plane->StreamForPeer(sender);
  if (!*fallback) {
    XPS_HIP_CHECK(hipStreamCreateWithFlags(fallback, hipStreamNonBlocking));
  }
  return *fallback;
}

// pull-response lane: copies here overlap the next push kernel on the
// peer's lane 0; same-key ordering rides the handler's event chain
static hipStream_t PeerPullStream(Postoffice* po, int sender, hipStream_t* fallback) {
  auto* plane = ThePlane(po);
  if (plane) return plane->PullStreamForPeer(sender);
  if (!*fallback) {
    XPS_HIP_CHECK(hipStreamCreateWithFlags(fallback, hipStreamNonBlocking));
  }
  return *fallback;
}

static hipEvent_t EventAlloc(Postoffice* po) {
  if (auto* plane = ThePlane(po)) return plane->GetEvent();
  hipEvent_t ev;
  XPS_HIP_CHECK(hipEventCreateWithFlags(&ev, hipEventDisableTiming));
  return ev;
}

static void EventFree(Postoffice* po, hipEvent_t ev) {
  if (auto* plane = ThePlane(po)) {
    plane->PutEvent(ev);
  } else {
    (void)hipEventDestroy(ev);
  }
}

// ------------------------------------------------------------------ dense

static EventRef MakeEventRef(Postoffice* po, hipStream_t s) {
  hipEvent_t ev = EventAlloc(po);
  XPS_HIP_CHECK(hipEventRecord(ev, s));
  return EventRef(ev, [po](hipEvent_t e) { EventFree(po, e); });
}

GpuDenseHandler::GpuDenseHandler(Postoffice* po, DenseMode mode, DenseDtype dtype)
    : po_(po), mode_(mode), dtype_(dtype) {
  XPS_CHECK(HbmPool::Get()->initialized()) << "GpuDenseHandler needs the HBM pool";
  num_workers_ = std::max(1, po_->num_workers());
  // >1 workers: same-key kernels land on different peers' streams
  chain_ = num_workers_ > 1;
}

void GpuDenseHandler::SumKernel(void* dst, const void* src, size_t nbytes, hipStream_t s) {
  if (dtype_ == DenseDtype::kBf16) {
    kern::DenseSumBf16(static_cast<uint16_t*>(dst), static_cast<const uint16_t*>(src),
                       nbytes / 2, s);
  } else {
    kern::DenseSumF32(static_cast<float*>(dst), static_cast<const float*>(src), nbytes / 4, s);
  }
}

void GpuDenseHandler::BatchedSum(const kern::CopyDesc* descs, int n, hipStream_t s) {
  if (dtype_ == DenseDtype::kBf16) {
    kern::BatchedSumBf16(descs, n, s);
  } else {
    kern::BatchedSumF32(descs, n, s);
  }
}

bool GpuDenseHandler::NeedChain(int sender) {
  if (chain_) return true;
  // lane split for this peer (cross-device): push kernels on lane 0 and
  // pull copies on the pull lane touch the same entries — chain them
  auto* plane = ThePlane(po_);
  return plane && plane->lanes() > 1 &&
         plane->PullStreamForPeer(sender) != plane->StreamForPeer(sender);
}

GpuDenseHandler::~GpuDenseHandler() = default;

void GpuDenseHandler::OrderAfter(Entry* e, hipStream_t s) {
  if (e->last_ev) XPS_HIP_CHECK(hipStreamWaitEvent(s, e->last_ev.get(), 0));
}

hipStream_t GpuDenseHandler::Stream(int sender) {
  return PeerStream(po_, sender, &fallback_stream_);
}

hipStream_t GpuDenseHandler::PullStream(int sender) {
  return PeerPullStream(po_, sender, &fallback_stream_);
}

void GpuDenseHandler::operator()(const KVMeta& req, const KVPairs<float>& kvs,
                                 KVServer<float>* server) {
  if (req.push && req.pull) {
    // fused ZPushPull: apply the push silently, then answer with the
    // post-push values (same stream -> ordered). Reduce mode holds
    // pulls for round accounting and cannot fuse them into one message.
    XPS_CHECK(mode_ != DenseMode::kReduce)
        << "ZPushPull is not supported in reduce mode (pulls are held per round)";
    HandlePush(req, kvs, server, /*respond=*/false);
    HandlePull(req, kvs, server);
    return;
  }
  if (req.push) {
    HandlePush(req, kvs, server);
  } else if (req.pull) {
    HandlePull(req, kvs, server);
  } else {
    server->Response(req);
  }
}

GpuDenseHandler::Group* GpuDenseHandler::GroupFor(const SArray<Key>& keys) {
  // FNV-1a over the key bytes identifies the key-set
  uint64_t h = 1469598103934665603ull;
  const unsigned char* p = reinterpret_cast<const unsigned char*>(keys.data());
  for (size_t i = 0; i < keys.nbytes(); ++i) {
    h ^= p[i];
    h *= 1099511628211ull;
  }
  Group* g;
  {
    std::lock_guard<std::mutex> lk(mu_);
    g = &groups_[h];
  }
  if (g->keys.empty()) {
    g->keys.assign(keys.begin(), keys.end());
    g->ents.reserve(keys.size());
    std::lock_guard<std::mutex> lk(mu_);
    for (Key k : g->keys) g->ents.push_back(&store_[k]);
  } else {
    XPS_CHECK(g->keys.size() == keys.size() &&
              std::equal(g->keys.begin(), g->keys.end(), keys.begin()))
        << "reduce mode needs a consistent key->message grouping across workers "
           "(key-set hash collision or inconsistent bucketing)";
  }
  return g;
}

void GpuDenseHandler::HandleReducePush(const KVMeta& req, const KVPairs<float>& kvs,
                                       KVServer<float>* server) {
  XPS_STAGE(reduce_push);
  size_t n = kvs.keys.size();
  Group* g = GroupFor(kvs.keys);
  if (g->pushes >= num_workers_) {
    // a fast worker started the next round before this round's pulls
    // drained: defer (the KVPairs copy keeps the remote buffer alive)
    g->waiting_pushes.emplace_back(req, kvs);
    return;
  }
  XPS_CHECK(kvs.vals.on_device()) << "reduce mode needs device vals (pool buffers)";
  if (g->lens.size() != n) {
    g->lens.resize(n);
    for (size_t i = 0; i < n; ++i) {
      g->lens[i] = kvs.lens.empty() ? kvs.vals.nbytes() / n
                                    : static_cast<size_t>(kvs.lens[i]) * sizeof(float);
    }
  }
  hipStream_t stream = Stream(req.sender);
  XPS_HIP_CHECK(hipSetDevice(HbmPool::Get()->device()));
  if (g->last_ev) XPS_HIP_CHECK(hipStreamWaitEvent(stream, g->last_ev.get(), 0));
  bool first = g->pushes == 0;
  if (first) {
    // the previous round's pull copies must finish before we overwrite
    for (auto& ev : g->pull_guard) {
      XPS_HIP_CHECK(hipStreamWaitEvent(stream, ev.get(), 0));
    }
    g->pull_guard.clear();
  }
  // one batched kernel chain for the whole key-set (3 launches for 169
  // rn50 buckets at kMaxBatch=64), one event per push
  std::vector<kern::CopyDesc> descs;
  descs.reserve(n);
  size_t off = 0;
  bool aligned = true;
  for (size_t i = 0; i < n; ++i) {
    size_t len = g->lens[i];
    Entry* e = g->ents[i];
    if (e->buf.size() < len) e->buf = HbmPool::Get()->AllocArray(len);
    descs.push_back({e->buf.data(), reinterpret_cast<const char*>(kvs.vals.data()) + off, len});
    aligned = aligned && (len % 16 == 0);
    off += len;
  }
  if (aligned) {
    if (first) {
      kern::BatchedAssign(descs.data(), static_cast<int>(n), stream);
    } else {
      BatchedSum(descs.data(), static_cast<int>(n), stream);
    }
  } else {
    for (auto& d : descs) {
      if (first) {
        kern::DenseAssign(d.dst, d.src, d.nbytes, stream);
      } else {
        SumKernel(d.dst, d.src, d.nbytes, stream);
      }
    }
  }
  g->pushes++;
  EventRef ev = MakeEventRef(po_, stream);
  g->round_events.push_back(ev);
  g->last_ev = ev;
  server->Response(req);
  if (g->pushes >= num_workers_) {
    std::vector<KVMeta> waiting;
    waiting.swap(g->waiting_pulls);
    for (auto& w : waiting) RespondPull(w, g, server);
  }
}

void GpuDenseHandler::HandlePush(const KVMeta& req, const KVPairs<float>& kvs,
                                 KVServer<float>* server, bool respond) {
  XPS_STAGE(dense_push);
  size_t n = kvs.keys.size();
  XPS_CHECK_GT(n, 0u);
  if (mode_ == DenseMode::kReduce) {
    HandleReducePush(req, kvs, server);
    return;
  }
  if (req.option & kOptInPlace) {
    // one-sided push: the worker's kernel already wrote our store entry
    // (the notification is ordered after the write's completion event);
    // nothing to launch — ack immediately
    if (respond) server->Response(req);
    return;
  }
  hipStream_t stream = Stream(req.sender);
  XPS_HIP_CHECK(hipSetDevice(HbmPool::Get()->device()));
  bool chain = NeedChain(req.sender);
  bool sum_all = mode_ == DenseMode::kAssign ? req.cmd == kCmdSum : req.cmd != kCmdAssign;
  // multi-key device push: one batched kernel launch for all segments
  if (kvs.vals.on_device() && n > 1) {
    std::vector<kern::CopyDesc> descs;
    std::vector<Entry*> ents;
    descs.reserve(n);
    ents.reserve(n);
    size_t boff = 0;
    bool aligned = true;
    for (size_t i = 0; i < n; ++i) {
      size_t len = kvs.lens.empty() ? kvs.vals.nbytes() / n
                                    : static_cast<size_t>(kvs.lens[i]) * sizeof(float);
      Entry* e;
      {
        std::lock_guard<std::mutex> lk(mu_);
        e = &store_[kvs.keys[i]];
        if (e->buf.size() < len) {
          e->buf = HbmPool::Get()->AllocArray(len);
          XPS_HIP_CHECK(hipMemset(e->buf.data(), 0, len));
        }
      }
      ents.push_back(e);
      descs.push_back({e->buf.data(),
                       reinterpret_cast<const char*>(kvs.vals.data()) + boff, len});
      aligned = aligned && (len % 16 == 0) && (boff % 16 == 0);
      boff += len;
    }
    if (aligned) {
      for (Entry* e : ents) OrderAfter(e, stream);
      if (sum_all) {
        BatchedSum(descs.data(), static_cast<int>(n), stream);
      } else {
        kern::BatchedAssign(descs.data(), static_cast<int>(n), stream);
      }
      if (chain) {
        EventRef ev = MakeEventRef(po_, stream);  // one event covers the batch
        for (Entry* e : ents) e->last_ev = ev;
      }
      auto* plane0 = po_->van() ? po_->van()->plane() : nullptr;
      if (!plane0) XPS_HIP_CHECK(hipStreamSynchronize(stream));
      if (respond) server->Response(req);
      return;
    }
  }
  size_t off = 0;  // bytes into vals
  bool synced = false;
  Entry* last_e = nullptr;
  for (size_t i = 0; i < n; ++i) {
    size_t len = kvs.lens.empty() ? kvs.vals.nbytes() / n
                                  : static_cast<size_t>(kvs.lens[i]) * sizeof(float);
    Entry* e;
    {
      std::lock_guard<std::mutex> lk(mu_);
      e = &store_[kvs.keys[i]];
      if (e->buf.size() < len) {
        // retire (never free) a smaller buffer: workers may hold its
        // offset in their one-sided entry caches — a stale write must
        // land in dead-but-owned memory, not a reused pool region
        if (!e->buf.empty()) e->retired.push_back(e->buf);
        e->buf = HbmPool::Get()->AllocArray(len);
        // zero before publishing: another peer's stream may accumulate
        // into this entry concurrently with our first push
        XPS_HIP_CHECK(hipMemset(e->buf.data(), 0, len));
      }
    }
    last_e = e;
    bool sum = sum_all;
    const char* src = reinterpret_cast<const char*>(kvs.vals.data()) + off;
    if (kvs.vals.on_device()) {
      OrderAfter(e, stream);
      {
        XPS_STAGE(push_kernel_launch);
        if (sum) {
          SumKernel(e->buf.data(), src, len, stream);
        } else {
          kern::DenseAssign(e->buf.data(), src, len, stream);
        }
      }
      if (chain) {
        XPS_STAGE(push_chain_event);
        e->last_ev = MakeEventRef(po_, stream);
      }
    } else {
      // host vals land via synchronous copies below: drain the entry's
      // outstanding cross-stream kernel first
      if (chain && e->last_ev) {
        XPS_HIP_CHECK(hipEventSynchronize(e->last_ev.get()));
        e->last_ev.reset();
      }
      // host vals (TCP-staged path): correctness-first synchronous route
      if (sum) {
        SArray<char> scratch = HbmPool::Get()->AllocArray(len);
        XPS_HIP_CHECK(hipMemcpy(scratch.data(), src, len, hipMemcpyHostToDevice));
        SumKernel(e->buf.data(), scratch.data(), len, stream);
        XPS_HIP_CHECK(hipStreamSynchronize(stream));
        synced = true;
      } else {
        XPS_HIP_CHECK(hipMemcpy(e->buf.data(), src, len, hipMemcpyHostToDevice));
      }
    }
    off += len;
  }
  auto* plane = po_->van() ? po_->van()->plane() : nullptr;
  if (!plane && !synced) XPS_HIP_CHECK(hipStreamSynchronize(stream));
  if (!respond) return;
  // assign mode, single key: advertise the store entry's pool offset so
  // this worker's next pushes of the key go one-sided (kOptEntryPush)
  if (mode_ == DenseMode::kAssign && n == 1 && last_e && plane) {
    uint64_t eoff = 0;
    if (HbmPool::Get()->OffsetOf(last_e->buf.data(), &eoff)) {
      KVMeta r = req;
      r.addr = eoff;
      r.val_len = static_cast<int64_t>(last_e->buf.size());
      r.option |= kOptEntryAddr;
      server->Response(r);
      return;
    }
  }
  server->Response(req);
}

void GpuDenseHandler::RespondPull(const KVMeta& req, Group* g, KVServer<float>* server) {
  XPS_STAGE(reduce_respond_pull);
  hipStream_t stream = PullStream(req.sender);
  for (auto& ev : g->round_events) {
    XPS_HIP_CHECK(hipStreamWaitEvent(stream, ev.get(), 0));
  }
  size_t n = g->keys.size();
  size_t total = 0;
  for (size_t len : g->lens) total += len;
  SArray<Key> keys(g->keys);
  SArray<int> lens(n);
  for (size_t i = 0; i < n; ++i) lens[i] = static_cast<int>(g->lens[i] / sizeof(float));
  bool responded = false;
  // fast path: write the whole group straight into the requester's
  // advertised pool destination (one batched kernel, meta-only response)
  if (req.option & kOptPullAddr) {
    if (auto* plane = ThePlane(po_)) {
      std::vector<kern::CopyDesc> descs;
      descs.reserve(n);
      uint64_t off = 0;
      bool ok = true;
      for (size_t i = 0; i < n && ok; ++i) {
        size_t len = g->lens[i];
        char* dst = plane->PeerDst(req.sender, req.addr + off, len);
        ok = dst != nullptr && len % 16 == 0;
        if (ok) {
          descs.push_back({dst, g->ents[i]->buf.data(), len});
          off += len;
        }
      }
      if (ok) {
        kern::BatchedAssign(descs.data(), static_cast<int>(n), stream);
        KVMeta r2 = req;
        r2.option |= kOptInPlace | kOptPullLane;
        r2.val_len = static_cast<int64_t>(total);
        KVPairs<float> res2;
        res2.keys = keys;
        res2.lens = lens;
        server->Response(r2, res2);  // plane defers the meta on `stream`
        responded = true;
      }
    }
  }
  if (!responded) {
    // staging path (TCP fallback / no advertised destination)
    KVPairs<float> res;
    res.keys = keys;
    res.lens = lens;
    if (n == 1) {
      res.vals = SArray<float>::View(g->ents[0]->buf);
    } else {
      SArray<char> tmp = HbmPool::Get()->AllocArray(total);
      std::vector<kern::CopyDesc> descs;
      descs.reserve(n);
      size_t off = 0;
      bool aligned = true;
      for (size_t i = 0; i < n; ++i) {
        descs.push_back({tmp.data() + off, g->ents[i]->buf.data(), g->lens[i]});
        aligned = aligned && (g->lens[i] % 16 == 0) && (off % 16 == 0);
        off += g->lens[i];
      }
      if (aligned) {
        kern::BatchedAssign(descs.data(), static_cast<int>(n), stream);
      } else {
        for (auto& d : descs) kern::DenseAssign(d.dst, d.src, d.nbytes, stream);
      }
      res.vals = SArray<float>::View(tmp);  // plane keeps it alive until sent
    }
    if (!(req.option & kOptPullAddr)) XPS_HIP_CHECK(hipStreamSynchronize(stream));
    KVMeta r = req;
    r.option |= kOptPullLane;  // ordering prepared on the pull lane
    server->Response(r, res);  // plane enqueues the in-place read on `stream`
  }
  EventRef pe = MakeEventRef(po_, stream);
  g->pull_guard.push_back(pe);
  g->last_ev = pe;
  g->pulled_senders.insert(req.sender);
  g->pulls++;
  if (g->pulls >= num_workers_) {
    // round over: reset, then replay deferred next-round pushes + pulls
    g->pushes = 0;
    g->pulls = 0;
    g->pulled_senders.clear();
    g->round_events.clear();  // refs drop back to the event pool
    std::vector<std::pair<KVMeta, KVPairs<float>>> dpush;
    dpush.swap(g->waiting_pushes);
    for (auto& d : dpush) HandleReducePush(d.first, d.second, server);
    std::vector<KVMeta> dpull;
    dpull.swap(g->waiting_next_pulls);
    for (auto& d : dpull) {
      if (g->pushes >= num_workers_) {
        RespondPull(d, g, server);
      } else {
        g->waiting_pulls.push_back(d);
      }
    }
  }
}

void GpuDenseHandler::HandlePull(const KVMeta& req, const KVPairs<float>& kvs,
                                 KVServer<float>* server) {
  XPS_STAGE(dense_pull);
  size_t n = kvs.keys.size();
  XPS_CHECK_GT(n, 0u);
  XPS_HIP_CHECK(hipSetDevice(HbmPool::Get()->device()));
  if (mode_ == DenseMode::kReduce) {
    // a pull may legitimately precede the round's pushes (it just
    // waits); GroupFor creates the round group on demand. NOTE: no lock
    // is held across RespondPull — its round-reset path replays deferred
    // pushes through HandleReducePush (which locks mu_ for store_
    // access); Group state itself is serialized by the handler lane.
    Group* g = GroupFor(kvs.keys);
    if (g->pulled_senders.count(req.sender)) {
      // this sender already pulled the current round: a NEXT-round pull
      g->waiting_next_pulls.push_back(req);
      return;
    }
    if (g->pushes < num_workers_) {
      g->waiting_pulls.push_back(req);  // released by the round's last push
      return;
    }
    RespondPull(req, g, server);
    return;
  }
  hipStream_t stream = PullStream(req.sender);
  bool chain = NeedChain(req.sender);
  // multi-key pull with an in-place destination: batched copy of every
  // store entry straight into the requester's mapped pool (no staging
  // buffer, meta-only response)
  if (n > 1 && (req.option & kOptPullAddr)) {
    if (auto* plane = ThePlane(po_)) {
      std::vector<kern::CopyDesc> descs;
      std::vector<Entry*> ents;
      SArray<int> lens2(n);
      uint64_t off = 0;
      bool ok = true;
      {
        std::lock_guard<std::mutex> lk(mu_);
        for (size_t i = 0; i < n && ok; ++i) {
          auto it = store_.find(kvs.keys[i]);
          XPS_CHECK(it != store_.end()) << "pull of unknown key " << kvs.keys[i];
          size_t len = it->second.buf.size();
          char* dst = plane->PeerDst(req.sender, req.addr + off, len);
          ok = dst != nullptr && len % 16 == 0 && off % 16 == 0;
          if (ok) {
            ents.push_back(&it->second);
            descs.push_back({dst, it->second.buf.data(), len});
            lens2[i] = static_cast<int>(len / sizeof(float));
            off += len;
          }
        }
      }
      if (ok) {
        for (Entry* e : ents) OrderAfter(e, stream);
        kern::BatchedAssign(descs.data(), static_cast<int>(n), stream);
        KVMeta r2 = req;
        r2.option |= kOptInPlace | kOptPullLane;
        r2.val_len = static_cast<int64_t>(off);
        KVPairs<float> res2;
        res2.keys = kvs.keys;
        res2.lens = lens2;
        server->Response(r2, res2);  // meta+keys/lens only; plane defers on `stream`
        if (chain) {
          EventRef ev = MakeEventRef(po_, stream);  // pushes must wait these reads
          for (Entry* e : ents) e->last_ev = ev;
        }
        return;
      }
    }
  }
  KVPairs<float> res;
  res.keys = kvs.keys;
  SArray<int> lens(n);
  std::vector<Entry*> touched;
  if (n == 1) {
    SArray<char> entry;
    {
      std::lock_guard<std::mutex> lk(mu_);
      auto it = store_.find(kvs.keys[0]);
      XPS_CHECK(it != store_.end()) << "pull of unknown key " << kvs.keys[0];
      entry = it->second.buf;
      touched.push_back(&it->second);
    }
    // the plane's in-place write (or the sync below) reads buf on
    // `stream`: order it behind the last writer
    {
      XPS_STAGE(pull_order_after);
      OrderAfter(touched[0], stream);
    }
    res.vals = SArray<float>::View(entry);  // zero-copy store view
    lens[0] = static_cast<int>(entry.size() / sizeof(float));
  } else {
    size_t total = 0;
    std::vector<SArray<char>> entries(n);
    {
      std::lock_guard<std::mutex> lk(mu_);
      for (size_t i = 0; i < n; ++i) {
        auto it = store_.find(kvs.keys[i]);
        XPS_CHECK(it != store_.end()) << "pull of unknown key " << kvs.keys[i];
        entries[i] = it->second.buf;
        touched.push_back(&it->second);
        lens[i] = static_cast<int>(entries[i].size() / sizeof(float));
        total += entries[i].size();
      }
    }
    for (Entry* e : touched) OrderAfter(e, stream);
    SArray<char> tmp = HbmPool::Get()->AllocArray(total);
    size_t off = 0;
    std::vector<kern::CopyDesc> descs;
    descs.reserve(n);
    bool aligned = true;
    for (size_t i = 0; i < n; ++i) {
      descs.push_back({tmp.data() + off, entries[i].data(), entries[i].size()});
      aligned = aligned && (entries[i].size() % 16 == 0) && (off % 16 == 0);
      off += entries[i].size();
    }
    if (aligned) {
      kern::BatchedAssign(descs.data(), static_cast<int>(n), stream);
    } else {
      for (auto& d : descs) kern::DenseAssign(d.dst, d.src, d.nbytes, stream);
    }
    res.vals = SArray<float>::View(tmp);  // plane keeps it alive until sent
  }
  res.lens = lens;
  KVMeta r = req;
  // the advertised destination can be SMALLER than the store entry
  // (entry grew since the worker sized its buffer): an in-place write
  // would overrun the requester's pool region — force the staging path,
  // whose worker-side merge checks the overflow loudly
  if ((r.option & kOptPullAddr) &&
      static_cast<int64_t>(res.vals.nbytes()) > req.val_len && req.val_len > 0) {
    r.option &= ~kOptPullAddr;
  }
  // TCP fallback (no in-place destination): the staging D2H copy below in
  // the van is stream-unaware — drain our stream first
  if (!(r.option & kOptPullAddr)) XPS_HIP_CHECK(hipStreamSynchronize(stream));
  r.option |= kOptPullLane;  // ordering prepared on the pull lane
  server->Response(r, res);  // plane enqueues the in-place read on `stream`
  if (chain) {
    EventRef ev = MakeEventRef(po_, stream);  // pushes must wait these reads
    for (Entry* e : touched) e->last_ev = ev;
  }
}

void GpuDenseHandler::RegisterEntry(Key key, void* ptr, size_t nbytes, int device) {
  XPS_CHECK(ptr && nbytes) << "RegisterEntry needs a real buffer";
  std::lock_guard<std::mutex> lk(mu_);
  Entry& e = store_[key];
  if (!e.buf.empty()) e.retired.push_back(e.buf);  // keep old offsets owned
  // non-owning view: the app keeps the buffer alive
  e.buf = SArray<char>(static_cast<char*>(ptr), nbytes, device);
}

void GpuDenseHandler::Save(const std::string& path) {
  FILE* f = fopen(path.c_str(), "wb");
  XPS_CHECK(f) << "cannot open checkpoint " << path;
  std::lock_guard<std::mutex> lk(mu_);
  uint64_t n = store_.size();
  fwrite(&n, 8, 1, f);
  std::vector<char> host;
  for (auto& kv : store_) {
    uint64_t key = kv.first, len = kv.second.buf.size();
    fwrite(&key, 8, 1, f);
    fwrite(&len, 8, 1, f);
    host.resize(len);
    XPS_HIP_CHECK(hipMemcpy(host.data(), kv.second.buf.data(), len, hipMemcpyDeviceToHost));
    fwrite(host.data(), 1, len, f);
  }
  fclose(f);
}

void GpuDenseHandler::Load(const std::string& path) {
  FILE* f = fopen(path.c_str(), "rb");
  XPS_CHECK(f) << "cannot open checkpoint " << path;
  std::lock_guard<std::mutex> lk(mu_);
  uint64_t n = 0;
  XPS_CHECK_EQ(fread(&n, 8, 1, f), 1u);
  std::vector<char> host;
  for (uint64_t i = 0; i < n; ++i) {
    uint64_t key, len;
    XPS_CHECK_EQ(fread(&key, 8, 1, f), 1u);
    XPS_CHECK_EQ(fread(&len, 8, 1, f), 1u);
    host.resize(len);
    XPS_CHECK_EQ(fread(host.data(), 1, len, f), len);
    auto& e = store_[key];
    if (e.buf.size() < len) e.buf = HbmPool::Get()->AllocArray(len);
    XPS_HIP_CHECK(hipMemcpy(e.buf.data(), host.data(), len, hipMemcpyHostToDevice));
  }
  fclose(f);
}

// ----------------------------------------------------------------- sparse

GpuSparseHandler::GpuSparseHandler(Postoffice* po, size_t rows, size_t row_len, bool accumulate,
                                   int key_shift)
    : po_(po), rows_(rows), row_len_(row_len), accumulate_(accumulate), key_shift_(key_shift) {
  auto* pool = HbmPool::Get();
  XPS_CHECK(pool->initialized()) << "GpuSparseHandler needs the HBM pool";
  // concurrent workers scatter on different peer streams: accumulate
  // must be element-atomic or overlapping rows lose updates
  atomic_ = po_->num_workers() > 1;
  if (key_shift_ > 0) {
    int rank = po_->my_rank();
    row_base_ = po_->GetServerKeyRanges()[rank].begin >> key_shift_;
  }
  XPS_HIP_CHECK(hipSetDevice(pool->device()));
  table_ = pool->AllocArray(rows * row_len * sizeof(float));
  XPS_HIP_CHECK(hipMemset(table_.data(), 0, table_.size()));
}

hipStream_t GpuSparseHandler::Stream(int sender) {
  return PeerStream(po_, sender, &fallback_stream_);
}

const uint64_t* GpuSparseHandler::DeviceKeys(const SArray<Key>& keys, int sender,
                                             hipStream_t s) {
  if (keys.on_device()) return keys.data();
  // host keys: stage synchronously into the per-sender scratch (ordered
  // before any kernel we launch afterwards on `s`)
  size_t bytes = keys.nbytes();
  SArray<char> scratch;
  {
    std::lock_guard<std::mutex> lk(mu_);
    auto& sc = key_scratch_[sender];
    if (sc.size() < bytes) sc = HbmPool::Get()->AllocArray(std::max<size_t>(bytes, 1 << 16));
    scratch = sc;
  }
  XPS_HIP_CHECK(hipStreamSynchronize(s));  // prior kernels still reading the scratch
  XPS_HIP_CHECK(hipMemcpy(scratch.data(), keys.data(), bytes, hipMemcpyHostToDevice));
  return reinterpret_cast<const uint64_t*>(scratch.data());
}

void GpuSparseHandler::operator()(const KVMeta& req, const KVPairs<float>& kvs,
                                  KVServer<float>* server) {
  auto* pool = HbmPool::Get();
  XPS_HIP_CHECK(hipSetDevice(pool->device()));
  hipStream_t stream = Stream(req.sender);
  size_t n = kvs.keys.size();
  XPS_CHECK_GT(n, 0u);
  float* table = reinterpret_cast<float*>(table_.data());
  if (req.push) {
    XPS_CHECK(kvs.vals.on_device()) << "sparse push needs device vals (pool buffers)";
    XPS_CHECK_EQ(kvs.vals.size(), n * row_len_);
    const uint64_t* rows = DeviceKeys(kvs.keys, req.sender, stream);
    if (accumulate_ || req.cmd == kCmdSum) {
      kern::SparseScatterAddF32(table, rows, n, row_len_, kvs.vals.data(), atomic_,
                                stream, key_shift_, row_base_, rows_);
    } else {
      kern::SparseScatterAssignF32(table, rows, n, row_len_, kvs.vals.data(), stream, key_shift_,
                                   row_base_, rows_);
    }
    if (req.pull) {
      // fused round (ZPushPull): gather the post-scatter rows on the
      // SAME stream (ordered behind the scatter) and answer in one trip
      SArray<char> out = pool->AllocArray(n * row_len_ * sizeof(float));
      kern::SparseGatherF32(table, rows, n, row_len_, reinterpret_cast<float*>(out.data()),
                            stream, key_shift_, row_base_, rows_);
      KVPairs<float> res;
      res.vals = SArray<float>::View(out);
      SArray<int> lens(1);
      lens[0] = static_cast<int>(n * row_len_);
      res.lens = lens;
      if (!(req.option & kOptPullAddr)) XPS_HIP_CHECK(hipStreamSynchronize(stream));
      server->Response(req, res);
      return;
    }
    auto* plane = po_->van() ? po_->van()->plane() : nullptr;
    if (!plane) XPS_HIP_CHECK(hipStreamSynchronize(stream));
    server->Response(req);
  } else if (req.pull) {
    const uint64_t* rows = DeviceKeys(kvs.keys, req.sender, stream);
    SArray<char> out = pool->AllocArray(n * row_len_ * sizeof(float));
    kern::SparseGatherF32(table, rows, n, row_len_, reinterpret_cast<float*>(out.data()), stream,
                          key_shift_, row_base_, rows_);
    KVPairs<float> res;
    // keys stay meta-only: device keys must not be dereferenced host-side
    res.vals = SArray<float>::View(out);
    SArray<int> lens(1);
    lens[0] = static_cast<int>(n * row_len_);
    res.lens = lens;
    if (!(req.option & kOptPullAddr)) XPS_HIP_CHECK(hipStreamSynchronize(stream));
    server->Response(req, res);
  }
}

}  // namespace xps

namespace xps {

void GpuSparseHandler::Save(const std::string& path) {
  FILE* f = fopen(path.c_str(), "wb");
  XPS_CHECK(f) << "cannot open checkpoint " << path;
  uint64_t hdr[3] = {rows_, row_len_, static_cast<uint64_t>(key_shift_)};
  fwrite(hdr, 8, 3, f);
  std::vector<char> host(table_.size());
  XPS_HIP_CHECK(hipMemcpy(host.data(), table_.data(), table_.size(), hipMemcpyDeviceToHost));
  fwrite(host.data(), 1, host.size(), f);
  fclose(f);
}

void GpuSparseHandler::Load(const std::string& path) {
  FILE* f = fopen(path.c_str(), "rb");
  XPS_CHECK(f) << "cannot open checkpoint " << path;
  uint64_t hdr[3];
  XPS_CHECK_EQ(fread(hdr, 8, 3, f), 3u);
  XPS_CHECK_EQ(hdr[0], rows_);
  XPS_CHECK_EQ(hdr[1], row_len_);
  std::vector<char> host(table_.size());
  XPS_CHECK_EQ(fread(host.data(), 1, host.size(), f), host.size());
  XPS_HIP_CHECK(hipMemcpy(table_.data(), host.data(), table_.size(), hipMemcpyHostToDevice));
  fclose(f);
}

}  // namespace xps
