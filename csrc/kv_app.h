// KVWorker / KVServer: sharded key-value push/pull over the Van.
//
// Reference parity: ps-lite include/ps/kv_app.h (KVPairs, KVWorker
// Push/Pull/ZPush/ZPull/Wait :66-318, DefaultSlicer :566, KVServer
// Process/Response :346-564, KVServerDefaultHandle :431). Re-designed:
// pull requests advertise the destination HBM-pool offset (meta.addr) so
// the MI355X data plane can write responses in place over xGMI
// (zero-copy pull, the rdma_transport.h:369-398 analog); responses
// delivered in place carry kOptInPlace and skip the worker-side merge.
#pragma once

#include <unistd.h>

#include <algorithm>
#include <atomic>
#include <chrono>
#include <condition_variable>
#include <functional>
#include <mutex>
#include <set>
#include <unordered_map>
#include <vector>

#include "gpu_plane.h"
#include "hip_pool.h"
#include "hip_util.h"
#include "host_par.h"
#include "host_pool.h"
#include "simple_app.h"

namespace xps {

// worker-side cmd values (meta.head) understood by the built-in handlers
static const int kCmdDefault = 0;  // handler's configured default op
static const int kCmdAssign = 1;
static const int kCmdSum = 2;

template <typename V>
struct KVPairs {
  SArray<Key> keys;      // host keys (slicing always runs host-side)
  SArray<V> vals;
  SArray<int> lens;
  // optional device mirror of `keys` (sparse path): when set, the wire
  // message carries this blob by pool reference so server kernels can
  // index the table without a host->device staging copy
  SArray<Key> keys_dev;
};

struct KVMeta {
  int cmd = 0;
  bool push = false;
  bool pull = false;
  int sender = kEmptyNodeID;
  int timestamp = -1;
  int customer_id = 0;
  Key key = 0;          // fast-path single key
  uint64_t addr = 0;    // pull: requester's destination pool offset
  int64_t val_len = 0;
  int option = 0;
  int src_dev = kCPU;
  int dst_dev = kCPU;
};

template <typename V>
class KVWorker : public SimpleApp {
 public:
  using Callback = std::function<void()>;
  using SlicedKVs = std::vector<std::pair<bool, KVPairs<V>>>;
  using Slicer = std::function<void(const KVPairs<V>&, const std::vector<Range>&, SlicedKVs*)>;

  explicit KVWorker(int app_id, int customer_id, int instance_idx = 0)
      : SimpleApp(Postoffice::GetWorker(instance_idx), DeferCustomer{}) {
    slicer_ = [this](const KVPairs<V>& s, const std::vector<Range>& r, SlicedKVs* out) {
      DefaultSlicer(s, r, out);
    };
    obj_.reset(new Customer(app_id, customer_id,
                            [this](const Message& m) { Process(m); }, po_));
  }

  int Push(const std::vector<Key>& keys, const std::vector<V>& vals,
           const std::vector<int>& lens = {}, int cmd = 0, const Callback& cb = nullptr) {
    return ZPush(SArray<Key>(keys), SArray<V>(vals), SArray<int>(lens), cmd, cb);
  }

  int Pull(const std::vector<Key>& keys, std::vector<V>* vals, std::vector<int>* lens = nullptr,
           int cmd = 0, const Callback& cb = nullptr) {
    SArray<Key> skeys(keys);
    int ts = AddPullRequest_(skeys, vals, lens, cmd, cb);
    return ts;
  }

  // zero-copy push: keys/vals/lens remain owned by the caller until done
  int ZPush(const SArray<Key>& keys, const SArray<V>& vals, const SArray<int>& lens = {},
            int cmd = 0, const Callback& cb = nullptr, const SArray<Key>& keys_dev = {}) {
    int ts = obj_->NewRequest(kServerGroup);
    AddCallback(ts, cb);
    KVPairs<V> kvs;
    kvs.keys = keys;
    kvs.vals = vals;
    kvs.lens = lens;
    kvs.keys_dev = keys_dev;
    Send(ts, true, false, cmd, kvs);
    return ts;
  }

  // Fused push+pull: ONE request per server that carries the pushed
  // vals AND advertises the pull destination; the server applies the
  // push and answers with the post-push values in the same trip. Halves
  // the round trips of the push-then-pull pattern (sparse embedding
  // lookup+update, BASELINE config #5). push vals and the pull dst
  // share the same keys/lens geometry.
  int ZPushPull(const SArray<Key>& keys, const SArray<V>& vals, SArray<V>* outs,
                const SArray<int>& lens = {}, int cmd = 0, const Callback& cb = nullptr,
                const SArray<Key>& keys_dev = {}) {
    XPS_CHECK(outs && outs->size()) << "ZPushPull needs a pre-sized outs buffer";
    int ts = obj_->NewRequest(kServerGroup);
    {
      std::lock_guard<std::mutex> lk(mu_);
      pull_dst_[ts] = {outs, nullptr};
    }
    AddCallback(ts, cb);
    KVPairs<V> kvs;
    kvs.keys = keys;
    kvs.vals = vals;
    kvs.lens = lens;
    kvs.keys_dev = keys_dev;
    Send(ts, true, true, cmd, kvs, outs);
    return ts;
  }

  // zero-copy pull into a pre-allocated vals buffer (device or host)
  int ZPull(const SArray<Key>& keys, SArray<V>* vals, SArray<int>* lens = nullptr, int cmd = 0,
            const Callback& cb = nullptr, const SArray<Key>& keys_dev = {}) {
    XPS_CHECK(vals && vals->size()) << "ZPull needs a pre-sized vals buffer";
    int ts = obj_->NewRequest(kServerGroup);
    {
      std::lock_guard<std::mutex> lk(mu_);
      pull_dst_[ts] = {vals, lens};
    }
    AddCallback(ts, cb);
    KVPairs<V> kvs;
    kvs.keys = keys;
    kvs.vals = *vals;  // carried for slicing geometry; not sent in requests
    if (lens) kvs.lens = *lens;
    kvs.keys_dev = keys_dev;
    Send(ts, false, true, cmd, kvs);
    return ts;
  }

  void Wait(int timestamp) { obj_->WaitRequest(timestamp); }

  void set_slicer(Slicer s) { slicer_ = std::move(s); }

  // Default: binary-search keys into per-server ranges; vals/lens are
  // zero-copy segments (ps-lite kv_app.h:566-621 behavior).
  void DefaultSlicer(const KVPairs<V>& send, const std::vector<Range>& ranges, SlicedKVs* sliced) {
    sliced->resize(ranges.size());
    size_t n = send.keys.size();
    std::vector<size_t> pos(ranges.size() + 1);
    const Key* begin = send.keys.begin();
    const Key* end = send.keys.end();
    for (size_t i = 0; i < ranges.size(); ++i) {
      pos[i] = std::lower_bound(begin, end, ranges[i].begin) - begin;
    }
    pos[ranges.size()] = n;
    // per-key val offsets (elements)
    std::vector<size_t> val_off(n + 1, 0);
    if (!send.lens.empty()) {
      XPS_CHECK_EQ(send.lens.size(), n);
      for (size_t i = 0; i < n; ++i) val_off[i + 1] = val_off[i] + send.lens[i];
    } else if (n > 0) {
      size_t k = send.vals.size() / n;
      for (size_t i = 0; i <= n; ++i) val_off[i] = i * k;
    }
    for (size_t i = 0; i < ranges.size(); ++i) {
      size_t a = pos[i], b = pos[i + 1];
      auto& out = (*sliced)[i];
      out.first = b > a;
      if (!out.first) continue;
      out.second.keys = send.keys.Segment(a, b);
      if (!send.keys_dev.empty()) {
        out.second.keys_dev = send.keys_dev.Segment(a, b);
      }
      if (!send.vals.empty()) {
        out.second.vals = send.vals.Segment(val_off[a], val_off[b]);
      }
      if (!send.lens.empty()) {
        out.second.lens = send.lens.Segment(a, b);
      }
    }
  }

 private:
  int AddPullRequest_(const SArray<Key>& keys, std::vector<V>* vals, std::vector<int>* lens,
                      int cmd, const Callback& cb) {
    // vector-based Pull: size the buffer by asking with lens unknown —
    // allocate after responses arrive. We use a two-phase: buffer grows in merge.
    int ts = obj_->NewRequest(kServerGroup);
    {
      std::lock_guard<std::mutex> lk(mu_);
      vec_pull_dst_[ts] = {vals, lens};
    }
    AddCallback(ts, cb);
    KVPairs<V> kvs;
    kvs.keys = keys;
    Send(ts, false, true, cmd, kvs);
    return ts;
  }

  void AddCallback(int ts, const Callback& cb) {
    if (!cb) return;
    std::lock_guard<std::mutex> lk(mu_);
    callbacks_[ts] = cb;
  }

  // dst: fused push+pull destination buffer (same geometry as kvs.vals)
  void Send(int ts, bool push, bool pull, int cmd, KVPairs<V>& kvs, SArray<V>* dst = nullptr) {
    SlicedKVs sliced;
    slicer_(kvs, po_->GetServerKeyRanges(), &sliced);
    int skipped = 0;
    for (auto& s : sliced)
      if (!s.first) ++skipped;
    if (skipped) obj_->AddResponse(ts, skipped);
    {
      std::lock_guard<std::mutex> lk(mu_);
      expected_[ts] = static_cast<int>(sliced.size()) - skipped;
    }
    if (static_cast<int>(sliced.size()) == skipped) {
      // nothing to send; run callback now
      RunCallback(ts);
      return;
    }
    // byte offset of each slice's vals within the original vals buffer
    // (for in-place pull destination advertisement)
    size_t val_elem_off = 0;
    for (size_t i = 0; i < sliced.size(); ++i) {
      if (!sliced[i].first) continue;
      auto& s = sliced[i].second;
      Message msg;
      msg.meta.app_id = obj_->app_id();
      msg.meta.customer_id = obj_->customer_id();
      msg.meta.request = true;
      msg.meta.push = push;
      msg.meta.pull = pull;
      msg.meta.head = cmd;
      msg.meta.timestamp = ts;
      msg.meta.recver = ServerRankToID(static_cast<int>(i));
      if (s.keys.size() >= 1) msg.meta.key = s.keys[0];
      msg.meta.src_dev = push ? s.vals.device() : kvs.vals.device();
      msg.meta.dst_dev = msg.meta.src_dev;
      if (!s.keys_dev.empty()) {
        msg.AddData(s.keys_dev);
      } else {
        msg.AddData(s.keys);
      }
      if (push) {
        msg.meta.val_len = static_cast<int64_t>(s.vals.nbytes());
        msg.AddData(s.vals);
        if (!s.lens.empty()) msg.AddData(s.lens);
        // one-sided steady state (reference rdma_van.h:486-508): a
        // prior ACK advertised this key's store-entry offset — ask the
        // plane to write vals there itself and send meta only. Assign
        // semantics only (sum needs the server's accumulate kernel).
        {
          std::lock_guard<std::mutex> lk(mu_);
          auto nit = entry_cache_.find(msg.meta.recver);
          if (nit != entry_cache_.end() && !nit->second.empty()) {
            if (!pull && s.keys.size() == 1 && cmd != kCmdSum && s.vals.on_device()) {
              auto it = nit->second.find(s.keys[0]);
              // uid guard: a RECOVERED server (same id, new process) has
              // a fresh pool — its old offsets must never be written
              if (it != nit->second.end() &&
                  it->second.len == static_cast<int64_t>(s.vals.nbytes()) &&
                  po_->van()->GetNode(msg.meta.recver).shm_uid == it->second.uid) {
                msg.meta.addr = it->second.off;
                msg.meta.option |= kOptEntryPush;
                it->second.one_sided = true;
              } else if (it != nit->second.end()) {
                it->second.one_sided = false;  // normal push re-orders via server
              }
            } else {
              // normal / multi-key push: any contained cached key loses
              // its one-sided ordering chain
              for (size_t ki = 0; ki < s.keys.size(); ++ki) {
                auto it = nit->second.find(s.keys[ki]);
                if (it != nit->second.end()) it->second.one_sided = false;
              }
            }
          }
        }
        if (pull && dst) {
          // fused round: advertise this slice's pull destination — the
          // dst buffer shares the vals geometry, so the slice's element
          // offset within kvs.vals locates its dst segment
          size_t elem_off = static_cast<size_t>(s.vals.data() - kvs.vals.data());
          const V* dptr = dst->data() + elem_off;
          uint64_t off = 0;
          if (dst->on_device()) {
            if (HbmPool::Get()->OffsetOf(dptr, &off)) {
              msg.meta.addr = off;
              msg.meta.option |= kOptPullAddr;
            }
          } else if (HostShmPool::Get()->OffsetOf(dptr, &off)) {
            msg.meta.addr = off;
            msg.meta.option |= kOptPullAddr | kOptHostAddr;
          }
        }
      } else {
        // One-sided pull (RDMA_READ analog, assign-mode steady state):
        // the entry offset is cached, OUR last push of the key was
        // one-sided (stream-ordered before this read), and the process
        // uid still matches — copy the server's entry into the dst with
        // our own kernel and complete the request with a synthetic
        // deferred response. No server round trip at all. Cross-worker
        // concurrent writes follow the reference's async-PS semantics
        // (an RDMA_READ racing another worker's RDMA_WRITE).
        if (s.keys.size() == 1 && cmd != kCmdSum && s.vals.on_device() &&
            s.vals.nbytes() > 0) {
          uint64_t eoff = 0;
          bool hit = false;
          {
            std::lock_guard<std::mutex> lk(mu_);
            auto nit = entry_cache_.find(msg.meta.recver);
            if (nit != entry_cache_.end()) {
              auto it = nit->second.find(s.keys[0]);
              if (it != nit->second.end() && it->second.one_sided &&
                  it->second.len == static_cast<int64_t>(s.vals.nbytes()) &&
                  po_->van()->GetNode(msg.meta.recver).shm_uid == it->second.uid) {
                eoff = it->second.off;
                hit = true;
              }
            }
          }
          if (hit) {
            auto* plane = dynamic_cast<GpuPlane*>(po_->van()->plane());
            if (plane) {
              Message resp;
              resp.meta.app_id = obj_->app_id();
              resp.meta.customer_id = obj_->customer_id();
              resp.meta.request = false;
              resp.meta.pull = true;
              resp.meta.head = cmd;
              resp.meta.timestamp = ts;
              resp.meta.sender = msg.meta.recver;
              resp.meta.recver = po_->node_id();
              resp.meta.key = s.keys[0];
              resp.meta.option = kOptInPlace;
              resp.meta.val_len = static_cast<int64_t>(s.vals.nbytes());
              if (plane->LocalPullRead(msg.meta.recver, s.vals.data(), eoff,
                                       s.vals.nbytes(), std::move(resp))) {
                continue;  // request satisfied by the local read
              }
            }
          }
        }
        // pull request: keys (+lens geometry) only; advertise destination
        msg.meta.val_len = static_cast<int64_t>(s.vals.nbytes());
        if (!s.vals.empty()) {
          uint64_t off = 0;
          if (s.vals.on_device()) {
            if (HbmPool::Get()->OffsetOf(s.vals.data(), &off)) {
              msg.meta.addr = off;
              msg.meta.option |= kOptPullAddr;  // in-place HBM response
            }
          } else if (HostShmPool::Get()->OffsetOf(s.vals.data(), &off)) {
            msg.meta.addr = off;
            msg.meta.option |= kOptPullAddr | kOptHostAddr;  // in-place host response
          }
        }
        if (!s.lens.empty()) msg.AddData(s.lens);
      }
      po_->van()->Send(msg);
      (void)val_elem_off;
    }
  }

  void Process(const Message& msg) {
    if (msg.meta.request) return;  // workers only receive responses
    int ts = msg.meta.timestamp;
    bool last = false;
    KVPairs<V> kvs;
    if (msg.meta.pull) {
      if (!msg.data.empty()) {
        kvs.keys = SArray<Key>::View(msg.data[0]);
        if (msg.data.size() > 1) kvs.vals = SArray<V>::View(msg.data[1]);
        if (msg.data.size() > 2) kvs.lens = SArray<int>::View(msg.data[2]);
      }
      if (kvs.keys.empty() || kvs.keys.on_device()) {
        // device keys (sparse path) are never dereferenced host-side
        kvs.keys = SArray<Key>({msg.meta.key});
      }
    }
    {
      std::lock_guard<std::mutex> lk(mu_);
      if (msg.meta.pull) {
        auto& got = recv_kvs_[ts];
        got.push_back(RecvSlice{kvs, msg.meta.option, msg.meta.val_len});
        last = static_cast<int>(got.size()) >= expected_[ts];
      } else {
        // a push ACK may advertise the server's entry offset for this
        // key: cache it so later assign pushes go one-sided
        if (msg.meta.option & kOptEntryAddr) {
          entry_cache_[msg.meta.sender][msg.meta.key] =
              EntryRec{msg.meta.addr, msg.meta.val_len,
                       po_->van()->GetNode(msg.meta.sender).shm_uid};
        }
        int n = ++push_acks_[ts];
        last = n >= expected_[ts];
      }
    }
    if (last) {
      if (msg.meta.pull) MergePull(ts);
      RunCallback(ts);
    }
  }

  void MergePull(int ts) {
    std::vector<RecvSlice> got;
    SArray<V>* dst_vals = nullptr;
    SArray<int>* dst_lens = nullptr;
    std::vector<V>* vec_vals = nullptr;
    std::vector<int>* vec_lens = nullptr;
    {
      std::lock_guard<std::mutex> lk(mu_);
      got.swap(recv_kvs_[ts]);
      recv_kvs_.erase(ts);
      auto it = pull_dst_.find(ts);
      if (it != pull_dst_.end()) {
        dst_vals = it->second.first;
        dst_lens = it->second.second;
        pull_dst_.erase(it);
      }
      auto it2 = vec_pull_dst_.find(ts);
      if (it2 != vec_pull_dst_.end()) {
        vec_vals = it2->second.first;
        vec_lens = it2->second.second;
        vec_pull_dst_.erase(it2);
      }
    }
    // order slices by their first key
    std::sort(got.begin(), got.end(), [](const RecvSlice& a, const RecvSlice& b) {
      Key ka = a.kvs.keys.empty() ? 0 : a.kvs.keys[0];
      Key kb = b.kvs.keys.empty() ? 0 : b.kvs.keys[0];
      return ka < kb;
    });
    if (vec_vals) {
      // vector Pull: concatenate
      vec_vals->clear();
      if (vec_lens) vec_lens->clear();
      for (auto& g : got) {
        vec_vals->insert(vec_vals->end(), g.kvs.vals.begin(), g.kvs.vals.end());
        if (vec_lens) {
          for (size_t i = 0; i < g.kvs.lens.size(); ++i) vec_lens->push_back(g.kvs.lens[i]);
        }
      }
      return;
    }
    if (!dst_vals) return;
    size_t off = 0;
    for (auto& g : got) {
      if (g.option & kOptInPlace) {
        // the data plane already wrote this slice into the destination
        off += static_cast<size_t>(g.val_len) / sizeof(V);
        continue;
      }
      size_t n = g.kvs.vals.size();
      if (n == 0) continue;
      XPS_CHECK_LE(off + n, dst_vals->size()) << "pull response overflows dst";
      if (dst_vals->on_device()) {
        gpu::CopyHostToDevice(dst_vals->data() + off, g.kvs.vals.data(), n * sizeof(V),
                              dst_vals->device());
      } else {
        memcpy(dst_vals->data() + off, g.kvs.vals.data(), n * sizeof(V));
      }
      off += n;
    }
    if (dst_lens && !got.empty()) {
      size_t loff = 0;
      for (auto& g : got) {
        for (size_t i = 0; i < g.kvs.lens.size() && loff < dst_lens->size(); ++i) {
          (*dst_lens)[loff++] = g.kvs.lens[i];
        }
      }
    }
  }

  void RunCallback(int ts) {
    Callback cb;
    {
      std::lock_guard<std::mutex> lk(mu_);
      auto it = callbacks_.find(ts);
      if (it != callbacks_.end()) {
        cb = it->second;
        callbacks_.erase(it);
      }
      expected_.erase(ts);
      push_acks_.erase(ts);
    }
    if (cb) cb();
  }

  struct RecvSlice {
    KVPairs<V> kvs;
    int option = 0;
    int64_t val_len = 0;
  };

  // server node id -> key -> advertised store-entry location, learned
  // from push ACKs; uid pins the advert to the server PROCESS that made
  // it (recovery = new process = new pool). Guarded by mu_.
  struct EntryRec {
    uint64_t off;
    int64_t len;
    uint64_t uid;
    // our LAST push of this key was one-sided (same-stream): only then
    // may a pull be served by a one-sided read — the stream orders the
    // read after our write. A normal (server-kernel) push clears it.
    bool one_sided = false;
  };
  std::unordered_map<int, std::unordered_map<Key, EntryRec>> entry_cache_;

  Slicer slicer_;
  std::mutex mu_;
  std::unordered_map<int, Callback> callbacks_;
  std::unordered_map<int, int> expected_;
  std::unordered_map<int, int> push_acks_;
  std::unordered_map<int, std::vector<RecvSlice>> recv_kvs_;
  std::unordered_map<int, std::pair<SArray<V>*, SArray<int>*>> pull_dst_;
  std::unordered_map<int, std::pair<std::vector<V>*, std::vector<int>*>> vec_pull_dst_;
};

template <typename V>
class KVServer : public SimpleApp {
 public:
  using ReqHandle = std::function<void(const KVMeta&, const KVPairs<V>&, KVServer<V>*)>;

  explicit KVServer(int app_id, int instance_idx = 0)
      : SimpleApp(Postoffice::GetServer(instance_idx), DeferCustomer{}) {
    obj_.reset(new Customer(app_id, app_id, [this](const Message& m) { Process(m); }, po_));
  }

  void set_request_handle(ReqHandle h) {
    {
      std::lock_guard<std::mutex> lk(handle_mu_);
      request_handle2_ = std::move(h);
      handle_set_.store(true, std::memory_order_release);
    }
    handle_cv_.notify_all();
  }

  void Response(const KVMeta& req, const KVPairs<V>& res = KVPairs<V>()) {
    Message msg;
    msg.meta.app_id = obj_->app_id();
    msg.meta.customer_id = req.customer_id;
    msg.meta.request = false;
    msg.meta.push = req.push;
    msg.meta.pull = req.pull;
    msg.meta.head = req.cmd;
    msg.meta.timestamp = req.timestamp;
    msg.meta.recver = req.sender;
    msg.meta.key = req.key;
    msg.meta.addr = req.addr;         // in-place pull destination (pool offset)
    msg.meta.option = req.option;
    msg.meta.val_len = req.val_len;   // in-place responses keep the request's byte count
    msg.meta.src_dev = res.vals.device();
    msg.meta.dst_dev = req.dst_dev;
    if (req.pull && (!res.vals.empty() || !res.keys.empty())) {
      if (!res.vals.empty()) msg.meta.val_len = static_cast<int64_t>(res.vals.nbytes());
      msg.AddData(res.keys);  // may be empty (sparse: meta.key suffices)
      msg.AddData(res.vals);
      if (!res.lens.empty()) msg.AddData(res.lens);
    }
    po_->van()->Send(msg);
  }

 private:
  void Process(const Message& msg) {
    if (msg.meta.simple_app) {
      SimpleApp::Process(msg);
      return;
    }
    KVMeta meta;
    meta.cmd = msg.meta.head;
    meta.push = msg.meta.push;
    meta.pull = msg.meta.pull;
    meta.sender = msg.meta.sender;
    meta.timestamp = msg.meta.timestamp;
    meta.customer_id = msg.meta.customer_id;
    meta.key = msg.meta.key;
    meta.addr = msg.meta.addr;
    meta.val_len = msg.meta.val_len;
    meta.option = msg.meta.option;
    meta.src_dev = msg.meta.src_dev;
    meta.dst_dev = msg.meta.dst_dev;
    KVPairs<V> kvs;
    if (!msg.data.empty()) {
      kvs.keys = SArray<Key>::View(msg.data[0]);
      if (msg.data.size() > 1 && msg.meta.push) kvs.vals = SArray<V>::View(msg.data[1]);
      size_t lens_idx = msg.meta.push ? 2 : 1;
      if (msg.data.size() > lens_idx) kvs.lens = SArray<int>::View(msg.data[lens_idx]);
    } else if (msg.meta.key || msg.meta.val_len) {
      kvs.keys = SArray<Key>({msg.meta.key});
    }
    // a worker's first request can arrive between construction and
    // set_request_handle — block (deterministically, on a cv) until the
    // app installs it
    if (!handle_set_.load(std::memory_order_acquire)) {
      std::unique_lock<std::mutex> lk(handle_mu_);
      bool ok = handle_cv_.wait_for(lk, std::chrono::seconds(60),
                                    [this] { return handle_set_.load(); });
      XPS_CHECK(ok) << "KVServer has no request handle after 60 s";
    }
    request_handle2_(meta, kvs, this);
  }

  ReqHandle request_handle2_;
  std::mutex handle_mu_;
  std::condition_variable handle_cv_;
  std::atomic<bool> handle_set_{false};
};

// Default CPU handle: store[key] op= vals (sum by default, memcpy for
// cmd=kCmdAssign); pull echoes the store (ps-lite kv_app.h:431-452
// behavior, plus the assign op the EmptyHandler benchmark semantics use).
inline void SumInto(float* dst, const float* src, size_t n) { HostPar::SumF32(dst, src, n); }
template <typename V>
inline void SumInto(V* dst, const V* src, size_t n) {
  for (size_t j = 0; j < n; ++j) dst[j] += src[j];
}

template <typename V>
struct KVServerDefaultHandle {
  void operator()(const KVMeta& req, const KVPairs<V>& kvs, KVServer<V>* server) {
    size_t n = kvs.keys.size();
    KVPairs<V> res;
    if (req.push) {
      XPS_CHECK(!kvs.vals.on_device()) << "default handle is CPU-only";
      size_t off = 0;
      for (size_t i = 0; i < n; ++i) {
        size_t len = kvs.lens.empty() ? kvs.vals.size() / n : kvs.lens[i];
        auto& entry = store[kvs.keys[i]];
        if (entry.size() < len) entry.resize(len, V(0));
        if (req.cmd == kCmdAssign) {
          HostPar::CopyBytes(entry.data(), kvs.vals.data() + off, len * sizeof(V));
        } else {
          SumInto(entry.data(), kvs.vals.data() + off, len);
        }
        off += len;
      }
    }
    // plain pull, or the pull half of a fused ZPushPull (push applied
    // above, so the response carries the post-push values)
    if (req.pull) {
      res.keys = kvs.keys;
      if (n == 1) {
        // zero-copy view of the store entry (stable: map nodes don't move)
        auto& entry = store[kvs.keys[0]];
        res.vals = SArray<V>(entry.data(), entry.size(), kCPU);
        SArray<int> lens(1);
        lens[0] = static_cast<int>(entry.size());
        res.lens = lens;
      } else {
        size_t total = 0;
        for (size_t i = 0; i < n; ++i) total += store[kvs.keys[i]].size();
        res.vals.Resize(total);
        SArray<int> lens(n);
        size_t off = 0;
        for (size_t i = 0; i < n; ++i) {
          auto& entry = store[kvs.keys[i]];
          std::copy(entry.begin(), entry.end(), res.vals.data() + off);
          off += entry.size();
          lens[i] = static_cast<int>(entry.size());
        }
        res.lens = lens;
      }
    }
    server->Response(req, res);
  }
  std::unordered_map<Key, std::vector<V>> store;
};

// CPU reduce-round handle: the BytePS round semantics of the GPU
// DenseMode::kReduce handler (server_handlers.cc) with synchronous host
// arithmetic — the first push of a round assigns, the rest accumulate,
// pulls are HELD until every worker pushed, the round resets after every
// worker pulled, and next-round traffic arriving early is deferred. Runs
// the same protocol the GPU path uses, minus streams/events, so the
// round logic is testable without a GPU (and `bench.py --cpu` can run
// the rn50 reduce mode).
template <typename V>
struct KVServerReduceHandle {
  explicit KVServerReduceHandle(int num_workers) : num_workers_(std::max(1, num_workers)) {}

  void operator()(const KVMeta& req, const KVPairs<V>& kvs, KVServer<V>* server) {
    XPS_CHECK(!(req.push && req.pull))
        << "ZPushPull is not supported in reduce mode (pulls are held per round)";
    if (req.push) {
      HandlePush(req, kvs, server);
    } else if (req.pull) {
      HandlePull(req, kvs, server);
    } else {
      server->Response(req);
    }
  }

  // Round state is per KEY-SET ("group"), exactly like the GPU reduce
  // handler: one multi-key message per worker per server per round
  // (BytePS bucketed DenseReduce); single-key messages are groups of 1.
  // All workers must use the same key->message grouping.
  struct Group {
    std::vector<Key> keys;
    std::vector<size_t> lens;  // elements per key
    std::vector<V> buf;        // concatenated reduce buffer
    int pushes = 0;
    int pulls = 0;
    std::vector<KVMeta> waiting_pulls;
    std::vector<KVMeta> waiting_next_pulls;
    std::vector<std::pair<KVMeta, KVPairs<V>>> waiting_pushes;
    std::set<int> pulled;
  };
  std::unordered_map<uint64_t, Group> store;
  int num_workers_;

  Group& GroupFor(const SArray<Key>& keys) {
    uint64_t h = 1469598103934665603ull;
    const unsigned char* p = reinterpret_cast<const unsigned char*>(keys.data());
    for (size_t i = 0; i < keys.nbytes(); ++i) {
      h ^= p[i];
      h *= 1099511628211ull;
    }
    Group& g = store[h];
    if (g.keys.empty()) {
      g.keys.assign(keys.begin(), keys.end());
    } else {
      XPS_CHECK(g.keys.size() == keys.size() &&
                std::equal(g.keys.begin(), g.keys.end(), keys.begin()))
          << "reduce mode needs a consistent key->message grouping across workers";
    }
    return g;
  }

 private:
  void HandlePush(const KVMeta& req, const KVPairs<V>& kvs, KVServer<V>* server) {
    XPS_CHECK(!kvs.vals.on_device()) << "KVServerReduceHandle is CPU-only";
    size_t n = kvs.keys.size();
    Group& g = GroupFor(kvs.keys);
    if (g.pushes >= num_workers_) {
      // a fast worker started the next round before this round's pulls
      // drained (the KVPairs copy keeps the payload alive)
      g.waiting_pushes.emplace_back(req, kvs);
      return;
    }
    if (g.lens.size() != n) {
      g.lens.resize(n);
      for (size_t i = 0; i < n; ++i) {
        g.lens[i] = kvs.lens.empty() ? kvs.vals.size() / n : static_cast<size_t>(kvs.lens[i]);
      }
    }
    size_t total = 0;
    for (size_t l : g.lens) total += l;
    if (g.buf.size() < total) g.buf.resize(total, V(0));
    const V* __restrict__ v = kvs.vals.data();
    V* __restrict__ b = g.buf.data();
    if (g.pushes == 0) {
      HostPar::CopyBytes(b, v, total * sizeof(V));
    } else {
      SumInto(b, v, total);
    }
    g.pushes++;
    server->Response(req);
    if (g.pushes >= num_workers_) {
      std::vector<KVMeta> waiting;
      waiting.swap(g.waiting_pulls);
      for (auto& w : waiting) RespondPull(w, g, server);
    }
  }

  void HandlePull(const KVMeta& req, const KVPairs<V>& kvs, KVServer<V>* server) {
    // a pull may precede the round's pushes (it just waits)
    Group& g = GroupFor(kvs.keys);
    if (g.pulled.count(req.sender)) {
      g.waiting_next_pulls.push_back(req);  // next-round pull, too early
      return;
    }
    if (g.pushes < num_workers_) {
      g.waiting_pulls.push_back(req);  // released by the round's last push
      return;
    }
    RespondPull(req, g, server);
  }

  void RespondPull(const KVMeta& req, Group& g, KVServer<V>* server) {
    KVPairs<V> res;
    res.keys = SArray<Key>(g.keys);
    // synchronous handler: the response is serialized before we return,
    // so a view of buf is safe (and zero-copy on the shm plane)
    res.vals = SArray<V>(g.buf.data(), g.buf.size(), kCPU);
    SArray<int> lens(g.keys.size());
    for (size_t i = 0; i < g.keys.size(); ++i) lens[i] = static_cast<int>(g.lens[i]);
    res.lens = lens;
    server->Response(req, res);
    g.pulled.insert(req.sender);
    g.pulls++;
    if (g.pulls >= num_workers_) {
      g.pushes = 0;
      g.pulls = 0;
      g.pulled.clear();
      std::vector<std::pair<KVMeta, KVPairs<V>>> dpush;
      dpush.swap(g.waiting_pushes);
      for (auto& d : dpush) HandlePush(d.first, d.second, server);
      std::vector<KVMeta> dpull;
      dpull.swap(g.waiting_next_pulls);
      for (auto& d : dpull) {
        if (g.pushes >= num_workers_) {
          RespondPull(d, g, server);
        } else {
          g.waiting_pulls.push_back(d);
        }
      }
    }
  }
};

}  // namespace xps
