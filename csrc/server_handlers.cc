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

static hipStream_t PeerStream(Postoffice* po, int sender, hipStream_t* fallback) {
  auto* plane = dynamic_cast<GpuPlane*>(po->van() ? po->van()->plane() : nullptr);
  if (plane) return plane->StreamForPeer(sender);
  if (!*fallback) {
    XPS_HIP_CHECK(hipStreamCreateWithFlags(fallback, hipStreamNonBlocking));
  }
  return *fallback;
}

// ------------------------------------------------------------------ dense

GpuDenseHandler::GpuDenseHandler(Postoffice* po, bool default_sum)
    : po_(po), default_sum_(default_sum) {
  XPS_CHECK(HbmPool::Get()->initialized()) << "GpuDenseHandler needs the HBM pool";
}

hipStream_t GpuDenseHandler::Stream(int sender) { return PeerStream(po_, sender, &fallback_stream_); }

void GpuDenseHandler::operator()(const KVMeta& req, const KVPairs<float>& kvs,
                                 KVServer<float>* server) {
  if (req.push) {
    HandlePush(req, kvs, server);
  } else if (req.pull) {
    HandlePull(req, kvs, server);
  } else {
    server->Response(req);
  }
}

void GpuDenseHandler::HandlePush(const KVMeta& req, const KVPairs<float>& kvs,
                                 KVServer<float>* server) {
  size_t n = kvs.keys.size();
  XPS_CHECK_GT(n, 0u);
  bool sum = req.cmd == kCmdSum || (req.cmd == kCmdDefault && default_sum_);
  hipStream_t stream = Stream(req.sender);
  XPS_HIP_CHECK(hipSetDevice(HbmPool::Get()->device()));
  size_t off = 0;  // bytes into vals
  bool synced = false;
  for (size_t i = 0; i < n; ++i) {
    size_t len = kvs.lens.empty() ? kvs.vals.nbytes() / n
                                  : static_cast<size_t>(kvs.lens[i]) * sizeof(float);
    SArray<char> entry;
    {
      std::lock_guard<std::mutex> lk(mu_);
      auto& e = store_[kvs.keys[i]];
      if (e.size() < len) {
        e = HbmPool::Get()->AllocArray(len);
        // zero before publishing: another peer's stream may accumulate
        // into this entry concurrently with our first push
        XPS_HIP_CHECK(hipMemset(e.data(), 0, len));
      }
      entry = e;
    }
    const char* src = reinterpret_cast<const char*>(kvs.vals.data()) + off;
    if (kvs.vals.on_device()) {
      if (sum) {
        kern::DenseSumF32(reinterpret_cast<float*>(entry.data()),
                          reinterpret_cast<const float*>(src), len / sizeof(float), stream);
      } else {
        kern::DenseAssign(entry.data(), src, len, stream);
      }
    } else {
      // host vals (TCP-staged path): correctness-first synchronous route
      if (sum) {
        SArray<char> scratch = HbmPool::Get()->AllocArray(len);
        XPS_HIP_CHECK(hipMemcpy(scratch.data(), src, len, hipMemcpyHostToDevice));
        kern::DenseSumF32(reinterpret_cast<float*>(entry.data()),
                          reinterpret_cast<const float*>(scratch.data()), len / sizeof(float),
                          stream);
        XPS_HIP_CHECK(hipStreamSynchronize(stream));
        synced = true;
      } else {
        XPS_HIP_CHECK(hipMemcpy(entry.data(), src, len, hipMemcpyHostToDevice));
      }
    }
    off += len;
  }
  // if the response cannot ride the plane (deferred behind this stream),
  // the worker may reuse its buffer as soon as the ack arrives — sync first
  auto* plane = po_->van() ? po_->van()->plane() : nullptr;
  if (!plane && !synced) XPS_HIP_CHECK(hipStreamSynchronize(stream));
  server->Response(req);
}

void GpuDenseHandler::HandlePull(const KVMeta& req, const KVPairs<float>& kvs,
                                 KVServer<float>* server) {
  size_t n = kvs.keys.size();
  XPS_CHECK_GT(n, 0u);
  hipStream_t stream = Stream(req.sender);
  XPS_HIP_CHECK(hipSetDevice(HbmPool::Get()->device()));
  KVPairs<float> res;
  res.keys = kvs.keys;
  SArray<int> lens(n);
  if (n == 1) {
    SArray<char> entry;
    {
      std::lock_guard<std::mutex> lk(mu_);
      auto it = store_.find(kvs.keys[0]);
      XPS_CHECK(it != store_.end()) << "pull of unknown key " << kvs.keys[0];
      entry = it->second;
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
        entries[i] = it->second;
        lens[i] = static_cast<int>(entries[i].size() / sizeof(float));
        total += entries[i].size();
      }
    }
    SArray<char> tmp = HbmPool::Get()->AllocArray(total);
    size_t off = 0;
    for (size_t i = 0; i < n; ++i) {
      kern::DenseAssign(tmp.data() + off, entries[i].data(), entries[i].size(), stream);
      off += entries[i].size();
    }
    res.vals = SArray<float>::View(tmp);  // plane keeps it alive until sent
  }
  res.lens = lens;
  // TCP fallback (no in-place destination): the staging D2H copy below in
  // the van is stream-unaware — drain our stream first
  if (!(req.option & kOptPullAddr)) XPS_HIP_CHECK(hipStreamSynchronize(stream));
  server->Response(req, res);
}

// ----------------------------------------------------------------- sparse

GpuSparseHandler::GpuSparseHandler(Postoffice* po, size_t rows, size_t row_len, bool accumulate,
                                   int key_shift)
    : po_(po), rows_(rows), row_len_(row_len), accumulate_(accumulate), key_shift_(key_shift) {
  auto* pool = HbmPool::Get();
  XPS_CHECK(pool->initialized()) << "GpuSparseHandler needs the HBM pool";
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
      kern::SparseScatterAddF32(table, rows, n, row_len_, kvs.vals.data(), /*atomic=*/false,
                                stream, key_shift_, row_base_);
    } else {
      kern::SparseScatterAssignF32(table, rows, n, row_len_, kvs.vals.data(), stream, key_shift_,
                                   row_base_);
    }
    auto* plane = po_->van() ? po_->van()->plane() : nullptr;
    if (!plane) XPS_HIP_CHECK(hipStreamSynchronize(stream));
    server->Response(req);
  } else if (req.pull) {
    const uint64_t* rows = DeviceKeys(kvs.keys, req.sender, stream);
    SArray<char> out = pool->AllocArray(n * row_len_ * sizeof(float));
    kern::SparseGatherF32(table, rows, n, row_len_, reinterpret_cast<float*>(out.data()), stream,
                          key_shift_, row_base_);
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
