#include "hip_pool.h"

#include <hip/hip_runtime.h>

#include "base.h"

namespace xps {

#define XPS_HIP_CHECK(cmd)                                                            \
  do {                                                                                \
    hipError_t e_ = (cmd);                                                            \
    XPS_CHECK(e_ == hipSuccess) << "HIP error: " << hipGetErrorString(e_) << " in " #cmd; \
  } while (0)

static const size_t kAlign = 512;

HbmPool* HbmPool::Get() {
  static HbmPool pool;
  return &pool;
}

void HbmPool::Init(int device, size_t capacity_bytes) {
  std::lock_guard<std::mutex> lk(mu_);
  if (!slabs_.empty()) {
    XPS_CHECK_EQ(device_, device) << "HbmPool already initialized on another device";
    return;
  }
  auto* env = Environment::Get();
  if (capacity_bytes == 0) {
    capacity_bytes = static_cast<size_t>(env->GetInt("XPS_POOL_GB", 8)) << 30;
  }
  slab_bytes_ = static_cast<size_t>(env->GetInt64("XPS_SLAB_BYTES", kDefaultSlabBytes));
  XPS_CHECK_LT(slab_bytes_, 2ull << 30)
      << "slabs must stay under 2 GiB (hipIpcOpenMemHandle hangs at >= 2 GiB)";
  size_t nslabs = (capacity_bytes + slab_bytes_ - 1) / slab_bytes_;
  XPS_HIP_CHECK(hipSetDevice(device));
  device_ = device;
  uint64_t global = 0;
  for (size_t i = 0; i < nslabs; ++i) {
    Slab s;
    s.capacity = slab_bytes_;
    XPS_HIP_CHECK(hipMalloc(&s.base, s.capacity));
    s.global_begin = global;
    global += s.capacity;
    static_assert(sizeof(hipIpcMemHandle_t) <= sizeof(s.ipc_handle), "ipc handle too large");
    hipIpcMemHandle_t h;
    hipError_t e = hipIpcGetMemHandle(&h, s.base);
    if (e == hipSuccess) {
      memcpy(s.ipc_handle, &h, sizeof(h));
    } else {
      XPS_LOG(Warning) << "hipIpcGetMemHandle failed (" << hipGetErrorString(e)
                       << "); cross-process zero-copy disabled for slab " << i;
      memset(s.ipc_handle, 0, sizeof(s.ipc_handle));
    }
    s.free_[0] = s.capacity;
    slabs_.push_back(std::move(s));
  }
  capacity_ = global;
  exported_slabs_ = slabs_.size();
  XPS_VLOG(1) << "HbmPool: " << (capacity_ >> 20) << " MiB in " << slabs_.size()
              << " slabs on device " << device_;
}

bool HbmPool::GrowLocked() {
  Slab s;
  s.capacity = slab_bytes_;
  if (hipSetDevice(device_) != hipSuccess) return false;
  hipError_t e = hipMalloc(&s.base, s.capacity);
  if (e != hipSuccess) {
    XPS_LOG(Warning) << "HbmPool growth failed: hipMalloc(" << s.capacity
                     << ") -> " << hipGetErrorString(e);
    return false;
  }
  // growth slabs live OUTSIDE the advertised global-offset space: peers
  // never learned their ipc handles, so OffsetOf must not match them
  // (by-ref sends from here fall back to TCP staging)
  s.global_begin = ~0ull;
  s.exported = false;
  s.free_[0] = s.capacity;
  slabs_.push_back(std::move(s));
  XPS_LOG(Warning) << "HbmPool grown by " << (slab_bytes_ >> 20)
                   << " MiB (local-only slab " << slabs_.size() - 1
                   << "; cross-process zero-copy uses the bootstrap slabs — raise "
                      "XPS_POOL_GB to keep everything on the fast path)";
  return true;
}

void* HbmPool::Alloc(size_t nbytes) {
  XPS_CHECK(!slabs_.empty()) << "HbmPool not initialized";
  nbytes = (nbytes + kAlign - 1) & ~(kAlign - 1);
  if (nbytes > slab_bytes_) {
    // oversized (> the <2 GiB hipIpc slab ceiling): dedicated local-only
    // slab — cannot be zero-copy-shared, but torch-allocator users may
    // hold multi-GiB tensors that never cross the wire
    XPS_LOG(Warning) << "pool allocation of " << (nbytes >> 20)
                     << " MiB exceeds the exported slab size (" << (slab_bytes_ >> 20)
                     << " MiB, a <2 GiB hipIpc limit): it gets a LOCAL-ONLY slab and "
                        "any wire traffic from it stages over TCP — split the buffer "
                        "into <= slab-sized pieces to stay zero-copy";
    std::lock_guard<std::mutex> lk(mu_);
    Slab s;
    s.capacity = nbytes;
    XPS_HIP_CHECK(hipSetDevice(device_));
    hipError_t e = hipMalloc(&s.base, s.capacity);
    XPS_CHECK(e == hipSuccess) << "hipMalloc(" << nbytes << ") failed: "
                               << hipGetErrorString(e);
    s.global_begin = ~0ull;
    s.exported = false;
    s.used_[0] = nbytes;
    slabs_.push_back(std::move(s));
    return slabs_.back().base;
  }
  std::lock_guard<std::mutex> lk(mu_);
  for (int attempt = 0; attempt < 2; ++attempt) {
    for (auto& slab : slabs_) {
      for (auto it = slab.free_.begin(); it != slab.free_.end(); ++it) {
        if (it->second >= nbytes) {
          size_t off = it->first;
          size_t rest = it->second - nbytes;
          slab.free_.erase(it);
          if (rest) slab.free_[off + nbytes] = rest;
          slab.used_[off] = nbytes;
          return static_cast<char*>(slab.base) + off;
        }
      }
    }
    // exhausted: grow by one local-only slab and retry once (288 GB of
    // HBM3E means the device almost always has room — only the
    // bootstrap-advertised zero-copy window is fixed)
    if (attempt == 0 && !GrowLocked()) break;
  }
  size_t in_use = 0;
  for (auto& slab : slabs_) {
    for (auto& kv : slab.used_) in_use += kv.second;
  }
  XPS_LOG(Fatal) << "HbmPool exhausted: want " << nbytes << " bytes, capacity " << capacity_
                 << " (in use " << in_use << ") and growth failed; raise XPS_POOL_GB";
  return nullptr;
}

void HbmPool::Free(void* p) {
  std::lock_guard<std::mutex> lk(mu_);
  for (auto& slab : slabs_) {
    const char* b = static_cast<const char*>(slab.base);
    const char* c = static_cast<const char*>(p);
    if (c < b || c >= b + slab.capacity) continue;
    size_t off = c - b;
    auto it = slab.used_.find(off);
    XPS_CHECK(it != slab.used_.end()) << "HbmPool::Free of unknown pointer";
    size_t size = it->second;
    slab.used_.erase(it);
    auto next = slab.free_.upper_bound(off);
    if (next != slab.free_.end() && off + size == next->first) {
      size += next->second;
      next = slab.free_.erase(next);
    }
    if (next != slab.free_.begin()) {
      auto prev = std::prev(next);
      if (prev->first + prev->second == off) {
        prev->second += size;
        return;
      }
    }
    slab.free_[off] = size;
    return;
  }
  XPS_LOG(Fatal) << "HbmPool::Free of pointer outside the pool";
}

bool HbmPool::OffsetOf(const void* p, uint64_t* global_off) const {
  const char* c = static_cast<const char*>(p);
  for (auto& slab : slabs_) {
    const char* b = static_cast<const char*>(slab.base);
    if (c >= b && c < b + slab.capacity) {
      // growth slabs are unknown to peers: no wire-referenceable offset
      if (!slab.exported) return false;
      *global_off = slab.global_begin + static_cast<uint64_t>(c - b);
      return true;
    }
  }
  return false;
}

SArray<char> HbmPool::AllocArray(size_t nbytes) {
  char* p = static_cast<char*>(Alloc(nbytes));
  return SArray<char>(p, nbytes, [](char* q) { HbmPool::Get()->Free(q); }, device_);
}

size_t HbmPool::bytes_in_use() const {
  std::lock_guard<std::mutex> lk(mu_);
  size_t total = 0;
  for (auto& slab : slabs_) {
    for (auto& kv : slab.used_) total += kv.second;
  }
  return total;
}

}  // namespace xps

// ---- torch pluggable-allocator hooks --------------------------------
// Installed via torch.cuda.memory.CUDAPluggableAllocator(_core.so,
// "xps_torch_alloc", "xps_torch_free"): every torch CUDA tensor then
// lives in the HbmPool, so plain torch tensors ride the zero-copy
// hipIpc plane (the reference's PinMemory/RegisterRecvBuffer use case —
// ucx_van.h:603-623, kv_app.h:488 — without per-buffer registration).
extern "C" {

void* xps_torch_alloc(ssize_t size, int device, hipStream_t stream) {
  (void)stream;
  if (size <= 0) return nullptr;
  auto* pool = xps::HbmPool::Get();
  if (!pool->initialized()) pool->Init(device);
  XPS_CHECK_EQ(pool->device(), device)
      << "HbmPool torch allocator: one pool per process (one process per GPU)";
  return pool->Alloc(static_cast<size_t>(size));
}

void xps_torch_free(void* ptr, ssize_t size, int device, hipStream_t stream) {
  (void)size;
  (void)device;
  if (!ptr) return;
  // stream-ordered free: the tensor's stream may still have kernels
  // reading this memory — release the region only once they finished
  if (stream) {
    hipError_t e = hipLaunchHostFunc(
        stream, [](void* p) { xps::HbmPool::Get()->Free(p); }, ptr);
    if (e == hipSuccess) return;
  }
  (void)hipDeviceSynchronize();
  xps::HbmPool::Get()->Free(ptr);
}

}  // extern "C"
