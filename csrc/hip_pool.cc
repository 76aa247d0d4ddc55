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
  if (base_) {
    XPS_CHECK_EQ(device_, device) << "HbmPool already initialized on another device";
    return;
  }
  if (capacity_bytes == 0) {
    capacity_bytes = static_cast<size_t>(Environment::Get()->GetInt("XPS_POOL_GB", 8)) << 30;
  }
  XPS_HIP_CHECK(hipSetDevice(device));
  XPS_HIP_CHECK(hipMalloc(&base_, capacity_bytes));
  device_ = device;
  capacity_ = capacity_bytes;
  static_assert(sizeof(hipIpcMemHandle_t) <= sizeof(ipc_handle_), "ipc handle too large");
  hipIpcMemHandle_t h;
  hipError_t e = hipIpcGetMemHandle(&h, base_);
  if (e == hipSuccess) {
    memcpy(ipc_handle_, &h, sizeof(h));
  } else {
    XPS_LOG(Warning) << "hipIpcGetMemHandle failed (" << hipGetErrorString(e)
                     << "); cross-process zero-copy disabled";
    memset(ipc_handle_, 0, sizeof(ipc_handle_));
  }
  free_[0] = capacity_;
  XPS_VLOG(1) << "HbmPool: " << (capacity_ >> 20) << " MiB on device " << device_;
}

void* HbmPool::Alloc(size_t nbytes) {
  XPS_CHECK(base_) << "HbmPool not initialized";
  nbytes = (nbytes + kAlign - 1) & ~(kAlign - 1);
  std::lock_guard<std::mutex> lk(mu_);
  for (auto it = free_.begin(); it != free_.end(); ++it) {
    if (it->second >= nbytes) {
      size_t off = it->first;
      size_t rest = it->second - nbytes;
      free_.erase(it);
      if (rest) free_[off + nbytes] = rest;
      used_[off] = nbytes;
      return static_cast<char*>(base_) + off;
    }
  }
  XPS_LOG(Fatal) << "HbmPool exhausted: want " << nbytes << " bytes, capacity " << capacity_;
  return nullptr;
}

void HbmPool::Free(void* p) {
  std::lock_guard<std::mutex> lk(mu_);
  size_t off = static_cast<char*>(p) - static_cast<char*>(base_);
  auto it = used_.find(off);
  XPS_CHECK(it != used_.end()) << "HbmPool::Free of unknown pointer";
  size_t size = it->second;
  used_.erase(it);
  // coalesce with neighbors
  auto next = free_.upper_bound(off);
  if (next != free_.end() && off + size == next->first) {
    size += next->second;
    next = free_.erase(next);
  }
  if (next != free_.begin()) {
    auto prev = std::prev(next);
    if (prev->first + prev->second == off) {
      prev->second += size;
      return;
    }
  }
  free_[off] = size;
}

bool HbmPool::OffsetOf(const void* p, uint64_t* off) const {
  if (!base_) return false;
  const char* c = static_cast<const char*>(p);
  const char* b = static_cast<const char*>(base_);
  if (c < b || c >= b + capacity_) return false;
  *off = static_cast<uint64_t>(c - b);
  return true;
}

SArray<char> HbmPool::AllocArray(size_t nbytes) {
  char* p = static_cast<char*>(Alloc(nbytes));
  return SArray<char>(p, nbytes, [](char* q) { HbmPool::Get()->Free(q); }, device_);
}

size_t HbmPool::bytes_in_use() const {
  std::lock_guard<std::mutex> lk(mu_);
  size_t total = 0;
  for (auto& kv : used_) total += kv.second;
  return total;
}

}  // namespace xps
