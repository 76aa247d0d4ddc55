// HbmPool: the process-wide HBM arena every device SArray comes from.
//
// MI355X-native replacement for ps-lite's lazy per-pointer ibv_reg_mr
// cache (src/rdma_van.h:520-548) and the rdma_utils.h MemoryAllocator:
// one hipMalloc slab per process, exported ONCE via hipIpcMemHandle at
// bootstrap (it rides the ADD_NODE Node record), so steady-state
// transfers carry only {offset, len} — no registration, no rendezvous.
// Sized for 288 GB HBM3E per GPU (default pool 8 GiB, growable via
// XPS_POOL_GB before Start).
#pragma once

#include <cstddef>
#include <cstdint>
#include <map>
#include <mutex>

#include "sarray.h"

namespace xps {

class HbmPool {
 public:
  static HbmPool* Get();

  // Allocate the slab on `device` (idempotent; first call wins).
  void Init(int device, size_t capacity_bytes = 0);
  bool initialized() const { return base_ != nullptr; }
  int device() const { return device_; }
  void* base() const { return base_; }
  size_t capacity() const { return capacity_; }
  const char* ipc_handle() const { return ipc_handle_; }

  void* Alloc(size_t nbytes);
  void Free(void* p);
  // true iff p lies inside the pool; fills byte offset from base
  bool OffsetOf(const void* p, uint64_t* off) const;

  // SArray drawing from the pool (freed back on last release)
  SArray<char> AllocArray(size_t nbytes);
  template <typename V>
  SArray<V> AllocTyped(size_t count) {
    return SArray<V>::View(AllocArray(count * sizeof(V)));
  }

  size_t bytes_in_use() const;

 private:
  HbmPool() = default;
  int device_ = -1;
  void* base_ = nullptr;
  size_t capacity_ = 0;
  char ipc_handle_[64] = {0};
  mutable std::mutex mu_;
  std::map<size_t, size_t> free_;   // offset -> size (coalesced free list)
  std::map<size_t, size_t> used_;   // offset -> size
};

}  // namespace xps
