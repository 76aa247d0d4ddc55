// HbmPool: the process-wide HBM arena every device SArray comes from.
//
// MI355X-native replacement for ps-lite's lazy per-pointer ibv_reg_mr
// cache (src/rdma_van.h:520-548) and the rdma_utils.h MemoryAllocator:
// hipMalloc slabs exported ONCE via hipIpcMemHandle at bootstrap (they
// ride the ADD_NODE Node record), so steady-state transfers carry only
// {global offset, len} — no registration, no rendezvous.
//
// The pool is MULTI-SLAB: hipIpcOpenMemHandle deadlocks inside the ROCm
// runtime for allocations >= 2 GiB (measured on this stack: 2037 MiB
// imports instantly, 2048 MiB hangs), so the arena is carved from
// <= 1 GiB hipMalloc slabs, all allocated eagerly at Init so the slab
// table is complete before it is broadcast. Addresses on the wire are
// GLOBAL byte offsets over the concatenated slabs; a single allocation
// never spans slabs.
#pragma once

#include <cstddef>
#include <cstdint>
#include <map>
#include <mutex>
#include <vector>

#include "sarray.h"

namespace xps {

class HbmPool {
 public:
  static const size_t kDefaultSlabBytes = 1ull << 30;  // 1 GiB (< 2 GiB ipc limit)

  static HbmPool* Get();

  // Allocate all slabs on `device` (idempotent; first call wins).
  void Init(int device, size_t capacity_bytes = 0);
  bool initialized() const { return !slabs_.empty(); }
  int device() const { return device_; }
  size_t capacity() const { return capacity_; }

  // bootstrap-exported slabs only (growth slabs are local, see below)
  size_t slab_count() const { return exported_slabs_; }
  size_t slab_bytes() const { return slab_bytes_; }
  uint64_t slab_capacity(size_t i) const { return slabs_[i].capacity; }
  const char* slab_handle(size_t i) const { return slabs_[i].ipc_handle; }
  void* slab_base(size_t i) const { return slabs_[i].base; }

  void* Alloc(size_t nbytes);
  void Free(void* p);
  // true iff p lies inside the pool; fills the GLOBAL byte offset
  bool OffsetOf(const void* p, uint64_t* global_off) const;

  // SArray drawing from the pool (freed back on last release)
  SArray<char> AllocArray(size_t nbytes);
  template <typename V>
  SArray<V> AllocTyped(size_t count) {
    return SArray<V>::View(AllocArray(count * sizeof(V)));
  }

  size_t bytes_in_use() const;

 private:
  struct Slab {
    void* base = nullptr;
    uint64_t capacity = 0;
    uint64_t global_begin = 0;  // global offset of byte 0
    char ipc_handle[64] = {0};
    // slabs allocated AFTER bootstrap (on-demand growth) are not in any
    // peer's slab table, so they must never be referenced by global
    // offset on the wire — OffsetOf skips them and traffic from them
    // takes the TCP staging path (slow but correct)
    bool exported = true;
    std::map<size_t, size_t> free_;  // local offset -> size
    std::map<size_t, size_t> used_;
  };

  // allocate one more (non-exported) slab; called on exhaustion.
  // Returns false if hipMalloc fails. Caller holds mu_.
  bool GrowLocked();

  HbmPool() = default;
  int device_ = -1;
  size_t capacity_ = 0;
  size_t slab_bytes_ = kDefaultSlabBytes;
  size_t exported_slabs_ = 0;  // slabs advertised at bootstrap
  mutable std::mutex mu_;
  std::vector<Slab> slabs_;
};

}  // namespace xps
