// HostShmPool: the host-memory sibling of HbmPool — a POSIX-shm arena
// every same-host peer can map, so HOST payloads ride the data plane
// zero-copy too (ps-lite IPCTransport parity for CPU clusters: the
// BASELINE config-#1 path). Addresses on the wire are byte offsets; the
// segment is named from the owner's shm_uid nonce.
#pragma once

#include <cstddef>
#include <cstdint>
#include <map>
#include <mutex>

#include "sarray.h"

namespace xps {

class HostShmPool {
 public:
  static HostShmPool* Get();

  // create the arena (idempotent); named by `uid` (the node's shm_uid)
  void Init(uint64_t uid, size_t capacity_bytes = 0);
  bool initialized() const { return base_ != nullptr; }
  uint64_t uid() const { return uid_; }
  size_t capacity() const { return capacity_; }

  void* Alloc(size_t nbytes);
  void Free(void* p);
  bool OffsetOf(const void* p, uint64_t* off) const;
  SArray<char> AllocArray(size_t nbytes);

  // map a PEER's arena by uid (cached); returns nullptr on failure
  static void* MapPeer(uint64_t uid, size_t capacity);

  void Unlink();

  // unlink any /dev/shm xps segment (host pool or ring) whose recorded
  // owner process is dead — crashed runs (SIGKILL, test teardown) leak
  // segments that no teardown path can reach
  static void GcStaleSegments();

 private:
  HostShmPool() = default;
  uint64_t uid_ = 0;
  void* base_ = nullptr;
  size_t capacity_ = 0;
  mutable std::mutex mu_;
  std::map<size_t, size_t> free_;
  std::map<size_t, size_t> used_;
};

}  // namespace xps
