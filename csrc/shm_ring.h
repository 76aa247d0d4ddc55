// ShmRing: lock-free cross-process MPMC-producer / single-consumer ring
// in POSIX shared memory — the same-host data-plane metadata channel.
// Every node owns ONE inbound ring (created at van start, named by its
// shm_uid nonce); all same-host peers produce into it. Payload slots
// carry the packed Meta plus small inline blobs; device payloads travel
// as {pool offset, len} references resolved against the sender's
// hipIpc-mapped pool.
//
// Reference parity: replaces the ps-lite IPCTransport POSIX-shm path
// (src/rdma_transport.h:591-617) and the per-message zmq identity frames.
// Enqueue is Vyukov-style: ticket from an atomic head, per-slot sequence
// numbers; the consumer is the single ring-poller thread.
#pragma once

#include <atomic>
#include <cstdint>
#include <string>

namespace xps {

struct RingSlot;

class ShmRing {
 public:
  static const uint32_t kSlotBytes = 12288;  // fits ~1.5K inline keys per slice
  static const uint32_t kSlots = 1024;       // 12 MiB per node

  ~ShmRing();

  // create the inbound ring (owner side); name derived from uid
  bool Create(uint64_t uid);
  // open a peer's ring (producer side)
  bool Open(uint64_t uid);
  void CloseAndUnlink();

  // producer: copy payload into the next slot. Returns false if payload
  // too large; spins briefly if the ring is full (consumer drains fast).
  bool Push(const void* payload, uint32_t len);
  // consumer: pop into buf (size >= max payload). Returns payload len or
  // 0 if empty.
  uint32_t Pop(void* buf);

  static uint32_t MaxPayload() { return kSlotBytes - 16; }
  bool ok() const { return mem_ != nullptr; }

 private:
  void* mem_ = nullptr;
  size_t bytes_ = 0;
  bool owner_ = false;
  std::string name_;
};

}  // namespace xps
