#include "shm_ring.h"

#include <fcntl.h>
#include <sys/mman.h>
#include <sys/stat.h>
#include <unistd.h>

#include <cstdio>
#include <cstring>
#include <thread>

#include "base.h"

namespace xps {

namespace {

struct RingHeader {
  alignas(64) std::atomic<uint64_t> head;  // producer position
  alignas(64) std::atomic<uint64_t> tail;  // consumer position
  alignas(64) uint64_t magic;
  alignas(64) uint64_t owner_pid;  // for stale-segment GC (host_pool.cc)
};
static const uint64_t kRingMagic = 0x587052696e673166ull;  // "XpRing1f"

}  // namespace

struct RingSlot {
  std::atomic<uint64_t> seq;
  uint32_t len;
  uint32_t pad;
  char payload[ShmRing::kSlotBytes - 16];
};

static RingHeader* Hdr(void* mem) { return static_cast<RingHeader*>(mem); }
static RingSlot* Slot(void* mem, uint64_t i) {
  return reinterpret_cast<RingSlot*>(static_cast<char*>(mem) + 4096 +
                                     (i % ShmRing::kSlots) * sizeof(RingSlot));
}

static std::string RingName(uint64_t uid) {
  char buf[64];
  snprintf(buf, sizeof(buf), "/xps_ring_%016llx", static_cast<unsigned long long>(uid));
  return buf;
}

ShmRing::~ShmRing() {
  if (mem_) munmap(mem_, bytes_);
  mem_ = nullptr;
}

bool ShmRing::Create(uint64_t uid) {
  name_ = RingName(uid);
  bytes_ = 4096 + static_cast<size_t>(kSlots) * sizeof(RingSlot);
  shm_unlink(name_.c_str());  // stale segment from a crashed run
  int fd = shm_open(name_.c_str(), O_CREAT | O_EXCL | O_RDWR, 0600);
  if (fd < 0) {
    XPS_LOG(Warning) << "shm_open create failed for " << name_;
    return false;
  }
  if (ftruncate(fd, bytes_) != 0) {
    close(fd);
    return false;
  }
  mem_ = mmap(nullptr, bytes_, PROT_READ | PROT_WRITE, MAP_SHARED, fd, 0);
  close(fd);
  if (mem_ == MAP_FAILED) {
    mem_ = nullptr;
    return false;
  }
  memset(mem_, 0, bytes_);
  for (uint64_t i = 0; i < kSlots; ++i) Slot(mem_, i)->seq.store(i, std::memory_order_relaxed);
  Hdr(mem_)->head.store(0);
  Hdr(mem_)->tail.store(0);
  Hdr(mem_)->owner_pid = static_cast<uint64_t>(getpid());
  __atomic_store_n(&Hdr(mem_)->magic, kRingMagic, __ATOMIC_RELEASE);
  owner_ = true;
  return true;
}

bool ShmRing::Open(uint64_t uid) {
  name_ = RingName(uid);
  bytes_ = 4096 + static_cast<size_t>(kSlots) * sizeof(RingSlot);
  int fd = -1;
  for (int i = 0; i < 200; ++i) {  // the owner may still be creating it
    fd = shm_open(name_.c_str(), O_RDWR, 0600);
    if (fd >= 0) break;
    usleep(10 * 1000);
  }
  if (fd < 0) return false;
  mem_ = mmap(nullptr, bytes_, PROT_READ | PROT_WRITE, MAP_SHARED, fd, 0);
  close(fd);
  if (mem_ == MAP_FAILED) {
    mem_ = nullptr;
    return false;
  }
  for (int i = 0; i < 200; ++i) {
    if (__atomic_load_n(&Hdr(mem_)->magic, __ATOMIC_ACQUIRE) == kRingMagic) return true;
    usleep(10 * 1000);
  }
  munmap(mem_, bytes_);
  mem_ = nullptr;
  return false;
}

void ShmRing::CloseAndUnlink() {
  if (mem_) {
    munmap(mem_, bytes_);
    mem_ = nullptr;
  }
  if (owner_ && !name_.empty()) shm_unlink(name_.c_str());
}

bool ShmRing::Push(const void* payload, uint32_t len) {
  if (len > MaxPayload() || !mem_) return false;
  // Vyukov MPMC enqueue: reserve a position ONLY once its slot is free
  // (seq == pos, then CAS head). A ticket fetch_add cannot back out on a
  // full ring: dropping would leave a hole the single consumer waits on
  // forever. With CAS, a bounded wait (~10 s) on a full ring can give up
  // without consuming a sequence number — the ring stays usable.
  RingHeader* h = Hdr(mem_);
  uint64_t pos = h->head.load(std::memory_order_relaxed);
  int spins = 0;
  RingSlot* s;
  for (;;) {
    s = Slot(mem_, pos);
    uint64_t seq = s->seq.load(std::memory_order_acquire);
    int64_t dif = static_cast<int64_t>(seq) - static_cast<int64_t>(pos);
    if (dif == 0) {
      if (h->head.compare_exchange_weak(pos, pos + 1, std::memory_order_relaxed)) break;
      // pos was reloaded by the failed CAS; retry
    } else if (dif < 0) {
      // ring full: the consumer has not freed this slot yet
      if (++spins > 1000) {
        std::this_thread::yield();
        if (spins > 10 * 1000 * 1000) {
          XPS_LOG(Warning) << "shm ring full for too long (consumer dead?); dropping message";
          return false;
        }
      }
      pos = h->head.load(std::memory_order_relaxed);
    } else {
      pos = h->head.load(std::memory_order_relaxed);
    }
  }
  memcpy(s->payload, payload, len);
  s->len = len;
  s->seq.store(pos + 1, std::memory_order_release);
  return true;
}

uint32_t ShmRing::Pop(void* buf) {
  if (!mem_) return 0;
  uint64_t pos = Hdr(mem_)->tail.load(std::memory_order_relaxed);
  RingSlot* s = Slot(mem_, pos);
  if (s->seq.load(std::memory_order_acquire) != pos + 1) return 0;  // empty
  uint32_t len = s->len;
  memcpy(buf, s->payload, len);
  s->seq.store(pos + kSlots, std::memory_order_release);  // free for lap head+kSlots
  Hdr(mem_)->tail.store(pos + 1, std::memory_order_relaxed);
  return len;
}

}  // namespace xps
