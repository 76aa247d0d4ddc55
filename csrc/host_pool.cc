#include "host_pool.h"

#include <dirent.h>
#include <fcntl.h>
#include <signal.h>
#include <sys/mman.h>
#include <unistd.h>

#include <cstdio>
#include <unordered_map>

#include "base.h"

namespace xps {

static const size_t kAlign = 64;
// 64-byte arena header: {magic, owner pid}; allocations start at offset
// 64 so peer-visible offsets stay consistent on both sides
static const size_t kHeaderBytes = 64;
static const uint64_t kPoolMagic = 0x587053686d506f31ull;  // "XpShmPo1"


static std::string PoolName(uint64_t uid) {
  char buf[64];
  snprintf(buf, sizeof(buf), "/xps_hostpool_%016llx", static_cast<unsigned long long>(uid));
  return buf;
}

HostShmPool* HostShmPool::Get() {
  static HostShmPool pool;
  return &pool;
}

void HostShmPool::Init(uint64_t uid, size_t capacity_bytes) {
  std::lock_guard<std::mutex> lk(mu_);
  if (base_) return;
  if (capacity_bytes == 0) {
    capacity_bytes = static_cast<size_t>(Environment::Get()->GetInt("XPS_HOST_POOL_GB", 1)) << 30;
  }
  std::string name = PoolName(uid);
  shm_unlink(name.c_str());
  int fd = shm_open(name.c_str(), O_CREAT | O_EXCL | O_RDWR, 0600);
  XPS_CHECK_GE(fd, 0) << "shm_open " << name;
  XPS_CHECK_EQ(ftruncate(fd, capacity_bytes), 0);
  base_ = mmap(nullptr, capacity_bytes, PROT_READ | PROT_WRITE, MAP_SHARED, fd, 0);
  close(fd);
  XPS_CHECK(base_ != MAP_FAILED) << "mmap host pool";
  uid_ = uid;
  capacity_ = capacity_bytes;
  uint64_t* hdr = static_cast<uint64_t*>(base_);
  hdr[1] = static_cast<uint64_t>(getpid());
  __atomic_store_n(&hdr[0], kPoolMagic, __ATOMIC_RELEASE);
  free_[kHeaderBytes] = capacity_ - kHeaderBytes;
  XPS_VLOG(1) << "HostShmPool: " << (capacity_ >> 20) << " MiB (" << name << ")";
}

void* HostShmPool::Alloc(size_t nbytes) {
  XPS_CHECK(base_) << "HostShmPool not initialized";
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
  XPS_LOG(Fatal) << "HostShmPool exhausted (capacity " << capacity_
                 << "); raise XPS_HOST_POOL_GB";
  return nullptr;
}

void HostShmPool::Free(void* p) {
  std::lock_guard<std::mutex> lk(mu_);
  size_t off = static_cast<char*>(p) - static_cast<char*>(base_);
  auto it = used_.find(off);
  XPS_CHECK(it != used_.end()) << "HostShmPool::Free of unknown pointer";
  size_t size = it->second;
  used_.erase(it);
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

bool HostShmPool::OffsetOf(const void* p, uint64_t* off) const {
  if (!base_) return false;
  const char* c = static_cast<const char*>(p);
  const char* b = static_cast<const char*>(base_);
  if (c < b || c >= b + capacity_) return false;
  *off = static_cast<uint64_t>(c - b);
  return true;
}

SArray<char> HostShmPool::AllocArray(size_t nbytes) {
  char* p = static_cast<char*>(Alloc(nbytes));
  return SArray<char>(p, nbytes, [](char* q) { HostShmPool::Get()->Free(q); }, kCPU);
}

void* HostShmPool::MapPeer(uint64_t uid, size_t capacity) {
  static std::mutex map_mu;
  static std::unordered_map<uint64_t, void*> mapped;
  std::lock_guard<std::mutex> lk(map_mu);
  auto it = mapped.find(uid);
  if (it != mapped.end()) return it->second;
  auto* own = Get();
  if (own->initialized() && own->uid() == uid) {
    mapped[uid] = own->base_;
    return own->base_;
  }
  int fd = shm_open(PoolName(uid).c_str(), O_RDWR, 0600);
  if (fd < 0) {
    mapped[uid] = nullptr;
    return nullptr;
  }
  void* base = mmap(nullptr, capacity, PROT_READ | PROT_WRITE, MAP_SHARED, fd, 0);
  close(fd);
  if (base == MAP_FAILED) base = nullptr;
  mapped[uid] = base;
  return base;
}

void HostShmPool::GcStaleSegments() {
  DIR* d = opendir("/dev/shm");
  if (!d) return;
  // ring header layout (shm_ring.cc): head@0, tail@64, magic@128, pid@192
  static const uint64_t kRingMagic = 0x587052696e673166ull;
  struct dirent* ent;
  while ((ent = readdir(d)) != nullptr) {
    bool pool = strncmp(ent->d_name, "xps_hostpool_", 13) == 0;
    bool ring = strncmp(ent->d_name, "xps_ring_", 9) == 0;
    if (!pool && !ring) continue;
    std::string name = std::string("/") + ent->d_name;
    int fd = shm_open(name.c_str(), O_RDONLY, 0600);
    if (fd < 0) continue;
    void* m = mmap(nullptr, 4096, PROT_READ, MAP_SHARED, fd, 0);
    close(fd);
    if (m == MAP_FAILED) continue;
    const uint64_t* w = static_cast<const uint64_t*>(m);
    uint64_t magic = pool ? w[0] : w[16];   // ring magic at byte 128
    uint64_t pid = pool ? w[1] : w[24];     // ring pid at byte 192
    bool dead = magic == (pool ? kPoolMagic : kRingMagic) && pid > 0 &&
                kill(static_cast<pid_t>(pid), 0) == -1 && errno == ESRCH;
    munmap(m, 4096);
    if (dead) {
      XPS_VLOG(1) << "GC stale shm segment " << name << " (owner " << pid << " dead)";
      shm_unlink(name.c_str());
    }
  }
  closedir(d);
}

void HostShmPool::Unlink() {
  if (uid_) shm_unlink(PoolName(uid_).c_str());
}

}  // namespace xps
