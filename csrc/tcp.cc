#include "tcp.h"

#include <arpa/inet.h>
#include <errno.h>
#include <ifaddrs.h>
#include <netdb.h>
#include <netinet/in.h>
#include <netinet/tcp.h>
#include <sys/socket.h>
#include <sys/uio.h>
#include <unistd.h>

#include <cstring>
#include <functional>

namespace xps {

TcpConn::TcpConn(int fd) : fd_(fd) {
  int one = 1;
  setsockopt(fd_, IPPROTO_TCP, TCP_NODELAY, &one, sizeof(one));
  int buf = 4 << 20;
  setsockopt(fd_, SOL_SOCKET, SO_SNDBUF, &buf, sizeof(buf));
  setsockopt(fd_, SOL_SOCKET, SO_RCVBUF, &buf, sizeof(buf));
}

TcpConn::~TcpConn() {
  // the only close(): by now every reader holding this conn has exited
  // (they keep a shared_ptr), so the fd number cannot be reused under a
  // blocked recv()
  int fd = fd_.exchange(-1);
  if (fd >= 0) {
    shutdown(fd, SHUT_RDWR);
    close(fd);
  }
}

void TcpConn::Close() {
  // wake any thread blocked in recv()/send() on this conn; the fd stays
  // open (and poisoned) until the destructor so the kernel cannot hand
  // the same number to an unrelated connection while a reader races
  closed_.store(true, std::memory_order_relaxed);
  int fd = fd_.load(std::memory_order_relaxed);
  if (fd >= 0) shutdown(fd, SHUT_RDWR);
}

bool TcpConn::SendAll(const void* p, size_t n) {
  const char* c = static_cast<const char*>(p);
  while (n > 0) {
    ssize_t w = send(fd_.load(std::memory_order_relaxed), c, n, MSG_NOSIGNAL);
    if (w < 0) {
      if (errno == EINTR) continue;
      return false;
    }
    c += w;
    n -= w;
  }
  return true;
}

bool TcpConn::RecvAll(void* p, size_t n) {
  char* c = static_cast<char*>(p);
  while (n > 0) {
    ssize_t r = recv(fd_.load(std::memory_order_relaxed), c, n, 0);
    if (r < 0) {
      if (errno == EINTR) continue;
      return false;
    }
    if (r == 0) return false;  // peer closed
    c += r;
    n -= r;
  }
  return true;
}

int64_t TcpConn::SendFrame(const std::string& meta, const std::vector<SArray<char>>& data) {
  std::lock_guard<std::mutex> lk(send_mu_);
  if (fd_.load(std::memory_order_relaxed) < 0) return -1;
  uint32_t ndata = static_cast<uint32_t>(data.size());
  std::string hdr;
  hdr.reserve(12 + 8 * ndata);
  hdr.append(reinterpret_cast<const char*>(&kFrameMagic), 4);
  uint32_t mlen = static_cast<uint32_t>(meta.size());
  hdr.append(reinterpret_cast<const char*>(&mlen), 4);
  hdr.append(reinterpret_cast<const char*>(&ndata), 4);
  int64_t total = 0;
  for (auto& d : data) {
    uint64_t len = d.size();
    hdr.append(reinterpret_cast<const char*>(&len), 8);
    total += len;
  }
  if (!SendAll(hdr.data(), hdr.size())) return -1;
  if (!SendAll(meta.data(), meta.size())) return -1;
  for (auto& d : data) {
    XPS_CHECK(!d.on_device()) << "TcpConn cannot send device memory; stage to host first";
    if (d.size() && !SendAll(d.data(), d.size())) return -1;
  }
  return static_cast<int64_t>(hdr.size() + meta.size()) + total;
}

int64_t TcpConn::RecvFrame(std::string* meta, std::vector<SArray<char>>* data) {
  char hdr[12];
  if (!RecvAll(hdr, 12)) return -1;
  uint32_t magic, mlen, ndata;
  memcpy(&magic, hdr, 4);
  memcpy(&mlen, hdr + 4, 4);
  memcpy(&ndata, hdr + 8, 4);
  if (magic != kFrameMagic) {
    XPS_LOG(Warning) << "bad frame magic " << magic;
    return -1;
  }
  if (ndata > 16 || mlen > (64u << 20)) return -1;
  std::vector<uint64_t> lens(ndata);
  if (ndata && !RecvAll(lens.data(), 8 * ndata)) return -1;
  // sanity-cap each blob: a corrupt 8-byte length must drop the
  // connection cleanly, not trigger a near-2^64 allocation
  static const uint64_t kMaxBlobBytes =
      static_cast<uint64_t>(Environment::Get()->GetInt("XPS_MAX_BLOB_GB", 32)) << 30;
  for (uint32_t i = 0; i < ndata; ++i) {
    if (lens[i] > kMaxBlobBytes) {
      XPS_LOG(Warning) << "frame blob " << i << " claims " << lens[i]
                       << " bytes (cap " << kMaxBlobBytes << "); dropping connection";
      return -1;
    }
  }
  meta->resize(mlen);
  if (mlen && !RecvAll(&(*meta)[0], mlen)) return -1;
  int64_t total = 12 + 8 * ndata + mlen;
  data->clear();
  for (uint32_t i = 0; i < ndata; ++i) {
    SArray<char> d(lens[i] ? lens[i] : 0);
    if (lens[i] && !RecvAll(d.data(), lens[i])) return -1;
    data->push_back(d);
    total += lens[i];
  }
  return total;
}

int TcpConnect(const std::string& host, int port, int retries, int retry_ms) {
  for (int attempt = 0; attempt < retries; ++attempt) {
    struct addrinfo hints, *res = nullptr;
    memset(&hints, 0, sizeof(hints));
    hints.ai_family = AF_INET;
    hints.ai_socktype = SOCK_STREAM;
    char portstr[16];
    snprintf(portstr, sizeof(portstr), "%d", port);
    if (getaddrinfo(host.c_str(), portstr, &hints, &res) != 0 || !res) {
      usleep(retry_ms * 1000);
      continue;
    }
    int fd = socket(res->ai_family, res->ai_socktype, res->ai_protocol);
    if (fd >= 0 && connect(fd, res->ai_addr, res->ai_addrlen) == 0) {
      freeaddrinfo(res);
      return fd;
    }
    if (fd >= 0) close(fd);
    freeaddrinfo(res);
    usleep(retry_ms * 1000);
  }
  return -1;
}

int TcpListener::Bind(int port, int retries) {
  int fd = socket(AF_INET, SOCK_STREAM, 0);
  listen_fd_ = fd;
  if (fd < 0) return -1;
  int one = 1;
  setsockopt(fd, SOL_SOCKET, SO_REUSEADDR, &one, sizeof(one));
  struct sockaddr_in addr;
  memset(&addr, 0, sizeof(addr));
  addr.sin_family = AF_INET;
  addr.sin_addr.s_addr = htonl(INADDR_ANY);
  for (int attempt = 0; attempt < retries; ++attempt) {
    addr.sin_port = htons(static_cast<uint16_t>(port));
    if (::bind(fd, reinterpret_cast<struct sockaddr*>(&addr), sizeof(addr)) == 0) {
      socklen_t len = sizeof(addr);
      getsockname(fd, reinterpret_cast<struct sockaddr*>(&addr), &len);
      port_ = ntohs(addr.sin_port);
      if (listen(fd, 128) == 0) return port_;
      return -1;
    }
    if (port != 0) port += 1;  // probe next port like ps-lite's bind retry
    usleep(100 * 1000);
  }
  return -1;
}

void TcpListener::StartAccepting(std::function<void(int)> cb) {
  accept_thread_ = std::thread([this, cb]() {
    while (!stop_.load()) {
      struct sockaddr_in peer;
      socklen_t len = sizeof(peer);
      int fd = accept(listen_fd_, reinterpret_cast<struct sockaddr*>(&peer), &len);
      if (fd < 0) {
        if (stop_.load()) break;
        if (errno == EINTR) continue;
        break;
      }
      cb(fd);
    }
  });
}

void TcpListener::Stop() {
  if (stop_.exchange(true)) return;
  // shutdown wakes the blocked accept(); close only after the accept
  // thread joined (no accept can race a reused fd number)
  int fd = listen_fd_.load(std::memory_order_relaxed);
  if (fd >= 0) shutdown(fd, SHUT_RDWR);
  if (accept_thread_.joinable()) accept_thread_.join();
  fd = listen_fd_.exchange(-1);
  if (fd >= 0) close(fd);
}

uint64_t HostHash() {
  char host[256] = {0};
  gethostname(host, sizeof(host) - 1);
  // FNV-1a over hostname + boot id when available
  std::string s(host);
  FILE* f = fopen("/proc/sys/kernel/random/boot_id", "r");
  if (f) {
    char buf[64] = {0};
    if (fgets(buf, sizeof(buf), f)) s += buf;
    fclose(f);
  }
  uint64_t h = 1469598103934665603ull;
  for (char c : s) {
    h ^= static_cast<unsigned char>(c);
    h *= 1099511628211ull;
  }
  return h;
}

std::string LocalIP() {
  auto* env = Environment::Get();
  std::string host = env->GetStr("DMLC_NODE_HOST");
  if (!host.empty()) return host;
  std::string iface = env->GetStr("DMLC_INTERFACE");
  struct ifaddrs* ifs = nullptr;
  std::string found = "127.0.0.1";
  if (getifaddrs(&ifs) == 0) {
    for (auto* p = ifs; p; p = p->ifa_next) {
      if (!p->ifa_addr || p->ifa_addr->sa_family != AF_INET) continue;
      char ip[INET_ADDRSTRLEN];
      auto* sin = reinterpret_cast<struct sockaddr_in*>(p->ifa_addr);
      inet_ntop(AF_INET, &sin->sin_addr, ip, sizeof(ip));
      std::string name = p->ifa_name ? p->ifa_name : "";
      if (!iface.empty()) {
        if (name == iface) {
          found = ip;
          break;
        }
      } else if (name != "lo" && strcmp(ip, "127.0.0.1") != 0) {
        found = ip;
        break;
      }
    }
    freeifaddrs(ifs);
  }
  return found;
}

}  // namespace xps
