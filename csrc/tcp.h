// Minimal framed TCP layer for the control/side channel and the CPU data
// path. No external deps (no zmq): blocking sockets, one recv thread per
// accepted connection, length-prefixed frames.
//
// Reference parity: ps-lite src/zmq_van.h (ROUTER/DEALER multipart
// messages) and src/network_utils.h (interface/ip/port discovery) —
// re-designed on raw POSIX sockets.
#pragma once

#include <atomic>
#include <functional>
#include <string>
#include <thread>
#include <unordered_map>
#include <vector>

#include "base.h"
#include "sarray.h"

namespace xps {

// frame layout: u32 magic | u32 meta_len | u32 ndata | u64 data_len[ndata] | meta | data...
static const uint32_t kFrameMagic = 0x58505331;  // "XPS1"

class TcpConn {
 public:
  explicit TcpConn(int fd);
  ~TcpConn();
  TcpConn(const TcpConn&) = delete;

  // thread-safe framed send; data blobs must be host memory. Returns bytes or -1.
  int64_t SendFrame(const std::string& meta, const std::vector<SArray<char>>& data);
  // blocking framed recv (call from a single reader thread). Returns bytes or -1 on close/error.
  int64_t RecvFrame(std::string* meta, std::vector<SArray<char>>* data);

  void Close();
  int fd() const { return fd_.load(std::memory_order_relaxed); }
  // true once Close() ran: the fd stays open (deferred close in the
  // destructor) but the connection is dead for routing purposes
  bool closed() const { return closed_.load(std::memory_order_relaxed); }

 private:
  bool SendAll(const void* p, size_t n);
  bool RecvAll(void* p, size_t n);
  // atomic: Close() runs from the van's Stop path to WAKE a reader
  // thread blocked in recv() on this same fd (shutdown-to-unblock);
  // exchange also makes destructor-vs-Stop close idempotent
  std::atomic<int> fd_;
  std::atomic<bool> closed_{false};
  std::mutex send_mu_;
};

// Connect to host:port with retry; returns connected fd or -1.
int TcpConnect(const std::string& host, int port, int retries = 40, int retry_ms = 250);

class TcpListener {
 public:
  // Bind to port (0 = ephemeral). Returns the bound port, or -1.
  int Bind(int port, int retries = 40);
  // Start the accept loop; cb is invoked with each new connection fd.
  void StartAccepting(std::function<void(int fd)> cb);
  void Stop();
  int port() const { return port_; }
  ~TcpListener() { Stop(); }

 private:
  std::atomic<int> listen_fd_{-1};  // Stop() closes it under the accept loop
  int port_ = -1;
  std::atomic<bool> stop_{false};
  std::thread accept_thread_;
};

// Hash identifying this physical host (for same-host shm/hipIpc fast path).
uint64_t HostHash();
// Best local IP for peers to connect to (DMLC_INTERFACE/DMLC_NODE_HOST aware).
std::string LocalIP();

}  // namespace xps
