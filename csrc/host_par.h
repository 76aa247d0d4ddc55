// Tiny fork-join pool for big HOST copies/reductions. A single core
// sustains ~8-15 GB/s of memcpy; the host-shm data plane (config #1)
// is bound by exactly that on its poll thread. Large copies fan out to
// a few persistent workers (caller participates; one fork-join at a
// time — callers are the plane poll thread and the server customer
// thread, never many at once).
#pragma once

#include <atomic>
#include <condition_variable>
#include <cstring>
#include <functional>
#include <mutex>
#include <thread>
#include <vector>

#include "base.h"

namespace xps {

class HostPar {
 public:
  static HostPar* Get() {
    static HostPar p;
    return &p;
  }

  static constexpr size_t kMinBytes = 4 << 20;  // below this: plain single-thread
  // (a ~1 MB copy is faster serial than two cv wake round-trips)

  // fn(lo, hi) over [0, n) split across workers + the caller.
  void For(size_t n, const std::function<void(size_t, size_t)>& fn) {
    if (n == 0) return;
    std::lock_guard<std::mutex> serial(serial_mu_);
    size_t parts = threads_.size() + 1;
    size_t chunk = (n + parts - 1) / parts;
    {
      std::lock_guard<std::mutex> lk(mu_);
      fn_ = &fn;
      n_ = n;
      chunk_ = chunk;
      pending_ = static_cast<int>(threads_.size());
      epoch_++;
    }
    cv_.notify_all();
    fn(0, std::min(chunk, n));  // caller takes the first chunk
    std::unique_lock<std::mutex> lk(mu_);
    done_cv_.wait(lk, [this] { return pending_ == 0; });
    fn_ = nullptr;
  }

  static void CopyBytes(void* dst, const void* src, size_t n) {
    if (n < kMinBytes) {
      memcpy(dst, src, n);
      return;
    }
    Get()->For(n, [dst, src](size_t lo, size_t hi) {
      memcpy(static_cast<char*>(dst) + lo, static_cast<const char*>(src) + lo, hi - lo);
    });
  }

  static void SumF32(float* dst, const float* src, size_t n) {
    if (n * sizeof(float) < kMinBytes) {
      for (size_t j = 0; j < n; ++j) dst[j] += src[j];
      return;
    }
    Get()->For(n, [dst, src](size_t lo, size_t hi) {
      float* __restrict__ d = dst + lo;
      const float* __restrict__ s = src + lo;
      for (size_t j = 0; j < hi - lo; ++j) d[j] += s[j];
    });
  }

 private:
  HostPar() {
    int nw = Environment::Get()->GetInt("XPS_HOST_COPY_THREADS", 3);
    for (int i = 1; i <= nw; ++i) {
      threads_.emplace_back([this, i] { Worker(static_cast<size_t>(i)); });
    }
  }
  ~HostPar() {
    {
      std::lock_guard<std::mutex> lk(mu_);
      stop_ = true;
      epoch_++;
    }
    cv_.notify_all();
    for (auto& t : threads_) t.join();
  }

  void Worker(size_t idx) {
    uint64_t seen = 0;
    while (true) {
      const std::function<void(size_t, size_t)>* fn;
      size_t lo, hi;
      {
        std::unique_lock<std::mutex> lk(mu_);
        cv_.wait(lk, [&] { return stop_ || epoch_ != seen; });
        if (stop_) return;
        seen = epoch_;
        fn = fn_;
        lo = std::min(idx * chunk_, n_);
        hi = std::min(lo + chunk_, n_);
      }
      if (fn && lo < hi) (*fn)(lo, hi);
      {
        std::lock_guard<std::mutex> lk(mu_);
        if (--pending_ == 0) done_cv_.notify_all();
      }
    }
  }

  std::mutex serial_mu_;  // one For() at a time
  std::mutex mu_;
  std::condition_variable cv_, done_cv_;
  std::vector<std::thread> threads_;
  const std::function<void(size_t, size_t)>* fn_ = nullptr;
  size_t n_ = 0, chunk_ = 0;
  int pending_ = 0;
  uint64_t epoch_ = 0;
  bool stop_ = false;
};

}  // namespace xps
