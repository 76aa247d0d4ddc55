// Thread-safe queues.
//
// Reference parity: ps-lite include/ps/internal/threadsafe_queue.h and
// spsc_queue.h. The MPMC queue is mutex+cv; a lock-free SPSC ring backs
// the shm data-plane rings (shm_ring.h) instead of an in-process queue.
#pragma once

#include <condition_variable>
#include <mutex>
#include <queue>
#include <utility>

namespace xps {

template <typename T>
class ThreadsafeQueue {
 public:
  void Push(T v) {
    {
      std::lock_guard<std::mutex> lk(mu_);
      q_.push(std::move(v));
    }
    cv_.notify_one();
  }

  void WaitAndPop(T* out) {
    // brief spin before sleeping: saves a ~5-10 µs cv wakeup per hop on
    // the hot path (customer recv thread), matching the data plane's
    // spin-then-sleep pollers
    for (int i = 0; i < 2000; ++i) {
      if (TryPop(out)) return;
    }
    std::unique_lock<std::mutex> lk(mu_);
    cv_.wait(lk, [this] { return !q_.empty(); });
    *out = std::move(q_.front());
    q_.pop();
  }

  bool TryPop(T* out) {
    std::lock_guard<std::mutex> lk(mu_);
    if (q_.empty()) return false;
    *out = std::move(q_.front());
    q_.pop();
    return true;
  }

  size_t Size() const {
    std::lock_guard<std::mutex> lk(mu_);
    return q_.size();
  }

 private:
  mutable std::mutex mu_;
  std::condition_variable cv_;
  std::queue<T> q_;
};

}  // namespace xps
