// Customer: per-app request tracker + receiving thread.
//
// Reference parity: ps-lite include/ps/internal/customer.h,
// src/customer.cc (tracker_ of (expected, received) pairs, recv thread
// draining a queue into the app's handle).
#pragma once

#include <atomic>
#include <condition_variable>
#include <deque>
#include <functional>
#include <mutex>
#include <thread>
#include <utility>

#include "message.h"
#include "queue.h"

namespace xps {

class Postoffice;

// true while the calling thread is inside any Customer handler frame
// (see the direct-delivery deadlock note in customer.cc)
bool InCustomerHandler();

class Customer {
 public:
  using RecvHandle = std::function<void(const Message&)>;

  Customer(int app_id, int customer_id, RecvHandle handle, Postoffice* po);
  ~Customer();
  Customer(const Customer&) = delete;

  int app_id() const { return app_id_; }
  int customer_id() const { return customer_id_; }

  // open a request slot destined for `recver` (node id or group mask);
  // returns the timestamp used to track its responses.
  int NewRequest(int recver);
  // block until all expected responses for `ts` arrived.
  void WaitRequest(int ts);
  int NumResponse(int ts);
  void AddResponse(int ts, int num = 1);
  bool IsFinished(int ts);

  // called by the Van for every data message addressed to this customer
  void Accept(Message msg) { queue_.Push(std::move(msg)); }

  // Inline delivery (the data plane's single poll thread): runs the
  // handler + tracker update on the CALLER's thread, serialized against
  // the queue thread by handle_mu_ — saves the queue hop + cv wakeup on
  // the hot path.
  void ProcessInline(Message& msg);

 private:
  // per-request slot: `received` is written lock-free by the delivery
  // thread and spun on by WaitRequest (mu_ only guards deque growth and
  // the cv sleep path)
  struct Slot {
    int expected = 0;
    std::atomic<int> received{0};
  };

  void Receiving();
  void RunHandle(Message& msg);
  Slot* GetSlot(int ts);

  int app_id_;
  int customer_id_;
  RecvHandle handle_;
  Postoffice* po_;
  ThreadsafeQueue<Message> queue_;
  std::thread thread_;
  std::mutex handle_mu_;
  std::mutex mu_;
  std::condition_variable cv_;
  std::deque<Slot> tracker_;  // stable addresses; indexed by timestamp
};

}  // namespace xps
