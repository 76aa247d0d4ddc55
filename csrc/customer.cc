#include "customer.h"

#include <chrono>

#include "postoffice.h"

namespace xps {

Customer::Customer(int app_id, int customer_id, RecvHandle handle, Postoffice* po)
    : app_id_(app_id), customer_id_(customer_id), handle_(std::move(handle)), po_(po) {
  po_->AddCustomer(this);
  thread_ = std::thread([this] { Receiving(); });
}

Customer::~Customer() {
  po_->RemoveCustomer(this);
  Message term;
  term.meta.control.cmd = Control::TERMINATE;
  queue_.Push(std::move(term));
  if (thread_.joinable()) thread_.join();
}

int Customer::NewRequest(int recver) {
  std::lock_guard<std::mutex> lk(mu_);
  int expected = recver >= 8 ? 1 : static_cast<int>(po_->GetNodeIDs(recver).size());
  tracker_.emplace_back();
  tracker_.back().expected = expected;
  return static_cast<int>(tracker_.size()) - 1;
}

Customer::Slot* Customer::GetSlot(int ts) {
  std::lock_guard<std::mutex> lk(mu_);
  XPS_CHECK_LT(static_cast<size_t>(ts), tracker_.size());
  return &tracker_[ts];  // deque: stable address across growth
}

void Customer::WaitRequest(int ts) {
  Slot* s = GetSlot(ts);
  // Lock-free spin first: data-plane responses land in ~10 µs while a
  // cv sleep/wake costs ~5-10 µs per side. The old spin re-took mu_
  // every iteration and fought the delivery thread's tracker update for
  // the same cache line — atomics keep the delivery path untouched.
  auto deadline = std::chrono::steady_clock::now() + std::chrono::microseconds(150);
  while (s->received.load(std::memory_order_acquire) < s->expected) {
    if (std::chrono::steady_clock::now() >= deadline) break;
  }
  if (s->received.load(std::memory_order_acquire) >= s->expected) return;
  // cv sleep path. XPS_WAIT_TIMEOUT_S > 0 bounds the wait: a responder
  // that died mid-round (crashed server, dead ring consumer) must raise
  // an error the app can act on, never a silent forever-stall.
  static const int timeout_s = Environment::Get()->GetInt("XPS_WAIT_TIMEOUT_S", 0);
  std::unique_lock<std::mutex> lk(mu_);
  auto done = [s] { return s->received.load(std::memory_order_acquire) >= s->expected; };
  if (timeout_s <= 0) {
    cv_.wait(lk, done);
    return;
  }
  if (!cv_.wait_for(lk, std::chrono::seconds(timeout_s), done)) {
    XPS_LOG(Fatal) << "request ts=" << ts << " (app " << app_id_ << ") got "
                   << s->received.load() << "/" << s->expected << " responses after "
                   << timeout_s << " s — responder dead? (XPS_WAIT_TIMEOUT_S)";
  }
}

bool Customer::IsFinished(int ts) {
  Slot* s = GetSlot(ts);
  return s->received.load(std::memory_order_acquire) >= s->expected;
}

int Customer::NumResponse(int ts) {
  return GetSlot(ts)->received.load(std::memory_order_acquire);
}

void Customer::AddResponse(int ts, int num) {
  Slot* s = GetSlot(ts);
  int got = s->received.fetch_add(num, std::memory_order_acq_rel) + num;
  if (got >= s->expected) {
    // empty critical section pairs with the waiter's predicate check:
    // without it a waiter could test-and-sleep between our update and
    // notify (lost wakeup)
    std::lock_guard<std::mutex> lk(mu_);
    cv_.notify_all();
  }
}

// Depth of customer-handler frames on this thread. The same-process
// direct-delivery path consults it: a response generated INSIDE a
// handler must be queued rather than delivered inline, or two threads
// could take two customers' handle_mu_ in opposite orders (server
// handler -> respond inline -> worker mutex, vs worker callback ->
// request inline -> server mutex) and deadlock.
thread_local int g_in_handler = 0;

bool InCustomerHandler() { return g_in_handler > 0; }

void Customer::RunHandle(Message& msg) {
  g_in_handler++;
  handle_(msg);
  g_in_handler--;
  if (!msg.meta.request) AddResponse(msg.meta.timestamp, 1);
}

void Customer::ProcessInline(Message& msg) {
  XPS_STAGE(customer_process);
  std::lock_guard<std::mutex> lk(handle_mu_);
  RunHandle(msg);
}

void Customer::Receiving() {
  while (true) {
    Message msg;
    queue_.WaitAndPop(&msg);
    if (msg.meta.control.cmd == Control::TERMINATE) break;
    std::lock_guard<std::mutex> lk(handle_mu_);
    RunHandle(msg);
  }
}

}  // namespace xps
