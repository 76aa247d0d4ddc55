#include "customer.h"

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
  tracker_.emplace_back(expected, 0);
  return static_cast<int>(tracker_.size()) - 1;
}

void Customer::WaitRequest(int ts) {
  // brief spin first: responses on the data plane land in ~10 µs and a
  // cv sleep/wake costs ~5-10 µs per side
  for (int i = 0; i < 4000; ++i) {
    std::lock_guard<std::mutex> lk(mu_);
    if (tracker_[ts].first == tracker_[ts].second) return;
  }
  std::unique_lock<std::mutex> lk(mu_);
  cv_.wait(lk, [this, ts] { return tracker_[ts].first == tracker_[ts].second; });
}

bool Customer::IsFinished(int ts) {
  std::lock_guard<std::mutex> lk(mu_);
  return tracker_[ts].first == tracker_[ts].second;
}

int Customer::NumResponse(int ts) {
  std::lock_guard<std::mutex> lk(mu_);
  return tracker_[ts].second;
}

void Customer::AddResponse(int ts, int num) {
  std::lock_guard<std::mutex> lk(mu_);
  tracker_[ts].second += num;
  if (tracker_[ts].second >= tracker_[ts].first) cv_.notify_all();
}

void Customer::RunHandle(Message& msg) {
  handle_(msg);
  if (!msg.meta.request) {
    std::lock_guard<std::mutex> lk(mu_);
    tracker_[msg.meta.timestamp].second++;
    if (tracker_[msg.meta.timestamp].second >= tracker_[msg.meta.timestamp].first) {
      cv_.notify_all();
    }
  }
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
