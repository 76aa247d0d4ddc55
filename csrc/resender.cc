#include "resender.h"

#include "postoffice.h"
#include "van.h"

namespace xps {

Resender::Resender(int timeout_ms, int max_retries, Van* van)
    : timeout_ms_(timeout_ms), max_retries_(max_retries), van_(van) {
  monitor_ = std::thread([this] { Monitoring(); });
}

Resender::~Resender() {
  stop_ = true;
  if (monitor_.joinable()) monitor_.join();
}

uint64_t Resender::Signature(const Meta& m) const {
  // pack (app, customer, ts, sender, recver, req|push|pull, key) into 64 bits
  uint64_t h = 1469598103934665603ull;
  auto mix = [&h](uint64_t v) {
    h ^= v;
    h *= 1099511628211ull;
  };
  mix(static_cast<uint64_t>(m.app_id) << 32 | static_cast<uint32_t>(m.customer_id));
  mix(static_cast<uint64_t>(m.timestamp));
  mix(static_cast<uint64_t>(m.sender) << 32 | static_cast<uint32_t>(m.recver));
  mix((m.request ? 1u : 0u) | (m.push ? 2u : 0u) | (m.pull ? 4u : 0u));
  mix(m.key);
  return h ? h : 1;  // 0 means "no signature"
}

void Resender::AddOutgoing(Message& msg) {
  msg.meta.msg_sig = Signature(msg.meta);
  std::lock_guard<std::mutex> lk(mu_);
  auto& e = outgoing_[msg.meta.msg_sig];
  e.msg = msg;
  e.sent = std::chrono::steady_clock::now();
}

bool Resender::AddIncoming(const Message& msg) {
  if (msg.meta.msg_sig == 0) return false;
  bool dup;
  {
    std::lock_guard<std::mutex> lk(mu_);
    dup = !seen_.insert(msg.meta.msg_sig).second;
    if (seen_.size() > 1u << 20) seen_.clear();  // bounded memory; sig collision window
  }
  Message ack;
  ack.meta.control.cmd = Control::ACK;
  ack.meta.control.msg_sig = msg.meta.msg_sig;
  ack.meta.recver = msg.meta.sender;
  van_->Send(ack);
  return dup;
}

void Resender::HandleAck(uint64_t sig) {
  std::lock_guard<std::mutex> lk(mu_);
  outgoing_.erase(sig);
}

void Resender::Monitoring() {
  while (!stop_.load()) {
    std::this_thread::sleep_for(std::chrono::milliseconds(timeout_ms_));
    std::vector<Message> resend;
    {
      std::lock_guard<std::mutex> lk(mu_);
      auto now = std::chrono::steady_clock::now();
      for (auto& kv : outgoing_) {
        auto age =
            std::chrono::duration_cast<std::chrono::milliseconds>(now - kv.second.sent).count();
        if (age >= timeout_ms_) {
          XPS_CHECK_LT(kv.second.retries, max_retries_)
              << "message lost after " << max_retries_ << " retries: "
              << kv.second.msg.DebugString();
          kv.second.retries++;
          kv.second.sent = now;
          resend.push_back(kv.second.msg);
        }
      }
    }
    for (auto& m : resend) {
      XPS_VLOG(1) << "resending " << m.DebugString();
      van_->Send(m);
    }
  }
}

}  // namespace xps
