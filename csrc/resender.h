// Resender: optional ACK + timeout retransmission with duplicate
// suppression (enabled by PS_RESEND=1, timeout PS_RESEND_TIMEOUT ms).
//
// Reference parity: ps-lite src/resender.h (AddOutgoing/AddIncomming/
// Monitoring; 64-bit signature dedup).
#pragma once

#include <atomic>
#include <chrono>
#include <thread>
#include <unordered_map>
#include <unordered_set>

#include "message.h"

namespace xps {

class Van;

class Resender {
 public:
  Resender(int timeout_ms, int max_retries, Van* van);
  ~Resender();

  // tag an outgoing data message with a signature and remember it
  void AddOutgoing(Message& msg);
  // record an incoming data message, send the ACK; returns true if duplicate
  bool AddIncoming(const Message& msg);
  void HandleAck(uint64_t sig);

 private:
  void Monitoring();
  uint64_t Signature(const Meta& m) const;

  int timeout_ms_;
  int max_retries_;
  Van* van_;
  std::mutex mu_;
  struct Entry {
    Message msg;
    std::chrono::steady_clock::time_point sent;
    int retries = 0;
  };
  std::unordered_map<uint64_t, Entry> outgoing_;
  std::unordered_set<uint64_t> seen_;
  std::atomic<bool> stop_{false};
  std::thread monitor_;
};

}  // namespace xps
