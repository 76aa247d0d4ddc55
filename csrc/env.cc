#include <unistd.h>

#include "base.h"

namespace xps {

long LogMessage::Pid() {
  static long pid = getpid();
  return pid;
}

Environment* Environment::Get() {
  static Environment inst;
  return &inst;
}

void Environment::Init(const std::map<std::string, std::string>& kv) {
  std::lock_guard<std::mutex> lk(mu_);
  for (auto& p : kv) kv_[p.first] = p.second;
}

void Environment::Set(const std::string& k, const std::string& v) {
  std::lock_guard<std::mutex> lk(mu_);
  kv_[k] = v;
}

const char* Environment::Find(const std::string& k) const {
  std::lock_guard<std::mutex> lk(mu_);
  auto it = kv_.find(k);
  if (it != kv_.end()) return it->second.c_str();
  return getenv(k.c_str());
}

std::string Environment::GetStr(const std::string& k, const std::string& dflt) const {
  const char* v = Find(k);
  return v ? std::string(v) : dflt;
}

int Environment::GetInt(const std::string& k, int dflt) const {
  const char* v = Find(k);
  return v && *v ? atoi(v) : dflt;
}

int64_t Environment::GetInt64(const std::string& k, int64_t dflt) const {
  const char* v = Find(k);
  return v && *v ? atoll(v) : dflt;
}

int VerboseLevel() {
  static int lvl = Environment::Get()->GetInt("PS_VERBOSE", 0);
  return lvl;
}

}  // namespace xps
