#include <unistd.h>

#include <vector>

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

static std::mutex g_stage_mu;
static std::vector<StageStat*>* g_stages = nullptr;

StageStat::StageStat(const char* nm) : name(nm) {
  std::lock_guard<std::mutex> lk(g_stage_mu);
  if (!g_stages) g_stages = new std::vector<StageStat*>();
  g_stages->push_back(this);
}

bool TimingEnabled() {
  static bool on = Environment::Get()->GetInt("XPS_TIMING", 0) != 0;
  return on;
}

void PrintStageStats(const char* tag) {
  if (!TimingEnabled()) return;
  static std::atomic<bool> printed{false};
  if (printed.exchange(true)) return;  // stats are process-global: once
  std::lock_guard<std::mutex> lk(g_stage_mu);
  if (!g_stages) return;
  std::ostringstream os;
  os << "stage timing [" << tag << "] pid " << getpid() << ":\n";
  for (auto* s : *g_stages) {
    uint64_t n = s->n.load(), ns = s->ns.load();
    if (!n) continue;
    char line[160];
    snprintf(line, sizeof(line), "  %-24s n=%10llu total=%9.3f ms avg=%8.2f us\n", s->name,
             (unsigned long long)n, ns / 1e6, ns / 1e3 / n);
    os << line;
  }
  std::cerr << os.str();
}

int VerboseLevel() {
  static int lvl = Environment::Get()->GetInt("PS_VERBOSE", 0);
  return lvl;
}

}  // namespace xps
