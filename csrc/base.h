// xps — MI355X-native parameter-server communication library.
// Core types, logging, environment.
//
// Reference parity (behavior, not code): ps-lite include/ps/base.h (Key,
// node-group ids), include/dmlc/logging.h (CHECK/LOG macros),
// include/ps/internal/env.h (Environment singleton). See SURVEY.md §L1.
#pragma once

#include <atomic>
#include <chrono>
#include <cstdint>
#include <cstdlib>
#include <cstring>
#include <iostream>
#include <limits>
#include <map>
#include <memory>
#include <mutex>
#include <sstream>
#include <stdexcept>
#include <string>

namespace xps {

// ---------------------------------------------------------------- key space
using Key = uint64_t;
static const Key kMaxKey = std::numeric_limits<Key>::max();

// node groups (bitmask-combinable, same scheme as ps-lite base.h:15-25)
static const int kScheduler = 1;
static const int kServerGroup = 2;
static const int kWorkerGroup = 4;

// node id scheme: scheduler=1; server rank r -> 8+2r (even >=8);
// worker rank r -> 9+2r (odd >=9). (ps-lite postoffice.h:106-193)
inline int WorkerRankToID(int rank) { return rank * 2 + 9; }
inline int ServerRankToID(int rank) { return rank * 2 + 8; }
inline int IDtoRank(int id) { return (id - 8) / 2; }
inline bool IsWorkerID(int id) { return id >= 8 && (id & 1) == 1; }
inline bool IsServerID(int id) { return id >= 8 && (id & 1) == 0; }

static const int kEmptyNodeID = -1;

// ------------------------------------------------------------------ logging
enum class LogLevel { kDebug = 0, kInfo = 1, kWarning = 2, kError = 3, kFatal = 4 };

class LogMessage {
 public:
  LogMessage(const char* file, int line, LogLevel lvl) : lvl_(lvl) {
    const char* base = strrchr(file, '/');
    ss_ << "[xps " << LevelStr(lvl) << " p" << Pid() << " " << (base ? base + 1 : file) << ":"
        << line << "] ";
  }
  static long Pid();
  ~LogMessage() noexcept(false) {
    ss_ << "\n";
    std::cerr << ss_.str() << std::flush;
    if (lvl_ == LogLevel::kFatal) {
      throw std::runtime_error(ss_.str());
    }
  }
  std::ostream& stream() { return ss_; }

 private:
  static const char* LevelStr(LogLevel l) {
    switch (l) {
      case LogLevel::kDebug: return "D";
      case LogLevel::kInfo: return "I";
      case LogLevel::kWarning: return "W";
      case LogLevel::kError: return "E";
      default: return "F";
    }
  }
  std::ostringstream ss_;
  LogLevel lvl_;
};

// A sink that swallows the stream (for disabled verbose levels).
class NullStream {
 public:
  template <typename T>
  NullStream& operator<<(const T&) { return *this; }
};

int VerboseLevel();  // cached PS_VERBOSE

#define XPS_LOG(lvl) ::xps::LogMessage(__FILE__, __LINE__, ::xps::LogLevel::k##lvl).stream()
#define XPS_VLOG(n) \
  if (::xps::VerboseLevel() >= (n)) XPS_LOG(Info)

#define XPS_CHECK(x) \
  if (!(x)) XPS_LOG(Fatal) << "Check failed: " #x " "
#define XPS_CHECK_EQ(a, b) XPS_CHECK((a) == (b)) << "(" << (a) << " vs " << (b) << ") "
#define XPS_CHECK_NE(a, b) XPS_CHECK((a) != (b)) << "(" << (a) << " vs " << (b) << ") "
#define XPS_CHECK_LT(a, b) XPS_CHECK((a) < (b)) << "(" << (a) << " vs " << (b) << ") "
#define XPS_CHECK_LE(a, b) XPS_CHECK((a) <= (b)) << "(" << (a) << " vs " << (b) << ") "
#define XPS_CHECK_GT(a, b) XPS_CHECK((a) > (b)) << "(" << (a) << " vs " << (b) << ") "
#define XPS_CHECK_GE(a, b) XPS_CHECK((a) >= (b)) << "(" << (a) << " vs " << (b) << ") "
#define XPS_CHECK_NOTNULL(p) XPS_CHECK((p) != nullptr)

// -------------------------------------------------------------- environment
// Env-var singleton with programmatic overrides (ps-lite env.h behavior).
class Environment {
 public:
  static Environment* Get();
  // set programmatic overrides (wins over getenv)
  void Init(const std::map<std::string, std::string>& kv);
  void Set(const std::string& k, const std::string& v);
  // returns nullptr-equivalent empty string if unset
  const char* Find(const std::string& k) const;
  std::string GetStr(const std::string& k, const std::string& dflt = "") const;
  int GetInt(const std::string& k, int dflt = 0) const;
  int64_t GetInt64(const std::string& k, int64_t dflt = 0) const;

 private:
  Environment() = default;
  mutable std::mutex mu_;
  std::map<std::string, std::string> kv_;
};

inline int GetEnvInt(const char* k, int dflt) { return Environment::Get()->GetInt(k, dflt); }

// ------------------------------------------------------------ stage timing
// XPS_TIMING=1 accumulates wall time per named hot-path stage (send,
// serialize, poll parse, handler, deferred release, ...) and prints one
// table per process at plane shutdown. Off by default: a single branch
// on a bool per scope.
struct StageStat {
  const char* name;
  std::atomic<uint64_t> ns{0};
  std::atomic<uint64_t> n{0};
  explicit StageStat(const char* nm);
};
bool TimingEnabled();
void PrintStageStats(const char* tag);

class StageScope {
 public:
  explicit StageScope(StageStat* s) : s_(TimingEnabled() ? s : nullptr) {
    if (s_) t0_ = std::chrono::steady_clock::now();
  }
  ~StageScope() {
    if (s_) {
      auto dt = std::chrono::steady_clock::now() - t0_;
      s_->ns.fetch_add(std::chrono::duration_cast<std::chrono::nanoseconds>(dt).count(),
                       std::memory_order_relaxed);
      s_->n.fetch_add(1, std::memory_order_relaxed);
    }
  }

 private:
  StageStat* s_;
  std::chrono::steady_clock::time_point t0_;
};

#define XPS_STAGE(nm)                         \
  static ::xps::StageStat xps_stage_##nm(#nm); \
  ::xps::StageScope xps_scope_##nm(&xps_stage_##nm)

}  // namespace xps
