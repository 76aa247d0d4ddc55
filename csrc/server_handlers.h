// Built-in GPU request handlers for KVServer<float>.
//
// Reference parity: ps-lite ships only CPU handlers — the tests'
// EmptyHandler (tests/test_benchmark.cc:132-203, stores pushed buffers,
// echoes on pull) and KVServerDefaultHandle (kv_app.h:431-452,
// store[key] += vals). Here both behaviors are CDNA4 kernels on the
// requester's per-peer HIP stream: push = DenseAssign / DenseSumF32
// reading the worker's HBM directly (hipIpc-mapped, reads ride xGMI);
// pull = zero-copy store view handed to the plane's in-place xGMI write.
// The sparse handler owns a rows x row_len fp32 embedding table and uses
// the gather/scatter kernels (BASELINE config #5).
#pragma once

#include <hip/hip_runtime.h>

#include <memory>
#include <set>
#include <type_traits>
#include <unordered_map>

#include "kernels.h"
#include "kv_app.h"

namespace xps {

// A pooled hipEvent_t with shared ownership: freed back to the plane's
// event pool (or destroyed) when the last reference drops. Shared
// ownership lets one recorded event serve as the ordering point for
// every key touched by a batched kernel.
using EventRef = std::shared_ptr<std::remove_pointer<hipEvent_t>::type>;

// handler modes
enum class DenseMode {
  kAssign,  // push overwrites the store (EmptyHandler / pure-goodput)
  kSum,     // push accumulates forever (KVServerDefaultHandle)
  kReduce,  // BytePS round semantics: first push of a round assigns, the
            // rest accumulate; pulls are HELD until all num_workers
            // pushes of the round arrived (cross-stream ordered via
            // events), and the round resets after num_workers pulls.
};

// payload dtype of the accumulate kernels (the wire/pools are bytes;
// bf16 halves bytes moved on the bandwidth-bound dense path). Assign
// mode is dtype-agnostic byte copies either way.
enum class DenseDtype { kF32, kBf16 };

class GpuDenseHandler {
 public:
  GpuDenseHandler(Postoffice* po, DenseMode mode, DenseDtype dtype = DenseDtype::kF32);
  ~GpuDenseHandler();
  void operator()(const KVMeta& req, const KVPairs<float>& kvs, KVServer<float>* server);

  // checkpoint / resume of the server KV state (an aux subsystem the
  // reference lacks entirely — SURVEY.md §5.4)
  void Save(const std::string& path);
  void Load(const std::string& path);

  // RegisterRecvBufferWithRank parity (reference kv_app.h:488): make an
  // APP-owned device buffer this key's store entry — pushes land
  // straight in it (zero-copy into user memory), pulls serve from it.
  // The buffer must outlive the handler; if it is pool-resident (e.g.
  // a torch-allocator tensor) the one-sided steady state applies too.
  void RegisterEntry(Key key, void* ptr, size_t nbytes, int device);

 private:
  struct Entry {
    SArray<char> buf;
    // last kernel that touched buf, on WHICHEVER peer stream: with >1
    // workers, concurrent senders use different streams, so every
    // kernel touching this entry waits on last_ev and publishes a new
    // one — otherwise two senders' sum kernels race (lost updates) and
    // a pull can read a half-written store
    EventRef last_ev;
    // superseded (smaller) buffers, kept alive forever: workers may
    // still hold their offsets in one-sided entry caches; stale writes
    // must hit dead-but-owned memory (they self-heal — the mismatched
    // length makes the worker fall back and re-learn the new offset)
    std::vector<SArray<char>> retired;
  };

  // Reduce-mode round state, per KEY-SET: a worker's buckets for this
  // server ride ONE multi-key message per round (BytePS bucketed
  // DenseReduce — collapses 2x169 per-key messages/step to 2 per
  // server), so round accounting is per key-set with one batched
  // sum/assign kernel chain and one event per push. Single-key messages
  // are just groups of size 1. CONTRACT: all workers must use the same
  // key->message grouping (the BytePS layout guarantees it); a key must
  // not appear in two different groups.
  struct Group {
    std::vector<Key> keys;
    std::vector<Entry*> ents;   // store_ entries (node-stable pointers)
    std::vector<size_t> lens;   // byte length per key (set by first push)
    int pushes = 0;
    int pulls = 0;
    std::vector<KVMeta> waiting_pulls;
    // pushes of the NEXT round arriving before this round's pulls drain
    // (KVPairs holds the remote buffer alive until we process + ack)
    std::vector<std::pair<KVMeta, KVPairs<float>>> waiting_pushes;
    // pulls from senders that already pulled this round (next-round pulls)
    std::vector<KVMeta> waiting_next_pulls;
    std::set<int> pulled_senders;
    std::vector<EventRef> round_events;  // one per pusher stream
    std::vector<EventRef> pull_guard;    // pull copies the next round must wait on
    EventRef last_ev;                    // last kernel touching the group
  };

  void HandlePush(const KVMeta& req, const KVPairs<float>& kvs, KVServer<float>* server,
                  bool respond = true);
  void HandleReducePush(const KVMeta& req, const KVPairs<float>& kvs, KVServer<float>* server);
  void HandlePull(const KVMeta& req, const KVPairs<float>& kvs, KVServer<float>* server);
  void RespondPull(const KVMeta& req, Group* g, KVServer<float>* server);
  Group* GroupFor(const SArray<Key>& keys);  // find-or-create round group
  hipStream_t Stream(int sender);      // lane 0: push/accumulate kernels
  hipStream_t PullStream(int sender);  // pull lane: response copies
  // same-key event chaining needed for this sender? (>1 workers, or the
  // peer's lanes actually split = cross-device)
  bool NeedChain(int sender);
  void OrderAfter(Entry* e, hipStream_t s);  // wait the entry's last_ev

  // dtype-dispatched accumulate (nbytes, not elements)
  void SumKernel(void* dst, const void* src, size_t nbytes, hipStream_t s);
  void BatchedSum(const kern::CopyDesc* descs, int n, hipStream_t s);

  Postoffice* po_;
  DenseMode mode_;
  DenseDtype dtype_;
  int num_workers_ = 1;
  bool chain_ = false;  // >1 workers: cross-stream same-key ordering needed
  std::mutex mu_;
  std::unordered_map<Key, Entry> store_;
  std::unordered_map<uint64_t, Group> groups_;  // keyset-hash -> round group
  hipStream_t fallback_stream_ = nullptr;
};

// NOTE on key uniqueness: a single sparse push message must not repeat a
// row (ps-lite's sorted-unique key contract); duplicate rows across
// MESSAGES are fine (atomic scatters with >1 workers; handler-serialized
// otherwise).
class GpuSparseHandler {
 public:
  // Allocates (and zeroes) a rows x row_len fp32 table shard in the pool.
  // key_shift: workers send key = global_row << key_shift so the key-range
  // slicer shards rows evenly; this server's local row =
  // (key >> key_shift) - row_base, row_base derived from its key range.
  GpuSparseHandler(Postoffice* po, size_t rows, size_t row_len, bool accumulate,
                   int key_shift = 0);
  void operator()(const KVMeta& req, const KVPairs<float>& kvs, KVServer<float>* server);
  uintptr_t table_ptr() const { return reinterpret_cast<uintptr_t>(table_.data()); }
  void Save(const std::string& path);
  void Load(const std::string& path);

 private:
  hipStream_t Stream(int sender);
  // keys usable by a kernel: device keys pass through; host keys are
  // staged synchronously into a per-peer scratch row-id buffer
  const uint64_t* DeviceKeys(const SArray<Key>& keys, int sender, hipStream_t s);

  Postoffice* po_;
  size_t rows_;
  size_t row_len_;
  bool accumulate_;
  bool atomic_ = false;  // >1 workers: concurrent scatters need atomics
  int key_shift_ = 0;
  uint64_t row_base_ = 0;
  SArray<char> table_;
  std::mutex mu_;
  std::unordered_map<int, SArray<char>> key_scratch_;  // per sender
  hipStream_t fallback_stream_ = nullptr;
};

}  // namespace xps
