// Pure-C++ smoke/benchmark of the xps core — the framework stands alone
// without Python (the reference's tests/test_benchmark.cc role).
//
// Runs a scheduler + server + worker in ONE process (each with its own
// Van on localhost), does correctness-checked push/pull, then times a
// PUSH_PULL loop over the shm data plane.
//
// Build: `make cppbench`  ->  build/bench_kv [msg_bytes] [keys] [iters]
#include <chrono>
#include <cstdio>
#include <thread>

#include "../../csrc/kv_app.h"
#include "../../csrc/ps.h"

using namespace xps;

int main(int argc, char** argv) {
  size_t msg_bytes = argc > 1 ? atoll(argv[1]) : (1 << 20);
  int num_keys = argc > 2 ? atoi(argv[2]) : 8;
  int iters = argc > 3 ? atoi(argv[3]) : 50;

  Environment::Get()->Init({
      {"DMLC_NUM_WORKER", "1"},
      {"DMLC_NUM_SERVER", "1"},
      {"DMLC_PS_ROOT_URI", "127.0.0.1"},
      {"DMLC_PS_ROOT_PORT", "9311"},
  });

  std::thread sched([] { Start(0, "scheduler", -1, true, -1); });
  std::thread server_start([] { Start(0, "server", -1, true, -1); });
  Start(0, "worker", -1, true, -1);
  sched.join();
  server_start.join();

  KVServer<float> server(0);
  KVServerDefaultHandle<float> handle;
  server.set_request_handle(
      [&handle](const KVMeta& m, const KVPairs<float>& kvs, KVServer<float>* s) {
        handle(m, kvs, s);
      });

  KVWorker<float> worker(0, 0);
  size_t n = msg_bytes / sizeof(float);

  // host-pool buffers ride the same-host shm plane zero-copy
  std::vector<SArray<float>> push_bufs, pull_bufs;
  std::vector<SArray<Key>> key_arrs;
  SArray<int> lens(1);
  lens[0] = static_cast<int>(n);
  for (int k = 0; k < num_keys; ++k) {
    auto pb = SArray<float>::View(HostShmPool::Get()->AllocArray(msg_bytes));
    for (size_t i = 0; i < n; ++i) pb[i] = static_cast<float>(k + 1);
    push_bufs.push_back(pb);
    pull_bufs.push_back(SArray<float>::View(HostShmPool::Get()->AllocArray(msg_bytes)));
    key_arrs.push_back(SArray<Key>({static_cast<Key>(k + 1)}));
  }

  auto round = [&]() {
    std::vector<int> ts;
    for (int k = 0; k < num_keys; ++k) {
      ts.push_back(worker.ZPush(key_arrs[k], push_bufs[k], lens, kCmdAssign));
      ts.push_back(worker.ZPull(key_arrs[k], &pull_bufs[k], nullptr, kCmdAssign));
    }
    for (int t : ts) worker.Wait(t);
  };

  round();  // warmup + correctness
  for (int k = 0; k < num_keys; ++k) {
    for (size_t i = 0; i < n; i += n / 7 + 1) {
      XPS_CHECK_EQ(pull_bufs[k][i], static_cast<float>(k + 1)) << "pull mismatch";
    }
  }
  printf("correctness OK (%d keys x %zu B)\n", num_keys, msg_bytes);

  auto t0 = std::chrono::steady_clock::now();
  for (int it = 0; it < iters; ++it) round();
  auto t1 = std::chrono::steady_clock::now();
  double sec = std::chrono::duration<double>(t1 - t0).count();
  double gb = 2.0 * msg_bytes * num_keys * iters / 1e9;
  printf("push+pull: %.2f GB/s (%.3f ms/round)\n", gb / sec, sec / iters * 1e3);

  // fused ZPushPull (one trip applies the push and returns the post-push
  // values): with the default sum handle, round r must return r * vals
  {
    SArray<Key> fk({static_cast<Key>(1000)});
    auto fin = SArray<float>::View(HostShmPool::Get()->AllocArray(msg_bytes));
    auto fout = SArray<float>::View(HostShmPool::Get()->AllocArray(msg_bytes));
    for (size_t i = 0; i < n; ++i) fin[i] = 2.0f;
    for (int r = 1; r <= 3; ++r) {
      worker.Wait(worker.ZPushPull(fk, fin, &fout, lens, kCmdSum));
      for (size_t i = 0; i < n; i += n / 5 + 1) {
        XPS_CHECK_EQ(fout[i], 2.0f * r) << "fused round mismatch at r=" << r;
      }
    }
    printf("fused ZPushPull OK (3 rounds)\n");
  }

  std::thread sfin([] { Finalize(0, "server", true); });
  std::thread schedfin([] { Finalize(0, "scheduler", true); });
  Finalize(0, "worker", true);
  sfin.join();
  schedfin.join();
  printf("DONE\n");
  return 0;
}
