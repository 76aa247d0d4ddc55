// Multi-threaded sorted key matching / merging and parallel sort
// utilities over SArray.
//
// Reference parity: ps-lite include/ps/internal/parallel_kv_match.h
// (ParallelOrderedMatch) and parallel_sort.h (ParallelSort), plus the
// assign-op enum of internal/assign_op.h (whose reference bodies were
// dead code; these compile and are tested).
#pragma once

#include <algorithm>
#include <thread>
#include <vector>

#include "sarray.h"

namespace xps {

enum class AssignOp { kAssign, kPlus, kMinus, kTimes, kDivide, kAnd, kOr, kXor };

template <typename T>
inline void AssignFunc(T* left, AssignOp op, const T& right) {
  switch (op) {
    case AssignOp::kAssign: *left = right; break;
    case AssignOp::kPlus: *left += right; break;
    case AssignOp::kMinus: *left -= right; break;
    case AssignOp::kTimes: *left *= right; break;
    case AssignOp::kDivide: *left /= right; break;
    default: XPS_LOG(Fatal) << "bitwise assign op on non-integral use";
  }
}

// Match sorted src_keys into sorted dst_keys: for every dst key present
// in src, copy (op=assign) or accumulate (op=plus) its k-width value
// block. Splits the dst range over `nthreads`. Returns #matched keys.
template <typename K, typename V>
size_t ParallelOrderedMatch(const SArray<K>& src_keys, const SArray<V>& src_vals,
                            const SArray<K>& dst_keys, SArray<V>* dst_vals,
                            size_t k = 1, AssignOp op = AssignOp::kAssign,
                            int nthreads = 4) {
  XPS_CHECK_EQ(src_keys.size() * k, src_vals.size());
  if (dst_vals->size() != dst_keys.size() * k) {
    dst_vals->Resize(dst_keys.size() * k);
    std::fill(dst_vals->begin(), dst_vals->end(), V(0));
  }
  if (dst_keys.empty() || src_keys.empty()) return 0;
  nthreads = std::max(1, std::min<int>(nthreads, static_cast<int>(dst_keys.size())));
  std::vector<size_t> matched(nthreads, 0);
  std::vector<std::thread> threads;
  size_t chunk = (dst_keys.size() + nthreads - 1) / nthreads;
  for (int t = 0; t < nthreads; ++t) {
    size_t begin = t * chunk;
    size_t end = std::min(dst_keys.size(), begin + chunk);
    if (begin >= end) break;
    threads.emplace_back([&, t, begin, end]() {
      const K* sk = std::lower_bound(src_keys.begin(), src_keys.end(), dst_keys[begin]);
      for (size_t i = begin; i < end; ++i) {
        while (sk != src_keys.end() && *sk < dst_keys[i]) ++sk;
        if (sk == src_keys.end()) break;
        if (*sk == dst_keys[i]) {
          size_t si = sk - src_keys.begin();
          for (size_t j = 0; j < k; ++j) {
            AssignFunc(&(*dst_vals)[i * k + j], op, src_vals[si * k + j]);
          }
          ++matched[t];
        }
      }
    });
  }
  for (auto& th : threads) th.join();
  size_t total = 0;
  for (size_t m : matched) total += m;
  return total;
}

// Recursive-thread merge sort on an SArray (ps-lite parallel_sort.h).
template <typename T, typename Cmp = std::less<T>>
void ParallelSort(SArray<T>* arr, int nthreads = 4, Cmp cmp = Cmp()) {
  if (nthreads <= 1 || arr->size() < (1u << 14)) {
    std::sort(arr->begin(), arr->end(), cmp);
    return;
  }
  size_t mid = arr->size() / 2;
  SArray<T> left = arr->Segment(0, mid);
  SArray<T> right = arr->Segment(mid, arr->size());
  std::thread t([&]() { ParallelSort(&left, nthreads / 2, cmp); });
  ParallelSort(&right, nthreads - nthreads / 2, cmp);
  t.join();
  std::inplace_merge(arr->begin(), arr->begin() + mid, arr->end(), cmp);
}

}  // namespace xps
