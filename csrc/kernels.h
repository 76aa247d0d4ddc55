// Host-side launchers for the CDNA4 server kernels (kernels.hip).
//
// These implement the compute the ps-lite server handler API implies
// (dense accumulate/assign replacing KVServerDefaultHandle
// kv_app.h:441-448, sparse gather/scatter over a key-indexed table) —
// all-new HIP code, the reference has no kernels.
#pragma once

#include <cstddef>
#include <cstdint>

#include <hip/hip_runtime.h>

namespace xps {
namespace kern {

// dst[i] = src[i] (bytes; 16B-vectorized grid-stride copy kernel)
void DenseAssign(void* dst, const void* src, size_t nbytes, hipStream_t s);
// dst[i] += src[i] (fp32)
void DenseSumF32(float* dst, const float* src, size_t n, hipStream_t s);
// dst[i] += src[i] (bf16 payloads, fp32 accumulate in-register,
// round-to-nearest-even back to bf16 — halves bytes moved on the
// bandwidth-bound dense path)
void DenseSumBf16(uint16_t* dst, const uint16_t* src, size_t n, hipStream_t s);
void BatchedSumBf16(const struct CopyDesc* descs_host, int n, hipStream_t s);
// dst[i] += sum_j srcs[j][i], up to 8 sources in one pass (one read of
// dst, one write — HBM-optimal multi-worker reduction)
// Sparse ops: local row index = (rows[r] >> key_shift) - row_base, so a
// server can index its local table shard from globally-sharded keys
// (key = global_row << key_shift spreads rows over the PS key space).
// table_rows bounds every decoded row index: out-of-range rows (corrupt
// or misrouted keys) are skipped by scatters and read as zeros by
// gathers, never touching memory outside the table. The ~0 default
// disables the check (trusted keys).
// out[r][:] = table[row(r)][:] for r in [0, nrows)
void SparseGatherF32(const float* table, const uint64_t* rows_dev, size_t nrows, size_t row_len,
                     float* out, hipStream_t s, int key_shift = 0, uint64_t row_base = 0,
                     uint64_t table_rows = ~0ull);
// table[row(r)][:] += src[r][:]; atomic=true tolerates duplicate rows
void SparseScatterAddF32(float* table, const uint64_t* rows_dev, size_t nrows, size_t row_len,
                         const float* src, bool atomic, hipStream_t s, int key_shift = 0,
                         uint64_t row_base = 0, uint64_t table_rows = ~0ull);
// table[row(r)][:] = src[r][:]
void SparseScatterAssignF32(float* table, const uint64_t* rows_dev, size_t nrows, size_t row_len,
                            const float* src, hipStream_t s, int key_shift = 0,
                            uint64_t row_base = 0, uint64_t table_rows = ~0ull);

// Batched segmented copy/accumulate: one launch serving up to
// kMaxBatch (dst, src, nbytes) segments — the device-side slicing/merge
// of SURVEY.md §2.6 item 3, used for multi-key messages.
static const int kMaxBatch = 64;
struct CopyDesc {
  void* dst;
  const void* src;
  size_t nbytes;  // must be 16B-multiples for the batched kernels
};
void BatchedAssign(const CopyDesc* descs_host, int n, hipStream_t s);
void BatchedSumF32(const CopyDesc* descs_host, int n, hipStream_t s);

}  // namespace kern
}  // namespace xps
