// CDNA4 (gfx950) kernels for the KVServer request handlers.
//
// All kernels are HBM-bandwidth-bound streaming ops, so the design rules
// are (cdna_hip_programming.md §2): 16 B per lane per instruction
// (uint4/float4), 64-wide wavefronts, grid-stride loops sized >> 256
// workgroups to fill 8 XCDs, and a single pass over dst for the
// multi-source reduction. Inputs may be hipIpc-mapped peer-GPU memory —
// the loads then ride xGMI.
#include <hip/hip_runtime.h>

#include <algorithm>
#include <cstdlib>

#include "kernels.h"

namespace xps {
namespace kern {

namespace {

constexpr int kBlock = 256;

inline int GridFor(size_t work_items, int cap = 8192) {
  size_t g = (work_items + kBlock - 1) / kBlock;
  if (g > static_cast<size_t>(cap)) g = cap;
  if (g == 0) g = 1;
  return static_cast<int>(g);
}

// NOTE: nontemporal loads/stores were measured 3 % SLOWER end-to-end
// here — the 256 MiB Infinity Cache serves the overlapped pull's re-read
// of the just-written store, and nt stores bypass it. Keep plain float4.
__global__ void assign_kernel(uint4* __restrict__ dst, const uint4* __restrict__ src, size_t n4) {
  size_t i = blockIdx.x * static_cast<size_t>(blockDim.x) + threadIdx.x;
  size_t stride = static_cast<size_t>(gridDim.x) * blockDim.x;
  for (; i < n4; i += stride) dst[i] = src[i];
}

__global__ void assign_tail_kernel(char* __restrict__ dst, const char* __restrict__ src,
                                   size_t begin, size_t end) {
  size_t i = begin + blockIdx.x * static_cast<size_t>(blockDim.x) + threadIdx.x;
  if (i < end) dst[i] = src[i];
}

__global__ void sum_kernel_f32(float4* __restrict__ dst, const float4* __restrict__ src,
                               size_t n4) {
  size_t i = blockIdx.x * static_cast<size_t>(blockDim.x) + threadIdx.x;
  size_t stride = static_cast<size_t>(gridDim.x) * blockDim.x;
  for (; i < n4; i += stride) {
    float4 d = dst[i];
    float4 s = src[i];
    d.x += s.x;
    d.y += s.y;
    d.z += s.z;
    d.w += s.w;
    dst[i] = d;
  }
}

__global__ void sum_tail_f32(float* __restrict__ dst, const float* __restrict__ src, size_t begin,
                             size_t end) {
  size_t i = begin + blockIdx.x * static_cast<size_t>(blockDim.x) + threadIdx.x;
  if (i < end) dst[i] += src[i];
}

// bf16 += bf16 with fp32 accumulate in-register, 16 B per lane (8 bf16
// elements via uint4). bf16 -> fp32 is a 16-bit shift; fp32 -> bf16
// rounds to nearest-even (matches torch's bf16 semantics).
__device__ inline float bf16_to_f32(uint16_t h) {
  union {
    uint32_t u;
    float f;
  } c;
  c.u = static_cast<uint32_t>(h) << 16;
  return c.f;
}

__device__ inline uint16_t f32_to_bf16(float f) {
  union {
    uint32_t u;
    float f;
  } c;
  c.f = f;
  uint32_t lsb = (c.u >> 16) & 1u;
  return static_cast<uint16_t>((c.u + 0x7FFFu + lsb) >> 16);
}

__device__ inline uint32_t bf16x2_sum(uint32_t d, uint32_t s) {
  float lo = bf16_to_f32(static_cast<uint16_t>(d)) + bf16_to_f32(static_cast<uint16_t>(s));
  float hi = bf16_to_f32(static_cast<uint16_t>(d >> 16)) +
             bf16_to_f32(static_cast<uint16_t>(s >> 16));
  return static_cast<uint32_t>(f32_to_bf16(lo)) |
         (static_cast<uint32_t>(f32_to_bf16(hi)) << 16);
}

__global__ void sum_kernel_bf16(uint4* __restrict__ dst, const uint4* __restrict__ src,
                                size_t n8) {
  size_t i = blockIdx.x * static_cast<size_t>(blockDim.x) + threadIdx.x;
  size_t stride = static_cast<size_t>(gridDim.x) * blockDim.x;
  for (; i < n8; i += stride) {
    uint4 d = dst[i];
    uint4 s = src[i];
    d.x = bf16x2_sum(d.x, s.x);
    d.y = bf16x2_sum(d.y, s.y);
    d.z = bf16x2_sum(d.z, s.z);
    d.w = bf16x2_sum(d.w, s.w);
    dst[i] = d;
  }
}

__global__ void sum_tail_bf16(uint16_t* __restrict__ dst, const uint16_t* __restrict__ src,
                              size_t begin, size_t end) {
  size_t i = begin + blockIdx.x * static_cast<size_t>(blockDim.x) + threadIdx.x;
  if (i < end) dst[i] = f32_to_bf16(bf16_to_f32(dst[i]) + bf16_to_f32(src[i]));
}

// flat indexing: thread i handles element i of the CONCATENATED rows
// (row = i / row_len4), so narrow rows (e.g. 64 floats = 16 float4)
// still use every lane — a row-per-block mapping left 94 % of the block
// idle at width 64.
// Every kernel bounds-checks the decoded local row against tab_rows: a
// misrouted or corrupt key must not scribble over the shared pool (or a
// hipIpc-mapped peer). The unsigned compare also catches keys below
// row_base (the subtraction wraps to a huge value). Gather returns
// zeros for such rows (defined output); scatters skip them.
__global__ void gather_rows_f32(const float4* __restrict__ table, const uint64_t* __restrict__ rows,
                                size_t nrows, size_t row_len4, float4* __restrict__ out,
                                int shift, uint64_t base, uint64_t tab_rows) {
  size_t total = nrows * row_len4;
  size_t i = blockIdx.x * static_cast<size_t>(blockDim.x) + threadIdx.x;
  size_t stride = static_cast<size_t>(gridDim.x) * blockDim.x;
  for (; i < total; i += stride) {
    size_t r = i / row_len4;
    size_t c = i - r * row_len4;
    uint64_t row = (rows[r] >> shift) - base;
    out[i] = row < tab_rows ? table[row * row_len4 + c] : float4{0.f, 0.f, 0.f, 0.f};
  }
}

__global__ void scatter_add_rows_f32(float4* __restrict__ table, const uint64_t* __restrict__ rows,
                                     size_t nrows, size_t row_len4,
                                     const float4* __restrict__ src, int shift, uint64_t base,
                                     uint64_t tab_rows) {
  size_t total = nrows * row_len4;
  size_t i = blockIdx.x * static_cast<size_t>(blockDim.x) + threadIdx.x;
  size_t stride = static_cast<size_t>(gridDim.x) * blockDim.x;
  for (; i < total; i += stride) {
    size_t r = i / row_len4;
    size_t c = i - r * row_len4;
    uint64_t row = (rows[r] >> shift) - base;
    if (row >= tab_rows) continue;
    float4* dst = table + row * row_len4 + c;
    float4 d = *dst;
    float4 v = src[i];
    d.x += v.x;
    d.y += v.y;
    d.z += v.z;
    d.w += v.w;
    *dst = d;
  }
}

__global__ void gather_rows_scalar_f32(const float* __restrict__ table,
                                       const uint64_t* __restrict__ rows, size_t nrows,
                                       size_t row_len, float* __restrict__ out, int shift,
                                       uint64_t base, uint64_t tab_rows) {
  for (size_t r = blockIdx.x; r < nrows; r += gridDim.x) {
    uint64_t row = (rows[r] >> shift) - base;
    const float* src = table + row * row_len;
    float* dst = out + r * row_len;
    for (size_t c = threadIdx.x; c < row_len; c += blockDim.x) {
      dst[c] = row < tab_rows ? src[c] : 0.f;
    }
  }
}

__global__ void scatter_assign_rows_f32(float4* __restrict__ table,
                                        const uint64_t* __restrict__ rows, size_t nrows,
                                        size_t row_len4, const float4* __restrict__ src,
                                        int shift, uint64_t base, uint64_t tab_rows) {
  size_t total = nrows * row_len4;
  size_t i = blockIdx.x * static_cast<size_t>(blockDim.x) + threadIdx.x;
  size_t stride = static_cast<size_t>(gridDim.x) * blockDim.x;
  for (; i < total; i += stride) {
    size_t r = i / row_len4;
    size_t c = i - r * row_len4;
    uint64_t row = (rows[r] >> shift) - base;
    if (row < tab_rows) table[row * row_len4 + c] = src[i];
  }
}

__global__ void scatter_assign_rows_scalar_f32(float* __restrict__ table,
                                               const uint64_t* __restrict__ rows, size_t nrows,
                                               size_t row_len, const float* __restrict__ src,
                                               int shift, uint64_t base, uint64_t tab_rows) {
  for (size_t r = blockIdx.x; r < nrows; r += gridDim.x) {
    uint64_t row = (rows[r] >> shift) - base;
    if (row >= tab_rows) continue;
    float* dst = table + row * row_len;
    const float* s = src + r * row_len;
    for (size_t c = threadIdx.x; c < row_len; c += blockDim.x) dst[c] = s[c];
  }
}

__global__ void scatter_add_rows_atomic_f32(float* __restrict__ table,
                                            const uint64_t* __restrict__ rows, size_t nrows,
                                            size_t row_len, const float* __restrict__ src,
                                            int shift, uint64_t base, uint64_t tab_rows) {
  for (size_t r = blockIdx.x; r < nrows; r += gridDim.x) {
    uint64_t row = (rows[r] >> shift) - base;
    if (row >= tab_rows) continue;
    float* dst = table + row * row_len;
    const float* s = src + r * row_len;
    for (size_t c = threadIdx.x; c < row_len; c += blockDim.x) {
      atomicAdd(&dst[c], s[c]);
    }
  }
}

struct DescArray {
  CopyDesc d[kMaxBatch];
  int n;
};

// balanced variant: segments are concatenated into one virtual chunk
// space (prefix = exclusive 16-B-chunk offsets); each BLOCK owns
// contiguous tiles of that space, so work splits evenly across mixed
// segment sizes (the per-segment grid-stride left small buckets on a
// handful of blocks while every block swept all 64 segment headers).
// Within a tile lanes stay consecutive (coalesced); the tile's segment
// is found once and walked forward.
struct BalancedDescArray {
  CopyDesc d[kMaxBatch];
  unsigned long long prefix[kMaxBatch + 1];  // chunk offsets, prefix[n] = total
  int n;
};

// kOp: 0 = byte assign, 1 = fp32 sum, 2 = bf16 sum (fp32 accumulate)
template <int kOp>
__global__ void batched_balanced_kernel(BalancedDescArray da, size_t chunks_per_block) {
  size_t total = da.prefix[da.n];
  size_t tile_begin = blockIdx.x * chunks_per_block;
  size_t tile_end = tile_begin + chunks_per_block;
  if (tile_end > total) tile_end = total;
  if (tile_begin >= total) return;
  // locate the first segment of this tile (binary search once)
  int seg = 0;
  {
    int lo = 0, hi = da.n - 1;
    while (lo < hi) {
      int mid = (lo + hi + 1) >> 1;
      if (da.prefix[mid] <= tile_begin) {
        lo = mid;
      } else {
        hi = mid - 1;
      }
    }
    seg = lo;
  }
  for (size_t g = tile_begin + threadIdx.x; g < tile_end; g += blockDim.x) {
    while (g >= da.prefix[seg + 1]) ++seg;  // walk forward (tiles are contiguous)
    size_t local = g - da.prefix[seg];
    if (kOp == 1) {
      float4* dst = reinterpret_cast<float4*>(da.d[seg].dst) + local;
      const float4* src = reinterpret_cast<const float4*>(da.d[seg].src) + local;
      float4 a = *dst;
      float4 b = *src;
      a.x += b.x;
      a.y += b.y;
      a.z += b.z;
      a.w += b.w;
      *dst = a;
    } else if (kOp == 2) {
      uint4* dst = reinterpret_cast<uint4*>(da.d[seg].dst) + local;
      const uint4* src = reinterpret_cast<const uint4*>(da.d[seg].src) + local;
      uint4 a = *dst;
      uint4 b = *src;
      a.x = bf16x2_sum(a.x, b.x);
      a.y = bf16x2_sum(a.y, b.y);
      a.z = bf16x2_sum(a.z, b.z);
      a.w = bf16x2_sum(a.w, b.w);
      *dst = a;
    } else {
      reinterpret_cast<uint4*>(da.d[seg].dst)[local] =
          reinterpret_cast<const uint4*>(da.d[seg].src)[local];
    }
  }
}

// every block strides over every segment's 16B chunks (grid sized for
// the concatenated total, so all segments together fill the 8 XCDs)
// (segment loop per block; fine for <= kMaxBatch segments)
__global__ void batched_assign_kernel(DescArray da) {
  for (int seg = 0; seg < da.n; ++seg) {
    uint4* dst = reinterpret_cast<uint4*>(da.d[seg].dst);
    const uint4* src = reinterpret_cast<const uint4*>(da.d[seg].src);
    size_t n4 = da.d[seg].nbytes / 16;
    size_t i = blockIdx.x * static_cast<size_t>(blockDim.x) + threadIdx.x;
    size_t stride = static_cast<size_t>(gridDim.x) * blockDim.x;
    for (; i < n4; i += stride) dst[i] = src[i];
  }
}

__global__ void batched_sum_kernel_bf16(DescArray da) {
  for (int seg = 0; seg < da.n; ++seg) {
    uint4* dst = reinterpret_cast<uint4*>(da.d[seg].dst);
    const uint4* src = reinterpret_cast<const uint4*>(da.d[seg].src);
    size_t n8 = da.d[seg].nbytes / 16;
    size_t i = blockIdx.x * static_cast<size_t>(blockDim.x) + threadIdx.x;
    size_t stride = static_cast<size_t>(gridDim.x) * blockDim.x;
    for (; i < n8; i += stride) {
      uint4 d = dst[i];
      uint4 s = src[i];
      d.x = bf16x2_sum(d.x, s.x);
      d.y = bf16x2_sum(d.y, s.y);
      d.z = bf16x2_sum(d.z, s.z);
      d.w = bf16x2_sum(d.w, s.w);
      dst[i] = d;
    }
  }
}

__global__ void batched_sum_kernel_f32(DescArray da) {
  for (int seg = 0; seg < da.n; ++seg) {
    float4* dst = reinterpret_cast<float4*>(da.d[seg].dst);
    const float4* src = reinterpret_cast<const float4*>(da.d[seg].src);
    size_t n4 = da.d[seg].nbytes / 16;
    size_t i = blockIdx.x * static_cast<size_t>(blockDim.x) + threadIdx.x;
    size_t stride = static_cast<size_t>(gridDim.x) * blockDim.x;
    for (; i < n4; i += stride) {
      float4 d = dst[i];
      float4 v = src[i];
      d.x += v.x;
      d.y += v.y;
      d.z += v.z;
      d.w += v.w;
      dst[i] = d;
    }
  }
}

}  // namespace

namespace {

bool BalancedBatchEnabled() {
  static const bool on = [] {
    const char* v = getenv("XPS_BALANCED_BATCH");
    return !v || atoi(v) != 0;  // default on
  }();
  return on;
}

template <int kOp>
void LaunchBalanced(const CopyDesc* descs_host, int n, hipStream_t s) {
  for (int off = 0; off < n; off += kMaxBatch) {
    BalancedDescArray da{};
    da.n = std::min(n - off, kMaxBatch);
    unsigned long long acc = 0;
    for (int i = 0; i < da.n; ++i) {
      da.d[i] = descs_host[off + i];
      da.prefix[i] = acc;
      acc += da.d[i].nbytes / 16;
    }
    da.prefix[da.n] = acc;
    if (acc == 0) continue;
    // tile size: fill ~8192 blocks (8 XCDs want >> 256 workgroups),
    // whole multiples of the block so lanes sweep full strides
    size_t cpb = (acc + 8191) / 8192;
    cpb = ((cpb + kBlock - 1) / kBlock) * kBlock;
    int grid = static_cast<int>((acc + cpb - 1) / cpb);
    hipLaunchKernelGGL((batched_balanced_kernel<kOp>), dim3(grid), dim3(kBlock), 0, s, da,
                       cpb);
  }
}

}  // namespace

void BatchedAssign(const CopyDesc* descs_host, int n, hipStream_t s) {
  if (BalancedBatchEnabled()) {
    LaunchBalanced<0>(descs_host, n, s);
    return;
  }
  size_t total = 0;
  for (int i = 0; i < n; ++i) total += descs_host[i].nbytes;
  DescArray da{};
  for (int off = 0; off < n; off += kMaxBatch) {
    da.n = std::min(n - off, kMaxBatch);
    for (int i = 0; i < da.n; ++i) da.d[i] = descs_host[off + i];
    hipLaunchKernelGGL(batched_assign_kernel, dim3(GridFor(total / 16)), dim3(kBlock), 0, s, da);
  }
}

void BatchedSumF32(const CopyDesc* descs_host, int n, hipStream_t s) {
  if (BalancedBatchEnabled()) {
    LaunchBalanced<1>(descs_host, n, s);
    return;
  }
  size_t total = 0;
  for (int i = 0; i < n; ++i) total += descs_host[i].nbytes;
  DescArray da{};
  for (int off = 0; off < n; off += kMaxBatch) {
    da.n = std::min(n - off, kMaxBatch);
    for (int i = 0; i < da.n; ++i) da.d[i] = descs_host[off + i];
    hipLaunchKernelGGL(batched_sum_kernel_f32, dim3(GridFor(total / 16)), dim3(kBlock), 0, s, da);
  }
}

void DenseAssign(void* dst, const void* src, size_t nbytes, hipStream_t s) {
  size_t n4 = nbytes / 16;
  if (n4) {
    hipLaunchKernelGGL(assign_kernel, dim3(GridFor(n4)), dim3(kBlock), 0, s,
                       static_cast<uint4*>(dst), static_cast<const uint4*>(src), n4);
  }
  if (nbytes % 16) {
    hipLaunchKernelGGL(assign_tail_kernel, dim3(1), dim3(kBlock), 0, s, static_cast<char*>(dst),
                       static_cast<const char*>(src), n4 * 16, nbytes);
  }
}

void DenseSumF32(float* dst, const float* src, size_t n, hipStream_t s) {
  size_t n4 = n / 4;
  if (n4) {
    hipLaunchKernelGGL(sum_kernel_f32, dim3(GridFor(n4)), dim3(kBlock), 0, s,
                       reinterpret_cast<float4*>(dst), reinterpret_cast<const float4*>(src), n4);
  }
  if (n % 4) {
    hipLaunchKernelGGL(sum_tail_f32, dim3(1), dim3(kBlock), 0, s, dst, src, n4 * 4, n);
  }
}

void DenseSumBf16(uint16_t* dst, const uint16_t* src, size_t n, hipStream_t s) {
  size_t n8 = n / 8;
  if (n8) {
    hipLaunchKernelGGL(sum_kernel_bf16, dim3(GridFor(n8)), dim3(kBlock), 0, s,
                       reinterpret_cast<uint4*>(dst), reinterpret_cast<const uint4*>(src), n8);
  }
  if (n % 8) {
    hipLaunchKernelGGL(sum_tail_bf16, dim3(1), dim3(kBlock), 0, s, dst, src, n8 * 8, n);
  }
}

void BatchedSumBf16(const CopyDesc* descs_host, int n, hipStream_t s) {
  if (BalancedBatchEnabled()) {
    LaunchBalanced<2>(descs_host, n, s);
    return;
  }
  size_t total = 0;
  for (int i = 0; i < n; ++i) total += descs_host[i].nbytes;
  DescArray da{};
  for (int off = 0; off < n; off += kMaxBatch) {
    da.n = std::min(n - off, kMaxBatch);
    for (int i = 0; i < da.n; ++i) da.d[i] = descs_host[off + i];
    hipLaunchKernelGGL(batched_sum_kernel_bf16, dim3(GridFor(total / 16)), dim3(kBlock), 0, s,
                       da);
  }
}

void SparseGatherF32(const float* table, const uint64_t* rows_dev, size_t nrows, size_t row_len,
                     float* out, hipStream_t s, int key_shift, uint64_t row_base,
                     uint64_t table_rows) {
  if (row_len % 4 == 0) {
    hipLaunchKernelGGL(gather_rows_f32, dim3(GridFor(nrows * (row_len / 4), 16384)), dim3(kBlock),
                       0, s, reinterpret_cast<const float4*>(table), rows_dev, nrows, row_len / 4,
                       reinterpret_cast<float4*>(out), key_shift, row_base, table_rows);
  } else {
    hipLaunchKernelGGL(gather_rows_scalar_f32, dim3(GridFor(nrows * row_len, 16384)), dim3(kBlock),
                       0, s, table, rows_dev, nrows, row_len, out, key_shift, row_base, table_rows);
  }
}

void SparseScatterAssignF32(float* table, const uint64_t* rows_dev, size_t nrows, size_t row_len,
                            const float* src, hipStream_t s, int key_shift, uint64_t row_base,
                            uint64_t table_rows) {
  if (row_len % 4 == 0) {
    hipLaunchKernelGGL(scatter_assign_rows_f32, dim3(GridFor(nrows * (row_len / 4), 16384)),
                       dim3(kBlock), 0, s, reinterpret_cast<float4*>(table), rows_dev, nrows,
                       row_len / 4, reinterpret_cast<const float4*>(src), key_shift, row_base,
                       table_rows);
  } else {
    hipLaunchKernelGGL(scatter_assign_rows_scalar_f32, dim3(GridFor(nrows * row_len, 16384)),
                       dim3(kBlock), 0, s, table, rows_dev, nrows, row_len, src, key_shift,
                       row_base, table_rows);
  }
}

void SparseScatterAddF32(float* table, const uint64_t* rows_dev, size_t nrows, size_t row_len,
                         const float* src, bool atomic, hipStream_t s, int key_shift,
                         uint64_t row_base, uint64_t table_rows) {
  if (atomic || row_len % 4 != 0) {
    hipLaunchKernelGGL(scatter_add_rows_atomic_f32, dim3(GridFor(nrows * row_len, 16384)),
                       dim3(kBlock), 0, s, table, rows_dev, nrows, row_len, src, key_shift,
                       row_base, table_rows);
  } else {
    hipLaunchKernelGGL(scatter_add_rows_f32, dim3(GridFor(nrows * (row_len / 4), 16384)),
                       dim3(kBlock), 0, s, reinterpret_cast<float4*>(table), rows_dev, nrows,
                       row_len / 4, reinterpret_cast<const float4*>(src), key_shift, row_base,
                       table_rows);
  }
}

}  // namespace kern
}  // namespace xps
