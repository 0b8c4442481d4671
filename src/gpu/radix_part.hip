// Radix partitioning of int64 keys by hash prefix (the probe-side half of
// the partitioned hash join; also the kernel under the Spark-style shuffle
// partition write).
//
// Reference analog: cudf's hash partition under spark-rapids' shuffle; here
// it exists to make the NDS join probe LLC-local: the i64 table indexes
// slots by the TOP hash bits (hashtable_i64.hip slot_of), so keys grouped by
// the same prefix probe a contiguous ~128 MiB table window that fits the
// 256 MiB Infinity Cache instead of walking 64 GB at random.
//
// Two kernels:
//   * part_hist: global histogram of partitions (LDS-staged atomics).
//   * part_scatter: each block takes a contiguous row slab, histograms it in
//     LDS, reserves per-partition output ranges with one atomicAdd per
//     (block, partition), then scatters (key, source row) pairs. Order
//     within a partition is arbitrary (join output order is unspecified).
#include "srj_common.hpp"

namespace srj {

__device__ inline uint64_t part_hash(long long k) { return mix64((uint64_t)k); }

__global__ void part_hist_kernel(const long long* __restrict__ keys, int64_t n,
                                 int32_t pbits,
                                 int64_t* __restrict__ hist) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  uint32_t* lh = reinterpret_cast<uint32_t*>(smem);
  int np = 1 << pbits;
  for (int i = threadIdx.x; i < np; i += blockDim.x) lh[i] = 0;
  __syncthreads();
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t row = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; row < n;
       row += stride) {
    uint32_t p = (uint32_t)(part_hash(keys[row]) >> (64 - pbits));
    atomicAdd(&lh[p], 1u);
  }
  __syncthreads();
  for (int i = threadIdx.x; i < np; i += blockDim.x)
    if (lh[i]) atomicAdd(reinterpret_cast<unsigned long long*>(&hist[i]),
                         (unsigned long long)lh[i]);
}

// rows per scatter block (contiguous slab); 256 threads x 32 rows
constexpr int SCATTER_ROWS = 8192;

__global__ void part_scatter_kernel(const long long* __restrict__ keys,
                                    int64_t n, int32_t pbits,
                                    int64_t* __restrict__ cursors,
                                    long long* __restrict__ out_keys,
                                    int32_t* __restrict__ out_idx) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  uint32_t* lh = reinterpret_cast<uint32_t*>(smem);           // [np] counts
  int64_t* lbase = reinterpret_cast<int64_t*>(smem + ((1 << pbits) + 1) / 2 * 8);
  int np = 1 << pbits;
  int64_t r0 = (int64_t)blockIdx.x * SCATTER_ROWS;
  int64_t r1 = r0 + SCATTER_ROWS < n ? r0 + SCATTER_ROWS : n;
  for (int i = threadIdx.x; i < np; i += blockDim.x) lh[i] = 0;
  __syncthreads();
  for (int64_t row = r0 + threadIdx.x; row < r1; row += blockDim.x) {
    uint32_t p = (uint32_t)(part_hash(keys[row]) >> (64 - pbits));
    atomicAdd(&lh[p], 1u);
  }
  __syncthreads();
  // reserve output ranges; reuse lh as the block-local running cursor
  for (int i = threadIdx.x; i < np; i += blockDim.x) {
    uint32_t c = lh[i];
    lbase[i] = c ? atomicAdd(reinterpret_cast<unsigned long long*>(&cursors[i]),
                             (unsigned long long)c)
                 : 0;
    lh[i] = 0;
  }
  __syncthreads();
  for (int64_t row = r0 + threadIdx.x; row < r1; row += blockDim.x) {
    long long k = keys[row];
    uint32_t p = (uint32_t)(part_hash(k) >> (64 - pbits));
    uint32_t o = atomicAdd(&lh[p], 1u);
    int64_t dst = lbase[p] + o;
    out_keys[dst] = k;
    out_idx[dst] = (int32_t)row;
  }
}

}  // namespace srj

using namespace srj;

extern "C" {

void srj_part_hist(const long long* keys, int64_t n, int32_t pbits,
                   int64_t* hist, hipStream_t stream) {
  int np = 1 << pbits;
  part_hist_kernel<<<grid_1d(n), DEFAULT_BLOCK, np * 4, stream>>>(keys, n,
                                                                  pbits, hist);
}

void srj_part_scatter(const long long* keys, int64_t n, int32_t pbits,
                      int64_t* cursors, long long* out_keys, int32_t* out_idx,
                      hipStream_t stream) {
  int np = 1 << pbits;
  int64_t nblocks = (n + SCATTER_ROWS - 1) / SCATTER_ROWS;
  // LDS: np u32 counts (rounded to 8B) + np i64 bases
  size_t lds = (size_t)((np + 1) / 2) * 8 + (size_t)np * 8;
  part_scatter_kernel<<<nblocks, DEFAULT_BLOCK, lds, stream>>>(
      keys, n, pbits, cursors, out_keys, out_idx);
}

}  // extern "C"
