// Copy primitives: gather (fixed-width / strings / validity) and hash
// partition (histogram + stable-enough scatter), the building blocks for
// join materialization, group-by key output, and shuffle.
//
// Reference parity: libcudf gather + cudf hash_partition as used by the
// reference's ops; negative gather-map entries produce nulls
// (the cudf convention the reference's make_left_outer/full_outer rely on).
//
// MI355X design notes:
//  * gather is one thread per output row, 64B+ coalesced writes; random reads
//    ride L2/L3 (Guideline: scatter/gather rely on cache hierarchy).
//  * partition scatter computes per-block LDS histograms and claims one
//    global cursor range per (block, partition) — one atomicAdd per partition
//    per block instead of one per row (Guideline 12).
#include "srj_common.hpp"

namespace srj {

// ---------------------------------------------------------------------------
// gather: fixed-width by element size
// ---------------------------------------------------------------------------
template <typename T>
__global__ void gather_fixed_kernel(const T* __restrict__ in,
                                    const uint8_t* __restrict__ in_valid,
                                    const int64_t* __restrict__ map, int64_t n,
                                    T* __restrict__ out,
                                    uint8_t* __restrict__ out_valid) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < ((n + WAVE - 1) & ~(int64_t)(WAVE - 1)); i += stride) {
    bool in_range = i < n;
    int64_t idx = in_range ? map[i] : -1;
    bool valid = in_range && idx >= 0 && is_valid(in_valid, idx);
    T v = T{};
    if (valid) v = in[idx];
    if (in_range) out[i] = v;
    if (out_valid) ballot_write_validity(out_valid, i, valid);
  }
}

__global__ void gather_str_lengths_kernel(const int32_t* __restrict__ offsets,
                                          const int64_t* __restrict__ map,
                                          int64_t n, int32_t* __restrict__ lens) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    int64_t idx = map[i];
    lens[i] = idx >= 0 ? offsets[idx + 1] - offsets[idx] : 0;
  }
}

__global__ void gather_str_chars_kernel(const char* __restrict__ in_chars,
                                        const int32_t* __restrict__ in_offsets,
                                        const uint8_t* __restrict__ in_valid,
                                        const int64_t* __restrict__ map,
                                        const int32_t* __restrict__ out_offsets,
                                        int64_t n, char* __restrict__ out_chars,
                                        uint8_t* __restrict__ out_valid) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < ((n + WAVE - 1) & ~(int64_t)(WAVE - 1)); i += stride) {
    bool in_range = i < n;
    int64_t idx = in_range ? map[i] : -1;
    bool valid = in_range && idx >= 0 && is_valid(in_valid, idx);
    if (in_range && idx >= 0) {
      int32_t s = in_offsets[idx], e = in_offsets[idx + 1];
      int32_t d = out_offsets[i];
      for (int32_t k = 0; k < e - s; ++k) out_chars[d + k] = in_chars[s + k];
    }
    if (out_valid) ballot_write_validity(out_valid, i, valid);
  }
}

// validity-only gather (STRUCT/LIST columns have no data buffer)
__global__ void gather_validity_kernel(const uint8_t* __restrict__ in_valid,
                                       const int64_t* __restrict__ map,
                                       int64_t n,
                                       uint8_t* __restrict__ out_valid) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < ((n + WAVE - 1) & ~(int64_t)(WAVE - 1)); i += stride) {
    bool in_range = i < n;
    int64_t idx = in_range ? map[i] : -1;
    bool valid = in_range && idx >= 0 && is_valid(in_valid, idx);
    ballot_write_validity(out_valid, i, valid);
  }
}

// ---------------------------------------------------------------------------
// hash partition
// ---------------------------------------------------------------------------
__global__ void partition_hist_kernel(const int32_t* __restrict__ parts,
                                      int64_t n, int32_t nparts,
                                      int64_t* __restrict__ hist) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  int32_t* lhist = reinterpret_cast<int32_t*>(smem);
  for (int32_t p = threadIdx.x; p < nparts; p += blockDim.x) lhist[p] = 0;
  __syncthreads();
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    atomicAdd(lhist + parts[i], 1);
  }
  __syncthreads();
  for (int32_t p = threadIdx.x; p < nparts; p += blockDim.x) {
    if (lhist[p])
      atomicAdd((unsigned long long*)(hist + p), (unsigned long long)lhist[p]);
  }
}

// scatter: produces gather map `perm` such that out[d] = in[perm[d]] groups
// rows by partition. cursors must be initialized to the exclusive-scan of the
// histogram.
__global__ void partition_scatter_kernel(const int32_t* __restrict__ parts,
                                         int64_t n, int32_t nparts,
                                         uint64_t* __restrict__ cursors,
                                         int64_t* __restrict__ perm) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  int32_t* lhist = reinterpret_cast<int32_t*>(smem);            // [nparts]
  int64_t* lbase = reinterpret_cast<int64_t*>(smem + ((nparts * 4 + 15) & ~15));
  int64_t chunk = (int64_t)gridDim.x * blockDim.x;
  for (int64_t base = (int64_t)blockIdx.x * blockDim.x; base < n; base += chunk) {
    int64_t i = base + threadIdx.x;
    for (int32_t p = threadIdx.x; p < nparts; p += blockDim.x) lhist[p] = 0;
    __syncthreads();
    int32_t p = -1, rank = 0;
    if (i < n) {
      p = parts[i];
      rank = atomicAdd(lhist + p, 1);
    }
    __syncthreads();
    for (int32_t q = threadIdx.x; q < nparts; q += blockDim.x) {
      lbase[q] = lhist[q]
                     ? (int64_t)atomicAdd((unsigned long long*)(cursors + q),
                                          (unsigned long long)lhist[q])
                     : 0;
    }
    __syncthreads();
    if (i < n) perm[lbase[p] + rank] = i;
    __syncthreads();
  }
}

// compute pmod(hash, nparts) like Spark's HashPartitioning
__global__ void pmod_kernel(const int32_t* __restrict__ hash, int64_t n,
                            int32_t nparts, int32_t* __restrict__ out) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    int32_t m = hash[i] % nparts;
    out[i] = m < 0 ? m + nparts : m;
  }
}

}  // namespace srj

using namespace srj;

extern "C" {

void srj_gather_fixed(const void* in, const uint8_t* in_valid, const int64_t* map,
                      int64_t n, void* out, uint8_t* out_valid, int32_t elem_size,
                      hipStream_t stream) {
  int64_t g = grid_1d(n);
  switch (elem_size) {
    case 1:
      gather_fixed_kernel<int8_t><<<g, DEFAULT_BLOCK, 0, stream>>>(
          (const int8_t*)in, in_valid, map, n, (int8_t*)out, out_valid);
      break;
    case 2:
      gather_fixed_kernel<int16_t><<<g, DEFAULT_BLOCK, 0, stream>>>(
          (const int16_t*)in, in_valid, map, n, (int16_t*)out, out_valid);
      break;
    case 4:
      gather_fixed_kernel<int32_t><<<g, DEFAULT_BLOCK, 0, stream>>>(
          (const int32_t*)in, in_valid, map, n, (int32_t*)out, out_valid);
      break;
    case 8:
      gather_fixed_kernel<int64_t><<<g, DEFAULT_BLOCK, 0, stream>>>(
          (const int64_t*)in, in_valid, map, n, (int64_t*)out, out_valid);
      break;
    case 16:
      gather_fixed_kernel<int4><<<g, DEFAULT_BLOCK, 0, stream>>>(
          (const int4*)in, in_valid, map, n, (int4*)out, out_valid);
      break;
  }
}

void srj_gather_validity(const uint8_t* in_valid, const int64_t* map,
                         int64_t n, uint8_t* out_valid, hipStream_t stream) {
  gather_validity_kernel<<<grid_1d(n), DEFAULT_BLOCK, 0, stream>>>(
      in_valid, map, n, out_valid);
}

void srj_gather_str_lengths(const int32_t* offsets, const int64_t* map, int64_t n,
                            int32_t* lens, hipStream_t stream) {
  gather_str_lengths_kernel<<<grid_1d(n), DEFAULT_BLOCK, 0, stream>>>(offsets, map,
                                                                      n, lens);
}

void srj_gather_str_chars(const char* in_chars, const int32_t* in_offsets,
                          const uint8_t* in_valid, const int64_t* map,
                          const int32_t* out_offsets, int64_t n, char* out_chars,
                          uint8_t* out_valid, hipStream_t stream) {
  gather_str_chars_kernel<<<grid_1d(n), DEFAULT_BLOCK, 0, stream>>>(
      in_chars, in_offsets, in_valid, map, out_offsets, n, out_chars, out_valid);
}

void srj_partition_hist(const int32_t* parts, int64_t n, int32_t nparts,
                        int64_t* hist, hipStream_t stream) {
  size_t smem = (size_t)nparts * 4;
  partition_hist_kernel<<<grid_1d(n), DEFAULT_BLOCK, smem, stream>>>(parts, n,
                                                                     nparts, hist);
}

void srj_partition_scatter(const int32_t* parts, int64_t n, int32_t nparts,
                           uint64_t* cursors, int64_t* perm, hipStream_t stream) {
  size_t smem = (size_t)((nparts * 4 + 15) & ~15) + (size_t)nparts * 8;
  partition_scatter_kernel<<<grid_1d(n), DEFAULT_BLOCK, smem, stream>>>(
      parts, n, nparts, cursors, perm);
}

void srj_pmod(const int32_t* hash, int64_t n, int32_t nparts, int32_t* out,
              hipStream_t stream) {
  pmod_kernel<<<grid_1d(n), DEFAULT_BLOCK, 0, stream>>>(hash, n, nparts, out);
}

}  // extern "C"
