// GPU hash-table primitives: multimap hash join (build/probe) and group-by
// hash aggregate.
//
// Reference parity: JoinPrimitives (join_primitives.hpp/cu — hash_inner_join,
// gather-map algebra feeds off these maps) and cudf-style hash aggregate; the
// implementation is a fresh MI355X design:
//   * slots are single u64 words: (fingerprint32 << 32) | (row + 1); 0 = empty.
//     One device-scope atomicCAS claims a slot — no key sentinel, works for
//     every key type, and the representative row index makes key equality
//     race-free (key columns are read-only inputs).
//   * linear probing, power-of-two capacity at 50% max load; probing is one
//     64B-line-granular random access per step — HBM/L2 bound, so the table
//     stores nothing but the one word per slot to keep the working set small
//     (1B build rows -> 16 GiB slot array, mostly L3-missing by design).
//   * join is a multimap: every build row claims its own slot; probe scans
//     until the first empty word, collecting fingerprint-confirmed matches.
//   * output uses wave-aggregated global cursors (one atomicAdd per wave,
//     Guideline 12) — gather-map order is unspecified, as in the reference.
//   * 64-wide waves throughout; grid-stride loops capped for 8-XCD fill.
#include "srj_common.hpp"
#include "agg_common.hpp"
#include "table_equal.hpp"

namespace srj {

__device__ inline uint32_t fingerprint(uint64_t h) {
  uint32_t fp = (uint32_t)(h >> 32);
  return fp ? fp : 1u;  // keep packed word nonzero even for row 0 safety
}

// ---------------------------------------------------------------------------
// join build: every non-null-key row inserts (fp|row+1) at first empty slot
// ---------------------------------------------------------------------------
__global__ void join_build_kernel(const ColDesc* __restrict__ cols,
                                  const int32_t* __restrict__ top, int32_t ntop,
                                  int64_t nrows, uint64_t* __restrict__ slots,
                                  uint64_t mask) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t row = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; row < nrows;
       row += stride) {
    if (row_has_null_key(cols, top, ntop, row)) continue;  // never matches inner
    uint64_t h = row_hash64(cols, top, ntop, row);
    uint64_t packed = ((uint64_t)fingerprint(h) << 32) | (uint64_t)(row + 1);
    uint64_t s = h & mask;
    while (true) {
      uint64_t prev = atomicCAS(reinterpret_cast<unsigned long long*>(slots + s),
                                0ull, (unsigned long long)packed);
      if (prev == 0) break;
      s = (s + 1) & mask;
    }
  }
}

// ---------------------------------------------------------------------------
// probe: two-phase (count then fill). COUNT uses one wave-level reduction +
// a single atomicAdd per wave. FILL appends match pairs at a global cursor.
// ---------------------------------------------------------------------------
constexpr int GPIPE = 8;  // probe rows in flight per lane (as hashtable_i64)

template <bool FILL>
__global__ void join_probe_kernel(
    const ColDesc* __restrict__ bcols, const int32_t* __restrict__ btop,
    const ColDesc* __restrict__ pcols, const int32_t* __restrict__ ptop,
    int32_t ntop, int64_t nprobe, const uint64_t* __restrict__ slots,
    uint64_t mask, uint64_t* __restrict__ counter,
    int32_t* __restrict__ out_build, int64_t* __restrict__ out_probe,
    int64_t out_capacity, uint8_t* __restrict__ build_matched) {
  // software-pipelined batches: hashes and first slot words for GPIPE rows
  // are issued before any chain resolves (the slot-word loads are THE
  // random accesses this kernel is bound by; same structure as the int64
  // fast path in hashtable_i64.hip)
  int64_t tid = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t nthreads = (int64_t)gridDim.x * blockDim.x;
  uint64_t local = 0;
  for (int64_t base = tid * GPIPE; base < nprobe; base += nthreads * GPIPE) {
    bool act[GPIPE];
    uint64_t h[GPIPE];
    uint64_t s0[GPIPE];
    uint64_t w[GPIPE];
#pragma unroll
    for (int b = 0; b < GPIPE; ++b) {
      int64_t row = base + b;
      act[b] = row < nprobe && !row_has_null_key(pcols, ptop, ntop, row);
      h[b] = act[b] ? row_hash64(pcols, ptop, ntop, row) : 0;
      s0[b] = h[b] & mask;
    }
#pragma unroll
    for (int b = 0; b < GPIPE; ++b) w[b] = slots[s0[b]];
#pragma unroll
    for (int b = 0; b < GPIPE; ++b) {
      if (!act[b]) continue;
      int64_t row = base + b;
      uint32_t fp = fingerprint(h[b]);
      uint64_t s = s0[b];
      uint64_t word = w[b];
      while (word != 0) {
        if ((uint32_t)(word >> 32) == fp) {
          int64_t brow = (int64_t)(word & 0xffffffffu) - 1;
          if (rows_equal(bcols, btop, ntop, brow, pcols, ptop, row)) {
            if (FILL) {
              uint64_t pos = atomicAdd((unsigned long long*)counter, 1ull);
              if ((int64_t)pos < out_capacity) {
                out_build[pos] = (int32_t)brow;
                out_probe[pos] = row;
              }
              if (build_matched) build_matched[brow] = 1;
            } else {
              ++local;
            }
          }
        }
        s = (s + 1) & mask;
        word = slots[s];
      }
    }
  }
  if (!FILL) {
    local = wave_sum(local);
    if ((threadIdx.x & (WAVE - 1)) == 0 && local)
      atomicAdd((unsigned long long*)counter, (unsigned long long)local);
  }
}

// semi/anti: emit each probe row at most once if it has (no) match
template <int MODE>  // 0 = semi, 1 = anti
__global__ void join_semi_kernel(
    const ColDesc* __restrict__ bcols, const int32_t* __restrict__ btop,
    const ColDesc* __restrict__ pcols, const int32_t* __restrict__ ptop,
    int32_t ntop, int64_t nprobe, const uint64_t* __restrict__ slots,
    uint64_t mask, uint64_t* __restrict__ counter,
    int64_t* __restrict__ out_probe, int64_t out_capacity) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t row = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; row < nprobe;
       row += stride) {
    bool has_match = false;
    if (!row_has_null_key(pcols, ptop, ntop, row)) {
      uint64_t h = row_hash64(pcols, ptop, ntop, row);
      uint32_t fp = fingerprint(h);
      uint64_t s = h & mask;
      while (true) {
        uint64_t word = slots[s];
        if (word == 0) break;
        if ((uint32_t)(word >> 32) == fp) {
          int64_t brow = (int64_t)(word & 0xffffffffu) - 1;
          if (rows_equal(bcols, btop, ntop, brow, pcols, ptop, row)) {
            has_match = true;
            break;
          }
        }
        s = (s + 1) & mask;
      }
    }
    bool emit = (MODE == 0) ? has_match : !has_match;
    // wave-aggregated append
    uint64_t ballot = __ballot(emit);
    int lane = threadIdx.x & (WAVE - 1);
    uint64_t base = 0;
    int nset = __popcll(ballot);
    if (lane == __ffsll((unsigned long long)ballot) - 1 && nset) {
      base = atomicAdd((unsigned long long*)counter, (unsigned long long)nset);
    }
    int leader = __ffsll((unsigned long long)ballot) - 1;
    base = __shfl(base, leader >= 0 ? leader : 0, WAVE);
    if (emit) {
      uint64_t pos = base + __popcll(ballot & ((1ull << lane) - 1));
      if ((int64_t)pos < out_capacity) out_probe[pos] = row;
    }
  }
}

// ---------------------------------------------------------------------------
// group-by aggregate
__global__ void groupby_kernel(const ColDesc* __restrict__ cols,
                               const int32_t* __restrict__ top, int32_t ntop,
                               int64_t nrows, uint64_t* __restrict__ slots,
                               uint64_t mask, const AggDesc* __restrict__ aggs,
                               int32_t naggs, int32_t* __restrict__ overflow) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t row = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; row < nrows;
       row += stride) {
    if (*overflow) return;  // undersized hint: host re-runs unhinted
    uint64_t h = row_hash64(cols, top, ntop, row);
    uint32_t fp = fingerprint(h);
    uint64_t packed = ((uint64_t)fp << 32) | (uint64_t)(row + 1);
    uint64_t s = h & mask;
    // insert-or-find, probe-bounded (a chain past 1024 means the table is
    // overloaded from a bad cardinality hint -- flag and bail)
    int64_t bound = (int64_t)mask < 1024 ? (int64_t)mask : 1024;
    bool found = false;
    for (int64_t probes = 0; probes <= bound; ++probes) {
      uint64_t word = slots[s];
      if (word == 0) {
        word = atomicCAS(reinterpret_cast<unsigned long long*>(slots + s), 0ull,
                         (unsigned long long)packed);
        if (word == 0) { found = true; break; }
      }
      if ((uint32_t)(word >> 32) == fp) {
        int64_t repr = (int64_t)(word & 0xffffffffu) - 1;
        if (repr == row || rows_equal(cols, top, ntop, repr, cols, top, row)) {
          found = true;
          break;
        }
      }
      s = (s + 1) & mask;
    }
    if (!found) {
      atomicOr(overflow, 1);
      continue;
    }
    agg_accumulate(aggs, naggs, row, (int64_t)s);
  }
}

// compact occupied slots: emit representative row + per-agg value
__global__ void groupby_compact_kernel(const uint64_t* __restrict__ slots,
                                       int64_t capacity,
                                       const AggDesc* __restrict__ aggs,
                                       int32_t naggs,
                                       uint64_t* __restrict__ counter,
                                       int64_t* __restrict__ out_repr,
                                       int64_t* __restrict__ out_agg_base,
                                       int64_t out_capacity) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t s = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; s < capacity;
       s += stride) {
    uint64_t word = slots[s];
    bool occ = word != 0;
    uint64_t ballot = __ballot(occ);
    int lane = threadIdx.x & (WAVE - 1);
    int nset = __popcll(ballot);
    int leader = __ffsll((unsigned long long)ballot) - 1;
    uint64_t base = 0;
    if (nset && lane == leader)
      base = atomicAdd((unsigned long long*)counter, (unsigned long long)nset);
    base = __shfl(base, leader >= 0 ? leader : 0, WAVE);
    if (occ) {
      uint64_t pos = base + __popcll(ballot & ((1ull << lane) - 1));
      if ((int64_t)pos < out_capacity) {
        out_repr[pos] = (int64_t)(word & 0xffffffffu) - 1;
        for (int32_t a = 0; a < naggs; ++a) {
          // out layout: agg a at out_agg_base + a*out_capacity (as i64 words)
          out_agg_base[(int64_t)a * out_capacity + (int64_t)pos] =
              reinterpret_cast<const int64_t*>(aggs[a].state)[s];
        }
      }
    }
  }
}

}  // namespace srj

using namespace srj;

extern "C" {

void srj_join_build(const void* cols, const int32_t* top, int32_t ntop,
                    int64_t nrows, uint64_t* slots, int64_t capacity,
                    hipStream_t stream) {
  join_build_kernel<<<grid_1d(nrows), DEFAULT_BLOCK, 0, stream>>>(
      reinterpret_cast<const ColDesc*>(cols), top, ntop, nrows, slots,
      (uint64_t)(capacity - 1));
}

void srj_join_probe_count(const void* bcols, const int32_t* btop,
                          const void* pcols, const int32_t* ptop, int32_t ntop,
                          int64_t nprobe, const uint64_t* slots, int64_t capacity,
                          uint64_t* counter, hipStream_t stream) {
  join_probe_kernel<false><<<grid_1d((nprobe + GPIPE - 1) / GPIPE),
                             DEFAULT_BLOCK, 0, stream>>>(
      reinterpret_cast<const ColDesc*>(bcols), btop,
      reinterpret_cast<const ColDesc*>(pcols), ptop, ntop, nprobe, slots,
      (uint64_t)(capacity - 1), counter, nullptr, nullptr, 0, nullptr);
}

void srj_join_probe_fill(const void* bcols, const int32_t* btop,
                         const void* pcols, const int32_t* ptop, int32_t ntop,
                         int64_t nprobe, const uint64_t* slots, int64_t capacity,
                         uint64_t* counter, int32_t* out_build, int64_t* out_probe,
                         int64_t out_capacity, uint8_t* build_matched,
                         hipStream_t stream) {
  join_probe_kernel<true><<<grid_1d(nprobe), DEFAULT_BLOCK, 0, stream>>>(
      reinterpret_cast<const ColDesc*>(bcols), btop,
      reinterpret_cast<const ColDesc*>(pcols), ptop, ntop, nprobe, slots,
      (uint64_t)(capacity - 1), counter, out_build, out_probe, out_capacity,
      build_matched);
}

void srj_join_semi(const void* bcols, const int32_t* btop, const void* pcols,
                   const int32_t* ptop, int32_t ntop, int64_t nprobe,
                   const uint64_t* slots, int64_t capacity, uint64_t* counter,
                   int64_t* out_probe, int64_t out_capacity, int32_t anti,
                   hipStream_t stream) {
  if (anti)
    join_semi_kernel<1><<<grid_1d(nprobe), DEFAULT_BLOCK, 0, stream>>>(
        reinterpret_cast<const ColDesc*>(bcols), btop,
        reinterpret_cast<const ColDesc*>(pcols), ptop, ntop, nprobe, slots,
        (uint64_t)(capacity - 1), counter, out_probe, out_capacity);
  else
    join_semi_kernel<0><<<grid_1d(nprobe), DEFAULT_BLOCK, 0, stream>>>(
        reinterpret_cast<const ColDesc*>(bcols), btop,
        reinterpret_cast<const ColDesc*>(pcols), ptop, ntop, nprobe, slots,
        (uint64_t)(capacity - 1), counter, out_probe, out_capacity);
}

void srj_groupby(const void* cols, const int32_t* top, int32_t ntop,
                 int64_t nrows, uint64_t* slots, int64_t capacity,
                 const void* aggs, int32_t naggs, int32_t* overflow,
                 hipStream_t stream) {
  groupby_kernel<<<grid_1d(nrows), DEFAULT_BLOCK, 0, stream>>>(
      reinterpret_cast<const ColDesc*>(cols), top, ntop, nrows, slots,
      (uint64_t)(capacity - 1), reinterpret_cast<const AggDesc*>(aggs), naggs,
      overflow);
}

void srj_groupby_compact(const uint64_t* slots, int64_t capacity, const void* aggs,
                         int32_t naggs, uint64_t* counter, int64_t* out_repr,
                         int64_t* out_agg, int64_t out_capacity,
                         hipStream_t stream) {
  groupby_compact_kernel<<<grid_1d(capacity), DEFAULT_BLOCK, 0, stream>>>(
      slots, capacity, reinterpret_cast<const AggDesc*>(aggs), naggs, counter,
      out_repr, out_agg, out_capacity);
}

}  // extern "C"
