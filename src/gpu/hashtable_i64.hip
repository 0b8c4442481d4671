// Specialized int64-key hash table (the NDS hot path: fact-table joins on
// int64 surrogate keys; BASELINE config[2]).
//
// Why a separate design from hashtable.hip's generic (fp|row) table: the
// generic probe is a dependent chain of TWO random loads per row (slot word,
// then the build key for equality). rocprof on the v1 kernel showed it
// latency-bound at ~3% of HBM bandwidth (profiles/r01_join_v1_kernel_stats).
// Here:
//   * 16-byte slots {key, row+1} — key inline, so a probe is ONE random
//     16-byte load (one 64B line) per slot visited;
//   * software-pipelined batches of PIPE rows per lane: hashes and first-slot
//     loads for the whole batch are issued independently before any resolve,
//     giving each lane PIPE outstanding loads (memory-level parallelism)
//     instead of one dependent chain;
//   * wave-aggregated output append (one atomicAdd per wave for the common
//     first-match case), per-match atomics only for rare duplicate matches.
// Claim protocol: atomicCAS on the row word (0 = empty), winner writes key;
// no sentinel key needed, build completes before probe launches.
#include "srj_common.hpp"
#include "agg_common.hpp"

namespace srj {

struct Slot64 {
  long long key;
  long long row1;  // low 62 bits: row + 1 (0 = empty); bit 62: chain bit,
                   // set when some LATER insert probed through this slot —
                   // a probe may stop at the first clear chain bit instead
                   // of loading slots until it sees an empty one (cuts the
                   // common case from 2 random loads to 1)
};

constexpr long long SLOT_CHAIN = 1ll << 62;
constexpr long long SLOT_ROW = SLOT_CHAIN - 1;

constexpr int PIPE = 8;

__device__ inline uint64_t i64_hash(long long k) { return mix64((uint64_t)k); }

// Slot from the TOP hash bits (mask = 2^k - 1). This orders the table by
// hash prefix; a radix-partitioned probe over it was built and measured NOT
// to pay on the 10Bx1B config (docs/PERF.md) — the layout is kept because it
// is equivalent in cost to low-bit indexing and keeps that option open.
__device__ inline uint64_t slot_of(uint64_t h, uint64_t mask) {
  return h >> __builtin_clzll(mask);
}

// ---------------------------------------------------------------------------
// build
// ---------------------------------------------------------------------------
__global__ void join_build_i64_kernel(const long long* __restrict__ keys,
                                      const uint8_t* __restrict__ valid,
                                      int64_t nrows, Slot64* __restrict__ slots,
                                      uint64_t mask) {
  int64_t tid = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t nthreads = (int64_t)gridDim.x * blockDim.x;
  // batch of PIPE strided rows per iteration
  for (int64_t base = tid * PIPE; base < nrows; base += nthreads * PIPE) {
    long long k[PIPE];
    uint64_t s[PIPE];
    bool act[PIPE];
    long long prev[PIPE];
#pragma unroll
    for (int b = 0; b < PIPE; ++b) {
      int64_t row = base + b;
      act[b] = row < nrows && is_valid(valid, row);
      k[b] = act[b] ? keys[row < nrows ? row : 0] : 0;
      s[b] = act[b] ? slot_of(i64_hash(k[b]), mask) : 0;
    }
    // first-slot claims issued back-to-back (independent atomics pipeline
    // with counted vmcnt); inactive lanes CAS 0->0 on slot 0, a no-op
#pragma unroll
    for (int b = 0; b < PIPE; ++b) {
      prev[b] = atomicCAS(
          reinterpret_cast<unsigned long long*>(&slots[s[b]].row1), 0ull,
          (unsigned long long)(act[b] ? base + b + 1 : 0));
    }
#pragma unroll
    for (int b = 0; b < PIPE; ++b) {
      if (!act[b]) continue;
      if (prev[b] == 0) {
        slots[s[b]].key = k[b];
        continue;
      }
      int64_t row = base + b;
      // mark the displaced-over slots so probes know the chain continues
      atomicOr(reinterpret_cast<unsigned long long*>(&slots[s[b]].row1),
               (unsigned long long)SLOT_CHAIN);
      uint64_t sl = (s[b] + 1) & mask;
      while (true) {
        long long p = atomicCAS(
            reinterpret_cast<unsigned long long*>(&slots[sl].row1), 0ull,
            (unsigned long long)(row + 1));
        if (p == 0) {
          slots[sl].key = k[b];
          break;
        }
        atomicOr(reinterpret_cast<unsigned long long*>(&slots[sl].row1),
                 (unsigned long long)SLOT_CHAIN);
        sl = (sl + 1) & mask;
      }
    }
  }
}

// ---------------------------------------------------------------------------
// probe (count / fill fused via template)
// ---------------------------------------------------------------------------
// Probe with count-then-emit per batch:
//  * stage 1: one validity byte covers the whole 8-row batch (PIPE == 8);
//    probe keys and first slots load unconditionally (clamped), back-to-back
//    — the compiler then emits 8 independent dwordx4 loads with counted
//    vmcnt, which is the MLP this kernel lives on.
//  * stage 2 (resolve): count this lane's matches; slot lines end L1-warm.
//  * stage 3 (emit, FILL only): ONE wave-wide atomicAdd reserves output space
//    (wave prefix sum gives per-lane bases), then a cache-warm rescan writes
//    the pairs with plain stores — no per-match atomics.
template <bool FILL, bool HAS_VALID, int PPIPE>
__global__ void join_probe_i64_kernel(
    const long long* __restrict__ probe, const uint8_t* __restrict__ pvalid,
    int64_t nprobe, const Slot64* __restrict__ slots, uint64_t mask,
    uint64_t* __restrict__ counter, int32_t* __restrict__ out_build,
    int64_t* __restrict__ out_probe, int64_t out_capacity,
    uint8_t* __restrict__ build_matched,
    const int32_t* __restrict__ idxmap /* source rows of partitioned keys */) {
  static_assert(PPIPE == 8 || PPIPE == 16, "validity bytes per batch");
  int64_t tid = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t nthreads = (int64_t)gridDim.x * blockDim.x;
  int lane = threadIdx.x & (WAVE - 1);
  uint64_t count_local = 0;
  int64_t nbatch = (nprobe + (int64_t)PPIPE - 1) / PPIPE;
  // pad so every lane of a wave runs the same iterations (wave shuffles)
  int64_t nbatch_pad = (nbatch + WAVE - 1) & ~(int64_t)(WAVE - 1);
  for (int64_t batch = tid; batch < nbatch_pad; batch += nthreads) {
    bool batch_ok = batch < nbatch;
    int64_t base = batch_ok ? batch * PPIPE : 0;
    uint32_t vbits = (1u << PPIPE) - 1u;
    if (HAS_VALID) {
      vbits = pvalid[base >> 3];
      if (PPIPE == 16) vbits |= (uint32_t)pvalid[(base >> 3) + 1] << 8;
    }
    if (base + PPIPE > nprobe) {
      int64_t tail = nprobe - base;
      vbits &= (uint32_t)((1u << (tail < 0 ? 0 : tail)) - 1u);
    }
    if (!batch_ok) vbits = 0;
    long long k[PPIPE];
    uint64_t s[PPIPE];
    Slot64 first[PPIPE];
#pragma unroll
    for (int b = 0; b < PPIPE; ++b) {
      int64_t row = base + b;
      k[b] = probe[row < nprobe ? row : 0];  // clamped, unconditional
      s[b] = slot_of(i64_hash(k[b]), mask);
    }
#pragma unroll
    for (int b = 0; b < PPIPE; ++b) first[b] = slots[s[b]];
    // resolve: count matches; carry the FIRST match's build row in a
    // register so the (overwhelmingly common) exactly-one-match case emits
    // without re-walking the slot chain
    uint32_t nm = 0;
    uint32_t nmb[PPIPE];
    long long hit1[PPIPE];
#pragma unroll
    for (int b = 0; b < PPIPE; ++b) {
      nmb[b] = 0;
      hit1[b] = 0;
      if (!((vbits >> b) & 1)) continue;
      Slot64 cur = first[b];
      uint64_t sl = s[b];
      while (cur.row1 != 0) {
        if (cur.key == k[b]) {
          if (nmb[b] == 0) hit1[b] = cur.row1 & SLOT_ROW;
          ++nmb[b];
        }
        if (!(cur.row1 & SLOT_CHAIN)) break;  // chain ends here
        sl = (sl + 1) & mask;
        cur = slots[sl];
      }
      nm += nmb[b];
    }
    if (!FILL) {
      count_local += nm;
      continue;
    }
    // emit: one atomic per wave; single-match rows write straight from the
    // carried register, only multi-match rows rescan (cache-warm)
    uint32_t incl = wave_prefix_incl(nm);
    uint32_t total = __shfl(incl, WAVE - 1, WAVE);
    uint64_t wave_base = 0;
    if (lane == WAVE - 1 && total)
      wave_base = atomicAdd((unsigned long long*)counter, (unsigned long long)total);
    wave_base = __shfl(wave_base, WAVE - 1, WAVE);
    int64_t pos = (int64_t)(wave_base + incl - nm);
    if (nm) {
#pragma unroll
      for (int b = 0; b < PPIPE; ++b) {
        if (nmb[b] == 0) continue;
        if (nmb[b] == 1) {
          if (pos < out_capacity) {
            out_build[pos] = (int32_t)(hit1[b] - 1);
            out_probe[pos] = idxmap ? (int64_t)idxmap[base + b] : base + b;
          }
          // matched flags are valid even when the pair output overflows
          // (capacity-0 mark-only probes depend on this)
          if (build_matched) build_matched[hit1[b] - 1] = 1;
          ++pos;
          continue;
        }
        Slot64 cur = first[b];
        uint64_t sl = s[b];
        while (cur.row1 != 0) {
          if (cur.key == k[b]) {
            long long r1 = cur.row1 & SLOT_ROW;
            if (pos < out_capacity) {
              out_build[pos] = (int32_t)(r1 - 1);
              out_probe[pos] = idxmap ? (int64_t)idxmap[base + b] : base + b;
            }
            if (build_matched) build_matched[r1 - 1] = 1;
            ++pos;
          }
          if (!(cur.row1 & SLOT_CHAIN)) break;
          sl = (sl + 1) & mask;
          cur = slots[sl];
        }
      }
    }
  }
  if (!FILL) {
    count_local = wave_sum(count_local);
    if (lane == 0 && count_local)
      atomicAdd((unsigned long long*)counter, (unsigned long long)count_local);
  }
}


// ---------------------------------------------------------------------------
// specialized int64-key group-by (BASELINE config[1]: hash-aggregate over an
// int64 grouping key). Differences from the generic groupby_kernel:
//   * 16B slots {key, row1} with the key CLAIMED BY CAS (EMPTY_KEY sentinel),
//     so concurrent insert-or-find needs no representative-row re-load: one
//     random 16B load resolves the common hit case;
//   * software-pipelined batches of PIPE rows (hash + first-slot loads issued
//     back-to-back) — the same MLP structure that took the join probe from
//     1.2 to 7.5+ B rows/s;
//   * rows whose key equals the sentinel (INT64_MIN) fall through to a single
//     reserved slot at index `capacity` (aggregated with plain atomics).
// Aggregation itself is the shared agg_accumulate switch.
// ---------------------------------------------------------------------------
constexpr long long GB_EMPTY_KEY = 0x8000000000000000ll;  // INT64_MIN

// returns slot index, or -1 when the table is FULL (undersized cardinality
// hint) — the caller sets the overflow flag and the host re-runs unhinted
__device__ inline int64_t gb64_find_or_claim(Slot64* __restrict__ slots,
                                             uint64_t mask, int64_t capacity,
                                             long long k, int64_t row) {
  if (k == GB_EMPTY_KEY) {
    atomicCAS(reinterpret_cast<unsigned long long*>(&slots[capacity].row1),
              0ull, (unsigned long long)(row + 1));
    return capacity;
  }
  uint64_t sl = slot_of(i64_hash(k), mask);
  Slot64 cur = slots[sl];
  // longest linear-probe cluster at <=50% load is O(log n) (~100 at 4B
  // slots); a chain past 1024 means the table is overloaded (bad hint)
  int64_t bound = (int64_t)mask < 1024 ? (int64_t)mask : 1024;
  for (int64_t probes = 0; probes <= bound; ++probes) {
    if (cur.key == k) return (int64_t)sl;
    if (cur.key == GB_EMPTY_KEY) {
      long long prev = atomicCAS(
          reinterpret_cast<unsigned long long*>(&slots[sl].key),
          (unsigned long long)GB_EMPTY_KEY, (unsigned long long)k);
      if (prev == GB_EMPTY_KEY) {
        slots[sl].row1 = row + 1;
        return (int64_t)sl;
      }
      if (prev == k) return (int64_t)sl;
    }
    sl = (sl + 1) & mask;
    cur = slots[sl];
  }
  return -1;
}

// ---------------------------------------------------------------------------
// LDS pre-aggregated group-by for LOW-cardinality keys (planner hint):
// each workgroup keeps a 2048-slot {key, aggs} table in LDS and only merges
// per-block partials into the global table at the end — global atomic
// traffic drops from one-per-row to one-per-(block x group), fixing the
// hot-address contention that capped 100-group aggregation at 2.5 B rows/s.
// Keys overflowing the LDS table fall through to the direct global path.
// ---------------------------------------------------------------------------
constexpr int GB_LDS_CAP = 2048;   // slots (pow2)
constexpr int GB_LDS_MAX_AGGS = 3;

__global__ void groupby_i64_lds_kernel(
    const long long* __restrict__ keys, int64_t nrows,
    Slot64* __restrict__ slots, uint64_t mask,
    const AggDesc* __restrict__ aggs, int32_t naggs,
    const int64_t* __restrict__ identities,
    int32_t* __restrict__ overflow) {
  __shared__ long long lkey[GB_LDS_CAP];
  __shared__ long long lrep[GB_LDS_CAP];  // a row holding this key
  __shared__ long long lagg[GB_LDS_MAX_AGGS][GB_LDS_CAP];
  int64_t capacity = (int64_t)mask + 1;
  for (int i = threadIdx.x; i < GB_LDS_CAP; i += blockDim.x) {
    lkey[i] = GB_EMPTY_KEY;
    lrep[i] = 0;
    for (int a = 0; a < naggs; ++a) lagg[a][i] = identities[a];
  }
  __syncthreads();

  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t row = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       row < nrows; row += stride) {
    // on overflow: BREAK (not return) so every wave still reaches the
    // __syncthreads() before the flush — the host re-runs unhinted anyway
    if (*overflow) break;
    long long k = keys[row];
    int lidx = -1;
    if (k != GB_EMPTY_KEY) {
      uint32_t sl = (uint32_t)(i64_hash(k) >> 32) & (GB_LDS_CAP - 1);
      for (int probe = 0; probe < 32; ++probe) {  // bounded: full -> global
        long long cur = lkey[sl];
        if (cur == k) { lidx = (int)sl; break; }
        if (cur == GB_EMPTY_KEY) {
          long long prev = atomicCAS(
              reinterpret_cast<unsigned long long*>(&lkey[sl]),
              (unsigned long long)GB_EMPTY_KEY, (unsigned long long)k);
          if (prev == GB_EMPTY_KEY) {
            lrep[sl] = row;  // claimer records a representative row
            lidx = (int)sl;
            break;
          }
          if (prev == k) { lidx = (int)sl; break; }
        }
        sl = (sl + 1) & (GB_LDS_CAP - 1);
      }
    }
    if (lidx < 0) {
      // sentinel key or LDS table full: direct global accumulate
      int64_t gi = gb64_find_or_claim(slots, mask, capacity, k, row);
      if (gi < 0) { atomicOr(overflow, 1); continue; }
      agg_accumulate(aggs, naggs, row, gi);
      continue;
    }
    // representative row for the group: claim via global table (once per
    // block per group is fine - find_or_claim is idempotent)
    for (int32_t a = 0; a < naggs; ++a) {
      const AggDesc& g = aggs[a];
      switch (g.op) {
        case AGG_COUNT_ALL:
          atomicAdd(reinterpret_cast<unsigned long long*>(&lagg[a][lidx]),
                    1ull);
          break;
        case AGG_COUNT_VALID:
          if (is_valid(g.valid, row))
            atomicAdd(reinterpret_cast<unsigned long long*>(&lagg[a][lidx]),
                      1ull);
          break;
        case AGG_SUM_INT64:
          if (is_valid(g.valid, row))
            atomicAdd(reinterpret_cast<unsigned long long*>(&lagg[a][lidx]),
                      (unsigned long long)fetch_int64(g.data, g.in_dtype, row));
          break;
        case AGG_SUM_FLOAT64:
          if (is_valid(g.valid, row))
            atomicAdd(reinterpret_cast<double*>(&lagg[a][lidx]),
                      fetch_double(g.data, g.in_dtype, row));
          break;
        case AGG_MIN_INT64:
          if (is_valid(g.valid, row))
            atomicMin(reinterpret_cast<long long*>(&lagg[a][lidx]),
                      (long long)fetch_int64(g.data, g.in_dtype, row));
          break;
        case AGG_MAX_INT64:
          if (is_valid(g.valid, row))
            atomicMax(reinterpret_cast<long long*>(&lagg[a][lidx]),
                      (long long)fetch_int64(g.data, g.in_dtype, row));
          break;
        case AGG_MIN_FLOAT64:
          if (is_valid(g.valid, row))
            atomic_min_f64(reinterpret_cast<double*>(&lagg[a][lidx]),
                           fetch_double(g.data, g.in_dtype, row));
          break;
        case AGG_MAX_FLOAT64:
          if (is_valid(g.valid, row))
            atomic_max_f64(reinterpret_cast<double*>(&lagg[a][lidx]),
                           fetch_double(g.data, g.in_dtype, row));
          break;
      }
    }
  }
  __syncthreads();
  // flush block partials into the global table (merge semantics)
  for (int i = threadIdx.x; i < GB_LDS_CAP; i += blockDim.x) {
    long long k = lkey[i];
    if (k == GB_EMPTY_KEY) continue;
    int64_t gi = gb64_find_or_claim(slots, mask, capacity, k, lrep[i]);
    if (gi < 0) { atomicOr(overflow, 1); continue; }
    for (int32_t a = 0; a < naggs; ++a) {
      const AggDesc& g = aggs[a];
      long long part = lagg[a][i];
      if (part == identities[a]) continue;
      switch (g.op) {
        case AGG_COUNT_ALL:
        case AGG_COUNT_VALID:
        case AGG_SUM_INT64:
          atomicAdd((unsigned long long*)g.state + gi,
                    (unsigned long long)part);
          break;
        case AGG_SUM_FLOAT64: {
          double d;
          __builtin_memcpy(&d, &part, 8);
          atomicAdd(reinterpret_cast<double*>(g.state) + gi, d);
          break;
        }
        case AGG_MIN_INT64:
          atomic_min_i64(reinterpret_cast<int64_t*>(g.state) + gi, part);
          break;
        case AGG_MAX_INT64:
          atomic_max_i64(reinterpret_cast<int64_t*>(g.state) + gi, part);
          break;
        case AGG_MIN_FLOAT64: {
          double d;
          __builtin_memcpy(&d, &part, 8);
          atomic_min_f64(reinterpret_cast<double*>(g.state) + gi, d);
          break;
        }
        case AGG_MAX_FLOAT64: {
          double d;
          __builtin_memcpy(&d, &part, 8);
          atomic_max_f64(reinterpret_cast<double*>(g.state) + gi, d);
          break;
        }
      }
    }
  }
}

__global__ void groupby_i64_kernel(const long long* __restrict__ keys,
                                   int64_t nrows, Slot64* __restrict__ slots,
                                   uint64_t mask,
                                   const AggDesc* __restrict__ aggs,
                                   int32_t naggs,
                                   int32_t* __restrict__ overflow) {
  int64_t tid = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t nthreads = (int64_t)gridDim.x * blockDim.x;
  int64_t capacity = (int64_t)mask + 1;
  for (int64_t base = tid * PIPE; base < nrows; base += nthreads * PIPE) {
    if (*overflow) return;  // undersized hint: host re-runs unhinted anyway
    long long k[PIPE];
    uint64_t s[PIPE];
    Slot64 first[PIPE];
#pragma unroll
    for (int b = 0; b < PIPE; ++b) {
      int64_t row = base + b;
      k[b] = keys[row < nrows ? row : 0];  // clamped, unconditional
      s[b] = slot_of(i64_hash(k[b]), mask);
    }
#pragma unroll
    for (int b = 0; b < PIPE; ++b) first[b] = slots[s[b]];
#pragma unroll
    for (int b = 0; b < PIPE; ++b) {
      int64_t row = base + b;
      if (row >= nrows) continue;
      int64_t idx;
      if (k[b] == GB_EMPTY_KEY) {
        // reserved overflow slot for the sentinel key value
        idx = capacity;
        atomicCAS(reinterpret_cast<unsigned long long*>(&slots[capacity].row1),
                  0ull, (unsigned long long)(row + 1));
      } else {
        idx = -1;
        Slot64 cur = first[b];
        uint64_t sl = s[b];
        int64_t bound = (int64_t)mask < 1024 ? (int64_t)mask : 1024;
        for (int64_t probes = 0; probes <= bound; ++probes) {
          if (cur.key == k[b]) { idx = (int64_t)sl; break; }
          if (cur.key == GB_EMPTY_KEY) {
            long long prev = atomicCAS(
                reinterpret_cast<unsigned long long*>(&slots[sl].key),
                (unsigned long long)GB_EMPTY_KEY, (unsigned long long)k[b]);
            if (prev == GB_EMPTY_KEY) {
              slots[sl].row1 = row + 1;  // winner records the representative
              idx = (int64_t)sl;
              break;
            }
            if (prev == k[b]) { idx = (int64_t)sl; break; }
            // lost to a different key: fall through and advance
          }
          sl = (sl + 1) & mask;
          cur = slots[sl];
        }
        if (idx < 0) {  // table FULL: undersized hint — host re-runs unhinted
          atomicOr(overflow, 1);
          continue;
        }
      }
      agg_accumulate(aggs, naggs, row, idx);
    }
  }
}

// compact the Slot64 table: same contract as groupby_compact_kernel but with
// 16B slots; scans capacity+1 entries (the reserved sentinel slot included).
// Two-pass compaction: pass 1 counts occupied slots per BLOCK (no global
// atomics), torch.cumsum turns the 2048-entry count array into exclusive
// bases, pass 2 assigns dense output indices from a per-block LDS cursor.
// The single-pass variant's wave-leader atomicAdd on ONE global counter was
// the bottleneck on large tables (512M slots -> 8M same-address atomics,
// 101 ms; two passes run at slot-scan bandwidth instead).
__global__ void groupby_compact_i64_count_kernel(
    const Slot64* __restrict__ slots, int64_t capacity1,
    int64_t* __restrict__ blk_counts) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  int64_t cnt = 0;
  for (int64_t s = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       s < ((capacity1 + WAVE - 1) & ~(int64_t)(WAVE - 1)); s += stride) {
    bool in_range = s < capacity1;
    Slot64 word = {GB_EMPTY_KEY, 0};
    if (in_range) word = slots[s];
    bool occ = in_range &&
               (s == capacity1 - 1 ? word.row1 != 0 : word.key != GB_EMPTY_KEY);
    uint64_t ballot = __ballot(occ);
    if ((threadIdx.x & (WAVE - 1)) == 0) cnt += __popcll(ballot);
  }
  __shared__ int64_t red[DEFAULT_BLOCK / WAVE];
  if ((threadIdx.x & (WAVE - 1)) == 0) red[threadIdx.x / WAVE] = cnt;
  __syncthreads();
  if (threadIdx.x == 0) {
    int64_t t = 0;
    for (int w = 0; w < (int)(blockDim.x / WAVE); ++w) t += red[w];
    blk_counts[blockIdx.x] = t;
  }
}

__global__ void groupby_compact_i64_fill_kernel(
    const Slot64* __restrict__ slots, int64_t capacity1,
    const AggDesc* __restrict__ aggs, int32_t naggs,
    const int64_t* __restrict__ blk_bases, int64_t* __restrict__ out_repr,
    int64_t* __restrict__ out_agg_base, int64_t out_capacity) {
  __shared__ unsigned long long cursor;
  if (threadIdx.x == 0) cursor = (unsigned long long)blk_bases[blockIdx.x];
  __syncthreads();
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t s = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       s < ((capacity1 + WAVE - 1) & ~(int64_t)(WAVE - 1)); s += stride) {
    bool in_range = s < capacity1;
    Slot64 word = {GB_EMPTY_KEY, 0};
    if (in_range) word = slots[s];
    bool occ = in_range &&
               (s == capacity1 - 1 ? word.row1 != 0 : word.key != GB_EMPTY_KEY);
    uint64_t ballot = __ballot(occ);
    int lane = threadIdx.x & (WAVE - 1);
    int nset = __popcll(ballot);
    int leader = __ffsll((unsigned long long)ballot) - 1;
    uint64_t bs = 0;
    if (nset && lane == leader)
      bs = atomicAdd(&cursor, (unsigned long long)nset);  // LDS atomic
    bs = __shfl(bs, leader >= 0 ? leader : 0, WAVE);
    if (occ) {
      uint64_t pos = bs + __popcll(ballot & ((1ull << lane) - 1));
      if ((int64_t)pos < out_capacity) {
        out_repr[pos] = word.row1 - 1;
        for (int32_t a = 0; a < naggs; ++a) {
          out_agg_base[(int64_t)a * out_capacity + (int64_t)pos] =
              reinterpret_cast<const int64_t*>(aggs[a].state)[s];
        }
      }
    }
  }
}

__global__ void groupby_compact_i64_kernel(
    const Slot64* __restrict__ slots, int64_t capacity1,
    const AggDesc* __restrict__ aggs, int32_t naggs,
    uint64_t* __restrict__ counter, int64_t* __restrict__ out_repr,
    int64_t* __restrict__ out_agg_base, int64_t out_capacity) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t s = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       s < ((capacity1 + WAVE - 1) & ~(int64_t)(WAVE - 1)); s += stride) {
    bool in_range = s < capacity1;
    Slot64 word = {GB_EMPTY_KEY, 0};
    if (in_range) word = slots[s];
    // normal slots: occupied iff key claimed; the reserved last slot keeps
    // the sentinel key and is occupied iff row1 was set
    bool occ = in_range &&
               (s == capacity1 - 1 ? word.row1 != 0 : word.key != GB_EMPTY_KEY);
    uint64_t ballot = __ballot(occ);
    int lane = threadIdx.x & (WAVE - 1);
    int nset = __popcll(ballot);
    int leader = __ffsll((unsigned long long)ballot) - 1;
    uint64_t bs = 0;
    if (nset && lane == leader)
      bs = atomicAdd((unsigned long long*)counter, (unsigned long long)nset);
    bs = __shfl(bs, leader >= 0 ? leader : 0, WAVE);
    if (occ) {
      uint64_t pos = bs + __popcll(ballot & ((1ull << lane) - 1));
      if ((int64_t)pos < out_capacity) {
        out_repr[pos] = word.row1 - 1;
        for (int32_t a = 0; a < naggs; ++a) {
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

void srj_join_build_i64(const long long* keys, const uint8_t* valid, int64_t nrows,
                        void* slots, int64_t capacity, hipStream_t stream) {
  int64_t nthreads_needed = (nrows + PIPE - 1) / PIPE;
  join_build_i64_kernel<<<grid_1d(nthreads_needed), DEFAULT_BLOCK, 0, stream>>>(
      keys, valid, nrows, reinterpret_cast<Slot64*>(slots),
      (uint64_t)(capacity - 1));
}

constexpr int PROBE_PIPE = 8;  // 16 measured slower (12.5 vs 15.8 B rows/s): occupancy drop

void srj_join_probe_i64(const long long* probe, const uint8_t* pvalid,
                        int64_t nprobe, const void* slots, int64_t capacity,
                        uint64_t* counter, int32_t* out_build, int64_t* out_probe,
                        int64_t out_capacity, uint8_t* build_matched, int32_t fill,
                        const int32_t* idxmap, hipStream_t stream) {
  int64_t nthreads_needed = (nprobe + PROBE_PIPE - 1) / PROBE_PIPE;
  int64_t g = grid_1d(nthreads_needed);
  const Slot64* sl = reinterpret_cast<const Slot64*>(slots);
  uint64_t mask = (uint64_t)(capacity - 1);
  if (fill) {
    if (pvalid)
      join_probe_i64_kernel<true, true, PROBE_PIPE><<<g, DEFAULT_BLOCK, 0, stream>>>(
          probe, pvalid, nprobe, sl, mask, counter, out_build, out_probe,
          out_capacity, build_matched, idxmap);
    else
      join_probe_i64_kernel<true, false, PROBE_PIPE><<<g, DEFAULT_BLOCK, 0, stream>>>(
          probe, pvalid, nprobe, sl, mask, counter, out_build, out_probe,
          out_capacity, build_matched, idxmap);
  } else {
    if (pvalid)
      join_probe_i64_kernel<false, true, PROBE_PIPE><<<g, DEFAULT_BLOCK, 0, stream>>>(
          probe, pvalid, nprobe, sl, mask, counter, nullptr, nullptr, 0,
          nullptr, nullptr);
    else
      join_probe_i64_kernel<false, false, PROBE_PIPE><<<g, DEFAULT_BLOCK, 0, stream>>>(
          probe, pvalid, nprobe, sl, mask, counter, nullptr, nullptr, 0,
          nullptr, nullptr);
  }
}

void srj_groupby_i64_lds(const long long* keys, int64_t nrows, void* slots,
                         int64_t capacity, const void* aggs, int32_t naggs,
                         const int64_t* identities, int32_t* overflow,
                         hipStream_t stream) {
  groupby_i64_lds_kernel<<<grid_1d(nrows), DEFAULT_BLOCK, 0, stream>>>(
      keys, nrows, reinterpret_cast<Slot64*>(slots), (uint64_t)(capacity - 1),
      reinterpret_cast<const AggDesc*>(aggs), naggs, identities, overflow);
}

void srj_groupby_i64(const long long* keys, int64_t nrows, void* slots,
                     int64_t capacity, const void* aggs, int32_t naggs,
                     int32_t* overflow, hipStream_t stream) {
  int64_t nthreads_needed = (nrows + PIPE - 1) / PIPE;
  groupby_i64_kernel<<<grid_1d(nthreads_needed), DEFAULT_BLOCK, 0, stream>>>(
      keys, nrows, reinterpret_cast<Slot64*>(slots), (uint64_t)(capacity - 1),
      reinterpret_cast<const AggDesc*>(aggs), naggs, overflow);
}

void srj_groupby_compact_i64(const void* slots, int64_t capacity1,
                             const void* aggs, int32_t naggs, uint64_t* counter,
                             int64_t* out_repr, int64_t* out_agg,
                             int64_t out_capacity, hipStream_t stream) {
  groupby_compact_i64_kernel<<<grid_1d(capacity1), DEFAULT_BLOCK, 0, stream>>>(
      reinterpret_cast<const Slot64*>(slots), capacity1,
      reinterpret_cast<const AggDesc*>(aggs), naggs, counter, out_repr, out_agg,
      out_capacity);
}

void srj_groupby_compact_i64_count(const void* slots, int64_t capacity1,
                                   int64_t* blk_counts, hipStream_t stream) {
  groupby_compact_i64_count_kernel<<<grid_1d(capacity1), DEFAULT_BLOCK, 0,
                                     stream>>>(
      reinterpret_cast<const Slot64*>(slots), capacity1, blk_counts);
}

void srj_groupby_compact_i64_fill(const void* slots, int64_t capacity1,
                                  const void* aggs, int32_t naggs,
                                  const int64_t* blk_bases, int64_t* out_repr,
                                  int64_t* out_agg, int64_t out_capacity,
                                  hipStream_t stream) {
  groupby_compact_i64_fill_kernel<<<grid_1d(capacity1), DEFAULT_BLOCK, 0,
                                    stream>>>(
      reinterpret_cast<const Slot64*>(slots), capacity1,
      reinterpret_cast<const AggDesc*>(aggs), naggs, blk_bases, out_repr,
      out_agg, out_capacity);
}

}  // extern "C"
