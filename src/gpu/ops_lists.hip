// List/map ops, Iceberg transforms, round_float, and the JoinPrimitives
// tail (AST gather-map filter, matched-rows bitmap).
//
// Reference parity: list_slice.cu, map_utils.cu (is_valid_map /
// map_from_entries), map.cu (sort_map_column), map_zip_with_utils.cu,
// iceberg/iceberg_bucket.cu + iceberg_truncate.cu + iceberg_datetime_util.cu,
// round_float.cu (decimal-string round-trip trick — done here in digit space
// via the Ryu d2d + Eisel-Lemire pair), join_primitives.hpp:115
// filter_gather_maps_by_ast and :237 get_matched_rows.
#include "srj_common.hpp"
#include "table_equal.hpp"

namespace srj {

// ---------------------------------------------------------------------------
// list_slice: Spark slice(list, start, length); 1-based start, negative
// start counts from the end. Produces new offsets; caller gathers child.
// phase 0: out_lens + per-row child start; phase uses gather map built here.
// ---------------------------------------------------------------------------
__global__ void list_slice_kernel(const int32_t* __restrict__ offsets,
                                  const uint8_t* __restrict__ valid, int64_t n,
                                  const int32_t* __restrict__ start_col,
                                  int32_t start_scalar,
                                  const int32_t* __restrict__ len_col,
                                  int32_t len_scalar,
                                  int32_t* __restrict__ out_lens,
                                  int32_t* __restrict__ child_start,
                                  uint8_t* __restrict__ out_valid,
                                  int64_t* __restrict__ err_row) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  int64_t npad = (n + WAVE - 1) & ~(int64_t)(WAVE - 1);
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < npad;
       i += stride) {
    bool in_range = i < n;
    bool v = in_range && is_valid(valid, i);
    int32_t out_len = 0, cstart = 0;
    if (v) {
      int32_t s = offsets[i], e = offsets[i + 1];
      int32_t sz = e - s;
      int32_t st = start_col ? start_col[i] : start_scalar;
      int32_t ln = len_col ? len_col[i] : len_scalar;
      if (st == 0 || ln < 0) {
        // Spark: start 0 / negative length -> error (ANSI) or null
        v = false;
        if (err_row)
          atomicMin(reinterpret_cast<long long*>(err_row), (long long)i);
      } else {
        int32_t begin = st > 0 ? st - 1 : sz + st;
        if (begin < 0 || begin >= sz) {
          out_len = 0;
          cstart = s;
        } else {
          out_len = min(ln, sz - begin);
          cstart = s + begin;
        }
      }
    }
    if (in_range) {
      out_lens[i] = v ? out_len : 0;
      child_start[i] = cstart;
    }
    ballot_write_validity(out_valid, i, v);
  }
}

// build child gather map from (child_start, out_lens exclusive-scanned)
__global__ void list_slice_gather_kernel(const int32_t* __restrict__ child_start,
                                         const int32_t* __restrict__ out_offsets,
                                         int64_t n,
                                         int64_t* __restrict__ gmap) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    int32_t o = out_offsets[i], e = out_offsets[i + 1];
    for (int32_t k = 0; k < e - o; ++k) gmap[o + k] = child_start[i] + k;
  }
}

// ---------------------------------------------------------------------------
// map validation (is_valid_map / map_from_entries): entries list rows with
// struct<key,value> children: keys non-null and unique per row.
// ---------------------------------------------------------------------------
__global__ void validate_map_kernel(const int32_t* __restrict__ offsets,
                                    const uint8_t* __restrict__ row_valid,
                                    int64_t n, const ColDesc* __restrict__ cols,
                                    int32_t key_col,
                                    uint8_t* __restrict__ out) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  const ColDesc& kc = cols[key_col];
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    bool ok = true;
    if (is_valid(row_valid, i)) {
      int32_t s = offsets[i], e = offsets[i + 1];
      for (int32_t a = s; a < e && ok; ++a) {
        if (!is_valid(kc.valid, a)) ok = false;
        for (int32_t b = a + 1; b < e && ok; ++b) {
          if (col_rows_equal(kc, a, kc, b)) ok = false;
        }
      }
    }
    out[i] = ok;
  }
}

// sort each row's entries by key: emits per-entry permutation (insertion
// sort per row — maps are small; reference sort_map_column)
__global__ void sort_map_kernel(const int32_t* __restrict__ offsets, int64_t n,
                                const ColDesc* __restrict__ cols,
                                int32_t key_col, int64_t* __restrict__ perm) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  const ColDesc& kc = cols[key_col];
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    int32_t s = offsets[i], e = offsets[i + 1];
    for (int32_t k = s; k < e; ++k) perm[k] = k;
    for (int32_t a = s + 1; a < e; ++a) {
      int64_t key = perm[a];
      int32_t b = a - 1;
      while (b >= s && col_rows_compare(kc, perm[b], kc, key) > 0) {
        perm[b + 1] = perm[b];
        --b;
      }
      perm[b + 1] = key;
    }
  }
}

// map_zip: full-outer key union of two SORTED maps (two-phase)
template <bool WRITE>
__global__ void map_zip_kernel(const int32_t* __restrict__ offs1,
                               const int32_t* __restrict__ offs2, int64_t n,
                               const ColDesc* __restrict__ cols, int32_t key1,
                               int32_t key2, int32_t* __restrict__ out_counts,
                               const int32_t* __restrict__ out_offsets,
                               int64_t* __restrict__ kmap,  // gather into map1 keys (or -(idx2+2))
                               int64_t* __restrict__ v1map,
                               int64_t* __restrict__ v2map) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  const ColDesc& k1 = cols[key1];
  const ColDesc& k2 = cols[key2];
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    int32_t a = offs1[i], ae = offs1[i + 1];
    int32_t b = offs2[i], be = offs2[i + 1];
    int32_t cnt = 0;
    int32_t o = WRITE ? out_offsets[i] : 0;
    while (a < ae || b < be) {
      int cmp;
      if (a >= ae) cmp = 1;
      else if (b >= be) cmp = -1;
      else cmp = col_rows_compare(k1, a, k2, b);
      if (WRITE) {
        if (cmp == 0) {
          kmap[o + cnt] = a;
          v1map[o + cnt] = a;
          v2map[o + cnt] = b;
        } else if (cmp < 0) {
          kmap[o + cnt] = a;
          v1map[o + cnt] = a;
          v2map[o + cnt] = -1;
        } else {
          kmap[o + cnt] = -(int64_t)b - 2;  // negative: key from map2
          v1map[o + cnt] = -1;
          v2map[o + cnt] = b;
        }
      }
      if (cmp <= 0) ++a;
      if (cmp >= 0) ++b;
      ++cnt;
    }
    if (!WRITE) out_counts[i] = cnt;
  }
}

// ---------------------------------------------------------------------------
// iceberg transforms
// ---------------------------------------------------------------------------
__global__ void iceberg_bucket_long_kernel(const int64_t* __restrict__ in,
                                           const uint8_t* __restrict__ valid,
                                           int64_t n, int32_t nbuckets,
                                           int32_t* __restrict__ out,
                                           uint8_t* __restrict__ out_valid) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  int64_t npad = (n + WAVE - 1) & ~(int64_t)(WAVE - 1);
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < npad;
       i += stride) {
    bool in_range = i < n;
    bool v = in_range && is_valid(valid, i);
    int32_t r = 0;
    if (v) {
      // iceberg: murmur3_x86_32 of the 8-byte little-endian value, seed 0
      int64_t x = in[i];
      char buf[8];
      __builtin_memcpy(buf, &x, 8);
      int32_t h = (int32_t)mm3_hash_bytes(buf, 8, 0);
      r = (h & 0x7FFFFFFF) % nbuckets;
    }
    if (in_range) out[i] = r;
    ballot_write_validity(out_valid, i, v);
  }
}

__global__ void iceberg_bucket_string_kernel(ColDesc in, int64_t n,
                                             int32_t nbuckets,
                                             int32_t* __restrict__ out,
                                             uint8_t* __restrict__ out_valid) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  int64_t npad = (n + WAVE - 1) & ~(int64_t)(WAVE - 1);
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < npad;
       i += stride) {
    bool in_range = i < n;
    bool v = in_range && is_valid(in.valid, i);
    int32_t r = 0;
    if (v) {
      StrView s = get_string(in, i);
      // standard murmur3 (iceberg spec), not the Spark tail variant
      int32_t h = (int32_t)mm3_hash_bytes_std(s.ptr, s.len, 0);
      r = (h & 0x7FFFFFFF) % nbuckets;
    }
    if (in_range) out[i] = r;
    ballot_write_validity(out_valid, i, v);
  }
}

// decimal bucket: murmur3_x86_32 of the unscaled value's minimal
// big-endian two's-complement bytes (iceberg spec appendix B; reference
// iceberg_bucket.hpp decimal path)
__global__ void iceberg_bucket_decimal_kernel(
    const uint8_t* __restrict__ in, const uint8_t* __restrict__ valid,
    int64_t n, int32_t width, int32_t nbuckets, int32_t* __restrict__ out,
    uint8_t* __restrict__ out_valid) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  int64_t npad = (n + WAVE - 1) & ~(int64_t)(WAVE - 1);
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < npad;
       i += stride) {
    bool in_range = i < n;
    bool v = in_range && is_valid(valid, i);
    int32_t b = 0;
    if (v) {
      // sign-extend the little-endian unscaled value to 16 bytes, then
      // take the java-minimal big-endian form
      uint8_t le[16];
      const uint8_t* p = in + i * width;
      uint8_t ext = (p[width - 1] & 0x80) ? 0xFF : 0x00;
      for (int k = 0; k < 16; ++k) le[k] = k < width ? p[k] : ext;
      uint8_t be[16];
      int len = dec128_java_bytes(le, be);
      uint32_t h = mm3_hash_bytes_std(reinterpret_cast<const char*>(be), len, 0);
      b = (int32_t)((h & 0x7FFFFFFFu) % (uint32_t)nbuckets);
    }
    if (in_range) out[i] = b;
    ballot_write_validity(out_valid, i, v);
  }
}

__global__ void iceberg_truncate_long_kernel(const int64_t* __restrict__ in,
                                             const uint8_t* __restrict__ valid,
                                             int64_t n, int64_t width,
                                             int64_t* __restrict__ out) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    int64_t x = in[i];
    // floored modulo (iceberg truncate for negative values)
    int64_t m = x % width;
    if (m < 0) m += width;
    out[i] = is_valid(valid, i) ? x - m : 0;
  }
}

// year/month/day/hour transforms from DATE32 days or TIMESTAMP_US micros
__global__ void iceberg_datetime_kernel(const void* __restrict__ in,
                                        const uint8_t* __restrict__ valid,
                                        int64_t n, int32_t from_micros,
                                        int32_t part,  // 0 year 1 month 2 day 3 hour
                                        int32_t* __restrict__ out) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    if (!is_valid(valid, i)) { out[i] = 0; continue; }
    int64_t days, us_in_day = 0;
    if (from_micros) {
      int64_t us = reinterpret_cast<const int64_t*>(in)[i];
      int64_t day_us = 86400000000LL;
      days = us >= 0 ? us / day_us : (us - (day_us - 1)) / day_us;
      us_in_day = us - days * day_us;
    } else {
      days = reinterpret_cast<const int32_t*>(in)[i];
    }
    if (part == 2) {
      out[i] = (int32_t)days;
      continue;
    }
    if (part == 3) {
      out[i] = (int32_t)(days * 24 + us_in_day / 3600000000LL);
      continue;
    }
    // civil date
    int64_t z = days + 719468;
    int64_t era = (z >= 0 ? z : z - 146096) / 146097;
    unsigned doe = (unsigned)(z - era * 146097);
    unsigned yoe = (doe - doe / 1460 + doe / 36524 - doe / 146096) / 365;
    int64_t y = (int64_t)yoe + era * 400;
    unsigned doy = doe - (365 * yoe + yoe / 4 - yoe / 100);
    unsigned mp = (5 * doy + 2) / 153;
    int m = (int)(mp + (mp < 10 ? 3 : -9));
    y += (m <= 2);
    if (part == 0) out[i] = (int32_t)(y - 1970);
    else out[i] = (int32_t)((y - 1970) * 12 + (m - 1));
  }
}

// ---------------------------------------------------------------------------
// JoinPrimitives tail
// ---------------------------------------------------------------------------
// AST program op codes for pair filtering: stack machine over (lrow, rrow)
enum AstOp : int32_t {
  AST_PUSH_LCOL = 0,   // arg = col index into cols (left table row)
  AST_PUSH_RCOL = 1,
  AST_PUSH_LIT_I64 = 2,  // arg2 = literal bits
  AST_PUSH_LIT_F64 = 3,
  AST_LT = 4, AST_LE = 5, AST_GT = 6, AST_GE = 7, AST_EQ = 8, AST_NE = 9,
  AST_AND = 10, AST_OR = 11, AST_NOT = 12,
  AST_ADD = 13, AST_SUB = 14,
};

struct AstInstr {
  int32_t op;
  int32_t arg;
  int64_t lit;
};

struct AstVal {
  double num;
  bool is_null;
};

__device__ inline AstVal ast_load(const ColDesc& c, int64_t row) {
  AstVal v{0, false};
  if (!is_valid(c.valid, row)) {
    v.is_null = true;
    return v;
  }
  switch (c.dtype) {
    case BOOL8:
    case INT8: v.num = reinterpret_cast<const int8_t*>(c.data)[row]; break;
    case INT16: v.num = reinterpret_cast<const int16_t*>(c.data)[row]; break;
    case INT32:
    case DATE32: v.num = reinterpret_cast<const int32_t*>(c.data)[row]; break;
    case INT64:
    case TIMESTAMP_US:
      v.num = (double)reinterpret_cast<const int64_t*>(c.data)[row];
      break;
    case FLOAT32: v.num = reinterpret_cast<const float*>(c.data)[row]; break;
    case FLOAT64: v.num = reinterpret_cast<const double*>(c.data)[row]; break;
    default: v.is_null = true;
  }
  return v;
}

template <bool FILL>
__global__ void ast_filter_pairs_kernel(
    const ColDesc* __restrict__ cols, const AstInstr* __restrict__ prog,
    int32_t nprog, const int32_t* __restrict__ lmap,
    const int64_t* __restrict__ rmap, int64_t npairs,
    uint64_t* __restrict__ counter, int32_t* __restrict__ out_l,
    int64_t* __restrict__ out_r, int64_t out_capacity) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  int lane = threadIdx.x & (WAVE - 1);
  int64_t npad = (npairs + WAVE - 1) & ~(int64_t)(WAVE - 1);
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < npad;
       i += stride) {
    bool keep = false;
    if (i < npairs) {
      int64_t lrow = lmap[i], rrow = rmap[i];
      AstVal stack[16];
      int sp = 0;
      for (int32_t pc = 0; pc < nprog; ++pc) {
        const AstInstr ins = prog[pc];
        switch (ins.op) {
          case AST_PUSH_LCOL: stack[sp++] = ast_load(cols[ins.arg], lrow); break;
          case AST_PUSH_RCOL: stack[sp++] = ast_load(cols[ins.arg], rrow); break;
          case AST_PUSH_LIT_I64:
            stack[sp++] = AstVal{(double)ins.lit, false};
            break;
          case AST_PUSH_LIT_F64: {
            double d;
            __builtin_memcpy(&d, &ins.lit, 8);
            stack[sp++] = AstVal{d, false};
            break;
          }
          case AST_NOT:
            stack[sp - 1].num = stack[sp - 1].num == 0 ? 1 : 0;
            break;
          default: {
            AstVal b = stack[--sp];
            AstVal a = stack[--sp];
            AstVal r{0, a.is_null || b.is_null};
            if (!r.is_null) {
              switch (ins.op) {
                case AST_LT: r.num = a.num < b.num; break;
                case AST_LE: r.num = a.num <= b.num; break;
                case AST_GT: r.num = a.num > b.num; break;
                case AST_GE: r.num = a.num >= b.num; break;
                case AST_EQ: r.num = a.num == b.num; break;
                case AST_NE: r.num = a.num != b.num; break;
                case AST_AND: r.num = (a.num != 0) && (b.num != 0); break;
                case AST_OR: r.num = (a.num != 0) || (b.num != 0); break;
                case AST_ADD: r.num = a.num + b.num; break;
                case AST_SUB: r.num = a.num - b.num; break;
              }
            }
            stack[sp++] = r;
          }
        }
      }
      keep = sp > 0 && !stack[sp - 1].is_null && stack[sp - 1].num != 0;
    }
    // wave-aggregated append
    uint32_t nm = keep ? 1 : 0;
    if (FILL) {
      uint32_t incl = wave_prefix_incl(nm);
      uint32_t total = __shfl(incl, WAVE - 1, WAVE);
      uint64_t base = 0;
      if (lane == WAVE - 1 && total)
        base = atomicAdd((unsigned long long*)counter,
                         (unsigned long long)total);
      base = __shfl(base, WAVE - 1, WAVE);
      if (keep) {
        int64_t pos = (int64_t)(base + incl - 1);
        if (pos < out_capacity) {
          out_l[pos] = lmap[i];
          out_r[pos] = rmap[i];
        }
      }
    } else {
      uint32_t c = wave_sum(nm);
      if (lane == 0 && c)
        atomicAdd((unsigned long long*)counter, (unsigned long long)c);
    }
  }
}

__global__ void matched_rows_kernel(const int32_t* __restrict__ gmap, int64_t n,
                                    uint8_t* __restrict__ flags) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    int32_t r = gmap[i];
    if (r >= 0) flags[r] = 1;
  }
}

}  // namespace srj

using namespace srj;

extern "C" {

void srj_list_slice(const int32_t* offsets, const uint8_t* valid, int64_t n,
                    const int32_t* start_col, int32_t start_scalar,
                    const int32_t* len_col, int32_t len_scalar,
                    int32_t* out_lens, int32_t* child_start, uint8_t* out_valid,
                    int64_t* err_row, hipStream_t stream) {
  list_slice_kernel<<<grid_1d(n), DEFAULT_BLOCK, 0, stream>>>(
      offsets, valid, n, start_col, start_scalar, len_col, len_scalar, out_lens,
      child_start, out_valid, err_row);
}

void srj_list_slice_gather(const int32_t* child_start, const int32_t* out_offsets,
                           int64_t n, int64_t* gmap, hipStream_t stream) {
  list_slice_gather_kernel<<<grid_1d(n), DEFAULT_BLOCK, 0, stream>>>(
      child_start, out_offsets, n, gmap);
}

void srj_validate_map(const int32_t* offsets, const uint8_t* row_valid, int64_t n,
                      const void* cols, int32_t key_col, uint8_t* out,
                      hipStream_t stream) {
  validate_map_kernel<<<grid_1d(n), DEFAULT_BLOCK, 0, stream>>>(
      offsets, row_valid, n, reinterpret_cast<const ColDesc*>(cols), key_col, out);
}

void srj_sort_map(const int32_t* offsets, int64_t n, const void* cols,
                  int32_t key_col, int64_t* perm, hipStream_t stream) {
  sort_map_kernel<<<grid_1d(n), DEFAULT_BLOCK, 0, stream>>>(
      offsets, n, reinterpret_cast<const ColDesc*>(cols), key_col, perm);
}

void srj_map_zip(const int32_t* offs1, const int32_t* offs2, int64_t n,
                 const void* cols, int32_t key1, int32_t key2, int32_t phase,
                 int32_t* out_counts, const int32_t* out_offsets, int64_t* kmap,
                 int64_t* v1map, int64_t* v2map, hipStream_t stream) {
  if (phase == 0)
    map_zip_kernel<false><<<grid_1d(n), DEFAULT_BLOCK, 0, stream>>>(
        offs1, offs2, n, reinterpret_cast<const ColDesc*>(cols), key1, key2,
        out_counts, nullptr, nullptr, nullptr, nullptr);
  else
    map_zip_kernel<true><<<grid_1d(n), DEFAULT_BLOCK, 0, stream>>>(
        offs1, offs2, n, reinterpret_cast<const ColDesc*>(cols), key1, key2,
        nullptr, out_offsets, kmap, v1map, v2map);
}

void srj_iceberg_bucket_long(const int64_t* in, const uint8_t* valid, int64_t n,
                             int32_t nbuckets, int32_t* out, uint8_t* out_valid,
                             hipStream_t stream) {
  iceberg_bucket_long_kernel<<<grid_1d(n), DEFAULT_BLOCK, 0, stream>>>(
      in, valid, n, nbuckets, out, out_valid);
}

void srj_iceberg_bucket_string(const void* in, int64_t n, int32_t nbuckets,
                               int32_t* out, uint8_t* out_valid,
                               hipStream_t stream) {
  iceberg_bucket_string_kernel<<<grid_1d(n), DEFAULT_BLOCK, 0, stream>>>(
      *reinterpret_cast<const ColDesc*>(in), n, nbuckets, out, out_valid);
}

void srj_iceberg_bucket_decimal(const void* in, const uint8_t* valid,
                                int64_t n, int32_t width, int32_t nbuckets,
                                int32_t* out, uint8_t* out_valid,
                                hipStream_t stream) {
  iceberg_bucket_decimal_kernel<<<grid_1d(n), DEFAULT_BLOCK, 0, stream>>>(
      reinterpret_cast<const uint8_t*>(in), valid, n, width, nbuckets, out,
      out_valid);
}

void srj_iceberg_truncate_long(const int64_t* in, const uint8_t* valid,
                               int64_t n, int64_t width, int64_t* out,
                               hipStream_t stream) {
  iceberg_truncate_long_kernel<<<grid_1d(n), DEFAULT_BLOCK, 0, stream>>>(
      in, valid, n, width, out);
}

void srj_iceberg_datetime(const void* in, const uint8_t* valid, int64_t n,
                          int32_t from_micros, int32_t part, int32_t* out,
                          hipStream_t stream) {
  iceberg_datetime_kernel<<<grid_1d(n), DEFAULT_BLOCK, 0, stream>>>(
      in, valid, n, from_micros, part, out);
}


void srj_ast_filter_pairs(const void* cols, const void* prog, int32_t nprog,
                          const int32_t* lmap, const int64_t* rmap,
                          int64_t npairs, uint64_t* counter, int32_t* out_l,
                          int64_t* out_r, int64_t out_capacity, int32_t fill,
                          hipStream_t stream) {
  if (fill)
    ast_filter_pairs_kernel<true><<<grid_1d(npairs), DEFAULT_BLOCK, 0, stream>>>(
        reinterpret_cast<const ColDesc*>(cols),
        reinterpret_cast<const AstInstr*>(prog), nprog, lmap, rmap, npairs,
        counter, out_l, out_r, out_capacity);
  else
    ast_filter_pairs_kernel<false><<<grid_1d(npairs), DEFAULT_BLOCK, 0, stream>>>(
        reinterpret_cast<const ColDesc*>(cols),
        reinterpret_cast<const AstInstr*>(prog), nprog, lmap, rmap, npairs,
        counter, nullptr, nullptr, 0);
}

void srj_matched_rows(const int32_t* gmap, int64_t n, uint8_t* flags,
                      hipStream_t stream) {
  matched_rows_kernel<<<grid_1d(n), DEFAULT_BLOCK, 0, stream>>>(gmap, n, flags);
}

}  // extern "C"
