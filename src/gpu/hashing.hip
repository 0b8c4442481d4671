// Spark-exact hash kernels: murmur3_32, xxhash64, hive_hash.
//
// Behavior parity with the reference (spark-rapids-jni):
//   murmur3:  src/main/cpp/src/hash/murmur_hash.cu (seed chained per column,
//             nulls pass the running hash through, floats normalized,
//             tail bytes processed one-at-a-time as full int blocks)
//   xxhash64: src/main/cpp/src/hash/xxhash64.cu (bool/int8/16 promoted to
//             4-byte int, int32 4 bytes, int64/decimal32/64 8 bytes)
//   hive:     src/main/cpp/src/hash/hive_hash.cu (Java hashCode semantics,
//             31*h + col_hash combine, null -> 0)
//
// MI355X design: one thread per row, 64-wide waves, grid-stride over rows with
// a ~2048-block cap (fills 8 XCDs); per-row column loop is wave-uniform so
// there is no divergence from the type dispatch.
#include "srj_common.hpp"

namespace srj {


// ---------------------------------------------------------------------------
// murmur3 (Spark Murmur3_x86_32)
// ---------------------------------------------------------------------------
__device__ uint32_t murmur3_col_row(const ColDesc* cols, const ColDesc& c,
                                    int64_t row, uint32_t seed, int depth);

// struct/list members hash their elements in order with the running seed.
__device__ uint32_t murmur3_children(const ColDesc* cols, const ColDesc& c,
                                     int64_t row, uint32_t seed, int depth) {
  if (depth > 8) return seed;  // matches reference MAX_STACK_DEPTH-style cap
  if (c.dtype == STRUCT) {
    uint32_t h = seed;
    for (int k = 0; k < c.num_children; ++k) {
      h = murmur3_col_row(cols, cols[c.child0 + k], row, h, depth);
    }
    return h;
  }
  if (c.dtype == LIST) {
    uint32_t h = seed;
    const ColDesc& child = cols[c.child0];
    for (int32_t j = c.offsets[row]; j < c.offsets[row + 1]; ++j) {
      h = murmur3_col_row(cols, child, j, h, depth);
    }
    return h;
  }
  return seed;
}

__device__ uint32_t murmur3_col_row(const ColDesc* cols, const ColDesc& c,
                                    int64_t row, uint32_t seed, int depth = 0) {
  if (!is_valid(c.valid, row)) return seed;  // null: seed passes through
  switch (c.dtype) {
    case BOOL8: return mm3_hash_int(reinterpret_cast<const int8_t*>(c.data)[row] != 0, seed);
    case INT8: return mm3_hash_int(reinterpret_cast<const int8_t*>(c.data)[row], seed);
    case INT16: return mm3_hash_int(reinterpret_cast<const int16_t*>(c.data)[row], seed);
    case INT32:
    case DATE32: return mm3_hash_int(reinterpret_cast<const int32_t*>(c.data)[row], seed);
    // Spark hashes small decimals' unscaled value as a long (hashLong), so
    // DECIMAL32 sign-extends to 64 bits (ref murmur_hash.cuh:185-197).
    case DECIMAL32:
      return mm3_hash_long((int64_t)reinterpret_cast<const int32_t*>(c.data)[row], seed);
    case INT64:
    case TIMESTAMP_US:
    case DECIMAL64: return mm3_hash_long(reinterpret_cast<const int64_t*>(c.data)[row], seed);
    case DECIMAL128: {
      uint8_t buf[16];
      int n = dec128_java_bytes(reinterpret_cast<const uint8_t*>(c.data) + row * 16, buf);
      return mm3_hash_bytes(reinterpret_cast<const char*>(buf), n, seed);
    }
    case FLOAT32:
      return mm3_hash_int(norm_float_bits(reinterpret_cast<const float*>(c.data)[row]), seed);
    case FLOAT64:
      return mm3_hash_long(norm_double_bits(reinterpret_cast<const double*>(c.data)[row]), seed);
    case STRING: {
      StrView s = get_string(c, row);
      return mm3_hash_bytes(s.ptr, s.len, seed);
    }
    case STRUCT:
    case LIST: return murmur3_children(cols, c, row, seed, depth + 1);
    default: return seed;
  }
}

__global__ void murmur3_kernel(const ColDesc* __restrict__ cols,
                               const int32_t* __restrict__ top, int32_t ntop,
                               int64_t nrows, uint32_t seed,
                               int32_t* __restrict__ out) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t row = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; row < nrows;
       row += stride) {
    uint32_t h = seed;
    for (int32_t i = 0; i < ntop; ++i) h = murmur3_col_row(cols, cols[top[i]], row, h);
    out[row] = (int32_t)h;
  }
}

// ---------------------------------------------------------------------------
// xxhash64
// ---------------------------------------------------------------------------
__device__ uint64_t xxhash64_col_row(const ColDesc* cols, const ColDesc& c,
                                     int64_t row, uint64_t seed, int depth);

__device__ uint64_t xxhash64_children(const ColDesc* cols, const ColDesc& c,
                                      int64_t row, uint64_t seed, int depth) {
  if (depth > 8) return seed;
  if (c.dtype == STRUCT) {
    uint64_t h = seed;
    for (int k = 0; k < c.num_children; ++k)
      h = xxhash64_col_row(cols, cols[c.child0 + k], row, h, depth);
    return h;
  }
  if (c.dtype == LIST) {
    uint64_t h = seed;
    const ColDesc& child = cols[c.child0];
    for (int32_t j = c.offsets[row]; j < c.offsets[row + 1]; ++j)
      h = xxhash64_col_row(cols, child, j, h, depth);
    return h;
  }
  return seed;
}

__device__ uint64_t xxhash64_col_row(const ColDesc* cols, const ColDesc& c,
                                     int64_t row, uint64_t seed, int depth = 0) {
  if (!is_valid(c.valid, row)) return seed;
  switch (c.dtype) {
    case BOOL8:
      return xxhash64_fixed((uint32_t)(reinterpret_cast<const int8_t*>(c.data)[row] != 0), 4, seed);
    case INT8:
      return xxhash64_fixed((uint32_t)reinterpret_cast<const int8_t*>(c.data)[row], 4, seed);
    case INT16:
      return xxhash64_fixed((uint32_t)reinterpret_cast<const int16_t*>(c.data)[row], 4, seed);
    case INT32:
    case DATE32:
      return xxhash64_fixed((uint32_t)reinterpret_cast<const int32_t*>(c.data)[row], 4, seed);
    case INT64:
    case TIMESTAMP_US:
      return xxhash64_fixed((uint64_t)reinterpret_cast<const int64_t*>(c.data)[row], 8, seed);
    case DECIMAL32:
      return xxhash64_fixed((uint64_t)reinterpret_cast<const int32_t*>(c.data)[row], 8, seed);
    case DECIMAL64:
      return xxhash64_fixed((uint64_t)reinterpret_cast<const int64_t*>(c.data)[row], 8, seed);
    case DECIMAL128: {
      uint8_t buf[16];
      int n = dec128_java_bytes(reinterpret_cast<const uint8_t*>(c.data) + row * 16, buf);
      return xxhash64_bytes(reinterpret_cast<const char*>(buf), n, seed);
    }
    case FLOAT32: {
      int32_t b = norm_float_bits(reinterpret_cast<const float*>(c.data)[row]);
      return xxhash64_fixed((uint32_t)b, 4, seed);
    }
    case FLOAT64: {
      int64_t b = norm_double_bits(reinterpret_cast<const double*>(c.data)[row]);
      return xxhash64_fixed((uint64_t)b, 8, seed);
    }
    case STRING: {
      StrView s = get_string(c, row);
      return xxhash64_bytes(s.ptr, s.len, seed);
    }
    case STRUCT:
    case LIST: return xxhash64_children(cols, c, row, seed, depth + 1);
    default: return seed;
  }
}

__global__ void xxhash64_kernel(const ColDesc* __restrict__ cols,
                                const int32_t* __restrict__ top, int32_t ntop,
                                int64_t nrows, uint64_t seed,
                                int64_t* __restrict__ out) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t row = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; row < nrows;
       row += stride) {
    uint64_t h = seed;
    for (int32_t i = 0; i < ntop; ++i) h = xxhash64_col_row(cols, cols[top[i]], row, h);
    out[row] = (int64_t)h;
  }
}

// ---------------------------------------------------------------------------
// hive hash (Java Object.hashCode semantics; combine: h = 31*h + colhash)
// ---------------------------------------------------------------------------
__device__ int32_t hive_col_row(const ColDesc* cols, const ColDesc& c, int64_t row,
                                int depth);

__device__ int32_t hive_children(const ColDesc* cols, const ColDesc& c, int64_t row,
                                 int depth) {
  if (depth > 8) return 0;
  if (c.dtype == STRUCT) {
    int32_t h = 0;
    for (int k = 0; k < c.num_children; ++k)
      h = 31 * h + hive_col_row(cols, cols[c.child0 + k], row, depth);
    return h;
  }
  if (c.dtype == LIST) {
    int32_t h = 0;
    const ColDesc& child = cols[c.child0];
    for (int32_t j = c.offsets[row]; j < c.offsets[row + 1]; ++j)
      h = 31 * h + hive_col_row(cols, child, j, depth);
    return h;
  }
  return 0;
}

__device__ int32_t hive_col_row(const ColDesc* cols, const ColDesc& c, int64_t row,
                                int depth = 0) {
  if (!is_valid(c.valid, row)) return 0;
  switch (c.dtype) {
    case BOOL8: return reinterpret_cast<const int8_t*>(c.data)[row] != 0 ? 1 : 0;
    case INT8: return reinterpret_cast<const int8_t*>(c.data)[row];
    case INT16: return reinterpret_cast<const int16_t*>(c.data)[row];
    case INT32:
    case DATE32: return reinterpret_cast<const int32_t*>(c.data)[row];
    case INT64: {
      int64_t v = reinterpret_cast<const int64_t*>(c.data)[row];
      return (int32_t)(v ^ ((uint64_t)v >> 32));
    }
    case TIMESTAMP_US: {
      // Hive TimestampWritableV2 hash: seconds ^ (seconds >>> 32) combined with nanos
      int64_t us = reinterpret_cast<const int64_t*>(c.data)[row];
      int64_t sec = us / 1000000;
      int64_t sub = us % 1000000;
      if (sub < 0) { sub += 1000000; sec -= 1; }
      int64_t nanos = sub * 1000;
      int64_t v = sec * 1000000000LL + nanos;
      return (int32_t)(v ^ ((uint64_t)v >> 32));
    }
    case FLOAT32: {
      float f = reinterpret_cast<const float*>(c.data)[row];
      int32_t b;
      if (f != f) b = 0x7fc00000;
      else { if (f == 0.0f) f = 0.0f; __builtin_memcpy(&b, &f, 4); }
      return b;
    }
    case FLOAT64: {
      double d = reinterpret_cast<const double*>(c.data)[row];
      int64_t b;
      if (d != d) b = 0x7ff8000000000000LL;
      else { if (d == 0.0) d = 0.0; __builtin_memcpy(&b, &d, 8); }
      return (int32_t)(b ^ ((uint64_t)b >> 32));
    }
    case STRING: {
      StrView s = get_string(c, row);
      int32_t h = 0;
      for (int32_t i = 0; i < s.len; ++i) h = 31 * h + (int8_t)s.ptr[i];
      return h;
    }
    case STRUCT:
    case LIST: return hive_children(cols, c, row, depth + 1);
    default: return 0;
  }
}

__global__ void hive_hash_kernel(const ColDesc* __restrict__ cols,
                                 const int32_t* __restrict__ top, int32_t ntop,
                                 int64_t nrows, int32_t* __restrict__ out) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t row = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; row < nrows;
       row += stride) {
    int32_t h = 0;
    for (int32_t i = 0; i < ntop; ++i) h = 31 * h + hive_col_row(cols, cols[top[i]], row);
    out[row] = h;
  }
}

// ---------------------------------------------------------------------------
// misc: null count
// ---------------------------------------------------------------------------
__global__ void count_unset_bits_kernel(const uint64_t* __restrict__ words,
                                        int64_t n, int64_t nwords,
                                        uint64_t* __restrict__ out) {
  uint64_t local = 0;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t w = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; w < nwords;
       w += stride) {
    local += __popcll(word_with_tail_masked(words, w, n));
  }
  local = wave_sum(local);
  if ((threadIdx.x & (WAVE - 1)) == 0 && local) atomicAdd(out, local);
}

}  // namespace srj

using namespace srj;

extern "C" {

void srj_murmur3(const ColDesc* cols_dev, const int32_t* top_dev, int32_t ntop,
                 int64_t nrows, int32_t seed, int32_t* out, hipStream_t stream) {
  murmur3_kernel<<<grid_1d(nrows), DEFAULT_BLOCK, 0, stream>>>(
      cols_dev, top_dev, ntop, nrows, (uint32_t)seed, out);
}

void srj_xxhash64(const ColDesc* cols_dev, const int32_t* top_dev, int32_t ntop,
                  int64_t nrows, int64_t seed, int64_t* out, hipStream_t stream) {
  xxhash64_kernel<<<grid_1d(nrows), DEFAULT_BLOCK, 0, stream>>>(
      cols_dev, top_dev, ntop, nrows, (uint64_t)seed, out);
}

void srj_hive_hash(const ColDesc* cols_dev, const int32_t* top_dev, int32_t ntop,
                   int64_t nrows, int32_t* out, hipStream_t stream) {
  hive_hash_kernel<<<grid_1d(nrows), DEFAULT_BLOCK, 0, stream>>>(
      cols_dev, top_dev, ntop, nrows, out);
}

// out must be a zeroed uint64 device scalar; result = count of SET bits.
void srj_count_set_bits(const uint8_t* mask, int64_t n, uint64_t* out,
                        hipStream_t stream) {
  int64_t nwords = (n + 63) / 64;
  count_unset_bits_kernel<<<grid_1d(nwords), DEFAULT_BLOCK, 0, stream>>>(
      reinterpret_cast<const uint64_t*>(mask), n, nwords, out);
}

}  // extern "C"
