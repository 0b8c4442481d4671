// Row hashing + null-safe row equality shared by the hash-table kernels
// (join build/probe, group-by aggregate) and sort.
//
// Spark semantics: NaN == NaN and -0.0 == 0.0 for join/group-by keys
// (normalized before hash and compare); null keys group together (null-safe
// equality) but never match in joins (caller filters).
#pragma once
#include "srj_common.hpp"

namespace srj {

// 64-bit row hash over key columns (internal — not Spark-visible): chained
// xxhash64, nulls pass through, floats normalized. Equal rows (null-safe,
// normalized) always produce equal hashes.
__device__ inline uint64_t row_hash64(const ColDesc* cols, const int32_t* top,
                                      int32_t ntop, int64_t row) {
  uint64_t h = 42;
  for (int32_t i = 0; i < ntop; ++i) {
    const ColDesc& c = cols[top[i]];
    if (!is_valid(c.valid, row)) continue;
    switch (c.dtype) {
      case BOOL8:
      case INT8:
        h = xxhash64_fixed((uint64_t)(int64_t)reinterpret_cast<const int8_t*>(c.data)[row], 8, h);
        break;
      case INT16:
        h = xxhash64_fixed((uint64_t)(int64_t)reinterpret_cast<const int16_t*>(c.data)[row], 8, h);
        break;
      case INT32:
      case DATE32:
      case DECIMAL32:
        h = xxhash64_fixed((uint64_t)(int64_t)reinterpret_cast<const int32_t*>(c.data)[row], 8, h);
        break;
      case INT64:
      case TIMESTAMP_US:
      case DECIMAL64:
        h = xxhash64_fixed((uint64_t)reinterpret_cast<const int64_t*>(c.data)[row], 8, h);
        break;
      case FLOAT32: {
        // normalize then widen to double so 1.0f-keyed and 1.0-keyed tables
        // are independent but consistent within a column
        float f = reinterpret_cast<const float*>(c.data)[row];
        int32_t b = norm_float_bits(f);
        h = xxhash64_fixed((uint64_t)(int64_t)b, 8, h);
        break;
      }
      case FLOAT64: {
        double d = reinterpret_cast<const double*>(c.data)[row];
        h = xxhash64_fixed((uint64_t)norm_double_bits(d), 8, h);
        break;
      }
      case STRING: {
        StrView s = get_string(c, row);
        h = xxhash64_bytes(s.ptr, s.len, h);
        break;
      }
      default:
        break;
    }
  }
  return h;
}

__device__ inline bool col_rows_equal(const ColDesc& a, int64_t ra,
                                      const ColDesc& b, int64_t rb) {
  bool va = is_valid(a.valid, ra), vb = is_valid(b.valid, rb);
  if (va != vb) return false;
  if (!va) return true;  // null-safe: both null -> equal
  switch (a.dtype) {
    case BOOL8:
      return (reinterpret_cast<const int8_t*>(a.data)[ra] != 0) ==
             (reinterpret_cast<const int8_t*>(b.data)[rb] != 0);
    case INT8:
      return reinterpret_cast<const int8_t*>(a.data)[ra] ==
             reinterpret_cast<const int8_t*>(b.data)[rb];
    case INT16:
      return reinterpret_cast<const int16_t*>(a.data)[ra] ==
             reinterpret_cast<const int16_t*>(b.data)[rb];
    case INT32:
    case DATE32:
    case DECIMAL32:
      return reinterpret_cast<const int32_t*>(a.data)[ra] ==
             reinterpret_cast<const int32_t*>(b.data)[rb];
    case INT64:
    case TIMESTAMP_US:
    case DECIMAL64:
      return reinterpret_cast<const int64_t*>(a.data)[ra] ==
             reinterpret_cast<const int64_t*>(b.data)[rb];
    case FLOAT32:
      return norm_float_bits(reinterpret_cast<const float*>(a.data)[ra]) ==
             norm_float_bits(reinterpret_cast<const float*>(b.data)[rb]);
    case FLOAT64:
      return norm_double_bits(reinterpret_cast<const double*>(a.data)[ra]) ==
             norm_double_bits(reinterpret_cast<const double*>(b.data)[rb]);
    case STRING: {
      StrView sa = get_string(a, ra), sb = get_string(b, rb);
      if (sa.len != sb.len) return false;
      for (int32_t i = 0; i < sa.len; ++i)
        if (sa.ptr[i] != sb.ptr[i]) return false;
      return true;
    }
    default:
      return false;
  }
}

__device__ inline bool rows_equal(const ColDesc* ca, const int32_t* ta,
                                  int32_t ntop, int64_t ra, const ColDesc* cb,
                                  const int32_t* tb, int64_t rb) {
  for (int32_t i = 0; i < ntop; ++i)
    if (!col_rows_equal(ca[ta[i]], ra, cb[tb[i]], rb)) return false;
  return true;
}

__device__ inline bool row_has_null_key(const ColDesc* cols, const int32_t* top,
                                        int32_t ntop, int64_t row) {
  for (int32_t i = 0; i < ntop; ++i)
    if (!is_valid(cols[top[i]].valid, row)) return true;
  return false;
}

// three-way compare for sort (nulls first, Spark default null ordering for
// ascending); floats compare with NaN greatest (Spark sorts NaN last asc).
__device__ inline int col_rows_compare(const ColDesc& a, int64_t ra,
                                       const ColDesc& b, int64_t rb) {
  bool va = is_valid(a.valid, ra), vb = is_valid(b.valid, rb);
  if (!va || !vb) return (int)va - (int)vb;  // nulls first (Spark asc default)
  switch (a.dtype) {
    case BOOL8:
    case INT8: {
      int8_t x = reinterpret_cast<const int8_t*>(a.data)[ra];
      int8_t y = reinterpret_cast<const int8_t*>(b.data)[rb];
      return (x > y) - (x < y);
    }
    case INT16: {
      int16_t x = reinterpret_cast<const int16_t*>(a.data)[ra];
      int16_t y = reinterpret_cast<const int16_t*>(b.data)[rb];
      return (x > y) - (x < y);
    }
    case INT32:
    case DATE32:
    case DECIMAL32: {
      int32_t x = reinterpret_cast<const int32_t*>(a.data)[ra];
      int32_t y = reinterpret_cast<const int32_t*>(b.data)[rb];
      return (x > y) - (x < y);
    }
    case INT64:
    case TIMESTAMP_US:
    case DECIMAL64: {
      int64_t x = reinterpret_cast<const int64_t*>(a.data)[ra];
      int64_t y = reinterpret_cast<const int64_t*>(b.data)[rb];
      return (x > y) - (x < y);
    }
    case FLOAT32: {
      float x = reinterpret_cast<const float*>(a.data)[ra];
      float y = reinterpret_cast<const float*>(b.data)[rb];
      bool nx = x != x, ny = y != y;
      if (nx || ny) return (int)nx - (int)ny;
      if (x == 0.0f && y == 0.0f) return 0;
      return (x > y) - (x < y);
    }
    case FLOAT64: {
      double x = reinterpret_cast<const double*>(a.data)[ra];
      double y = reinterpret_cast<const double*>(b.data)[rb];
      bool nx = x != x, ny = y != y;
      if (nx || ny) return (int)nx - (int)ny;
      if (x == 0.0 && y == 0.0) return 0;
      return (x > y) - (x < y);
    }
    case STRING: {
      StrView sa = get_string(a, ra), sb = get_string(b, rb);
      int32_t n = sa.len < sb.len ? sa.len : sb.len;
      for (int32_t i = 0; i < n; ++i) {
        uint8_t x = (uint8_t)sa.ptr[i], y = (uint8_t)sb.ptr[i];
        if (x != y) return x < y ? -1 : 1;
      }
      return (sa.len > sb.len) - (sa.len < sb.len);
    }
    default:
      return 0;
  }
}

}  // namespace srj
