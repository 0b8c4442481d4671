// Shared group-by aggregation machinery (AggDesc ABI + accumulate switch),
// used by the generic groupby_kernel (hashtable.hip) and the specialized
// int64-key path (hashtable_i64.hip). ABI must match ops/aggregate.py
// _AGGDESC_FMT = "<iiQQQ".
#pragma once
#include "srj_common.hpp"

namespace srj {

enum AggOp : int32_t {
  AGG_COUNT_ALL = 0,
  AGG_COUNT_VALID = 1,
  AGG_SUM_INT64 = 2,
  AGG_SUM_FLOAT64 = 3,
  AGG_MIN_INT64 = 4,
  AGG_MAX_INT64 = 5,
  AGG_MIN_FLOAT64 = 6,
  AGG_MAX_FLOAT64 = 7,
};

struct AggDesc {
  int32_t op;
  int32_t in_dtype;         // source column dtype (pre-upcast)
  const void* data;         // may be null for COUNT_ALL
  const uint8_t* valid;
  void* state;              // int64* or double* [capacity]
};

__device__ inline int64_t fetch_int64(const void* data, int32_t dt, int64_t row) {
  switch (dt) {
    case BOOL8:
    case INT8: return reinterpret_cast<const int8_t*>(data)[row];
    case INT16: return reinterpret_cast<const int16_t*>(data)[row];
    case INT32:
    case DATE32:
    case DECIMAL32: return reinterpret_cast<const int32_t*>(data)[row];
    default: return reinterpret_cast<const int64_t*>(data)[row];
  }
}

__device__ inline double fetch_double(const void* data, int32_t dt, int64_t row) {
  if (dt == FLOAT32) return reinterpret_cast<const float*>(data)[row];
  if (dt == FLOAT64) return reinterpret_cast<const double*>(data)[row];
  return (double)fetch_int64(data, dt, row);
}

__device__ inline void atomic_min_i64(int64_t* p, int64_t v) {
  atomicMin(reinterpret_cast<long long*>(p), (long long)v);
}
__device__ inline void atomic_max_i64(int64_t* p, int64_t v) {
  atomicMax(reinterpret_cast<long long*>(p), (long long)v);
}
__device__ inline void atomic_min_f64(double* p, double v) {
  // CAS loop; identity is +inf so NaN handling follows Spark MIN (NaN > all)
  unsigned long long* u = reinterpret_cast<unsigned long long*>(p);
  unsigned long long old = *u;
  while (true) {
    double cur = __longlong_as_double(old);
    double nv = (v < cur || cur != cur) ? v : cur;
    if (nv == cur && !(cur != cur)) return;
    unsigned long long assumed = old;
    old = atomicCAS(u, assumed, __double_as_longlong(nv));
    if (old == assumed) return;
  }
}
__device__ inline void atomic_max_f64(double* p, double v) {
  unsigned long long* u = reinterpret_cast<unsigned long long*>(p);
  unsigned long long old = *u;
  while (true) {
    double cur = __longlong_as_double(old);
    double nv = (v > cur || cur != cur) ? v : cur;
    if (nv == cur && !(cur != cur)) return;
    unsigned long long assumed = old;
    old = atomicCAS(u, assumed, __double_as_longlong(nv));
    if (old == assumed) return;
  }
}


__device__ inline void agg_accumulate(const AggDesc* __restrict__ aggs,
                                      int32_t naggs, int64_t row, int64_t s) {
  for (int32_t a = 0; a < naggs; ++a) {
    const AggDesc& g = aggs[a];
    switch (g.op) {
      case AGG_COUNT_ALL:
        atomicAdd((unsigned long long*)g.state + s, 1ull);
        break;
      case AGG_COUNT_VALID:
        if (is_valid(g.valid, row))
          atomicAdd((unsigned long long*)g.state + s, 1ull);
        break;
      case AGG_SUM_INT64:
        if (is_valid(g.valid, row))
          atomicAdd((unsigned long long*)g.state + s,
                    (unsigned long long)fetch_int64(g.data, g.in_dtype, row));
        break;
      case AGG_SUM_FLOAT64:
        if (is_valid(g.valid, row))
          atomicAdd(reinterpret_cast<double*>(g.state) + s,
                    fetch_double(g.data, g.in_dtype, row));
        break;
      case AGG_MIN_INT64:
        if (is_valid(g.valid, row))
          atomic_min_i64(reinterpret_cast<int64_t*>(g.state) + s,
                         fetch_int64(g.data, g.in_dtype, row));
        break;
      case AGG_MAX_INT64:
        if (is_valid(g.valid, row))
          atomic_max_i64(reinterpret_cast<int64_t*>(g.state) + s,
                         fetch_int64(g.data, g.in_dtype, row));
        break;
      case AGG_MIN_FLOAT64:
        if (is_valid(g.valid, row))
          atomic_min_f64(reinterpret_cast<double*>(g.state) + s,
                         fetch_double(g.data, g.in_dtype, row));
        break;
      case AGG_MAX_FLOAT64:
        if (is_valid(g.valid, row))
          atomic_max_f64(reinterpret_cast<double*>(g.state) + s,
                         fetch_double(g.data, g.in_dtype, row));
        break;
    }
  }
}

}  // namespace srj
