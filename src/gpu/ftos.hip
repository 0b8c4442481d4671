// Shortest-round-trip float/double -> string (Ryu algorithm, Ulf Adams,
// PLDI 2018), formatted to match Java Double.toString / Float.toString —
// the Spark CAST(floating AS STRING) semantics.
//
// Reference parity: ftos_converter.cuh (1,493 LoC) + cast_float_to_string.cu
// + format_float.cu. This is a fresh implementation of the same published
// algorithm; tables generated into ryu_tables.inc.
//
// Java formatting rules: NaN/Infinity/-Infinity literals; decimal notation
// for 1e-3 <= |x| < 1e7 with at least one fraction digit ("1.0"); otherwise
// scientific d.dddE±x (no '+' sign, e.g. "1.0E10", "-4.9E-324").
#include "srj_common.hpp"
#include "ryu_tables.inc"
#include "exact_fp.inc"

namespace srj {

#include "ryu_format.inc"

template <typename T, bool WRITE>
__global__ void float_to_string_kernel(const T* __restrict__ in,
                                       const uint8_t* __restrict__ valid,
                                       int64_t nrows, int32_t* __restrict__ lens,
                                       const int32_t* __restrict__ offsets,
                                       char* __restrict__ chars,
                                       uint8_t* __restrict__ out_valid) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  int64_t npad = (nrows + WAVE - 1) & ~(int64_t)(WAVE - 1);
  for (int64_t row = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; row < npad;
       row += stride) {
    bool in_range = row < nrows;
    bool v = in_range && is_valid(valid, row);
    char buf[32];
    int len = 0;
    if (v) {
      if constexpr (sizeof(T) == 8) len = format_double(in[row], buf);
      else len = format_float(in[row], buf);
    }
    if (WRITE) {
      if (v) {
        int32_t o = offsets[row];
        for (int k = 0; k < len; ++k) chars[o + k] = buf[k];
      }
      ballot_write_validity(out_valid, row, v);
    } else {
      if (in_range) lens[row] = len;
    }
  }
}

// ---------------------------------------------------------------------------
// round_float: Spark round/bround on doubles at decimal scale d, exact via
// shortest decimal digits (Ryu) -> decimal-space rounding -> exact re-parse
// (Eisel-Lemire). HALF_UP (round) or HALF_EVEN (bround).
// ---------------------------------------------------------------------------
__global__ void round_double_kernel(const double* __restrict__ in,
                                    const uint8_t* __restrict__ valid, int64_t n,
                                    int32_t scale, int32_t half_even,
                                    double* __restrict__ out) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    double x = in[i];
    if (!is_valid(valid, i)) { out[i] = 0; continue; }
    uint64_t bits;
    __builtin_memcpy(&bits, &x, 8);
    uint32_t e = (uint32_t)((bits >> 52) & 0x7FF);
    uint64_t m = bits & ((1ull << 52) - 1);
    bool neg = bits >> 63;
    if (e == 0x7FF || x == 0.0) { out[i] = x; continue; }
    Dec64 d = ryu_d2d(m, e);
    // round digits*10^exponent at decimal position -scale
    int32_t drop = -scale - d.exponent;  // digits to drop from the right
    if (drop <= 0) { out[i] = x; continue; }
    uint64_t v = d.digits;
    int nd = 0;
    for (uint64_t t = v; t; t /= 10) ++nd;
    if (drop >= nd + 1) { out[i] = neg ? -0.0 : 0.0; continue; }
    uint64_t p10 = 1;
    for (int k = 0; k < drop; ++k) p10 *= 10;
    uint64_t q = v / p10;
    uint64_t r = v - q * p10;
    uint64_t half = p10 / 2;
    bool up;
    if (half_even) {
      up = r > half || (r == half && (q & 1));
    } else {
      up = r >= half;
    }
    if (up) ++q;
    double res;
    if (!eisel_lemire(q, d.exponent + drop, neg, &res)) {
      // fallback: scale by double arithmetic (ambiguous cases are rare)
      res = (neg ? -1.0 : 1.0) * (double)q;
      int32_t ee = d.exponent + drop;
      while (ee > 0) { res *= 10.0; --ee; }
      while (ee < 0) { res /= 10.0; ++ee; }
    }
    out[i] = res;
  }
}



// ---------------------------------------------------------------------------
// format_number (reference format_float.cu): Spark format_number(col, d) =
// java.text.DecimalFormat "#,###,###.##" — Java rounds the SHORTEST decimal
// representation (Ryu digits) HALF_EVEN at d places, then groups the integer
// part with commas. NaN -> "NaN", +-Infinity -> "Infinity"/"-Infinity".
// ---------------------------------------------------------------------------
template <bool WRITE>
__global__ void format_number_kernel(const double* __restrict__ in,
                                     const uint8_t* __restrict__ valid,
                                     int64_t n, int32_t d,
                                     int32_t* __restrict__ lens,
                                     const int32_t* __restrict__ offsets,
                                     char* __restrict__ chars,
                                     uint8_t* __restrict__ out_valid) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  int64_t npad = (n + WAVE - 1) & ~(int64_t)(WAVE - 1);
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < npad;
       i += stride) {
    bool in_range = i < n;
    bool ok = in_range && is_valid(valid, i);
    char buf[448];
    int len = 0;
    if (ok) {
      double x = in[i];
      uint64_t bits;
      __builtin_memcpy(&bits, &x, 8);
      uint32_t e = (uint32_t)((bits >> 52) & 0x7FF);
      uint64_t m = bits & ((1ull << 52) - 1);
      bool neg = bits >> 63;
      if (e == 0x7FF) {
        const char* lit = m ? "NaN" : (neg ? "-Infinity" : "Infinity");
        while (lit[len]) { buf[len] = lit[len]; ++len; }
      } else {
        uint64_t digs;
        int32_t exp10;
        if (x == 0.0) {
          digs = 0; exp10 = 0; neg = false;
        } else {
          Dec64 dec = ryu_d2d(m, e);
          digs = dec.digits; exp10 = dec.exponent;
        }
        // round HALF_EVEN at -d
        int32_t drop = -d - exp10;
        if (drop > 0) {
          int nd = 0;
          for (uint64_t t = digs; t; t /= 10) ++nd;
          if (drop >= nd + 1) { digs = 0; exp10 = -d; }
          else {
            uint64_t p10 = 1;
            for (int k = 0; k < drop; ++k) p10 *= 10;
            uint64_t q = digs / p10, r = digs - (digs / p10) * p10;
            uint64_t half = p10 / 2;
            if (r > half || (r == half && (q & 1))) ++q;
            digs = q; exp10 = -d;
          }
        }
        if (digs == 0) neg = false;  // -0.00 prints as 0.00 (Java)
        // digit string of digs
        char ds[20];
        int nd = 0;
        if (digs == 0) ds[nd++] = '0';
        for (uint64_t t = digs; t; t /= 10) ds[nd++] = '0' + (t % 10);
        // value = ds (reversed) * 10^exp10 ; int digits count:
        int32_t int_digits = nd + exp10;  // may be <= 0
        if (neg) buf[len++] = '-';
        if (int_digits <= 0) {
          buf[len++] = '0';
        } else {
          for (int32_t k = 0; k < int_digits; ++k) {
            int src = nd - 1 - k;  // index into reversed ds
            buf[len++] = src >= 0 ? ds[src] : '0';
            int32_t remaining = int_digits - 1 - k;
            if (remaining > 0 && remaining % 3 == 0) buf[len++] = ',';
          }
        }
        if (d > 0) {
          buf[len++] = '.';
          for (int32_t k = 0; k < d; ++k) {
            // fractional digit k: overall digit index int_digits + k
            int32_t di = int_digits + k;
            int src = nd - 1 - di;
            buf[len++] = (di >= 0 && src >= 0) ? ds[src] : '0';
          }
        }
      }
    }
    if (WRITE) {
      if (ok) {
        char* o = chars + offsets[i];
        for (int k = 0; k < len; ++k) o[k] = buf[k];
      }
      ballot_write_validity(out_valid, i, ok);
    } else if (in_range) {
      lens[i] = ok ? len : 0;
    }
  }
}

}  // namespace srj

using namespace srj;

extern "C" {

void srj_float_to_string(const void* in, const uint8_t* valid, int64_t nrows,
                         int32_t width, int32_t phase, int32_t* lens,
                         const int32_t* offsets, char* chars, uint8_t* out_valid,
                         hipStream_t stream) {
  int64_t g = grid_1d(nrows);
  if (width == 8) {
    if (phase == 0)
      float_to_string_kernel<double, false><<<g, DEFAULT_BLOCK, 0, stream>>>(
          (const double*)in, valid, nrows, lens, nullptr, nullptr, nullptr);
    else
      float_to_string_kernel<double, true><<<g, DEFAULT_BLOCK, 0, stream>>>(
          (const double*)in, valid, nrows, nullptr, offsets, chars, out_valid);
  } else {
    if (phase == 0)
      float_to_string_kernel<float, false><<<g, DEFAULT_BLOCK, 0, stream>>>(
          (const float*)in, valid, nrows, lens, nullptr, nullptr, nullptr);
    else
      float_to_string_kernel<float, true><<<g, DEFAULT_BLOCK, 0, stream>>>(
          (const float*)in, valid, nrows, nullptr, offsets, chars, out_valid);
  }
}

void srj_round_double(const double* in, const uint8_t* valid, int64_t n,
                      int32_t scale, int32_t half_even, double* out,
                      hipStream_t stream) {
  round_double_kernel<<<grid_1d(n), DEFAULT_BLOCK, 0, stream>>>(
      in, valid, n, scale, half_even, out);
}

void srj_format_number(const double* in, const uint8_t* valid, int64_t n,
                       int32_t d, int32_t phase, int32_t* lens,
                       const int32_t* offsets, char* chars, uint8_t* out_valid,
                       hipStream_t stream) {
  if (phase == 0)
    format_number_kernel<false><<<grid_1d(n), DEFAULT_BLOCK, 0, stream>>>(
        in, valid, n, d, lens, nullptr, nullptr, nullptr);
  else
    format_number_kernel<true><<<grid_1d(n), DEFAULT_BLOCK, 0, stream>>>(
        in, valid, n, d, nullptr, offsets, chars, out_valid);
}

}  // extern "C"
