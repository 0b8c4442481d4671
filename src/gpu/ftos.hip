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

struct Dec64 {
  uint64_t digits;  // decimal significand, no trailing zeros
  int32_t exponent; // value = digits * 10^exponent
};

__device__ inline uint32_t pow5_factor64(uint64_t v) {
  uint32_t count = 0;
  while (v > 0) {
    if (v % 5 != 0) return count;
    v /= 5;
    ++count;
  }
  return 0;
}

__device__ inline bool multiple_of_pow5_64(uint64_t v, uint32_t p) {
  return pow5_factor64(v) >= p;
}

__device__ inline bool multiple_of_pow2_64(uint64_t v, uint32_t p) {
  return (v & ((1ull << p) - 1)) == 0;
}

__device__ inline uint64_t mulshift64(uint64_t m, const unsigned long long* mul,
                                      int32_t j) {
  // (m * (mul[1]:mul[0])) >> j, j in (64, 128)
  uint64_t b0_lo = m * mul[0];
  uint64_t b0_hi = __umul64hi(m, mul[0]);
  uint64_t b2_lo = m * mul[1];
  uint64_t b2_hi = __umul64hi(m, mul[1]);
  uint64_t mid = b0_hi + b2_lo;
  uint64_t hi = b2_hi + (mid < b0_hi);
  // result = (hi:mid) >> (j - 64)
  int32_t s = j - 64;
  return (mid >> s) | (hi << (64 - s));
}

__device__ inline int32_t log10pow2(int32_t e) { return (e * 78913) >> 18; }
__device__ inline int32_t log10pow5(int32_t e) { return (e * 732923) >> 20; }
__device__ inline int32_t pow5bits(int32_t e) {
  return ((e * 1217359) >> 19) + 1;
}

// core Ryu d2d: ieee mantissa (52 explicit bits) + biased exponent
__device__ Dec64 ryu_d2d(uint64_t ieee_mantissa, uint32_t ieee_exponent) {
  int32_t e2;
  uint64_t m2;
  if (ieee_exponent == 0) {
    e2 = 1 - 1023 - 52 - 2;
    m2 = ieee_mantissa;
  } else {
    e2 = (int32_t)ieee_exponent - 1023 - 52 - 2;
    m2 = (1ull << 52) | ieee_mantissa;
  }
  bool even = (m2 & 1) == 0;
  bool accept_bounds = even;

  uint64_t mv = 4 * m2;
  uint32_t mm_shift = ieee_mantissa != 0 || ieee_exponent <= 1;
  uint64_t vr, vp, vm;
  int32_t e10;
  bool vm_trailing = false, vr_trailing = false;
  if (e2 >= 0) {
    uint32_t q = log10pow2(e2) - (e2 > 3);
    e10 = (int32_t)q;
    int32_t k = 125 + pow5bits((int32_t)q) - 1;
    int32_t i = -e2 + (int32_t)q + k;
    vr = mulshift64(mv, RYU_D_POW5_INV[q], i);
    vp = mulshift64(mv + 2, RYU_D_POW5_INV[q], i);
    vm = mulshift64(mv - 1 - mm_shift, RYU_D_POW5_INV[q], i);
    if (q <= 21) {
      if (mv % 5 == 0) {
        vr_trailing = multiple_of_pow5_64(mv, q);
      } else if (accept_bounds) {
        vm_trailing = multiple_of_pow5_64(mv - 1 - mm_shift, q);
      } else {
        vp -= multiple_of_pow5_64(mv + 2, q);
      }
    }
  } else {
    uint32_t q = log10pow5(-e2) - (-e2 > 1);
    e10 = (int32_t)q + e2;
    int32_t i = -e2 - (int32_t)q;
    int32_t k = pow5bits(i) - 125;
    int32_t j = (int32_t)q - k;
    vr = mulshift64(mv, RYU_D_POW5[i], j);
    vp = mulshift64(mv + 2, RYU_D_POW5[i], j);
    vm = mulshift64(mv - 1 - mm_shift, RYU_D_POW5[i], j);
    if (q <= 1) {
      vr_trailing = true;
      if (accept_bounds) {
        vm_trailing = mm_shift == 1;
      } else {
        --vp;
      }
    } else if (q < 63) {
      vr_trailing = multiple_of_pow2_64(mv, q);
    }
  }

  int32_t removed = 0;
  uint8_t last_removed = 0;
  uint64_t output;
  if (vm_trailing || vr_trailing) {
    while (vp / 10 > vm / 10) {
      vm_trailing &= vm % 10 == 0;
      vr_trailing &= last_removed == 0;
      last_removed = (uint8_t)(vr % 10);
      vr /= 10; vp /= 10; vm /= 10;
      ++removed;
    }
    if (vm_trailing) {
      while (vm % 10 == 0) {
        vr_trailing &= last_removed == 0;
        last_removed = (uint8_t)(vr % 10);
        vr /= 10; vp /= 10; vm /= 10;
        ++removed;
      }
    }
    if (vr_trailing && last_removed == 5 && vr % 2 == 0) {
      last_removed = 4;  // round even
    }
    output = vr + ((vr == vm && (!accept_bounds || !vm_trailing)) ||
                   last_removed >= 5);
  } else {
    bool round_up = false;
    if (vp / 100 > vm / 100) {
      round_up = vr % 100 >= 50;
      vr /= 100; vp /= 100; vm /= 100;
      removed += 2;
    }
    while (vp / 10 > vm / 10) {
      round_up = vr % 10 >= 5;
      vr /= 10; vp /= 10; vm /= 10;
      ++removed;
    }
    output = vr + (vr == vm || round_up);
  }
  return Dec64{output, e10 + removed};
}

__device__ Dec64 ryu_f2d(uint32_t ieee_mantissa, uint32_t ieee_exponent) {
  int32_t e2;
  uint32_t m2;
  if (ieee_exponent == 0) {
    e2 = 1 - 127 - 23 - 2;
    m2 = ieee_mantissa;
  } else {
    e2 = (int32_t)ieee_exponent - 127 - 23 - 2;
    m2 = (1u << 23) | ieee_mantissa;
  }
  bool even = (m2 & 1) == 0;
  bool accept_bounds = even;
  uint32_t mv = 4 * m2;
  uint32_t mp = mv + 2;
  uint32_t mm_shift = ieee_mantissa != 0 || ieee_exponent <= 1;
  uint32_t mm = mv - 1 - mm_shift;
  uint32_t vr, vp, vm;
  int32_t e10;
  bool vm_trailing = false, vr_trailing = false;
  uint8_t last_removed = 0;

  auto mulshift32 = [](uint32_t m, uint64_t factor, int32_t shift) {
    uint64_t lo = (uint64_t)m * (uint32_t)factor;
    uint64_t hi = (uint64_t)m * (uint32_t)(factor >> 32);
    uint64_t sum = (lo >> 32) + hi;
    return (uint32_t)(sum >> (shift - 32));
  };

  if (e2 >= 0) {
    uint32_t q = (uint32_t)log10pow2(e2);
    e10 = (int32_t)q;
    int32_t k = 59 + pow5bits((int32_t)q) - 1;
    int32_t i = -e2 + (int32_t)q + k;
    vr = mulshift32(mv, RYU_F_POW5_INV[q], i);
    vp = mulshift32(mp, RYU_F_POW5_INV[q], i);
    vm = mulshift32(mm, RYU_F_POW5_INV[q], i);
    if (q != 0 && (vp - 1) / 10 <= vm / 10) {
      int32_t l = 59 + pow5bits((int32_t)q - 1) - 1;
      last_removed =
          (uint8_t)(mulshift32(mv, RYU_F_POW5_INV[q - 1], -e2 + (int32_t)q - 1 + l) % 10);
    }
    if (q <= 9) {
      if (mv % 5 == 0) vr_trailing = multiple_of_pow5_64(mv, q);
      else if (accept_bounds) vm_trailing = multiple_of_pow5_64(mm, q);
      else vp -= multiple_of_pow5_64(mp, q);
    }
  } else {
    uint32_t q = (uint32_t)log10pow5(-e2);
    e10 = (int32_t)q + e2;
    int32_t i = -e2 - (int32_t)q;
    int32_t k = pow5bits(i) - 61;
    int32_t j = (int32_t)q - k;
    vr = mulshift32(mv, RYU_F_POW5[i], j);
    vp = mulshift32(mp, RYU_F_POW5[i], j);
    vm = mulshift32(mm, RYU_F_POW5[i], j);
    if (q != 0 && (vp - 1) / 10 <= vm / 10) {
      int32_t jj = (int32_t)q - 1 - (pow5bits(i + 1) - 61);
      last_removed = (uint8_t)(mulshift32(mv, RYU_F_POW5[i + 1], jj) % 10);
    }
    if (q <= 1) {
      vr_trailing = true;
      if (accept_bounds) vm_trailing = mm_shift == 1;
      else --vp;
    } else if (q < 31) {
      vr_trailing = multiple_of_pow2_64(mv, q - 1);
    }
  }

  int32_t removed = 0;
  uint32_t output;
  if (vm_trailing || vr_trailing) {
    while (vp / 10 > vm / 10) {
      vm_trailing &= vm % 10 == 0;
      vr_trailing &= last_removed == 0;
      last_removed = (uint8_t)(vr % 10);
      vr /= 10; vp /= 10; vm /= 10;
      ++removed;
    }
    if (vm_trailing) {
      while (vm % 10 == 0) {
        vr_trailing &= last_removed == 0;
        last_removed = (uint8_t)(vr % 10);
        vr /= 10; vp /= 10; vm /= 10;
        ++removed;
      }
    }
    if (vr_trailing && last_removed == 5 && vr % 2 == 0) last_removed = 4;
    output = vr + ((vr == vm && (!accept_bounds || !vm_trailing)) ||
                   last_removed >= 5);
  } else {
    while (vp / 10 > vm / 10) {
      last_removed = (uint8_t)(vr % 10);
      vr /= 10; vp /= 10; vm /= 10;
      ++removed;
    }
    output = vr + (vr == vm || last_removed >= 5);
  }
  return Dec64{output, e10 + removed};
}

// ---------------------------------------------------------------------------
// Java-style formatting of (digits, exponent)
// ---------------------------------------------------------------------------
__device__ inline int format_java(uint64_t digits, int32_t exp10, bool neg,
                                  char* buf) {
  // value = digits * 10^exp10, digits has no trailing zeros
  char dig[20];
  int nd = 0;
  uint64_t d = digits;
  do {
    dig[nd++] = (char)('0' + d % 10);
    d /= 10;
  } while (d);
  // dig is reversed; decimal point position after first digit: value =
  // 0.d1d2... * 10^(nd + exp10)
  int32_t point = nd + exp10;  // digits before the decimal point
  int len = 0;
  if (neg) buf[len++] = '-';
  if (point > 0 && point <= 7) {
    // plain decimal, Java style
    for (int i = 0; i < point; ++i)
      buf[len++] = i < nd ? dig[nd - 1 - i] : '0';
    buf[len++] = '.';
    if (point >= nd) {
      buf[len++] = '0';
    } else {
      for (int i = point; i < nd; ++i) buf[len++] = dig[nd - 1 - i];
    }
  } else if (point <= 0 && point > -3) {
    buf[len++] = '0';
    buf[len++] = '.';
    for (int i = 0; i < -point; ++i) buf[len++] = '0';
    for (int i = 0; i < nd; ++i) buf[len++] = dig[nd - 1 - i];
  } else {
    // scientific: d.dddE[-]x with exponent = point - 1
    buf[len++] = dig[nd - 1];
    buf[len++] = '.';
    if (nd == 1) buf[len++] = '0';
    else
      for (int i = 1; i < nd; ++i) buf[len++] = dig[nd - 1 - i];
    buf[len++] = 'E';
    int32_t e = point - 1;
    if (e < 0) {
      buf[len++] = '-';
      e = -e;
    }
    char ebuf[4];
    int en = 0;
    do {
      ebuf[en++] = (char)('0' + e % 10);
      e /= 10;
    } while (e);
    for (int i = en - 1; i >= 0; --i) buf[len++] = ebuf[i];
  }
  return len;
}

__device__ inline int format_double(double v, char* buf) {
  uint64_t bits;
  __builtin_memcpy(&bits, &v, 8);
  bool neg = bits >> 63;
  uint32_t e = (uint32_t)((bits >> 52) & 0x7FF);
  uint64_t m = bits & ((1ull << 52) - 1);
  if (e == 0x7FF) {
    const char* s = m ? "NaN" : (neg ? "-Infinity" : "Infinity");
    int n = 0;
    while (s[n]) { buf[n] = s[n]; ++n; }
    return n;
  }
  if (e == 0 && m == 0) {
    int len = 0;
    if (neg) buf[len++] = '-';
    buf[len++] = '0'; buf[len++] = '.'; buf[len++] = '0';
    return len;
  }
  Dec64 dec = ryu_d2d(m, e);
  return format_java(dec.digits, dec.exponent, neg, buf);
}

__device__ inline int format_float(float v, char* buf) {
  uint32_t bits;
  __builtin_memcpy(&bits, &v, 4);
  bool neg = bits >> 31;
  uint32_t e = (bits >> 23) & 0xFF;
  uint32_t m = bits & ((1u << 23) - 1);
  if (e == 0xFF) {
    const char* s = m ? "NaN" : (neg ? "-Infinity" : "Infinity");
    int n = 0;
    while (s[n]) { buf[n] = s[n]; ++n; }
    return n;
  }
  if (e == 0 && m == 0) {
    int len = 0;
    if (neg) buf[len++] = '-';
    buf[len++] = '0'; buf[len++] = '.'; buf[len++] = '0';
    return len;
  }
  Dec64 dec = ryu_f2d(m, e);
  return format_java(dec.digits, dec.exponent, neg, buf);
}

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
