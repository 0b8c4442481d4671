// DECIMAL128 arithmetic with 256-bit intermediates and Spark semantics.
//
// Reference parity: decimal_utils.cu (chunked256 4xu64 bignum, multiply/
// divide/add/sub/integer-divide/remainder with Spark's result scale and
// HALF_UP rounding, overflow -> null or ANSI error) + DecimalUtils.java.
//
// Values are 128-bit two's complement (2 x int64 little-endian words per
// row). Intermediates use a 4 x u64 chunked 256-bit type like the
// reference's chunked256.
#include "srj_common.hpp"

namespace srj {

struct U256 {
  uint64_t w[4];  // little-endian
};

__device__ inline U256 u256_from_i128(__int128 v) {
  U256 r;
  unsigned __int128 u = (unsigned __int128)v;
  r.w[0] = (uint64_t)u;
  r.w[1] = (uint64_t)(u >> 64);
  // sign-extend
  uint64_t ext = v < 0 ? ~0ull : 0ull;
  r.w[2] = ext;
  r.w[3] = ext;
  return r;
}

__device__ inline bool u256_is_neg(const U256& a) { return a.w[3] >> 63; }

__device__ inline U256 u256_neg(const U256& a) {
  U256 r;
  unsigned __int128 carry = 1;
  for (int i = 0; i < 4; ++i) {
    unsigned __int128 s = (unsigned __int128)(~a.w[i]) + carry;
    r.w[i] = (uint64_t)s;
    carry = (uint64_t)(s >> 64);
  }
  return r;
}

// unsigned 128x128 -> 256 multiply
__device__ inline U256 u256_mul_u128(unsigned __int128 a, unsigned __int128 b) {
  uint64_t a0 = (uint64_t)a, a1 = (uint64_t)(a >> 64);
  uint64_t b0 = (uint64_t)b, b1 = (uint64_t)(b >> 64);
  unsigned __int128 p00 = (unsigned __int128)a0 * b0;
  unsigned __int128 p01 = (unsigned __int128)a0 * b1;
  unsigned __int128 p10 = (unsigned __int128)a1 * b0;
  unsigned __int128 p11 = (unsigned __int128)a1 * b1;
  U256 r{};
  r.w[0] = (uint64_t)p00;
  unsigned __int128 mid = (p00 >> 64) + (uint64_t)p01 + (uint64_t)p10;
  r.w[1] = (uint64_t)mid;
  unsigned __int128 hi = (mid >> 64) + (p01 >> 64) + (p10 >> 64) + (uint64_t)p11;
  r.w[2] = (uint64_t)hi;
  r.w[3] = (uint64_t)((hi >> 64) + (p11 >> 64));
  return r;
}

// divide |a| (256-bit) by small u64, returning quotient; remainder out
__device__ inline U256 u256_divmod_u64(const U256& a, uint64_t d,
                                       uint64_t* rem) {
  U256 q{};
  unsigned __int128 r = 0;
  for (int i = 3; i >= 0; --i) {
    r = (r << 64) | a.w[i];
    q.w[i] = (uint64_t)(r / d);
    r = r % d;
  }
  *rem = (uint64_t)r;
  return q;
}

__device__ inline bool u256_fits_i128(const U256& a) {
  // signed: w[2]/w[3] must be sign extension of bit 127
  uint64_t ext = (a.w[1] >> 63) ? ~0ull : 0ull;
  return a.w[2] == ext && a.w[3] == ext;
}

__device__ inline __int128 u256_to_i128(const U256& a) {
  return (__int128)(((unsigned __int128)a.w[1] << 64) | a.w[0]);
}

__device__ inline U256 u256_add(const U256& a, const U256& b) {
  U256 r;
  unsigned __int128 c = 0;
  for (int i = 0; i < 4; ++i) {
    unsigned __int128 s = (unsigned __int128)a.w[i] + b.w[i] + (uint64_t)c;
    r.w[i] = (uint64_t)s;
    c = s >> 64;
  }
  return r;
}

__device__ inline const uint64_t* pow10_u64_table() {
  static const uint64_t t[20] = {1ull,
                                 10ull,
                                 100ull,
                                 1000ull,
                                 10000ull,
                                 100000ull,
                                 1000000ull,
                                 10000000ull,
                                 100000000ull,
                                 1000000000ull,
                                 10000000000ull,
                                 100000000000ull,
                                 1000000000000ull,
                                 10000000000000ull,
                                 100000000000000ull,
                                 1000000000000000ull,
                                 10000000000000000ull,
                                 100000000000000000ull,
                                 1000000000000000000ull,
                                 10000000000000000000ull};
  return t;
}

// scale |v|(256) down by `drop` decimal digits with HALF_UP. Only the FIRST
// dropped digit decides HALF_UP, so: truncate (drop-1) digits chunked, then
// one divmod-10 with rem >= 5 rounding.
__device__ inline U256 u256_scale_down_half_up(U256 mag, int drop) {
  const uint64_t* P = pow10_u64_table();
  int trunc = drop - 1;
  while (trunc > 0) {
    int step = trunc > 19 ? 19 : trunc;
    uint64_t rem;
    mag = u256_divmod_u64(mag, P[step], &rem);
    trunc -= step;
  }
  uint64_t rem;
  U256 q = u256_divmod_u64(mag, 10, &rem);
  if (rem >= 5) {
    U256 one{};
    one.w[0] = 1;
    q = u256_add(q, one);
  }
  return q;
}

__device__ inline __int128 pow10_i128(int p) {
  __int128 r = 1;
  for (int i = 0; i < p; ++i) r *= 10;
  return r;
}

__device__ inline bool i128_precision_ok(__int128 v, int precision) {
  if (precision >= 39) return true;
  __int128 lim = pow10_i128(precision);
  if (v < 0) v = -v;
  return v < lim;
}

// ---------------------------------------------------------------------------
// multiply: result scale/precision per Spark (caller computes); product of
// unscaled values rescaled from (s1+s2) to out_scale with HALF_UP.
// ---------------------------------------------------------------------------
__global__ void dec128_mul_kernel(const __int128* __restrict__ a,
                                  const uint8_t* __restrict__ va,
                                  const __int128* __restrict__ b,
                                  const uint8_t* __restrict__ vb, int64_t n,
                                  int32_t scale_sum, int32_t out_scale,
                                  int32_t out_precision,
                                  __int128* __restrict__ out,
                                  uint8_t* __restrict__ out_valid,
                                  int64_t* __restrict__ err_row) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  int64_t npad = (n + WAVE - 1) & ~(int64_t)(WAVE - 1);
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < npad;
       i += stride) {
    bool in_range = i < n;
    bool valid = in_range && is_valid(va, i) && is_valid(vb, i);
    __int128 res = 0;
    if (valid) {
      __int128 x = a[i], y = b[i];
      bool neg = (x < 0) != (y < 0);
      unsigned __int128 ux = x < 0 ? (unsigned __int128)(-x) : (unsigned __int128)x;
      unsigned __int128 uy = y < 0 ? (unsigned __int128)(-y) : (unsigned __int128)y;
      U256 p = u256_mul_u128(ux, uy);
      int drop = scale_sum - out_scale;
      if (drop > 0) p = u256_scale_down_half_up(p, drop);
      if (!u256_fits_i128(p) || (p.w[1] >> 63)) {
        valid = false;  // magnitude exceeds int128
      } else {
        res = u256_to_i128(p);
        if (drop < 0) {
          // scale up (rare): out_scale > s1+s2
          for (int k = 0; k < -drop && valid; ++k) {
            __int128 nx = res * 10;
            if (nx / 10 != res) valid = false;
            res = nx;
          }
        }
        if (valid && !i128_precision_ok(res, out_precision)) valid = false;
        if (neg) res = -res;
      }
      if (!valid && err_row)
        atomicMin(reinterpret_cast<long long*>(err_row), (long long)i);
    }
    if (in_range) out[i] = res;
    ballot_write_validity(out_valid, i, valid);
  }
}

// ---------------------------------------------------------------------------
// divide: Spark DIVIDE: result = round_half_up(a / b at out_scale).
// numerator scaled up by (out_scale - s1 + s2 + 1) then divided by b with an
// extra digit for rounding (the reference's approach).
// ---------------------------------------------------------------------------
__device__ inline U256 u256_mul_small(const U256& a, uint64_t m,
                                      bool* ovf = nullptr) {
  U256 r{};
  unsigned __int128 carry = 0;
  for (int i = 0; i < 4; ++i) {
    unsigned __int128 p = (unsigned __int128)a.w[i] * m + (uint64_t)carry;
    r.w[i] = (uint64_t)p;
    carry = p >> 64;
  }
  if (ovf && carry) *ovf = true;
  return r;
}

// long division of 256-bit by 128-bit magnitude (shift-subtract, 256 steps)
__device__ inline U256 u256_div_u128(const U256& num, unsigned __int128 den) {
  U256 q{};
  unsigned __int128 rem = 0;
  for (int bit = 255; bit >= 0; --bit) {
    // rem = rem*2 + bit(num); overflow-safe: rem < den <= 2^128-1
    unsigned __int128 top = rem >> 127;
    rem = (rem << 1) | ((num.w[bit >> 6] >> (bit & 63)) & 1);
    if (top || rem >= den) {
      rem -= den;
      q.w[bit >> 6] |= 1ull << (bit & 63);
    }
  }
  return q;
}

__global__ void dec128_div_kernel(const __int128* __restrict__ a,
                                  const uint8_t* __restrict__ va,
                                  const __int128* __restrict__ b,
                                  const uint8_t* __restrict__ vb, int64_t n,
                                  int32_t s1, int32_t s2, int32_t out_scale,
                                  int32_t out_precision, int32_t integer_div,
                                  int32_t remainder,
                                  __int128* __restrict__ out,
                                  uint8_t* __restrict__ out_valid,
                                  int64_t* __restrict__ err_row) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  int64_t npad = (n + WAVE - 1) & ~(int64_t)(WAVE - 1);
  const uint64_t* P = pow10_u64_table();
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < npad;
       i += stride) {
    bool in_range = i < n;
    bool valid = in_range && is_valid(va, i) && is_valid(vb, i);
    __int128 res = 0;
    if (valid) {
      __int128 x = a[i], y = b[i];
      if (y == 0) {
        valid = false;
      } else if (remainder) {
        // remainder on aligned scales: s1/s2 carry the scale-up counts here
        bool ovf = false;
        for (int k = 0; k < s1 && !ovf; ++k) {
          __int128 nx = x * 10;
          if (nx / 10 != x) ovf = true;
          x = nx;
        }
        for (int k = 0; k < s2 && !ovf; ++k) {
          __int128 ny = y * 10;
          if (ny / 10 != y) ovf = true;
          y = ny;
        }
        if (ovf) valid = false;
        else res = x % y;
      } else if (integer_div) {
        // integral part of a/b accounting scales: (x * 10^s2) / (y * 10^s1)
        bool neg = (x < 0) != (y < 0);
        unsigned __int128 ux = x < 0 ? (unsigned __int128)(-x)
                                     : (unsigned __int128)x;
        unsigned __int128 uy = y < 0 ? (unsigned __int128)(-y)
                                     : (unsigned __int128)y;
        U256 num = u256_from_i128((__int128)0);
        num.w[0] = (uint64_t)ux;
        num.w[1] = (uint64_t)(ux >> 64);
        int up = s2;
        bool num_ovf = false;
        while (up > 0) {
          int step = up > 19 ? 19 : up;
          num = u256_mul_small(num, P[step], &num_ovf);
          up -= step;
        }
        if (num_ovf) valid = false;
        unsigned __int128 den = uy;
        int dup = s1;
        bool den_ovf = false;
        while (dup > 0) {
          int step = dup > 19 ? 19 : dup;
          unsigned __int128 nd = den * P[step];
          if (den != 0 && nd / den != P[step]) { den_ovf = true; break; }
          den = nd;
          dup -= step;
        }
        if (den_ovf) {
          res = 0;  // denominator astronomically large -> 0
        } else {
          U256 q = u256_div_u128(num, den);
          if (!u256_fits_i128(q) || (q.w[1] >> 63)) valid = false;
          else {
            res = u256_to_i128(q);
            if (neg) res = -res;
          }
        }
        if (valid && !i128_precision_ok(res, out_precision)) valid = false;
      } else {
        bool neg = (x < 0) != (y < 0);
        unsigned __int128 ux = x < 0 ? (unsigned __int128)(-x)
                                     : (unsigned __int128)x;
        unsigned __int128 uy = y < 0 ? (unsigned __int128)(-y)
                                     : (unsigned __int128)y;
        // numerator = |x| * 10^(out_scale - s1 + s2 + 1)
        int up = out_scale - s1 + s2 + 1;
        U256 num{};
        num.w[0] = (uint64_t)ux;
        num.w[1] = (uint64_t)(ux >> 64);
        bool ovf = up < 0;
        while (up > 0 && !ovf) {
          int step = up > 19 ? 19 : up;
          num = u256_mul_small(num, P[step], &ovf);
          up -= step;
        }
        if (ovf) {
          valid = false;
        } else {
          U256 q = u256_div_u128(num, uy);
          // round: q has one extra digit
          uint64_t rem;
          U256 qq = u256_divmod_u64(q, 10, &rem);
          if (rem >= 5) {
            U256 one{};
            one.w[0] = 1;
            qq = u256_add(qq, one);
          }
          if (!u256_fits_i128(qq) || (qq.w[1] >> 63)) valid = false;
          else {
            res = u256_to_i128(qq);
            if (neg) res = -res;
          }
          if (valid && !i128_precision_ok(res, out_precision)) valid = false;
        }
      }
      if (!valid && err_row)
        atomicMin(reinterpret_cast<long long*>(err_row), (long long)i);
    }
    if (in_range) out[i] = res;
    ballot_write_validity(out_valid, i, valid);
  }
}

// add/sub with scale alignment (caller passes scale-up factors)
__global__ void dec128_addsub_kernel(const __int128* __restrict__ a,
                                     const uint8_t* __restrict__ va,
                                     const __int128* __restrict__ b,
                                     const uint8_t* __restrict__ vb, int64_t n,
                                     int32_t up_a, int32_t up_b, int32_t sub,
                                     int32_t out_precision,
                                     __int128* __restrict__ out,
                                     uint8_t* __restrict__ out_valid,
                                     int64_t* __restrict__ err_row) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  int64_t npad = (n + WAVE - 1) & ~(int64_t)(WAVE - 1);
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < npad;
       i += stride) {
    bool in_range = i < n;
    bool valid = in_range && is_valid(va, i) && is_valid(vb, i);
    __int128 res = 0;
    if (valid) {
      __int128 x = a[i], y = b[i];
      bool ovf = false;
      for (int k = 0; k < up_a && !ovf; ++k) {
        __int128 nx = x * 10;
        if (nx / 10 != x) ovf = true;
        x = nx;
      }
      for (int k = 0; k < up_b && !ovf; ++k) {
        __int128 ny = y * 10;
        if (ny / 10 != y) ovf = true;
        y = ny;
      }
      if (sub) y = -y;
      res = x + y;
      // overflow: same-sign addends producing flipped sign
      if (!ovf && ((x >= 0) == (y >= 0)) && ((res >= 0) != (x >= 0)) &&
          !(x == 0 || y == 0))
        ovf = true;
      if (ovf || !i128_precision_ok(res, out_precision)) {
        valid = false;
        res = 0;
        if (err_row)
          atomicMin(reinterpret_cast<long long*>(err_row), (long long)i);
      }
    }
    if (in_range) out[i] = res;
    ballot_write_validity(out_valid, i, valid);
  }
}

}  // namespace srj

using namespace srj;

extern "C" {

void srj_dec128_mul(const void* a, const uint8_t* va, const void* b,
                    const uint8_t* vb, int64_t n, int32_t scale_sum,
                    int32_t out_scale, int32_t out_precision, void* out,
                    uint8_t* out_valid, int64_t* err_row, hipStream_t stream) {
  dec128_mul_kernel<<<grid_1d(n), DEFAULT_BLOCK, 0, stream>>>(
      (const __int128*)a, va, (const __int128*)b, vb, n, scale_sum, out_scale,
      out_precision, (__int128*)out, out_valid, err_row);
}

void srj_dec128_div(const void* a, const uint8_t* va, const void* b,
                    const uint8_t* vb, int64_t n, int32_t s1, int32_t s2,
                    int32_t out_scale, int32_t out_precision, int32_t integer_div,
                    int32_t remainder, void* out, uint8_t* out_valid,
                    int64_t* err_row, hipStream_t stream) {
  dec128_div_kernel<<<grid_1d(n), DEFAULT_BLOCK, 0, stream>>>(
      (const __int128*)a, va, (const __int128*)b, vb, n, s1, s2, out_scale,
      out_precision, integer_div, remainder, (__int128*)out, out_valid, err_row);
}

void srj_dec128_addsub(const void* a, const uint8_t* va, const void* b,
                       const uint8_t* vb, int64_t n, int32_t up_a, int32_t up_b,
                       int32_t sub, int32_t out_precision, void* out,
                       uint8_t* out_valid, int64_t* err_row, hipStream_t stream) {
  dec128_addsub_kernel<<<grid_1d(n), DEFAULT_BLOCK, 0, stream>>>(
      (const __int128*)a, va, (const __int128*)b, vb, n, up_a, up_b, sub,
      out_precision, (__int128*)out, out_valid, err_row);
}

}  // extern "C"
