// Second misc-op batch: HyperLogLog++ (Spark-exact sketch format),
// percentile-from-histogram, number_converter (conv), parse_uri,
// charset_decode (GBK->UTF-8).
//
// Reference parity: hyper_log_log_plus_plus.cu (XXHash64(42), idx = h >>
// (64-p), rho = clz(h << p | padding) + 1, registers packed 10 x 6 bits per
// long), histogram.cu, number_converter.cu, parse_uri.cu, charset_decode.cu
// + gbk_to_unicode_table.inc (table regenerated from the GBK codec).
#include "srj_common.hpp"
#include "gbk_table.inc"

namespace srj {

// ---------------------------------------------------------------------------
// HLL++ (Spark HyperLogLogPlusPlus)
// ---------------------------------------------------------------------------
constexpr int HLL_REG_PER_LONG = 10;  // 6 bits each (reference javadoc)

__global__ void hllpp_update_kernel(const int64_t* __restrict__ hashes,
                                    const uint8_t* __restrict__ valid,
                                    int64_t nrows, int32_t precision,
                                    int32_t* __restrict__ registers) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  int32_t idx_shift = 64 - precision;
  uint64_t padding = 1ull << (precision - 1);
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < nrows;
       i += stride) {
    if (!is_valid(valid, i)) continue;
    uint64_t h = (uint64_t)hashes[i];
    uint32_t reg = (uint32_t)(h >> idx_shift);
    uint64_t w = (h << precision) | padding;
    int32_t rho = (int32_t)__clzll((long long)w) + 1;
    atomicMax(registers + reg, rho);
  }
}

__global__ void hllpp_merge_kernel(const int32_t* __restrict__ src,
                                   int32_t* __restrict__ dst, int64_t nregs) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < nregs;
       i += stride)
    atomicMax(dst + i, src[i]);
}

// pack registers into Spark longs (10 x 6 bits, register r in long r/10 at
// bit 6*(r%10)), and unpack
__global__ void hllpp_pack_kernel(const int32_t* __restrict__ regs,
                                  int64_t nregs, int64_t* __restrict__ longs) {
  int64_t nlongs = (nregs + HLL_REG_PER_LONG - 1) / HLL_REG_PER_LONG;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < nlongs;
       i += stride) {
    uint64_t v = 0;
    for (int k = 0; k < HLL_REG_PER_LONG; ++k) {
      int64_t r = i * HLL_REG_PER_LONG + k;
      if (r < nregs) v |= ((uint64_t)(regs[r] & 0x3F)) << (6 * k);
    }
    longs[i] = (int64_t)v;
  }
}

__global__ void hllpp_unpack_kernel(const int64_t* __restrict__ longs,
                                    int64_t nregs, int32_t* __restrict__ regs) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t r = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; r < nregs;
       r += stride) {
    uint64_t v = (uint64_t)longs[r / HLL_REG_PER_LONG];
    regs[r] = (int32_t)((v >> (6 * (r % HLL_REG_PER_LONG))) & 0x3F);
  }
}

// ---------------------------------------------------------------------------
// percentile from histogram (reference histogram.cu percentileFromHistogram):
// histogram rows = LIST<STRUCT<value float64, freq int64>> sorted by value;
// percentiles with linear interpolation over cumulative frequency.
// ---------------------------------------------------------------------------
__global__ void percentile_kernel(const int32_t* __restrict__ offsets,
                                  const double* __restrict__ values,
                                  const int64_t* __restrict__ freqs,
                                  int64_t nrows,
                                  const double* __restrict__ percentages,
                                  int32_t npct, double* __restrict__ out,
                                  uint8_t* __restrict__ out_valid) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  int64_t total_out = nrows * npct;
  int64_t npad = (total_out + WAVE - 1) & ~(int64_t)(WAVE - 1);
  for (int64_t t = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; t < npad;
       t += stride) {
    bool in_range = t < total_out;
    int64_t row = in_range ? t / npct : 0;
    int32_t pi = in_range ? (int32_t)(t % npct) : 0;
    int32_t s = offsets[row], e = offsets[row + 1];
    bool valid = in_range && e > s;
    double result = 0;
    if (valid) {
      int64_t total = 0;
      for (int32_t j = s; j < e; ++j) total += freqs[j];
      if (total <= 0) {
        valid = false;
      } else {
        double pos = percentages[pi] * (double)(total - 1);
        int64_t lo_rank = (int64_t)pos;
        double frac = pos - (double)lo_rank;
        // find values at rank lo_rank and lo_rank+1 (0-based over expanded)
        int64_t cum = 0;
        double v_lo = 0, v_hi = 0;
        bool got_lo = false, got_hi = false;
        for (int32_t j = s; j < e && !(got_lo && got_hi); ++j) {
          cum += freqs[j];
          if (!got_lo && cum > lo_rank) {
            v_lo = values[j];
            got_lo = true;
          }
          if (!got_hi && cum > lo_rank + 1) {
            v_hi = values[j];
            got_hi = true;
          }
        }
        if (!got_hi) v_hi = v_lo;
        result = v_lo + frac * (v_hi - v_lo);
      }
    }
    if (in_range) out[t] = result;
    ballot_write_validity(out_valid, t, valid);
  }
}

// ---------------------------------------------------------------------------
// conv(num, from_base, to_base) — Hive/Spark semantics
// ---------------------------------------------------------------------------
__device__ inline int conv_digit(char c) {
  if (c >= '0' && c <= '9') return c - '0';
  char l = c | 32;
  if (l >= 'a' && l <= 'z') return l - 'a' + 10;
  return -1;
}

template <bool WRITE>
__global__ void conv_kernel(ColDesc in, int64_t nrows, int32_t from_base,
                            int32_t to_base, int32_t* __restrict__ lens,
                            const int32_t* __restrict__ offsets,
                            char* __restrict__ chars,
                            uint8_t* __restrict__ out_valid) {
  const char* digits = "0123456789ABCDEFGHIJKLMNOPQRSTUVWXYZ";
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  int64_t npad = (nrows + WAVE - 1) & ~(int64_t)(WAVE - 1);
  for (int64_t row = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; row < npad;
       row += stride) {
    bool in_range = row < nrows;
    bool valid = in_range && is_valid(in.valid, row);
    char buf[68];
    int len = 0;
    if (valid) {
      StrView s = get_string(in, row);
      // trim spaces (Hive trims blanks)
      while (s.len > 0 && s.ptr[0] == ' ') { ++s.ptr; --s.len; }
      while (s.len > 0 && s.ptr[s.len - 1] == ' ') --s.len;
      int i = 0;
      bool neg = false;
      if (s.len > 0 && (s.ptr[0] == '-' || s.ptr[0] == '+')) {
        neg = s.ptr[0] == '-';
        i = 1;
      }
      uint64_t v = 0;
      bool any = false, overflow = false;
      for (; i < s.len; ++i) {
        int d = conv_digit(s.ptr[i]);
        if (d < 0 || d >= from_base) break;  // stop at first invalid (Hive)
        any = true;
        uint64_t nv = v * (uint64_t)from_base + (uint64_t)d;
        if (nv < v || (v > (~0ull - d) / from_base)) overflow = true;
        v = nv;
      }
      if (!any) {
        valid = false;
      } else {
        if (overflow) v = ~0ull;
        if (neg) v = (uint64_t)(-(int64_t)v);
        bool out_neg = false;
        if (to_base < 0) {
          int64_t sv = (int64_t)v;
          if (sv < 0) {
            out_neg = true;
            v = (uint64_t)(-sv);
          }
        }
        int ab = to_base < 0 ? -to_base : to_base;
        char tmp[66];
        int n = 0;
        do {
          tmp[n++] = digits[v % ab];
          v /= ab;
        } while (v);
        if (out_neg) buf[len++] = '-';
        for (int k = n - 1; k >= 0; --k) buf[len++] = tmp[k];
      }
    }
    if (WRITE) {
      if (valid) {
        int32_t o = offsets[row];
        for (int k = 0; k < len; ++k) chars[o + k] = buf[k];
      }
      ballot_write_validity(out_valid, row, valid);
    } else if (in_range) {
      lens[row] = valid ? len : 0;
    }
  }
}

// ---------------------------------------------------------------------------
// parse_uri (Spark parse_url): extract PROTOCOL / HOST / PATH / QUERY /
// QUERY(key) with validation (invalid URI -> null).
// ---------------------------------------------------------------------------
enum UriPart : int32_t {
  URI_PROTOCOL = 0,
  URI_HOST = 1,
  URI_PATH = 2,
  URI_QUERY = 3,
  URI_QUERY_KEY = 4,
};

__device__ inline bool uri_scheme_char(char c, bool first) {
  bool alpha = (c | 32) >= 'a' && (c | 32) <= 'z';
  if (first) return alpha;
  return alpha || (c >= '0' && c <= '9') || c == '+' || c == '-' || c == '.';
}

template <bool WRITE>
__global__ void parse_uri_kernel(ColDesc in, int64_t nrows, int32_t part,
                                 const char* __restrict__ qkey, int32_t qkey_len,
                                 int32_t* __restrict__ lens,
                                 const int32_t* __restrict__ offsets,
                                 char* __restrict__ chars,
                                 uint8_t* __restrict__ out_valid) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  int64_t npad = (nrows + WAVE - 1) & ~(int64_t)(WAVE - 1);
  for (int64_t row = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; row < npad;
       row += stride) {
    bool in_range = row < nrows;
    bool valid = in_range && is_valid(in.valid, row);
    const char* out_p = nullptr;
    int32_t out_n = 0;
    if (valid) {
      StrView s = get_string(in, row);
      // scheme
      int i = 0;
      while (i < s.len && s.ptr[i] != ':') {
        if (!uri_scheme_char(s.ptr[i], i == 0)) { i = -1; break; }
        ++i;
      }
      if (i <= 0 || i >= s.len) {
        valid = false;
      } else {
        int scheme_end = i;  // ':' position
        int p = i + 1;
        int auth_start = -1, auth_end = -1;
        if (p + 1 < s.len && s.ptr[p] == '/' && s.ptr[p + 1] == '/') {
          auth_start = p + 2;
          auth_end = auth_start;
          while (auth_end < s.len && s.ptr[auth_end] != '/' &&
                 s.ptr[auth_end] != '?' && s.ptr[auth_end] != '#')
            ++auth_end;
          p = auth_end;
        }
        int path_start = p, path_end = p;
        while (path_end < s.len && s.ptr[path_end] != '?' &&
               s.ptr[path_end] != '#')
          ++path_end;
        p = path_end;
        int q_start = -1, q_end = -1;
        if (p < s.len && s.ptr[p] == '?') {
          q_start = p + 1;
          q_end = q_start;
          while (q_end < s.len && s.ptr[q_end] != '#') ++q_end;
        }
        // basic validation: no spaces/control chars anywhere
        for (int k = 0; k < s.len && valid; ++k) {
          unsigned char c = (unsigned char)s.ptr[k];
          if (c <= ' ' || c == '<' || c == '>' || c == '"' || c == '`' ||
              c == '{' || c == '}' || c == '|' || c == '\\' || c == '^')
            valid = false;
        }
        if (valid) {
          switch (part) {
            case URI_PROTOCOL:
              out_p = s.ptr;
              out_n = scheme_end;
              break;
            case URI_HOST: {
              if (auth_start < 0) { valid = false; break; }
              int hs = auth_start, he = auth_end;
              for (int k = auth_start; k < auth_end; ++k)
                if (s.ptr[k] == '@') hs = k + 1;
              // strip port (but not inside [])
              bool brac = hs < he && s.ptr[hs] == '[';
              if (brac) {
                int k = hs;
                while (k < he && s.ptr[k] != ']') ++k;
                out_p = s.ptr + hs;
                out_n = k + 1 <= he ? k + 1 - hs : he - hs;
              } else {
                int pe = he;
                for (int k = hs; k < he; ++k)
                  if (s.ptr[k] == ':') { pe = k; break; }
                out_p = s.ptr + hs;
                out_n = pe - hs;
                // validate host chars
                for (int k = hs; k < pe && valid; ++k) {
                  char c = s.ptr[k];
                  bool ok = (c | 32) >= 'a' && (c | 32) <= 'z';
                  ok = ok || (c >= '0' && c <= '9') || c == '.' || c == '-' ||
                       c == '_' || c == '%';
                  if (!ok) valid = false;
                }
              }
              if (out_n == 0) valid = false;
              break;
            }
            case URI_PATH:
              out_p = s.ptr + path_start;
              out_n = path_end - path_start;
              break;
            case URI_QUERY:
              if (q_start < 0) { valid = false; break; }
              out_p = s.ptr + q_start;
              out_n = q_end - q_start;
              break;
            case URI_QUERY_KEY: {
              if (q_start < 0) { valid = false; break; }
              valid = false;
              int k = q_start;
              while (k < q_end) {
                int amp = k;
                while (amp < q_end && s.ptr[amp] != '&') ++amp;
                int eq = k;
                while (eq < amp && s.ptr[eq] != '=') ++eq;
                if (eq - k == qkey_len) {
                  bool m = true;
                  for (int t2 = 0; t2 < qkey_len; ++t2)
                    if (s.ptr[k + t2] != qkey[t2]) { m = false; break; }
                  if (m && eq < amp) {
                    out_p = s.ptr + eq + 1;
                    out_n = amp - eq - 1;
                    valid = true;
                    break;
                  }
                }
                k = amp + 1;
              }
              break;
            }
          }
        }
      }
    }
    if (WRITE) {
      if (valid && out_n > 0) {
        int32_t o = offsets[row];
        for (int k = 0; k < out_n; ++k) chars[o + k] = out_p[k];
      }
      ballot_write_validity(out_valid, row, valid);
    } else if (in_range) {
      lens[row] = valid ? out_n : 0;
    }
  }
}

// ---------------------------------------------------------------------------
// charset_decode: GBK -> UTF-8 (reference charset_decode.cu; REPLACE mode
// emits U+FFFD, REPORT mode records first bad row)
// ---------------------------------------------------------------------------
__device__ inline int utf8_encode(uint32_t cp, char* out) {
  if (cp < 0x80) {
    out[0] = (char)cp;
    return 1;
  }
  if (cp < 0x800) {
    out[0] = (char)(0xC0 | (cp >> 6));
    out[1] = (char)(0x80 | (cp & 0x3F));
    return 2;
  }
  out[0] = (char)(0xE0 | (cp >> 12));
  out[1] = (char)(0x80 | ((cp >> 6) & 0x3F));
  out[2] = (char)(0x80 | (cp & 0x3F));
  return 3;
}

template <bool WRITE>
__global__ void gbk_decode_kernel(ColDesc in, int64_t nrows, int32_t report,
                                  int32_t* __restrict__ lens,
                                  const int32_t* __restrict__ offsets,
                                  char* __restrict__ chars,
                                  uint8_t* __restrict__ out_valid,
                                  int64_t* __restrict__ err_row) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  int64_t npad = (nrows + WAVE - 1) & ~(int64_t)(WAVE - 1);
  for (int64_t row = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; row < npad;
       row += stride) {
    bool in_range = row < nrows;
    bool valid = in_range && is_valid(in.valid, row);
    int32_t out_len = 0;
    if (valid) {
      StrView s = get_string(in, row);
      int32_t o = WRITE ? offsets[row] : 0;
      for (int32_t i = 0; i < s.len;) {
        uint8_t b = (uint8_t)s.ptr[i];
        char tmp[4];
        int n;
        if (b < 0x80) {
          tmp[0] = (char)b;
          n = 1;
          ++i;
        } else if (b >= 0x81 && b <= 0xFE && i + 1 < s.len) {
          uint8_t t2 = (uint8_t)s.ptr[i + 1];
          uint32_t cp = 0xFFFD;
          if (t2 >= 0x40 && t2 <= 0xFE)
            cp = SRJ_GBK_TABLE[(b - 0x81) * 191 + (t2 - 0x40)];
          if (cp == 0xFFFD && report) {
            if (err_row)
              atomicMin(reinterpret_cast<long long*>(err_row), (long long)row);
            valid = false;
            break;
          }
          n = utf8_encode(cp, tmp);
          i += 2;
        } else {
          if (report) {
            if (err_row)
              atomicMin(reinterpret_cast<long long*>(err_row), (long long)row);
            valid = false;
            break;
          }
          n = utf8_encode(0xFFFD, tmp);
          ++i;
        }
        if (WRITE)
          for (int k = 0; k < n; ++k) chars[o + out_len + k] = tmp[k];
        out_len += n;
      }
    }
    if (WRITE) {
      ballot_write_validity(out_valid, row, valid);
    } else if (in_range) {
      lens[row] = valid ? out_len : 0;
    }
  }
}


// ---------------------------------------------------------------------------
// timezone conversion: binary search pre-expanded transitions
// (reference timezones.cu convert_timestamp_to_utc / _to_timezone)
// ---------------------------------------------------------------------------
__global__ void tz_convert_kernel(const int64_t* __restrict__ in,
                                  const uint8_t* __restrict__ valid, int64_t n,
                                  const int64_t* __restrict__ utc_us,
                                  const int64_t* __restrict__ local_us,
                                  const int64_t* __restrict__ off_sec,
                                  const int32_t* __restrict__ zone_offsets,
                                  int32_t zone_idx, int32_t to_utc,
                                  int64_t* __restrict__ out) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  int32_t s = zone_offsets[zone_idx], e = zone_offsets[zone_idx + 1];
  const int64_t* arr = to_utc ? local_us : utc_us;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    int64_t t = in[i];
    if (!is_valid(valid, i)) { out[i] = t; continue; }
    int32_t lo = s, hi = e - 1;
    while (lo < hi) {  // last index with arr[idx] <= t
      int32_t mid = (lo + hi + 1) >> 1;
      if (arr[mid] <= t) lo = mid;
      else hi = mid - 1;
    }
    int64_t off = off_sec[lo] * 1000000LL;
    out[i] = to_utc ? t - off : t + off;
  }
}

}  // namespace srj

using namespace srj;

extern "C" {

void srj_hllpp_update(const int64_t* hashes, const uint8_t* valid, int64_t n,
                      int32_t precision, int32_t* registers, hipStream_t stream) {
  hllpp_update_kernel<<<grid_1d(n), DEFAULT_BLOCK, 0, stream>>>(
      hashes, valid, n, precision, registers);
}
void srj_hllpp_merge(const int32_t* src, int32_t* dst, int64_t nregs,
                     hipStream_t stream) {
  hllpp_merge_kernel<<<grid_1d(nregs), DEFAULT_BLOCK, 0, stream>>>(src, dst,
                                                                   nregs);
}
void srj_hllpp_pack(const int32_t* regs, int64_t nregs, int64_t* longs,
                    hipStream_t stream) {
  int64_t nlongs = (nregs + 9) / 10;
  hllpp_pack_kernel<<<grid_1d(nlongs), DEFAULT_BLOCK, 0, stream>>>(regs, nregs,
                                                                   longs);
}
void srj_hllpp_unpack(const int64_t* longs, int64_t nregs, int32_t* regs,
                      hipStream_t stream) {
  hllpp_unpack_kernel<<<grid_1d(nregs), DEFAULT_BLOCK, 0, stream>>>(longs, nregs,
                                                                    regs);
}

void srj_percentile_from_histogram(const int32_t* offsets, const double* values,
                                   const int64_t* freqs, int64_t nrows,
                                   const double* percentages, int32_t npct,
                                   double* out, uint8_t* out_valid,
                                   hipStream_t stream) {
  percentile_kernel<<<grid_1d(nrows * npct), DEFAULT_BLOCK, 0, stream>>>(
      offsets, values, freqs, nrows, percentages, npct, out, out_valid);
}

void srj_conv(const void* in, int64_t nrows, int32_t from_base, int32_t to_base,
              int32_t phase, int32_t* lens, const int32_t* offsets, char* chars,
              uint8_t* out_valid, hipStream_t stream) {
  ColDesc c = *reinterpret_cast<const ColDesc*>(in);
  if (phase == 0)
    conv_kernel<false><<<grid_1d(nrows), DEFAULT_BLOCK, 0, stream>>>(
        c, nrows, from_base, to_base, lens, nullptr, nullptr, nullptr);
  else
    conv_kernel<true><<<grid_1d(nrows), DEFAULT_BLOCK, 0, stream>>>(
        c, nrows, from_base, to_base, nullptr, offsets, chars, out_valid);
}

void srj_parse_uri(const void* in, int64_t nrows, int32_t part, const char* qkey,
                   int32_t qkey_len, int32_t phase, int32_t* lens,
                   const int32_t* offsets, char* chars, uint8_t* out_valid,
                   hipStream_t stream) {
  ColDesc c = *reinterpret_cast<const ColDesc*>(in);
  if (phase == 0)
    parse_uri_kernel<false><<<grid_1d(nrows), DEFAULT_BLOCK, 0, stream>>>(
        c, nrows, part, qkey, qkey_len, lens, nullptr, nullptr, nullptr);
  else
    parse_uri_kernel<true><<<grid_1d(nrows), DEFAULT_BLOCK, 0, stream>>>(
        c, nrows, part, qkey, qkey_len, nullptr, offsets, chars, out_valid);
}

void srj_gbk_decode(const void* in, int64_t nrows, int32_t report, int32_t phase,
                    int32_t* lens, const int32_t* offsets, char* chars,
                    uint8_t* out_valid, int64_t* err_row, hipStream_t stream) {
  ColDesc c = *reinterpret_cast<const ColDesc*>(in);
  if (phase == 0)
    gbk_decode_kernel<false><<<grid_1d(nrows), DEFAULT_BLOCK, 0, stream>>>(
        c, nrows, report, lens, nullptr, nullptr, nullptr, err_row);
  else
    gbk_decode_kernel<true><<<grid_1d(nrows), DEFAULT_BLOCK, 0, stream>>>(
        c, nrows, report, nullptr, offsets, chars, out_valid, err_row);
}

void srj_tz_convert(const int64_t* in, const uint8_t* valid, int64_t n,
                    const int64_t* utc_us, const int64_t* local_us,
                    const int64_t* off_sec, const int32_t* zone_offsets,
                    int32_t zone_idx, int32_t to_utc, int64_t* out,
                    hipStream_t stream) {
  tz_convert_kernel<<<grid_1d(n), DEFAULT_BLOCK, 0, stream>>>(
      in, valid, n, utc_us, local_us, off_sec, zone_offsets, zone_idx, to_utc,
      out);
}

}  // extern "C"
