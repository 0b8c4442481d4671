// Misc Spark-exact ops, one kernel family each:
//   case_when select_first_true_index   (reference case_when.cu)
//   bloom filter build/probe            (reference bloom_filter.cu — Spark
//                                        V1/V2 formats, BE headers, the
//                                        (word^1, bit^0x18) endian swizzle)
//   zorder interleave_bits + hilbert    (reference zorder.cu, Skilling/Moten)
//   hex / uuid / substring_index        (reference hex.cu, uuid.cu,
//                                        substring_index.cu)
//   literal_range_pattern               (reference regex_rewrite_utils.cu)
//   Aggregation64Utils chunk ops        (reference aggregation64_utils.cu)
//   ANSI overflow multiply              (reference multiply.cu +
//                                        exception_with_row_index)
//   datetime rebase + date_trunc        (reference datetime_rebase.cu,
//                                        datetime_truncate.cu)
#include "srj_common.hpp"

namespace srj {

// ---------------------------------------------------------------------------
// case_when: index of first true bool column, else num_columns
// ---------------------------------------------------------------------------
__global__ void select_first_true_kernel(const ColDesc* __restrict__ cols,
                                         const int32_t* __restrict__ top,
                                         int32_t ncols, int64_t nrows,
                                         int32_t* __restrict__ out) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t row = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; row < nrows;
       row += stride) {
    int32_t sel = ncols;
    for (int32_t c = 0; c < ncols; ++c) {
      const ColDesc& d = cols[top[c]];
      if (is_valid(d.valid, row) &&
          reinterpret_cast<const int8_t*>(d.data)[row] != 0) {
        sel = c;
        break;
      }
    }
    out[row] = sel;
  }
}

// ---------------------------------------------------------------------------
// bloom filter (Spark-compatible)
// ---------------------------------------------------------------------------
__device__ inline void bloom_bit_pos(int64_t bit_pos, int64_t& word_index,
                                     uint32_t& mask) {
  // buffer holds BE longs; as LE int32 words: swizzle word and bit
  word_index = (bit_pos / 32) ^ 0x1;
  mask = 1u << ((int32_t)(bit_pos % 32) ^ 0x18);
}

template <int VERSION, bool PROBE>
__global__ void bloom_filter_kernel(uint32_t* __restrict__ bits,
                                    int64_t filter_bits,
                                    const int64_t* __restrict__ input,
                                    const uint8_t* __restrict__ valid,
                                    int64_t nrows, int32_t num_hashes,
                                    int32_t seed, uint8_t* __restrict__ out,
                                    uint8_t* __restrict__ out_valid) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  int64_t npad = (nrows + WAVE - 1) & ~(int64_t)(WAVE - 1);
  for (int64_t row = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; row < npad;
       row += stride) {
    bool in_range = row < nrows;
    bool v = in_range && is_valid(valid, row);
    bool found = true;
    if (v) {
      int64_t el = input[row];
      int32_t hash_seed = VERSION == 1 ? 0 : seed;
      int32_t h1 = (int32_t)mm3_hash_long(el, (uint32_t)hash_seed);
      int32_t h2 = (int32_t)mm3_hash_long(el, (uint32_t)h1);
      if (VERSION == 1) {
        for (int32_t i = 1; i <= num_hashes; ++i) {
          int32_t combined = h1 + i * h2;
          int64_t pos = (int64_t)(combined < 0 ? ~combined : combined) %
                        filter_bits;
          int64_t w; uint32_t m;
          bloom_bit_pos(pos, w, m);
          if (PROBE) {
            if ((bits[w] & m) == 0) { found = false; break; }
          } else {
            atomicOr(bits + w, m);
          }
        }
      } else {
        int64_t combined = (int64_t)h1 * 0x7FFFFFFFLL;
        for (int32_t i = 0; i < num_hashes; ++i) {
          combined += h2;
          int64_t ci = combined < 0 ? ~combined : combined;
          int64_t pos = ci % filter_bits;
          int64_t w; uint32_t m;
          bloom_bit_pos(pos, w, m);
          if (PROBE) {
            if ((bits[w] & m) == 0) { found = false; break; }
          } else {
            atomicOr(bits + w, m);
          }
        }
      }
    }
    if (PROBE && in_range) {
      out[row] = v && found;
      // Spark bloom_filter_might_contain(null) -> null
    }
    if (PROBE) ballot_write_validity(out_valid, row, v);
  }
}

__global__ void bitwise_or_kernel(const uint32_t* __restrict__ src,
                                  uint32_t* __restrict__ dst, int64_t nwords) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < nwords;
       i += stride)
    dst[i] |= src[i];
}

// ---------------------------------------------------------------------------
// zorder: interleave bits (column 0 most significant), per-row fixed bytes
// ---------------------------------------------------------------------------
__global__ void interleave_bits_kernel(const ColDesc* __restrict__ cols,
                                       const int32_t* __restrict__ top,
                                       int32_t ncols, int32_t width,
                                       int64_t nrows, uint8_t* __restrict__ out) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  int32_t total_bytes = ncols * width;
  for (int64_t row = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; row < nrows;
       row += stride) {
    uint8_t* r = out + row * total_bytes;
    int32_t nbits = width * 8;
    // out bit j (0 = MSB of first byte): column j%ncols, bit j/ncols from MSB
    for (int32_t byte = 0; byte < total_bytes; ++byte) {
      uint8_t b = 0;
      for (int32_t k = 0; k < 8; ++k) {
        int32_t j = byte * 8 + k;
        int32_t c = j % ncols;
        int32_t bit_from_msb = j / ncols;
        const ColDesc& d = cols[top[c]];
        uint64_t val = 0;
        if (is_valid(d.valid, row)) {
          switch (width) {
            case 1: val = (uint8_t)reinterpret_cast<const int8_t*>(d.data)[row]; break;
            case 2: val = (uint16_t)reinterpret_cast<const int16_t*>(d.data)[row]; break;
            case 4: val = (uint32_t)reinterpret_cast<const int32_t*>(d.data)[row]; break;
            default: val = (uint64_t)reinterpret_cast<const int64_t*>(d.data)[row];
          }
        }
        int bit = (val >> (nbits - 1 - bit_from_msb)) & 1;
        b |= bit << (7 - k);
      }
      r[byte] = b;
    }
  }
}

// hilbert index (Skilling transform, matching the Moten library the
// reference credits at zorder.cu:65)
__global__ void hilbert_index_kernel(const ColDesc* __restrict__ cols,
                                     const int32_t* __restrict__ top,
                                     int32_t ncols, int32_t nbits, int64_t nrows,
                                     int64_t* __restrict__ out) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t row = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; row < nrows;
       row += stride) {
    uint32_t X[10];
    for (int32_t c = 0; c < ncols && c < 10; ++c) {
      const ColDesc& d = cols[top[c]];
      X[c] = is_valid(d.valid, row)
                 ? (uint32_t)reinterpret_cast<const int32_t*>(d.data)[row]
                 : 0u;
    }
    uint32_t M = 1u << (nbits - 1);
    // inverse undo
    for (uint32_t Q = M; Q > 1; Q >>= 1) {
      uint32_t P = Q - 1;
      for (int32_t i = 0; i < ncols; ++i) {
        if (X[i] & Q) {
          X[0] ^= P;
        } else {
          uint32_t t = (X[0] ^ X[i]) & P;
          X[0] ^= t;
          X[i] ^= t;
        }
      }
    }
    // gray encode
    for (int32_t i = 1; i < ncols; ++i) X[i] ^= X[i - 1];
    uint32_t t = 0;
    for (uint32_t Q = M; Q > 1; Q >>= 1)
      if (X[ncols - 1] & Q) t ^= Q - 1;
    for (int32_t i = 0; i < ncols; ++i) X[i] ^= t;
    // transpose -> index, MSB first
    uint64_t idx = 0;
    for (int32_t b = nbits - 1; b >= 0; --b)
      for (int32_t j = 0; j < ncols; ++j)
        idx = (idx << 1) | ((X[j] >> b) & 1);
    out[row] = (int64_t)idx;
  }
}

// ---------------------------------------------------------------------------
// hex (Spark hex(): uppercase, no padding for numbers; for binary input:
// two uppercase hex digits per byte)
// ---------------------------------------------------------------------------
template <bool WRITE>
__global__ void bytes_to_hex_kernel(ColDesc in, int64_t nrows,
                                    int32_t* __restrict__ lens,
                                    const int32_t* __restrict__ offsets,
                                    char* __restrict__ chars,
                                    uint8_t* __restrict__ out_valid) {
  const char* digits = "0123456789ABCDEF";
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  int64_t npad = (nrows + WAVE - 1) & ~(int64_t)(WAVE - 1);
  for (int64_t row = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; row < npad;
       row += stride) {
    bool in_range = row < nrows;
    bool valid = in_range && is_valid(in.valid, row);
    if (!WRITE) {
      if (in_range) {
        StrView s = valid ? get_string(in, row) : StrView{nullptr, 0};
        lens[row] = s.len * 2;
      }
    } else {
      if (valid) {
        StrView s = get_string(in, row);
        int32_t o = offsets[row];
        for (int32_t i = 0; i < s.len; ++i) {
          uint8_t b = (uint8_t)s.ptr[i];
          chars[o + 2 * i] = digits[b >> 4];
          chars[o + 2 * i + 1] = digits[b & 15];
        }
      }
      ballot_write_validity(out_valid, row, valid);
    }
  }
}

// ---------------------------------------------------------------------------
// uuid: random v4 UUID strings, 36 chars (reference uuid.cu:92)
// ---------------------------------------------------------------------------
__global__ void uuid_kernel(int64_t nrows, uint64_t seed,
                            char* __restrict__ chars) {
  const char* digits = "0123456789abcdef";
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t row = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; row < nrows;
       row += stride) {
    uint64_t hi = mix64(seed ^ (uint64_t)row * 0x9E3779B97F4A7C15ull);
    uint64_t lo = mix64(hi ^ 0xD1B54A32D192ED03ull);
    // set version 4 + variant bits
    hi = (hi & 0xFFFFFFFFFFFF0FFFull) | 0x0000000000004000ull;
    lo = (lo & 0x3FFFFFFFFFFFFFFFull) | 0x8000000000000000ull;
    char* r = chars + row * 36;
    int p = 0;
    for (int i = 0; i < 16; ++i) {
      uint64_t w = i < 8 ? hi : lo;
      int shift = 56 - 8 * (i & 7);
      uint8_t b = (uint8_t)(w >> shift);
      if (i == 4 || i == 6 || i == 8 || i == 10) r[p++] = '-';
      r[p++] = digits[b >> 4];
      r[p++] = digits[b & 15];
    }
  }
}

// ---------------------------------------------------------------------------
// substring_index (Spark semantics)
// ---------------------------------------------------------------------------
template <bool WRITE>
__global__ void substring_index_kernel(ColDesc in, const char* __restrict__ delim,
                                       int32_t delim_len, int32_t count,
                                       int64_t nrows, int32_t* __restrict__ lens,
                                       const int32_t* __restrict__ offsets,
                                       char* __restrict__ chars,
                                       uint8_t* __restrict__ out_valid) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  int64_t npad = (nrows + WAVE - 1) & ~(int64_t)(WAVE - 1);
  for (int64_t row = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; row < npad;
       row += stride) {
    bool in_range = row < nrows;
    bool valid = in_range && is_valid(in.valid, row);
    int32_t out_start = 0, out_len = 0;
    if (valid) {
      StrView s = get_string(in, row);
      if (delim_len == 0 || count == 0) {
        out_len = 0;
      } else if (count > 0) {
        int found = 0;
        int32_t pos = -1;
        for (int32_t i = 0; i + delim_len <= s.len; ++i) {
          bool m = true;
          for (int32_t k = 0; k < delim_len; ++k)
            if (s.ptr[i + k] != delim[k]) { m = false; break; }
          if (m) {
            ++found;
            if (found == count) { pos = i; break; }
            i += delim_len - 1;
          }
        }
        out_start = 0;
        out_len = pos >= 0 ? pos : s.len;
      } else {
        // from the end: count occurrences right-to-left
        int want = -count;
        int found = 0;
        int32_t pos = -1;
        for (int32_t i = s.len - delim_len; i >= 0; --i) {
          bool m = true;
          for (int32_t k = 0; k < delim_len; ++k)
            if (s.ptr[i + k] != delim[k]) { m = false; break; }
          if (m) {
            ++found;
            if (found == want) { pos = i; break; }
            i -= delim_len - 1;
          }
        }
        if (pos >= 0) {
          out_start = pos + delim_len;
          out_len = s.len - out_start;
        } else {
          out_start = 0;
          out_len = s.len;
        }
      }
    }
    if (!WRITE) {
      if (in_range) lens[row] = out_len;
    } else {
      if (valid && out_len > 0) {
        StrView s = get_string(in, row);
        int32_t o = offsets[row];
        for (int32_t k = 0; k < out_len; ++k) chars[o + k] = s.ptr[out_start + k];
      }
      ballot_write_validity(out_valid, row, valid);
    }
  }
}

// ---------------------------------------------------------------------------
// literal_range_pattern: does the string contain <literal><len chars in
// [range_start..range_end]> (regex-free rewrite of %lit[a-b]{n}% patterns)
// ---------------------------------------------------------------------------
// UTF-8 codepoint decode (advances i past the sequence); malformed bytes
// decode as U+FFFD one byte at a time
__device__ inline uint32_t lr_utf8_decode(const char* p, int32_t len,
                                          int32_t& i) {
  uint8_t c0 = (uint8_t)p[i];
  if (c0 < 0x80) { i += 1; return c0; }
  if ((c0 >> 5) == 0x6 && i + 1 < len) {
    uint32_t v = ((uint32_t)(c0 & 0x1F) << 6) | ((uint8_t)p[i + 1] & 0x3F);
    i += 2; return v;
  }
  if ((c0 >> 4) == 0xE && i + 2 < len) {
    uint32_t v = ((uint32_t)(c0 & 0x0F) << 12) |
                 (((uint8_t)p[i + 1] & 0x3F) << 6) |
                 ((uint8_t)p[i + 2] & 0x3F);
    i += 3; return v;
  }
  if ((c0 >> 3) == 0x1E && i + 3 < len) {
    uint32_t v = ((uint32_t)(c0 & 0x07) << 18) |
                 (((uint8_t)p[i + 1] & 0x3F) << 12) |
                 (((uint8_t)p[i + 2] & 0x3F) << 6) |
                 ((uint8_t)p[i + 3] & 0x3F);
    i += 4; return v;
  }
  i += 1;
  return 0xFFFD;
}

__global__ void literal_range_kernel(ColDesc in, const char* __restrict__ lit,
                                     int32_t lit_len, int32_t range_len,
                                     uint32_t range_start, uint32_t range_end,
                                     int64_t nrows, uint8_t* __restrict__ out,
                                     uint8_t* __restrict__ out_valid) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  int64_t npad = (nrows + WAVE - 1) & ~(int64_t)(WAVE - 1);
  for (int64_t row = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; row < npad;
       row += stride) {
    bool in_range = row < nrows;
    bool valid = in_range && is_valid(in.valid, row);
    bool found = false;
    if (valid) {
      // codepoint semantics (reference regex_rewrite_utils.cu operates on
      // cudf's codepoint iterator): the range chars after the literal are
      // CODEPOINTS in [range_start..range_end] — e.g. a CJK class — and
      // candidate starts advance codepoint-wise
      StrView s = get_string(in, row);
      int32_t i = 0;
      while (i + lit_len <= s.len && !found) {
        bool m = true;
        for (int32_t k = 0; k < lit_len && m; ++k)
          if (s.ptr[i + k] != lit[k]) m = false;
        if (m) {
          int32_t j = i + lit_len;
          for (int32_t k = 0; k < range_len && m; ++k) {
            if (j >= s.len) { m = false; break; }
            uint32_t cp = lr_utf8_decode(s.ptr, s.len, j);
            if (cp < range_start || cp > range_end) m = false;
          }
          found = m;
        }
        int32_t adv = i;
        lr_utf8_decode(s.ptr, s.len, adv);
        i = adv;
      }
    }
    if (in_range) out[row] = found;
    ballot_write_validity(out_valid, row, valid);
  }
}

// ---------------------------------------------------------------------------
// Aggregation64Utils (reference Aggregation64Utils.java:42-61)
// ---------------------------------------------------------------------------
__global__ void extract_chunk32_kernel(const int64_t* __restrict__ in,
                                       const uint8_t* __restrict__ valid,
                                       int64_t nrows, int32_t chunk,
                                       int64_t* __restrict__ out) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < nrows;
       i += stride) {
    uint64_t v = (uint64_t)in[i];
    int64_t c = (int64_t)((v >> (32 * chunk)) & 0xFFFFFFFFull);
    // sign-extend the high chunk so negative sums decompose correctly
    if (chunk == 1) c = (int64_t)(int32_t)c;
    out[i] = is_valid(valid, i) ? c : 0;
  }
}

__global__ void combine_chunks_kernel(const int64_t* __restrict__ lo,
                                      const int64_t* __restrict__ hi,
                                      int64_t nrows, int64_t* __restrict__ out,
                                      uint8_t* __restrict__ overflow) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < nrows;
       i += stride) {
    __int128 total = (__int128)lo[i] + ((__int128)hi[i] << 32);
    out[i] = (int64_t)total;
    overflow[i] = total > (__int128)0x7FFFFFFFFFFFFFFFLL ||
                  total < -((__int128)1 << 63);
  }
}

// ---------------------------------------------------------------------------
// ANSI overflow multiply (reference multiply.cu + exception_with_row_index)
// ---------------------------------------------------------------------------
__global__ void multiply_i64_kernel(const int64_t* __restrict__ a,
                                    const uint8_t* __restrict__ va,
                                    const int64_t* __restrict__ b,
                                    const uint8_t* __restrict__ vb,
                                    int64_t nrows, int64_t* __restrict__ out,
                                    uint8_t* __restrict__ out_valid,
                                    int64_t* __restrict__ err_row) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  int64_t npad = (nrows + WAVE - 1) & ~(int64_t)(WAVE - 1);
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < npad;
       i += stride) {
    bool in_range = i < nrows;
    bool valid = in_range && is_valid(va, i) && is_valid(vb, i);
    int64_t r = 0;
    if (valid) {
      __int128 p = (__int128)a[i] * b[i];
      r = (int64_t)p;
      if (p != (__int128)r) {
        valid = false;
        if (err_row)
          atomicMin(reinterpret_cast<long long*>(err_row), (long long)i);
      }
    }
    if (in_range) out[i] = r;
    ballot_write_validity(out_valid, i, valid);
  }
}

// ---------------------------------------------------------------------------
// datetime rebase: proleptic Gregorian <-> hybrid Julian (reference
// datetime_rebase.cu). Dates before 1582-10-15 differ; convert via civil
// triple in one calendar, re-serialize in the other.
// ---------------------------------------------------------------------------
__device__ inline void civil_from_days_greg(int64_t z, int& y, int& m, int& d) {
  z += 719468;
  int64_t era = (z >= 0 ? z : z - 146096) / 146097;
  unsigned doe = (unsigned)(z - era * 146097);
  unsigned yoe = (doe - doe / 1460 + doe / 36524 - doe / 146096) / 365;
  int64_t yy = (int64_t)yoe + era * 400;
  unsigned doy = doe - (365 * yoe + yoe / 4 - yoe / 100);
  unsigned mp = (5 * doy + 2) / 153;
  d = (int)(doy - (153 * mp + 2) / 5 + 1);
  m = (int)(mp + (mp < 10 ? 3 : -9));
  y = (int)(yy + (m <= 2));
}

__device__ inline int64_t days_from_civil_julian(int y, int m, int d) {
  // Julian calendar: every 4th year is a leap year. 4-year era = 1461 days;
  // constant verified: julian 1970-01-01 = gregorian day 13, julian
  // 1582-10-05 = gregorian 1582-10-15 (= GREG_START_DAYS).
  y -= m <= 2;
  int64_t era = (y >= 0 ? y : y - 3) / 4;
  int yoe = y - (int)era * 4;
  int doy = (153 * (m + (m > 2 ? -3 : 9)) + 2) / 5 + d - 1;
  int doe = yoe * 365 + yoe / 4 + doy;
  return era * 1461 + doe - 719470;
}

__device__ inline void civil_from_days_julian(int64_t z, int& y, int& m, int& d) {
  z += 719470;
  int64_t era = (z >= 0 ? z : z - 1460) / 1461;
  unsigned doe = (unsigned)(z - era * 1461);
  unsigned yoe = (doe - doe / 1460) / 365;
  int64_t yy = (int64_t)yoe + era * 4;
  unsigned doy = doe - (365 * yoe + yoe / 4);
  unsigned mp = (5 * doy + 2) / 153;
  d = (int)(doy - (153 * mp + 2) / 5 + 1);
  m = (int)(mp + (mp < 10 ? 3 : -9));
  y = (int)(yy + (m <= 2));
}

constexpr int64_t GREG_START_DAYS = -141427;  // 1582-10-15

template <bool TO_JULIAN>
__global__ void rebase_days_kernel(const int32_t* __restrict__ in,
                                   const uint8_t* __restrict__ valid,
                                   int64_t nrows, int32_t* __restrict__ out) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < nrows;
       i += stride) {
    int32_t days = in[i];
    if (is_valid(valid, i)) {
      if (TO_JULIAN) {
        if (days < GREG_START_DAYS) {
          int y, m, d;
          civil_from_days_greg(days, y, m, d);
          days = (int32_t)days_from_civil_julian(y, m, d);
        }
      } else {
        if (days < GREG_START_DAYS) {
          int y, m, d;
          civil_from_days_julian(days, y, m, d);
          // days in hybrid julian < cutover reinterpreted as gregorian
          days = (int32_t)days_from_civil(y, m, d);
        }
      }
    }
    out[i] = days;
  }
}

// date_trunc components (UTC)
enum TruncUnit : int32_t {
  TRUNC_YEAR = 0, TRUNC_QUARTER = 1, TRUNC_MONTH = 2, TRUNC_WEEK = 3,
  TRUNC_DAY = 4, TRUNC_HOUR = 5, TRUNC_MINUTE = 6, TRUNC_SECOND = 7,
  TRUNC_MILLISECOND = 8, TRUNC_MICROSECOND = 9,
};

__global__ void trunc_timestamp_kernel(const int64_t* __restrict__ in,
                                       const uint8_t* __restrict__ valid,
                                       int64_t nrows, int32_t unit,
                                       int64_t* __restrict__ out) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < nrows;
       i += stride) {
    int64_t us = in[i];
    if (is_valid(valid, i)) {
      int64_t day_us = 86400000000LL;
      int64_t days = us >= 0 ? us / day_us : (us - (day_us - 1)) / day_us;
      int64_t tod = us - days * day_us;
      switch (unit) {
        case TRUNC_MICROSECOND: break;
        case TRUNC_MILLISECOND: us = us - ((us % 1000 + 1000) % 1000); break;
        case TRUNC_SECOND: us = us - ((us % 1000000 + 1000000) % 1000000); break;
        case TRUNC_MINUTE: {
          int64_t m = 60000000LL;
          us = us - ((us % m + m) % m);
          break;
        }
        case TRUNC_HOUR: {
          int64_t h = 3600000000LL;
          us = us - ((us % h + h) % h);
          break;
        }
        case TRUNC_DAY: us = days * day_us; break;
        case TRUNC_WEEK: {
          // Monday start; 1970-01-01 was Thursday (dow=3 from Monday)
          int64_t dow = ((days + 3) % 7 + 7) % 7;
          us = (days - dow) * day_us;
          break;
        }
        default: {
          int y, m, d;
          civil_from_days_greg(days, y, m, d);
          if (unit == TRUNC_YEAR) m = 1;
          else if (unit == TRUNC_QUARTER) m = ((m - 1) / 3) * 3 + 1;
          // TRUNC_MONTH keeps m
          us = days_from_civil(y, m, 1) * day_us;
          break;
        }
      }
      (void)tod;
    }
    out[i] = us;
  }
}

}  // namespace srj

using namespace srj;

extern "C" {

void srj_select_first_true(const void* cols, const int32_t* top, int32_t ncols,
                           int64_t nrows, int32_t* out, hipStream_t stream) {
  select_first_true_kernel<<<grid_1d(nrows), DEFAULT_BLOCK, 0, stream>>>(
      reinterpret_cast<const ColDesc*>(cols), top, ncols, nrows, out);
}

void srj_bloom_filter(uint32_t* bits, int64_t filter_bits, const int64_t* input,
                      const uint8_t* valid, int64_t nrows, int32_t num_hashes,
                      int32_t seed, int32_t version, int32_t probe, uint8_t* out,
                      uint8_t* out_valid, hipStream_t stream) {
  int64_t g = grid_1d(nrows);
  if (version == 1) {
    if (probe)
      bloom_filter_kernel<1, true><<<g, DEFAULT_BLOCK, 0, stream>>>(
          bits, filter_bits, input, valid, nrows, num_hashes, seed, out, out_valid);
    else
      bloom_filter_kernel<1, false><<<g, DEFAULT_BLOCK, 0, stream>>>(
          bits, filter_bits, input, valid, nrows, num_hashes, seed, out, out_valid);
  } else {
    if (probe)
      bloom_filter_kernel<2, true><<<g, DEFAULT_BLOCK, 0, stream>>>(
          bits, filter_bits, input, valid, nrows, num_hashes, seed, out, out_valid);
    else
      bloom_filter_kernel<2, false><<<g, DEFAULT_BLOCK, 0, stream>>>(
          bits, filter_bits, input, valid, nrows, num_hashes, seed, out, out_valid);
  }
}

void srj_bitmask_or(const uint32_t* src, uint32_t* dst, int64_t nwords,
                    hipStream_t stream) {
  bitwise_or_kernel<<<grid_1d(nwords), DEFAULT_BLOCK, 0, stream>>>(src, dst,
                                                                   nwords);
}

void srj_interleave_bits(const void* cols, const int32_t* top, int32_t ncols,
                         int32_t width, int64_t nrows, uint8_t* out,
                         hipStream_t stream) {
  interleave_bits_kernel<<<grid_1d(nrows), DEFAULT_BLOCK, 0, stream>>>(
      reinterpret_cast<const ColDesc*>(cols), top, ncols, width, nrows, out);
}

void srj_hilbert_index(const void* cols, const int32_t* top, int32_t ncols,
                       int32_t nbits, int64_t nrows, int64_t* out,
                       hipStream_t stream) {
  hilbert_index_kernel<<<grid_1d(nrows), DEFAULT_BLOCK, 0, stream>>>(
      reinterpret_cast<const ColDesc*>(cols), top, ncols, nbits, nrows, out);
}

void srj_bytes_to_hex(const void* in, int64_t nrows, int32_t phase, int32_t* lens,
                      const int32_t* offsets, char* chars, uint8_t* out_valid,
                      hipStream_t stream) {
  ColDesc c = *reinterpret_cast<const ColDesc*>(in);
  if (phase == 0)
    bytes_to_hex_kernel<false><<<grid_1d(nrows), DEFAULT_BLOCK, 0, stream>>>(
        c, nrows, lens, nullptr, nullptr, nullptr);
  else
    bytes_to_hex_kernel<true><<<grid_1d(nrows), DEFAULT_BLOCK, 0, stream>>>(
        c, nrows, nullptr, offsets, chars, out_valid);
}

void srj_uuid(int64_t nrows, uint64_t seed, char* chars, hipStream_t stream) {
  uuid_kernel<<<grid_1d(nrows), DEFAULT_BLOCK, 0, stream>>>(nrows, seed, chars);
}

void srj_substring_index(const void* in, const char* delim, int32_t delim_len,
                         int32_t count, int64_t nrows, int32_t phase,
                         int32_t* lens, const int32_t* offsets, char* chars,
                         uint8_t* out_valid, hipStream_t stream) {
  ColDesc c = *reinterpret_cast<const ColDesc*>(in);
  if (phase == 0)
    substring_index_kernel<false><<<grid_1d(nrows), DEFAULT_BLOCK, 0, stream>>>(
        c, delim, delim_len, count, nrows, lens, nullptr, nullptr, nullptr);
  else
    substring_index_kernel<true><<<grid_1d(nrows), DEFAULT_BLOCK, 0, stream>>>(
        c, delim, delim_len, count, nrows, nullptr, offsets, chars, out_valid);
}

void srj_literal_range(const void* in, const char* lit, int32_t lit_len,
                       int32_t range_len, uint32_t range_start,
                       uint32_t range_end, int64_t nrows, uint8_t* out,
                       uint8_t* out_valid, hipStream_t stream) {
  literal_range_kernel<<<grid_1d(nrows), DEFAULT_BLOCK, 0, stream>>>(
      *reinterpret_cast<const ColDesc*>(in), lit, lit_len, range_len, range_start,
      range_end, nrows, out, out_valid);
}

void srj_extract_chunk32(const int64_t* in, const uint8_t* valid, int64_t nrows,
                         int32_t chunk, int64_t* out, hipStream_t stream) {
  extract_chunk32_kernel<<<grid_1d(nrows), DEFAULT_BLOCK, 0, stream>>>(
      in, valid, nrows, chunk, out);
}

void srj_combine_chunks(const int64_t* lo, const int64_t* hi, int64_t nrows,
                        int64_t* out, uint8_t* overflow, hipStream_t stream) {
  combine_chunks_kernel<<<grid_1d(nrows), DEFAULT_BLOCK, 0, stream>>>(
      lo, hi, nrows, out, overflow);
}

void srj_multiply_i64(const int64_t* a, const uint8_t* va, const int64_t* b,
                      const uint8_t* vb, int64_t nrows, int64_t* out,
                      uint8_t* out_valid, int64_t* err_row, hipStream_t stream) {
  multiply_i64_kernel<<<grid_1d(nrows), DEFAULT_BLOCK, 0, stream>>>(
      a, va, b, vb, nrows, out, out_valid, err_row);
}

void srj_rebase_days(const int32_t* in, const uint8_t* valid, int64_t nrows,
                     int32_t to_julian, int32_t* out, hipStream_t stream) {
  if (to_julian)
    rebase_days_kernel<true><<<grid_1d(nrows), DEFAULT_BLOCK, 0, stream>>>(
        in, valid, nrows, out);
  else
    rebase_days_kernel<false><<<grid_1d(nrows), DEFAULT_BLOCK, 0, stream>>>(
        in, valid, nrows, out);
}

void srj_trunc_timestamp(const int64_t* in, const uint8_t* valid, int64_t nrows,
                         int32_t unit, int64_t* out, hipStream_t stream) {
  trunc_timestamp_kernel<<<grid_1d(nrows), DEFAULT_BLOCK, 0, stream>>>(
      in, valid, nrows, unit, out);
}

}  // extern "C"
