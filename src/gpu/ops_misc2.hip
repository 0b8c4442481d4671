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

// Full RFC-shaped validation machine with the reference's exact semantics
// (reference parse_uri.cu:87-756): per-chunk character allowlists,
// percent-escape and UTF-8 validation with the unicode-whitespace
// blacklist, IPv4/IPv6/domain host validation (3-state: VALID / host-only
// INVALID / FATAL), userinfo/port splitting with the reference's quirky
// last_colon bookkeeping, opaque vs hierarchical URIs, and fragment
// validation. Bug-compat notes are marked inline.

struct USpan { const char* p = nullptr; int n = 0; };

__device__ inline bool u_alpha(char c) { return (c >= 'a' && c <= 'z') || (c >= 'A' && c <= 'Z'); }
__device__ inline bool u_num(char c) { return c >= '0' && c <= '9'; }
__device__ inline bool u_alnum(char c) { return u_alpha(c) || u_num(c); }
__device__ inline bool u_hex(char c) {
  return u_num(c) || (c >= 'a' && c <= 'f') || (c >= 'A' && c <= 'F');
}

// Validate (and skip) percent escapes and multibyte UTF-8 at position *i.
// Packed-value whitespace blacklist matches the reference's char_utf8
// comparisons (parse_uri.cu:101-140).
__device__ inline bool uri_skip_special(const char* s, int len, int* i,
                                        bool allow_bad_escape) {
  while (*i < len) {
    unsigned char b = (unsigned char)s[*i];
    if (b == '%' && !allow_bad_escape) {
      if (*i + 2 >= len) return false;
      if (!u_hex(s[*i + 1]) || !u_hex(s[*i + 2])) return false;
      *i += 3;
    } else if (b >= 0xC0) {
      int nb = (b & 0xE0) == 0xC0 ? 2 : (b & 0xF0) == 0xE0 ? 3
               : (b & 0xF8) == 0xF0 ? 4 : 1;
      if (nb == 1) return false;  // 0xF8+ invalid lead
      if (*i + nb > len) return false;
      uint32_t packed = b;
      for (int k = 1; k < nb; ++k)
        packed = (packed << 8) | (unsigned char)s[*i + k];
      if (nb > 1 && (packed & 0xC0) != 0x80) return false;
      if (nb > 2 && (packed & 0xC000) != 0x8000) return false;
      if (nb > 3 && (packed & 0xC00000) != 0x800000) return false;
      if ((packed >= 0xc280 && packed <= 0xc2a0) || packed == 0xe19a80 ||
          (packed >= 0xe28080 && packed <= 0xe2808a) || packed == 0xe280af ||
          packed == 0xe280a8 || packed == 0xe2819f || packed == 0xe38080)
        return false;
      *i += nb;
    } else {
      break;  // plain byte (incl. stray continuation): the allowlist decides
    }
  }
  return true;
}

template <typename Pred>
__device__ inline bool uri_chunk_ok(USpan s, Pred pred,
                                    bool allow_bad_escape = false) {
  int i = 0;
  if (!uri_skip_special(s.p, s.n, &i, allow_bad_escape)) return false;
  while (i < s.n) {
    if (!pred(s.p[i])) return false;
    ++i;
    if (!uri_skip_special(s.p, s.n, &i, allow_bad_escape)) return false;
  }
  return true;
}

__device__ inline bool uri_valid_scheme(USpan s) {
  if (s.n <= 0 || !u_alpha(s.p[0])) return false;
  for (int i = 1; i < s.n; ++i) {
    char c = s.p[i];
    if (!u_alnum(c) && c != '+' && c != '-' && c != '.') return false;
  }
  return true;
}

__device__ inline bool uri_valid_ipv6(USpan s) {
  constexpr int max_colons = 8;
  if (s.n < 2) return false;
  bool found_double_colon = false;
  int open_br = 0, close_br = 0, periods = 0, colons = 0, percents = 0;
  char prev = 0;
  int address = 0, addr_chars = 0;
  bool addr_hex = false;
  for (int i = 0; i < s.n; ++i) {
    char c = s.p[i];
    switch (c) {
      case '[':
        if (++open_br > 1) return false;
        break;
      case ']':
        if (++close_br > 1) return false;
        if (periods > 0 && (addr_hex || address > 255)) return false;
        break;
      case ':':
        ++colons;
        if (prev == ':') {
          if (found_double_colon) return false;
          found_double_colon = true;
        }
        address = 0; addr_hex = false; addr_chars = 0;
        if (colons > max_colons ||
            (colons == max_colons && !found_double_colon))
          return false;
        if (periods > 0 || percents > 0) return false;
        break;
      case '.':
        ++periods;
        if (percents > 0) return false;
        if (periods > 3) return false;
        if (addr_hex) return false;
        if (address > 255) return false;
        if (colons != 6 && !found_double_colon) return false;
        if (colons >= max_colons) return false;
        address = 0; addr_hex = false; addr_chars = 0;
        break;
      case '%':
        // zone id suffix (%eth0); anything goes after it
        if (++percents > 1) return false;
        if (periods > 0 && (addr_hex || address > 255)) return false;
        address = 0; addr_hex = false; addr_chars = 0;
        break;
      default:
        if (percents == 0) {
          if (addr_chars > 3) return false;
          ++addr_chars;
          address *= 10;
          if (c >= 'a' && c <= 'f') {
            address += 10 + (c - 'a');
            addr_hex = true;
          } else if (c >= 'A' && c <= 'Z') {
            // reference quirk: A..Z (not just A..F) count as hex digits
            address += 10 + (c - 'A');
            addr_hex = true;
          } else if (u_num(c)) {
            address += c - '0';
          } else {
            return false;
          }
        }
        break;
    }
    prev = c;
  }
  return true;
}

__device__ inline bool uri_valid_ipv4(USpan s) {
  int address = 0, addr_chars = 0, dots = 0;
  for (int i = 0; i < s.n; ++i) {
    char c = s.p[i];
    if (!u_num(c) && (i == 0 || c != '.')) return false;
    if (c == '.') {
      if (addr_chars == 0) return false;
      address = 0; addr_chars = 0; ++dots;
      continue;
    }
    ++addr_chars;
    address = address * 10 + (c - '0');
    if (address > 255) return false;
  }
  if (addr_chars == 0) return false;
  return dots == 3;
}

__device__ inline bool uri_valid_domain(USpan s) {
  bool last_dash = false, last_dot = false, numeric_start = false;
  int chars_before_dot = 0;
  for (int i = 0; i < s.n; ++i) {
    char c = s.p[i];
    if (!u_alnum(c) && c != '-' && c != '.') return false;
    numeric_start = last_dot && u_num(c);
    if (c == '-') {
      if (last_dot || i == 0 || i == s.n - 1) return false;
      last_dash = true;
      last_dot = false;
    } else if (c == '.') {
      if (last_dash || last_dot || chars_before_dot == 0) return false;
      last_dot = true;
      last_dash = false;
      chars_before_dot = 0;
    } else {
      last_dot = false;
      last_dash = false;
      ++chars_before_dot;
    }
  }
  return !numeric_start;
}

enum UriHostValidity : int { UHOST_FATAL = 0, UHOST_INVALID = 1, UHOST_VALID = 2 };

__device__ inline int uri_valid_host(USpan host) {
  if (host.n > 0 && host.p[0] == '[') {
    if (host.p[host.n - 1] != ']') return UHOST_FATAL;
    return uri_valid_ipv6(host) ? UHOST_VALID : UHOST_FATAL;
  }
  int last_open = -1, last_close = -1, last_dot = -1;
  for (int i = 0; i < host.n; ++i) {
    char c = host.p[i];
    if (c == '[') last_open = i;
    else if (c == ']') last_close = i;
    else if (c == '.') last_dot = i;
  }
  if (last_open >= 0 || last_close >= 0) return UHOST_FATAL;
  if (last_dot < 0 || last_dot == host.n - 1 || !u_num(host.p[last_dot + 1])) {
    if (uri_valid_domain(host)) return UHOST_VALID;
  } else if (uri_valid_ipv4(host)) {
    return UHOST_VALID;
  }
  return UHOST_INVALID;
}

struct UriQueryPred {
  __device__ bool operator()(char c) const {
    return c == '!' || c == '"' || c == '$' || (c >= '&' && c <= ';') ||
           c == '=' || (c >= '?' && c <= ']' && c != '\\') ||
           (c >= 'a' && c <= 'z') || c == '_' || c == '~';
  }
};
struct UriAuthorityPred {
  bool allow_bad_escape;
  __device__ bool operator()(char c) const {
    return c == '!' || c == '$' || (c >= '&' && c <= ';' && c != '/') ||
           c == '=' || (c >= '@' && c <= '_' && c != '^' && c != '\\') ||
           (c >= 'a' && c <= 'z') || c == '~' ||
           (allow_bad_escape && c == '%');
  }
};
struct UriUserinfoPred {
  __device__ bool operator()(char c) const { return c != '[' && c != ']'; }
};
struct UriPortPred {
  // bug-compat: the reference's condition (c < '0' && c > '9') can never
  // be true, so ports accept any character
  __device__ bool operator()(char c) const { return true; }
};
struct UriPathPred {
  __device__ bool operator()(char c) const {
    return c == '!' || c == '$' || (c >= '&' && c <= ';') || c == '=' ||
           (c >= '@' && c <= 'Z') || c == '_' || (c >= 'a' && c <= 'z') ||
           c == '~';
  }
};
struct UriOpaqueFragPred {
  __device__ bool operator()(char c) const {
    return c == '!' || c == '$' || (c >= '&' && c <= ';') || c == '=' ||
           (c >= '?' && c <= ']' && c != '\\') || c == '_' || c == '~' ||
           (c >= 'a' && c <= 'z');
  }
};

// query-parameter extraction (reference find_query_part)
__device__ inline bool uri_find_query_part(USpan q, const char* key,
                                           int key_len, USpan* out) {
  const char* h = q.p;
  const char* h_end = q.p + q.n;
  while (h + key_len < h_end) {
    bool match = true;
    for (int j = 0; j < key_len; ++j) {
      if (h[j] != key[j]) { match = false; break; }
    }
    if (match && h[key_len] == '=') {
      h += key_len + 1;
      const char* start = h;
      int n = 0;
      while (h < h_end && *h != '&') { ++n; ++h; }
      *out = {start, n};
      return true;
    }
    while (h + key_len < h_end && *h != '&') ++h;
    ++h;
  }
  return false;
}

struct UriParts {
  USpan scheme, host, authority, path, fragment, query, userinfo, port, opaque;
  uint32_t valid = 0;  // bit per chunk below
};
enum UriChunkBit : int {
  UB_PROTOCOL = 0, UB_HOST = 1, UB_AUTHORITY = 2, UB_PATH = 3,
  UB_FRAGMENT = 4, UB_QUERY = 5, UB_USERINFO = 6, UB_PORT = 7, UB_OPAQUE = 8,
};

__device__ inline UriParts uri_validate(const char* str, int len,
                                        const char* qkey, int qkey_len) {
  UriParts ret;
  const char* original = str;
  int col = -1, slash = -1, hash = -1, question = -1;
  for (int i = 0; i < len &&
       (col == -1 || slash == -1 || hash == -1 || question == -1); ++i) {
    switch (str[i]) {
      case ':': if (col == -1) col = i; break;
      case '/': if (slash == -1) slash = i; break;
      case '#': if (hash == -1) hash = i; break;
      case '?': if (question == -1) question = i; break;
      default: break;
    }
  }
  if (hash >= 0) {
    ret.fragment = {str + hash + 1, len - hash - 1};
    if (!uri_chunk_ok(ret.fragment, UriOpaqueFragPred{})) {
      ret.valid = 0;
      return ret;
    }
    ret.valid |= 1u << UB_FRAGMENT;
    len = hash;
    if (col > hash) col = -1;
    if (slash > hash) slash = -1;
    if (question > hash) question = -1;
  }
  bool const has_scheme = (col != -1) && (slash == -1 || col < slash) &&
                          (hash == -1 || col < hash);
  if (has_scheme) {
    ret.scheme = {str, col};
    if (!uri_valid_scheme(ret.scheme)) {
      ret.valid = 0;
      return ret;
    }
    ret.valid |= 1u << UB_PROTOCOL;
    int const skip = col + 1;
    str += skip;
    len -= skip;
    question -= skip;
    hash -= skip;
    slash -= skip;
  }
  if (len <= 0) {
    // scheme-only is invalid; empty/fragment-only counts as an empty path
    ret.valid = (has_scheme ? 0u : 1u) << UB_PATH;
    return ret;
  }
  bool const hierarchical = str[0] == '/' || str == original;
  if (hierarchical) {
    if (question >= 0) {
      ret.query = {str + question + 1, len - question - 1};
      if (!uri_chunk_ok(ret.query, UriQueryPred{})) {
        ret.valid = 0;
        return ret;
      }
      if (qkey != nullptr) {
        USpan part;
        if (!uri_find_query_part(ret.query, qkey, qkey_len, &part)) {
          ret.valid = 0;
          return ret;
        }
        ret.query = part;
      }
      ret.valid |= 1u << UB_QUERY;
    }
    int const path_len = question >= 0 ? question : len;
    if (len >= 2 && str[0] == '/' && str[1] == '/') {
      int next_slash = -1;
      for (int i = 2; i < path_len; ++i) {
        if (str[i] == '/') { next_slash = i; break; }
      }
      ret.authority = {str + 2,
                       next_slash == -1
                           ? (question < 0 ? len - 2 : question - 2)
                           : next_slash - 2};
      if (next_slash > 0) ret.path = {str + next_slash, path_len - next_slash};
      if (ret.authority.n > 0) {
        bool const ipv6_addr = ret.authority.n > 2 && ret.authority.p[0] == '[';
        if (!uri_chunk_ok(ret.authority, UriAuthorityPred{ipv6_addr},
                          ipv6_addr)) {
          ret.valid = 0;
          return ret;
        }
        ret.valid |= 1u << UB_AUTHORITY;
        const char* auth = ret.authority.p;
        int auth_size = ret.authority.n;
        int amp = -1, closing_bracket = -1, last_colon = -1;
        for (int i = 0; i < auth_size; ++i) {
          switch (auth[i]) {
            case '@':
              if (amp == -1) {
                amp = i;
                if (last_colon > 0) last_colon = -1;
                if (closing_bracket > 0) closing_bracket = -1;
              }
              break;
            case ':': last_colon = amp > 0 ? i - amp - 1 : i; break;
            case ']':
              if (closing_bracket == -1)
                closing_bracket = amp > 0 ? i - amp : i;
              break;
          }
        }
        if (amp > 0) {
          ret.userinfo = {auth, amp};
          if (!uri_chunk_ok(ret.userinfo, UriUserinfoPred{})) {
            ret.valid = 0;
            return ret;
          }
          ret.valid |= 1u << UB_USERINFO;
          ++amp;
          auth += amp;
          auth_size -= amp;
        }
        if (last_colon > 0 && last_colon > closing_bracket) {
          ret.port = {auth + last_colon + 1, auth_size - last_colon - 1};
          if (!uri_chunk_ok(ret.port, UriPortPred{})) {
            ret.valid = 0;
            return ret;
          }
          ret.valid |= 1u << UB_PORT;
          ret.host = {auth, last_colon};
        } else {
          ret.host = {auth, auth_size};
        }
        switch (uri_valid_host(ret.host)) {
          case UHOST_FATAL: ret.valid = 0; return ret;
          case UHOST_INVALID: ret.host = {nullptr, 0}; break;
          case UHOST_VALID: ret.valid |= 1u << UB_HOST; break;
        }
      }
    } else {
      ret.path = {str, path_len};
    }
    if (!uri_chunk_ok(ret.path, UriPathPred{})) {
      ret.valid = 0;
      return ret;
    }
    ret.valid |= 1u << UB_PATH;
  } else {
    ret.opaque = {str, len};
    if (!uri_chunk_ok(ret.opaque, UriOpaqueFragPred{})) {
      ret.valid = 0;
      return ret;
    }
    ret.valid |= 1u << UB_OPAQUE;
  }
  return ret;
}

template <bool WRITE>
__global__ void parse_uri_kernel(ColDesc in, int64_t nrows, int32_t part,
                                 const char* __restrict__ qkey, int32_t qkey_len,
                                 ColDesc qcol, int32_t has_qcol,
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
      bool want_key = part == URI_QUERY_KEY;
      const char* key = qkey;
      int32_t key_len = qkey_len;
      if (want_key && has_qcol) {
        // per-row key column (reference parseURIQueryWithColumn; a null
        // key row yields a null result)
        if (!is_valid(qcol.valid, row)) {
          valid = false;
          key = nullptr;
        } else {
          StrView k = get_string(qcol, row);
          key = k.ptr;
          key_len = k.len;
        }
      }
      UriParts parts{};
      if (valid)
        parts = uri_validate(s.ptr, s.len, want_key ? key : nullptr,
                             key_len);
      if (!valid) parts.valid = 0;
      USpan sel{nullptr, 0};
      int bit = -1;
      switch (part) {
        case URI_PROTOCOL: sel = parts.scheme; bit = UB_PROTOCOL; break;
        case URI_HOST: sel = parts.host; bit = UB_HOST; break;
        case URI_PATH: sel = parts.path; bit = UB_PATH; break;
        case URI_QUERY:
        case URI_QUERY_KEY: sel = parts.query; bit = UB_QUERY; break;
      }
      if (bit < 0 || !(parts.valid & (1u << bit))) {
        valid = false;
      } else {
        out_p = sel.p;
        out_n = sel.n;
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
                   int32_t qkey_len, const void* qcol, int32_t phase,
                   int32_t* lens, const int32_t* offsets, char* chars,
                   uint8_t* out_valid, hipStream_t stream) {
  ColDesc c = *reinterpret_cast<const ColDesc*>(in);
  ColDesc kc{};
  if (qcol) kc = *reinterpret_cast<const ColDesc*>(qcol);
  if (phase == 0)
    parse_uri_kernel<false><<<grid_1d(nrows), DEFAULT_BLOCK, 0, stream>>>(
        c, nrows, part, qkey, qkey_len, kc, qcol ? 1 : 0, lens, nullptr,
        nullptr, nullptr);
  else
    parse_uri_kernel<true><<<grid_1d(nrows), DEFAULT_BLOCK, 0, stream>>>(
        c, nrows, part, qkey, qkey_len, kc, qcol ? 1 : 0, nullptr, offsets,
        chars, out_valid);
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
