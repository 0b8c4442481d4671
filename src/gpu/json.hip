// JSON ops: device pull parser + get_json_object (JSONPath) + from_json.
//
// Reference parity: json_parser.cuh (1,710 LoC device pull parser),
// get_json_object.cu (1,262 — path instructions with named keys, indices and
// [*] wildcards; Hive/Spark semantics), from_json_to_structs.cu,
// from_json_to_raw_map.cu. Fresh MI355X implementation:
//   * one thread per row (JSON docs are row-sized; wave64 validity writes)
//   * two-phase string output (sizes -> cumsum -> write)
//   * Spark result semantics: single string match -> unescaped value;
//     container match -> raw JSON span; multiple wildcard matches ->
//     "[m1,m2,...]"; missing/invalid -> null.
#include "srj_common.hpp"
#include "ryu_tables.inc"

namespace srj {

#include "fp_parse.inc"
#include "ryu_format.inc"

constexpr int JSON_MAX_DEPTH = 64;   // reference json_parser max nesting
constexpr int MAX_PATH_DEPTH = 16;   // reference JSONUtils.MAX_PATH_DEPTH

struct JsonSpan {
  const char* p;
  int32_t len;
};

__device__ inline const char* j_skip_ws(const char* p, const char* e) {
  while (p < e && (*p == ' ' || *p == '\t' || *p == '\n' || *p == '\r')) ++p;
  return p;
}

// returns end of string token (after closing quote), or null on error;
// accepts " and ' quoting (Spark's lenient parser)
__device__ inline const char* j_skip_string(const char* p, const char* e) {
  // p points at opening quote
  char q = *p;
  ++p;
  while (p < e) {
    if (*p == '\\') {
      p += 2;
    } else if (*p == q) {
      return p + 1;
    } else {
      ++p;
    }
  }
  return nullptr;
}

// skip one JSON value starting at p; returns its end or null
__device__ const char* j_skip_value(const char* p, const char* e) {
  int depth = 0;
  p = j_skip_ws(p, e);
  if (p >= e) return nullptr;
  do {
    if (p >= e) return nullptr;
    char c = *p;
    if (c == '"' || c == '\'') {
      p = j_skip_string(p, e);
      if (!p) return nullptr;
    } else if (c == '{' || c == '[') {
      ++depth;
      if (depth > JSON_MAX_DEPTH) return nullptr;
      ++p;
    } else if (c == '}' || c == ']') {
      --depth;
      if (depth < 0) return nullptr;
      ++p;
    } else if (c == ',' || c == ':') {
      ++p;
    } else {
      // literal: number/true/false/null
      while (p < e && *p != ',' && *p != '}' && *p != ']' && *p != ' ' &&
             *p != '\t' && *p != '\n' && *p != '\r')
        ++p;
    }
    p = j_skip_ws(p, e);
  } while (depth > 0);
  return p;
}

__device__ inline bool j_hex4(const char* p, const char* e, uint32_t* out) {
  if (p + 4 > e) return false;
  uint32_t v = 0;
  for (int k = 0; k < 4; ++k) {
    char h = p[k];
    uint32_t d;
    if (h >= '0' && h <= '9') d = h - '0';
    else if ((h | 32) >= 'a' && (h | 32) <= 'f') d = (h | 32) - 'a' + 10;
    else return false;
    v = v * 16 + d;
  }
  *out = v;
  return true;
}

// compare an escaped key body (between quotes) against a plain UTF-8 path
// key (reference unescapes names before matching —
// GetJsonObjectTest_NamesWithEscapedCharacters)
__device__ inline bool j_key_equals(const char* kbody, int32_t klen,
                                    const char* key, int32_t key_len) {
  int32_t i = 0, j = 0;
  while (i < klen) {
    char outbuf[4];
    int nout = 0;
    char c = kbody[i];
    if (c == '\\' && i + 1 < klen) {
      char x = kbody[i + 1];
      i += 2;
      switch (x) {
        case 'n': outbuf[nout++] = '\n'; break;
        case 't': outbuf[nout++] = '\t'; break;
        case 'r': outbuf[nout++] = '\r'; break;
        case 'b': outbuf[nout++] = '\b'; break;
        case 'f': outbuf[nout++] = '\f'; break;
        case 'u': {
          uint32_t cp;
          if (!j_hex4(kbody + i, kbody + klen, &cp)) return false;
          i += 4;
          if (cp >= 0xD800 && cp < 0xDC00 && i + 6 <= klen &&
              kbody[i] == '\\' && kbody[i + 1] == 'u') {
            uint32_t lo;
            if (j_hex4(kbody + i + 2, kbody + klen, &lo) && lo >= 0xDC00 &&
                lo < 0xE000) {
              cp = 0x10000 + ((cp - 0xD800) << 10) + (lo - 0xDC00);
              i += 6;
            }
          }
          if (cp < 0x80) {
            outbuf[nout++] = (char)cp;
          } else if (cp < 0x800) {
            outbuf[nout++] = (char)(0xC0 | (cp >> 6));
            outbuf[nout++] = (char)(0x80 | (cp & 0x3F));
          } else if (cp < 0x10000) {
            outbuf[nout++] = (char)(0xE0 | (cp >> 12));
            outbuf[nout++] = (char)(0x80 | ((cp >> 6) & 0x3F));
            outbuf[nout++] = (char)(0x80 | (cp & 0x3F));
          } else {
            outbuf[nout++] = (char)(0xF0 | (cp >> 18));
            outbuf[nout++] = (char)(0x80 | ((cp >> 12) & 0x3F));
            outbuf[nout++] = (char)(0x80 | ((cp >> 6) & 0x3F));
            outbuf[nout++] = (char)(0x80 | (cp & 0x3F));
          }
          break;
        }
        default: outbuf[nout++] = x;
      }
    } else {
      outbuf[nout++] = c;
      ++i;
    }
    for (int k = 0; k < nout; ++k) {
      if (j >= key_len || key[j] != outbuf[k]) return false;
      ++j;
    }
  }
  return j == key_len;
}

// path instruction: kind 0 = key, 1 = index, 2 = wildcard ([*] or .*)
struct PathInstr {
  int32_t kind;
  int32_t key_off;   // into shared key chars
  int32_t key_len;
  int32_t index;
};

// output sink: counts in phase 0, writes in phase 1
struct Sink {
  char* buf;
  int32_t pos;
  bool write;

  __device__ void put(char c) {
    if (write) buf[pos] = c;
    ++pos;
  }
  __device__ void span(const char* p, int32_t n) {
    if (write)
      for (int32_t i = 0; i < n; ++i) buf[pos + i] = p[i];
    pos += n;
  }
  // unescape a string token body (between quotes)
  __device__ void unescaped(const char* p, int32_t n) {
    for (int32_t i = 0; i < n; ++i) {
      char c = p[i];
      if (c == '\\' && i + 1 < n) {
        ++i;
        char x = p[i];
        switch (x) {
          case 'n': put('\n'); break;
          case 't': put('\t'); break;
          case 'r': put('\r'); break;
          case 'b': put('\b'); break;
          case 'f': put('\f'); break;
          case '/': put('/'); break;
          case '"': put('"'); break;
          case '\\': put('\\'); break;
          case 'u': {
            if (i + 4 < n) {
              uint32_t cp = 0;
              for (int k = 1; k <= 4; ++k) {
                char h = p[i + k];
                cp = cp * 16 + (h <= '9' ? h - '0' : ((h | 32) - 'a' + 10));
              }
              i += 4;
              if (cp < 0x80) {
                put((char)cp);
              } else if (cp < 0x800) {
                put((char)(0xC0 | (cp >> 6)));
                put((char)(0x80 | (cp & 0x3F)));
              } else {
                put((char)(0xE0 | (cp >> 12)));
                put((char)(0x80 | ((cp >> 6) & 0x3F)));
                put((char)(0x80 | (cp & 0x3F)));
              }
            }
            break;
          }
          default: put(x);
        }
      } else {
        put(c);
      }
    }
  }
};

struct MatchCtx {
  const PathInstr* instrs;
  const char* keychars;
  int32_t ninstr;
  int32_t nmatches;
  JsonSpan matches[8];   // first few match spans (value extents incl quotes)
  bool overflow;
};

// recursive matcher: collect value spans matching instrs[step:] within value
// at p.. Returns false on malformed json.
__device__ bool j_match(const char* p, const char* e, MatchCtx& ctx,
                        int32_t step, int depth) {
  if (depth > MAX_PATH_DEPTH + 2) return false;
  p = j_skip_ws(p, e);
  if (step == ctx.ninstr) {
    const char* vend = j_skip_value(p, e);
    if (!vend) return false;
    if (ctx.nmatches < 8)
      ctx.matches[ctx.nmatches] = {p, (int32_t)(vend - p)};
    else
      ctx.overflow = true;
    ++ctx.nmatches;
    return true;
  }
  if (p >= e) return true;  // no match
  const PathInstr ins = ctx.instrs[step];
  if (*p == '{' && ins.kind == 0) {
    ++p;
    while (true) {
      p = j_skip_ws(p, e);
      if (p < e && *p == '}') return true;
      if (p >= e || (*p != '"' && *p != '\'')) return false;
      const char* kend = j_skip_string(p, e);
      if (!kend) return false;
      const char* kbody = p + 1;
      int32_t klen = (int32_t)(kend - p - 2);
      p = j_skip_ws(kend, e);
      if (p >= e || *p != ':') return false;
      ++p;
      p = j_skip_ws(p, e);
      bool match = j_key_equals(kbody, klen,
                                 ctx.keychars + ins.key_off, ins.key_len);
      if (match) {
        if (!j_match(p, e, ctx, step + 1, depth + 1)) return false;
      }
      const char* vend = j_skip_value(p, e);
      if (!vend) return false;
      p = j_skip_ws(vend, e);
      if (p < e && *p == ',') { ++p; continue; }
      if (p < e && *p == '}') return true;
      return false;
    }
  }
  if (*p == '[' && (ins.kind == 1 || ins.kind == 2)) {
    ++p;
    int32_t idx = 0;
    while (true) {
      p = j_skip_ws(p, e);
      if (p < e && *p == ']') return true;
      if (ins.kind == 2 || idx == ins.index) {
        if (!j_match(p, e, ctx, step + 1, depth + 1)) return false;
      }
      const char* vend = j_skip_value(p, e);
      if (!vend) return false;
      p = j_skip_ws(vend, e);
      ++idx;
      if (p < e && *p == ',') { ++p; continue; }
      if (p < e && *p == ']') return true;
      return false;
    }
  }
  // wildcard over object values (Hive "$.*")
  if (*p == '{' && ins.kind == 2) {
    ++p;
    while (true) {
      p = j_skip_ws(p, e);
      if (p < e && *p == '}') return true;
      if (p >= e || (*p != '"' && *p != '\'')) return false;
      const char* kend = j_skip_string(p, e);
      if (!kend) return false;
      p = j_skip_ws(kend, e);
      if (p >= e || *p != ':') return false;
      ++p;
      p = j_skip_ws(p, e);
      if (!j_match(p, e, ctx, step + 1, depth + 1)) return false;
      const char* vend = j_skip_value(p, e);
      if (!vend) return false;
      p = j_skip_ws(vend, e);
      if (p < e && *p == ',') { ++p; continue; }
      if (p < e && *p == '}') return true;
      return false;
    }
  }
  return true;  // structure mismatch: no match, not an error
}


// ---------------------------------------------------------------------------
// normalized re-serialization: the reference emits results through its JSON
// generator — whitespace stripped, strings double-quoted with canonical
// escapes (\uXXXX decoded to UTF-8, surrogate pairs combined), numbers
// through the exact parse + Java Double.toString pipeline (shared
// fp_parse.inc / ryu_format.inc), single-quoted input accepted, numbers
// with leading zeros rejected (GetJsonObjectTest_Number_Normalization,
// _Escape, _Test_leading_zeros).
// ---------------------------------------------------------------------------

__device__ inline void j_emit_char(Sink& sink, char c, bool as_json) {
  if (!as_json) {
    sink.put(c);
    return;
  }
  switch (c) {
    case '"': sink.put('\\'); sink.put('"'); return;
    case '\\': sink.put('\\'); sink.put('\\'); return;
    case '\n': sink.put('\\'); sink.put('n'); return;
    case '\t': sink.put('\\'); sink.put('t'); return;
    case '\r': sink.put('\\'); sink.put('r'); return;
    case '\b': sink.put('\\'); sink.put('b'); return;
    case '\f': sink.put('\\'); sink.put('f'); return;
    default:
      if ((unsigned char)c < 0x20) {
        const char* hx = "0123456789abcdef";
        sink.put('\\'); sink.put('u'); sink.put('0'); sink.put('0');
        sink.put(hx[((unsigned char)c >> 4) & 15]);
        sink.put(hx[(unsigned char)c & 15]);
      } else {
        sink.put(c);
      }
  }
}



// p at the opening quote; emits the normalized string (JSON-escaped when
// as_json, raw bytes for a top-level string result); advances p past the
// closing quote
__device__ bool j_str_norm(Sink& sink, const char*& p, const char* e,
                           bool as_json) {
  char q = *p;
  ++p;
  if (as_json) sink.put('"');
  while (p < e && *p != q) {
    char c = *p;
    if (c == '\\') {
      if (p + 1 >= e) return false;
      char x = p[1];
      p += 2;
      switch (x) {
        case 'n': j_emit_char(sink, '\n', as_json); break;
        case 't': j_emit_char(sink, '\t', as_json); break;
        case 'r': j_emit_char(sink, '\r', as_json); break;
        case 'b': j_emit_char(sink, '\b', as_json); break;
        case 'f': j_emit_char(sink, '\f', as_json); break;
        case '/': j_emit_char(sink, '/', as_json); break;
        case '"': j_emit_char(sink, '"', as_json); break;
        case '\'': j_emit_char(sink, '\'', as_json); break;
        case '\\': j_emit_char(sink, '\\', as_json); break;
        case 'u': {
          uint32_t cp;
          if (!j_hex4(p, e, &cp)) return false;
          p += 4;
          if (cp >= 0xD800 && cp < 0xDC00 && p + 6 <= e && p[0] == '\\' &&
              p[1] == 'u') {
            uint32_t lo;
            if (j_hex4(p + 2, e, &lo) && lo >= 0xDC00 && lo < 0xE000) {
              cp = 0x10000 + ((cp - 0xD800) << 10) + (lo - 0xDC00);
              p += 6;
            }
          }
          if (cp < 0x80) {
            j_emit_char(sink, (char)cp, as_json);
          } else if (cp < 0x800) {
            sink.put((char)(0xC0 | (cp >> 6)));
            sink.put((char)(0x80 | (cp & 0x3F)));
          } else if (cp < 0x10000) {
            sink.put((char)(0xE0 | (cp >> 12)));
            sink.put((char)(0x80 | ((cp >> 6) & 0x3F)));
            sink.put((char)(0x80 | (cp & 0x3F)));
          } else {
            sink.put((char)(0xF0 | (cp >> 18)));
            sink.put((char)(0x80 | ((cp >> 12) & 0x3F)));
            sink.put((char)(0x80 | ((cp >> 6) & 0x3F)));
            sink.put((char)(0x80 | (cp & 0x3F)));
          }
          break;
        }
        default: j_emit_char(sink, x, as_json);
      }
    } else {
      j_emit_char(sink, c, as_json);
      ++p;
    }
  }
  if (p >= e) return false;
  ++p;
  if (as_json) sink.put('"');
  return true;
}

// number token: integers verbatim (-0 -> 0; leading zeros invalid), anything
// with . or e/E through exact parse + Java Double.toString; +/-Infinity and
// NaN become quoted strings (Spark emits them as JSON strings)
__device__ bool j_num_norm(Sink& sink, const char*& p, const char* e) {
  const char* st = p;
  if (p < e && *p == '-') ++p;
  const char* ds = p;
  while (p < e && *p >= '0' && *p <= '9') ++p;
  int32_t nint = (int32_t)(p - ds);
  if (nint == 0) return false;
  if (nint > 1 && ds[0] == '0') return false;  // leading zeros invalid
  bool isflt = false;
  if (p < e && *p == '.') {
    isflt = true;
    ++p;
    const char* fs = p;
    while (p < e && *p >= '0' && *p <= '9') ++p;
    if (p == fs) return false;
  }
  if (p < e && (*p == 'e' || *p == 'E')) {
    isflt = true;
    ++p;
    if (p < e && (*p == '+' || *p == '-')) ++p;
    const char* es = p;
    while (p < e && *p >= '0' && *p <= '9') ++p;
    if (p == es) return false;
  }
  if (!isflt) {
    if (st[0] == '-' && nint == 1 && ds[0] == '0') {
      sink.put('0');  // "-0" normalizes to 0
      return true;
    }
    sink.span(st, (int32_t)(p - st));
    return true;
  }
  double d;
  StrView sv{st, (int32_t)(p - st)};
  if (!parse_double(sv, &d)) return false;
  char buf[40];
  int n = format_double(d, buf);
  bool special = buf[0] == 'N' || buf[n - 1] == 'y';  // NaN / ...Infinity
  if (special) {
    sink.put('"');
    sink.span(buf, n);
    sink.put('"');
  } else {
    sink.span(buf, n);
  }
  return true;
}

// whole-value re-serializer; advances p past the value
__device__ bool emit_norm_value(Sink& sink, const char*& p, const char* e,
                                int depth) {
  p = j_skip_ws(p, e);
  if (p >= e || depth > JSON_MAX_DEPTH) return false;
  char c = *p;
  if (c == '"' || c == '\'') return j_str_norm(sink, p, e, true);
  if (c == '{') {
    ++p;
    sink.put('{');
    p = j_skip_ws(p, e);
    if (p < e && *p == '}') { ++p; sink.put('}'); return true; }
    while (true) {
      p = j_skip_ws(p, e);
      if (p >= e || (*p != '"' && *p != '\'')) return false;
      if (!j_str_norm(sink, p, e, true)) return false;
      p = j_skip_ws(p, e);
      if (p >= e || *p != ':') return false;
      ++p;
      sink.put(':');
      if (!emit_norm_value(sink, p, e, depth + 1)) return false;
      p = j_skip_ws(p, e);
      if (p < e && *p == ',') { ++p; sink.put(','); continue; }
      if (p < e && *p == '}') { ++p; sink.put('}'); return true; }
      return false;
    }
  }
  if (c == '[') {
    ++p;
    sink.put('[');
    p = j_skip_ws(p, e);
    if (p < e && *p == ']') { ++p; sink.put(']'); return true; }
    while (true) {
      if (!emit_norm_value(sink, p, e, depth + 1)) return false;
      p = j_skip_ws(p, e);
      if (p < e && *p == ',') { ++p; sink.put(','); continue; }
      if (p < e && *p == ']') { ++p; sink.put(']'); return true; }
      return false;
    }
  }
  if (c == 't' && e - p >= 4 && p[1] == 'r' && p[2] == 'u' && p[3] == 'e') {
    sink.span(p, 4);
    p += 4;
    return true;
  }
  if (c == 'f' && e - p >= 5 && p[1] == 'a' && p[2] == 'l' && p[3] == 's' &&
      p[4] == 'e') {
    sink.span(p, 5);
    p += 5;
    return true;
  }
  if (c == 'n' && e - p >= 4 && p[1] == 'u' && p[2] == 'l' && p[3] == 'l') {
    sink.span(p, 4);
    p += 4;
    return true;
  }
  return j_num_norm(sink, p, e);
}

// emit one matched span through the normalizer; false = invalid token
__device__ bool emit_match(Sink& sink, JsonSpan m, bool as_element) {
  const char* p = m.p;
  const char* e = m.p + m.len;
  if (m.len >= 2 && (m.p[0] == '"' || m.p[0] == '\'') && !as_element) {
    return j_str_norm(sink, p, e, false);
  }
  return emit_norm_value(sink, p, e, 0);
}

// from_json paths keep the row and fall back to the raw span when the
// normalizer rejects a token (deterministic across the measure/write phases)
__device__ void emit_match_or_raw(Sink& sink, JsonSpan m, bool as_element) {
  Sink probe{nullptr, 0, false};
  if (emit_match(probe, m, as_element)) {
    emit_match(sink, m, as_element);
  } else if (m.len >= 2 && (m.p[0] == '"' || m.p[0] == '\'') && !as_element) {
    sink.unescaped(m.p + 1, m.len - 2);
  } else {
    sink.span(m.p, m.len);
  }
}

template <bool WRITE>
__global__ void get_json_object_kernel(ColDesc in, int64_t nrows,
                                       const PathInstr* __restrict__ instrs,
                                       const char* __restrict__ keychars,
                                       int32_t ninstr,
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
    int32_t out_len = 0;
    Sink sink{WRITE ? chars + (in_range && offsets ? offsets[row] : 0) : nullptr,
              0, WRITE};
    if (valid) {
      StrView s = get_string(in, row);
      MatchCtx ctx{instrs, keychars, ninstr, 0, {}, false};
      bool ok = j_match(s.ptr, s.ptr + s.len, ctx, 0, 0);
      if (!ok || ctx.nmatches == 0 || ctx.overflow) {
        valid = false;
      } else if (ctx.nmatches == 1) {
        // Spark/Hive: a matched JSON null yields SQL NULL
        JsonSpan m = ctx.matches[0];
        if (m.len == 4 && m.p[0] == 'n' && m.p[1] == 'u' && m.p[2] == 'l' &&
            m.p[3] == 'l') {
          valid = false;
        } else {
          Sink probe{nullptr, 0, false};
          if (!emit_match(probe, m, false)) {
            valid = false;
          } else {
            emit_match(sink, m, false);
            out_len = sink.pos;
          }
        }
      } else {
        Sink probe{nullptr, 0, false};
        bool eok = true;
        for (int32_t i = 0; i < ctx.nmatches && eok; ++i)
          eok = emit_match(probe, ctx.matches[i], true);
        if (!eok) {
          valid = false;
        } else {
          sink.put('[');
          for (int32_t i = 0; i < ctx.nmatches; ++i) {
            if (i) sink.put(',');
            emit_match(sink, ctx.matches[i], true);
          }
          sink.put(']');
          out_len = sink.pos;
        }
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
// multi-path shared scan (reference get_json_object.cu multi-path kernel with
// num_threads_per_row / JSONUtils.getJsonObjectMultiplePaths): every path
// instruction consumes exactly one nesting level, so ONE document walk can
// carry a bitmask of still-viable paths and match them all simultaneously —
// the doc is tokenized once instead of once per path.
// ---------------------------------------------------------------------------
constexpr int MAX_MULTI_PATHS = 8;
constexpr int MULTI_MATCH_CAP = 4;

struct MultiCtx {
  const PathInstr* instrs;   // concatenated
  const char* keychars;
  const int32_t* path_off;   // [npaths] start into instrs
  const int32_t* path_len;   // [npaths] instruction count
  int32_t npaths;
  uint8_t nmatches[MAX_MULTI_PATHS];
  bool overflow[MAX_MULTI_PATHS];
  JsonSpan matches[MAX_MULTI_PATHS][MULTI_MATCH_CAP];
};

__device__ inline void multi_record(MultiCtx& ctx, int i, const char* p,
                                    const char* vend) {
  if (ctx.nmatches[i] < MULTI_MATCH_CAP)
    ctx.matches[i][ctx.nmatches[i]] = {p, (int32_t)(vend - p)};
  else
    ctx.overflow[i] = true;
  ++ctx.nmatches[i];
}

// single walk carrying `mask` of paths viable at this nesting level
__device__ bool j_match_multi(const char* p, const char* e, MultiCtx& ctx,
                              int32_t depth, uint32_t mask) {
  if (depth > MAX_PATH_DEPTH + 2) return false;
  p = j_skip_ws(p, e);
  // terminal paths match this whole value
  uint32_t term = 0;
  for (int i = 0; i < ctx.npaths; ++i)
    if ((mask >> i) & 1 && ctx.path_len[i] == depth) term |= 1u << i;
  if (term) {
    const char* vend = j_skip_value(p, e);
    if (!vend) return false;
    for (int i = 0; i < ctx.npaths; ++i)
      if ((term >> i) & 1) multi_record(ctx, i, p, vend);
  }
  uint32_t desc = mask & ~term;
  if (!desc || p >= e) return true;
  if (*p == '{') {
    ++p;
    while (true) {
      p = j_skip_ws(p, e);
      if (p < e && *p == '}') return true;
      if (p >= e || (*p != '"' && *p != '\'')) return false;
      const char* kend = j_skip_string(p, e);
      if (!kend) return false;
      const char* kbody = p + 1;
      int32_t klen = (int32_t)(kend - p - 2);
      p = j_skip_ws(kend, e);
      if (p >= e || *p != ':') return false;
      ++p;
      p = j_skip_ws(p, e);
      uint32_t m2 = 0;
      for (int i = 0; i < ctx.npaths; ++i) {
        if (!((desc >> i) & 1)) continue;
        const PathInstr ins = ctx.instrs[ctx.path_off[i] + depth];
        if (ins.kind == 2) {
          m2 |= 1u << i;
        } else if (ins.kind == 0 &&
                   j_key_equals(kbody, klen, ctx.keychars + ins.key_off,
                                ins.key_len)) {
          m2 |= 1u << i;
        }
      }
      if (m2) {
        if (!j_match_multi(p, e, ctx, depth + 1, m2)) return false;
      }
      const char* vend = j_skip_value(p, e);
      if (!vend) return false;
      p = j_skip_ws(vend, e);
      if (p < e && *p == ',') { ++p; continue; }
      if (p < e && *p == '}') return true;
      return false;
    }
  }
  if (*p == '[') {
    ++p;
    int32_t idx = 0;
    while (true) {
      p = j_skip_ws(p, e);
      if (p < e && *p == ']') return true;
      uint32_t m2 = 0;
      for (int i = 0; i < ctx.npaths; ++i) {
        if (!((desc >> i) & 1)) continue;
        const PathInstr ins = ctx.instrs[ctx.path_off[i] + depth];
        if (ins.kind == 2 || (ins.kind == 1 && ins.index == idx)) m2 |= 1u << i;
      }
      if (m2) {
        if (!j_match_multi(p, e, ctx, depth + 1, m2)) return false;
      }
      const char* vend = j_skip_value(p, e);
      if (!vend) return false;
      p = j_skip_ws(vend, e);
      ++idx;
      if (p < e && *p == ',') { ++p; continue; }
      if (p < e && *p == ']') return true;
      return false;
    }
  }
  return true;  // structure mismatch: no match, not an error
}

// per-path output block pointers (lens / offsets / chars / validity)
struct MultiOut {
  int32_t* lens;
  const int32_t* offsets;
  char* chars;
  uint8_t* out_valid;
  int32_t* overflow;  // one flag per path (host triggers per-path fallback)
};

template <bool WRITE>
__global__ void get_json_multi_kernel(ColDesc in, int64_t nrows,
                                      const PathInstr* __restrict__ instrs,
                                      const char* __restrict__ keychars,
                                      const int32_t* __restrict__ path_off,
                                      const int32_t* __restrict__ path_len,
                                      int32_t npaths,
                                      const MultiOut* __restrict__ outs) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  int64_t npad = (nrows + WAVE - 1) & ~(int64_t)(WAVE - 1);
  for (int64_t row = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; row < npad;
       row += stride) {
    bool in_range = row < nrows;
    bool doc_valid = in_range && is_valid(in.valid, row);
    MultiCtx ctx;
    ctx.instrs = instrs;
    ctx.keychars = keychars;
    ctx.path_off = path_off;
    ctx.path_len = path_len;
    ctx.npaths = npaths;
    for (int i = 0; i < npaths; ++i) {
      ctx.nmatches[i] = 0;
      ctx.overflow[i] = false;
    }
    bool ok = false;
    if (doc_valid) {
      StrView s = get_string(in, row);
      ok = j_match_multi(s.ptr, s.ptr + s.len, ctx, 0,
                         (uint32_t)((1u << npaths) - 1));
    }
    for (int i = 0; i < npaths; ++i) {
      const MultiOut o = outs[i];
      bool valid = doc_valid && ok && ctx.nmatches[i] > 0 && !ctx.overflow[i];
      int32_t out_len = 0;
      Sink sink{WRITE && in_range && o.offsets
                    ? o.chars + o.offsets[row] : nullptr,
                0, WRITE};
      if (valid) {
        if (ctx.nmatches[i] == 1) {
          JsonSpan m = ctx.matches[i][0];
          if (m.len == 4 && m.p[0] == 'n' && m.p[1] == 'u' && m.p[2] == 'l' &&
              m.p[3] == 'l') {
            valid = false;
          } else {
            Sink probe{nullptr, 0, false};
            if (!emit_match(probe, m, false)) {
              valid = false;
            } else {
              emit_match(sink, m, false);
              out_len = sink.pos;
            }
          }
        } else {
          Sink probe{nullptr, 0, false};
          bool eok = true;
          for (int32_t k = 0; k < (int32_t)ctx.nmatches[i] && eok; ++k)
            eok = emit_match(probe, ctx.matches[i][k], true);
          if (!eok) {
            valid = false;
          } else {
            sink.put('[');
            for (int32_t k = 0; k < (int32_t)ctx.nmatches[i]; ++k) {
              if (k) sink.put(',');
              emit_match(sink, ctx.matches[i][k], true);
            }
            sink.put(']');
            out_len = sink.pos;
          }
        }
      }
      if (ctx.overflow[i] && in_range) atomicOr(o.overflow + 0, 1);
      if (WRITE) {
        ballot_write_validity(o.out_valid, row, valid);
      } else if (in_range) {
        o.lens[row] = valid ? out_len : 0;
      }
    }
  }
}

// ---------------------------------------------------------------------------
// from_json: top-level object field extraction -> string columns (one kernel
// run per schema field reusing the path machinery), and raw-map extraction.
// ---------------------------------------------------------------------------
template <bool WRITE>
__global__ void json_to_map_kernel(ColDesc in, int64_t nrows,
                                   int32_t* __restrict__ entry_counts,
                                   const int32_t* __restrict__ entry_offsets,
                                   int32_t* __restrict__ key_lens,
                                   int32_t* __restrict__ val_lens,
                                   const int32_t* __restrict__ key_offsets,
                                   const int32_t* __restrict__ val_offsets,
                                   char* __restrict__ key_chars,
                                   char* __restrict__ val_chars,
                                   uint8_t* __restrict__ out_valid) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  int64_t npad = (nrows + WAVE - 1) & ~(int64_t)(WAVE - 1);
  for (int64_t row = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; row < npad;
       row += stride) {
    bool in_range = row < nrows;
    bool valid = in_range && is_valid(in.valid, row);
    int32_t nent = 0;
    if (valid) {
      StrView s = get_string(in, row);
      const char* p = j_skip_ws(s.ptr, s.ptr + s.len);
      const char* e = s.ptr + s.len;
      if (p >= e || *p != '{') {
        valid = false;
      } else {
        ++p;
        int32_t ebase = WRITE && in_range ? entry_offsets[row] : 0;
        while (valid) {
          p = j_skip_ws(p, e);
          if (p < e && *p == '}') break;
          if (p >= e || (*p != '"' && *p != '\'')) { valid = false; break; }
          const char* kend = j_skip_string(p, e);
          if (!kend) { valid = false; break; }
          const char* kbody = p + 1;
          int32_t klen = (int32_t)(kend - p - 2);
          p = j_skip_ws(kend, e);
          if (p >= e || *p != ':') { valid = false; break; }
          ++p;
          p = j_skip_ws(p, e);
          const char* vstart = p;
          const char* vend = j_skip_value(p, e);
          if (!vend) { valid = false; break; }
          // value emitted unquoted for strings, raw otherwise
          if (WRITE) {
            int32_t ei = ebase + nent;
            Sink ks{key_chars + key_offsets[ei], 0, true};
            ks.unescaped(kbody, klen);
            Sink vs{val_chars + val_offsets[ei], 0, true};
            JsonSpan m{vstart, (int32_t)(vend - vstart)};
            emit_match_or_raw(vs, m, false);
          } else {
            // measure
            Sink ks{nullptr, 0, false};
            ks.unescaped(kbody, klen);
            Sink vs{nullptr, 0, false};
            JsonSpan m{vstart, (int32_t)(vend - vstart)};
            emit_match_or_raw(vs, m, false);
            if (key_lens) key_lens[in_range ? row : 0] = 0;  // placeholder
          }
          ++nent;
          p = j_skip_ws(vend, e);
          if (p < e && *p == ',') { ++p; continue; }
          if (p < e && *p == '}') break;
          valid = false;
        }
      }
    }
    if (!WRITE && in_range) entry_counts[row] = valid ? nent : 0;
    if (WRITE) ballot_write_validity(out_valid, row, valid);
  }
}

// measure per-entry key/value lengths (phase between count and write):
// entry_offsets gives each row's first entry slot.
__global__ void json_map_entry_lens_kernel(ColDesc in, int64_t nrows,
                                           const int32_t* __restrict__ entry_offsets,
                                           int32_t* __restrict__ key_lens,
                                           int32_t* __restrict__ val_lens) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t row = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; row < nrows;
       row += stride) {
    if (!is_valid(in.valid, row)) continue;
    StrView s = get_string(in, row);
    const char* p = j_skip_ws(s.ptr, s.ptr + s.len);
    const char* e = s.ptr + s.len;
    if (p >= e || *p != '{') continue;
    ++p;
    int32_t ei = entry_offsets[row];
    while (true) {
      p = j_skip_ws(p, e);
      if (p >= e || *p == '}') break;
      if (*p != '"' && *p != '\'') break;
      const char* kend = j_skip_string(p, e);
      if (!kend) break;
      Sink ks{nullptr, 0, false};
      ks.unescaped(p + 1, (int32_t)(kend - p - 2));
      p = j_skip_ws(kend, e);
      if (p >= e || *p != ':') break;
      ++p;
      p = j_skip_ws(p, e);
      const char* vend = j_skip_value(p, e);
      if (!vend) break;
      Sink vs{nullptr, 0, false};
      emit_match_or_raw(vs, {p, (int32_t)(vend - p)}, false);
      key_lens[ei] = ks.pos;
      val_lens[ei] = vs.pos;
      ++ei;
      p = j_skip_ws(vend, e);
      if (p < e && *p == ',') { ++p; continue; }
      break;
    }
  }
}

}  // namespace srj

using namespace srj;

extern "C" {

void srj_get_json_object(const void* in, int64_t nrows, const void* instrs,
                         const char* keychars, int32_t ninstr, int32_t phase,
                         int32_t* lens, const int32_t* offsets, char* chars,
                         uint8_t* out_valid, hipStream_t stream) {
  ColDesc c = *reinterpret_cast<const ColDesc*>(in);
  if (phase == 0)
    get_json_object_kernel<false><<<grid_1d(nrows), DEFAULT_BLOCK, 0, stream>>>(
        c, nrows, reinterpret_cast<const PathInstr*>(instrs), keychars, ninstr,
        lens, nullptr, nullptr, nullptr);
  else
    get_json_object_kernel<true><<<grid_1d(nrows), DEFAULT_BLOCK, 0, stream>>>(
        c, nrows, reinterpret_cast<const PathInstr*>(instrs), keychars, ninstr,
        nullptr, offsets, chars, out_valid);
}

void srj_get_json_multi(const void* in, int64_t nrows, const void* instrs,
                        const char* keychars, const int32_t* path_off,
                        const int32_t* path_len, int32_t npaths, int32_t phase,
                        const void* outs, hipStream_t stream) {
  ColDesc c = *reinterpret_cast<const ColDesc*>(in);
  if (phase == 0)
    get_json_multi_kernel<false><<<grid_1d(nrows), DEFAULT_BLOCK, 0, stream>>>(
        c, nrows, reinterpret_cast<const PathInstr*>(instrs), keychars,
        path_off, path_len, npaths, reinterpret_cast<const MultiOut*>(outs));
  else
    get_json_multi_kernel<true><<<grid_1d(nrows), DEFAULT_BLOCK, 0, stream>>>(
        c, nrows, reinterpret_cast<const PathInstr*>(instrs), keychars,
        path_off, path_len, npaths, reinterpret_cast<const MultiOut*>(outs));
}

void srj_json_map_count(const void* in, int64_t nrows, int32_t* entry_counts,
                        hipStream_t stream) {
  ColDesc c = *reinterpret_cast<const ColDesc*>(in);
  json_to_map_kernel<false><<<grid_1d(nrows), DEFAULT_BLOCK, 0, stream>>>(
      c, nrows, entry_counts, nullptr, nullptr, nullptr, nullptr, nullptr,
      nullptr, nullptr, nullptr);
}

void srj_json_map_entry_lens(const void* in, int64_t nrows,
                             const int32_t* entry_offsets, int32_t* key_lens,
                             int32_t* val_lens, hipStream_t stream) {
  ColDesc c = *reinterpret_cast<const ColDesc*>(in);
  json_map_entry_lens_kernel<<<grid_1d(nrows), DEFAULT_BLOCK, 0, stream>>>(
      c, nrows, entry_offsets, key_lens, val_lens);
}

void srj_json_map_write(const void* in, int64_t nrows,
                        const int32_t* entry_offsets, const int32_t* key_offsets,
                        const int32_t* val_offsets, char* key_chars,
                        char* val_chars, uint8_t* out_valid, hipStream_t stream) {
  ColDesc c = *reinterpret_cast<const ColDesc*>(in);
  json_to_map_kernel<true><<<grid_1d(nrows), DEFAULT_BLOCK, 0, stream>>>(
      c, nrows, nullptr, entry_offsets, nullptr, nullptr, key_offsets,
      val_offsets, key_chars, val_chars, out_valid);
}

}  // extern "C"
