// Parquet page decode for MI355X.
//
// The reference delegates page decode to libcudf (SURVEY.md §7 item 5 calls
// this the largest NEW kernel work). Design here: the host (Python) walks
// footers + page headers (Thrift, src/host/thrift_compact.cpp) and builds
// per-page descriptors; these kernels decode all pages of all row groups in
// a handful of launches:
//   * rle_decode           — Parquet RLE/bit-packed hybrid -> uint8/int32.
//                            One workgroup per page; lane 0 walks run
//                            headers into an LDS run table (batched), then
//                            all 256 threads expand element-parallel with a
//                            binary search over the LDS table.
//   * scatter_fixed        — PLAIN or dictionary values -> output rows, using
//                            a column-wide exclusive scan of def levels for
//                            null scatter (scan done via torch.cumsum).
//   * string_plain_index   — lane-0 walk of PLAIN byte-array pages producing
//                            per-value offsets + per-row lengths.
//   * string_copy_chars    — parallel char gather for string output.
//   * def_to_validity      — def-level bytes -> Arrow validity bits (wave64
//                            ballot).
// Scope v1: flat schemas (max def level 1, no rep levels), UNCOMPRESSED pages.
#include "srj_common.hpp"

namespace srj {

struct RleDesc {
  const uint8_t* src;
  int64_t src_len;
  void* out;         // uint8 (kind 0) or int32 (kind 1)
  int64_t num_out;
  int32_t bit_width;
  int32_t out_kind;
};

constexpr int RUN_BATCH = 256;

struct Run {
  int64_t out_pos;
  int32_t count;
  int32_t kind;      // 0 = rle, 1 = bit-packed
  uint32_t value;    // rle value
  int64_t src_off;   // bit-packed payload offset
};

__global__ void rle_decode_kernel(const RleDesc* __restrict__ descs,
                                  int32_t npages) {
  __shared__ Run runs[RUN_BATCH];
  __shared__ int batch_n;
  __shared__ int64_t batch_elems;
  for (int32_t page = blockIdx.x; page < npages; page += gridDim.x) {
    RleDesc d = descs[page];
    int64_t pos = 0;       // src byte position (lane 0 state, shared via LDS)
    int64_t out_pos = 0;
    __shared__ int64_t s_pos, s_out;
    if (threadIdx.x == 0) { s_pos = 0; s_out = 0; }
    __syncthreads();
    while (true) {
      if (threadIdx.x == 0) {
        pos = s_pos;
        out_pos = s_out;
        int n = 0;
        while (n < RUN_BATCH && pos < d.src_len && out_pos < d.num_out) {
          // varint header
          uint64_t h = 0;
          int shift = 0;
          while (pos < d.src_len) {
            uint8_t b = d.src[pos++];
            h |= (uint64_t)(b & 0x7F) << shift;
            if (!(b & 0x80)) break;
            shift += 7;
          }
          if (h & 1) {  // bit-packed: (h>>1) groups of 8
            int64_t cnt = (int64_t)(h >> 1) * 8;
            if (cnt > d.num_out - out_pos) cnt = d.num_out - out_pos;
            runs[n] = Run{out_pos, (int32_t)cnt, 1, 0, pos};
            pos += ((h >> 1) * d.bit_width);  // bytes = groups * bw
            out_pos += cnt;
          } else {  // rle run
            int64_t cnt = (int64_t)(h >> 1);
            uint32_t v = 0;
            int nb = (d.bit_width + 7) / 8;
            for (int b = 0; b < nb && pos < d.src_len; ++b)
              v |= (uint32_t)d.src[pos++] << (8 * b);
            if (cnt > d.num_out - out_pos) cnt = d.num_out - out_pos;
            runs[n] = Run{out_pos, (int32_t)cnt, 0, v, 0};
            out_pos += cnt;
          }
          ++n;
        }
        batch_n = n;
        batch_elems = n ? (runs[n - 1].out_pos + runs[n - 1].count - runs[0].out_pos)
                        : 0;
        s_pos = pos;
        s_out = out_pos;
      }
      __syncthreads();
      int bn = batch_n;
      if (bn == 0) break;
      int64_t base = runs[0].out_pos;
      int64_t total = batch_elems;
      for (int64_t e = threadIdx.x; e < total; e += blockDim.x) {
        int64_t tgt = base + e;
        // binary search run
        int lo = 0, hi = bn - 1;
        while (lo < hi) {
          int mid = (lo + hi + 1) >> 1;
          if (runs[mid].out_pos <= tgt) lo = mid;
          else hi = mid - 1;
        }
        const Run& r = runs[lo];
        int64_t i = tgt - r.out_pos;
        uint32_t v;
        if (r.kind == 0) {
          v = r.value;
        } else {
          int64_t bit = i * d.bit_width;
          int64_t byte = r.src_off + (bit >> 3);
          int sh = (int)(bit & 7);
          uint32_t w = d.src[byte];
          if (d.bit_width + sh > 8) w |= (uint32_t)d.src[byte + 1] << 8;
          if (d.bit_width + sh > 16) w |= (uint32_t)d.src[byte + 2] << 16;
          if (d.bit_width + sh > 24) w |= (uint32_t)d.src[byte + 3] << 24;
          v = (w >> sh) & ((d.bit_width >= 32) ? 0xFFFFFFFFu
                                               : ((1u << d.bit_width) - 1));
        }
        if (d.out_kind == 0)
          reinterpret_cast<uint8_t*>(d.out)[tgt] = (uint8_t)v;
        else
          reinterpret_cast<int32_t*>(d.out)[tgt] = (int32_t)v;
      }
      __syncthreads();  // next batch reuses the LDS table
    }
    __syncthreads();
  }
}

// ---------------------------------------------------------------------------
// fixed-width value scatter
// ---------------------------------------------------------------------------
struct ScatterDesc {
  const uint8_t* values;    // PLAIN data or decoded dict indices (int32)
  const uint8_t* dict;      // dictionary values (or null for PLAIN)
  const uint8_t* def;       // per-row def bytes (or null = all valid)
  const int64_t* vprefix;   // column-wide exclusive count of valid rows
  int64_t row_start;        // first output row of this page
  int64_t nrows;            // rows (levels) in this page
  int64_t value_base;       // vprefix value at row_start
  int32_t width;
  int32_t is_dict;
};

__global__ void scatter_fixed_kernel(const ScatterDesc* __restrict__ descs,
                                     int32_t npages, uint8_t* __restrict__ out) {
  for (int32_t page = blockIdx.x; page < npages; page += gridDim.x) {
    ScatterDesc d = descs[page];
    for (int64_t i = threadIdx.x; i < d.nrows; i += blockDim.x) {
      int64_t row = d.row_start + i;
      bool valid = d.def == nullptr || d.def[row] != 0;
      if (!valid) continue;
      int64_t vi = d.def == nullptr ? i : (d.vprefix[row] - d.value_base);
      const uint8_t* src;
      if (d.is_dict) {
        int32_t idx = reinterpret_cast<const int32_t*>(d.values)[vi];
        src = d.dict + (int64_t)idx * d.width;
      } else {
        src = d.values + vi * d.width;
      }
      uint8_t* dst = out + row * d.width;
      for (int b = 0; b < d.width; ++b) dst[b] = src[b];
    }
  }
}

// ---------------------------------------------------------------------------
// strings
// ---------------------------------------------------------------------------
struct StrIndexDesc {
  const uint8_t* src;      // PLAIN byte-array section
  int64_t src_len;
  int64_t num_values;      // non-null values in page
  int64_t* val_off;        // out: payload byte offset per value
  int32_t* val_len;        // out: length per value
};

__global__ void string_plain_index_kernel(const StrIndexDesc* __restrict__ descs,
                                          int32_t npages) {
  for (int32_t page = blockIdx.x; page < npages; page += gridDim.x) {
    if (threadIdx.x != 0) continue;
    StrIndexDesc d = descs[page];
    int64_t pos = 0;
    for (int64_t v = 0; v < d.num_values && pos + 4 <= d.src_len; ++v) {
      uint32_t len;
      __builtin_memcpy(&len, d.src + pos, 4);
      pos += 4;
      d.val_off[v] = pos;
      d.val_len[v] = (int32_t)len;
      pos += len;
    }
  }
}

struct StrCopyDesc {
  const uint8_t* src;       // byte-array section (or dict chars base)
  const int64_t* val_off;   // per-value offsets (page values or dict entries)
  const int32_t* val_len;
  const uint8_t* indices;   // dict indices (int32) or null for PLAIN
  const uint8_t* def;       // per-row def bytes or null
  const int64_t* vprefix;
  int64_t row_start;
  int64_t nrows;
  int64_t value_base;
};

// phase 0: lengths per row; phase 1: copy chars using column offsets
template <int PHASE>
__global__ void string_copy_kernel(const StrCopyDesc* __restrict__ descs,
                                   int32_t npages, int32_t* __restrict__ lens,
                                   const int32_t* __restrict__ offsets,
                                   uint8_t* __restrict__ chars) {
  for (int32_t page = blockIdx.x; page < npages; page += gridDim.x) {
    StrCopyDesc d = descs[page];
    for (int64_t i = threadIdx.x; i < d.nrows; i += blockDim.x) {
      int64_t row = d.row_start + i;
      bool valid = d.def == nullptr || d.def[row] != 0;
      if (!valid) {
        if (PHASE == 0) lens[row] = 0;
        continue;
      }
      int64_t vi = d.def == nullptr ? i : (d.vprefix[row] - d.value_base);
      int64_t entry = d.indices
                          ? reinterpret_cast<const int32_t*>(d.indices)[vi]
                          : vi;
      if (PHASE == 0) {
        lens[row] = d.val_len[entry];
      } else {
        int64_t so = d.val_off[entry];
        int32_t n = d.val_len[entry];
        int32_t o = offsets[row];
        for (int32_t k = 0; k < n; ++k) chars[o + k] = d.src[so + k];
      }
    }
  }
}

__global__ void def_to_validity_kernel(const uint8_t* __restrict__ def,
                                       int64_t nrows,
                                       uint8_t* __restrict__ validity) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  int64_t npad = (nrows + WAVE - 1) & ~(int64_t)(WAVE - 1);
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < npad;
       i += stride) {
    bool valid = i < nrows && def[i] != 0;
    ballot_write_validity(validity, i, valid);
  }
}


// ---------------------------------------------------------------------------
// snappy page decompression (device): one wave per page, all 64 lanes parse
// the tag stream redundantly in lockstep (identical control flow, no
// divergence) and cooperate on the literal/match copies. The reference
// delegates page decompression to nvcomp/libcudf; here it is ~80 lines of
// CDNA4 HIP. Overlapped matches (offset < wave width) fall back to lane 0.
// ---------------------------------------------------------------------------
struct SnapDesc {
  const uint8_t* src;
  int64_t src_len;
  uint8_t* dst;
  int64_t dst_len;
};

__global__ void snappy_decomp_kernel(const SnapDesc* __restrict__ descs,
                                     int32_t n) {
  int waves_per_block = blockDim.x / WAVE;
  int wid = blockIdx.x * waves_per_block + (int)(threadIdx.x / WAVE);
  int lane = threadIdx.x & (WAVE - 1);
  if (wid >= n) return;
  SnapDesc d = descs[wid];
  const uint8_t* p = d.src;
  const uint8_t* e = d.src + d.src_len;
  while (p < e && (*p & 0x80)) ++p;  // preamble: uncompressed length varint
  ++p;
  uint8_t* o = d.dst;
  uint8_t* oend = d.dst + d.dst_len;
  while (p < e && o < oend) {
    uint8_t tag = *p++;
    int op = tag & 3;
    if (op == 0) {  // literal run
      int64_t len = (tag >> 2) + 1;
      if (len > 60) {
        int nb = (int)len - 60;
        len = 0;
        for (int i = 0; i < nb; ++i) len |= (int64_t)p[i] << (8 * i);
        ++len;
        p += nb;
      }
      if (o + len > oend || p + len > e) return;  // corrupt
      for (int64_t k = lane; k < len; k += WAVE) o[k] = p[k];
      o += len;
      p += len;
    } else {  // back-reference
      int64_t len, off;
      if (op == 1) {
        if (p >= e) return;
        len = ((tag >> 2) & 7) + 4;
        off = ((int64_t)(tag >> 5) << 8) | p[0];
        p += 1;
      } else if (op == 2) {
        if (p + 2 > e) return;
        len = (tag >> 2) + 1;
        off = (int64_t)p[0] | ((int64_t)p[1] << 8);
        p += 2;
      } else {
        if (p + 4 > e) return;
        len = (tag >> 2) + 1;
        off = (int64_t)p[0] | ((int64_t)p[1] << 8) | ((int64_t)p[2] << 16) |
              ((int64_t)p[3] << 24);
        p += 4;
      }
      if (off <= 0 || o - off < d.dst || o + len > oend) return;  // corrupt
      const uint8_t* s2 = o - off;
      if (off >= len) {
        for (int64_t k = lane; k < len; k += WAVE) o[k] = s2[k];
      } else if (off == 1) {
        uint8_t b = s2[0];  // RLE fill
        for (int64_t k = lane; k < len; k += WAVE) o[k] = b;
      } else if (lane == 0) {
        // overlapped pattern: serial on one lane (rare)
        for (int64_t k = 0; k < len; ++k) o[k] = s2[k];
      }
      o += len;
    }
  }
}

// fetch single bytes at arbitrary device addresses (host planning needs the
// RLE bit-width byte / def-level length inside device-decompressed pages)
__global__ void gather_u8_at_kernel(const uint64_t* __restrict__ addrs,
                                    int64_t n, uint8_t* __restrict__ out) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) out[i] = *reinterpret_cast<const uint8_t*>(addrs[i]);
}


// FIXED_LEN_BYTE_ARRAY decimals: W-byte big-endian two's-complement values
// -> DECIMAL128 (2 x int64 little-endian words, sign-extended). Reuses the
// ScatterDesc grid (width = FLBA byte length, is_dict selects the source).
__global__ void flba_dec128_kernel(const ScatterDesc* __restrict__ descs,
                                   int32_t npages,
                                   int64_t* __restrict__ out) {
  for (int32_t page = blockIdx.x; page < npages; page += gridDim.x) {
    ScatterDesc d = descs[page];
    for (int64_t i = threadIdx.x; i < d.nrows; i += blockDim.x) {
      int64_t row = d.row_start + i;
      bool valid = d.def == nullptr || d.def[row] != 0;
      if (!valid) continue;
      int64_t vi = d.def == nullptr ? i : (d.vprefix[row] - d.value_base);
      int W = d.width;
      const uint8_t* src;
      if (d.is_dict) {
        int32_t idx = reinterpret_cast<const int32_t*>(d.values)[vi];
        src = d.dict + (int64_t)idx * W;
      } else {
        src = d.values + vi * W;
      }
      bool neg = src[0] & 0x80;
      uint64_t hi = neg ? ~0ull : 0ull, lo = neg ? ~0ull : 0ull;
      for (int k = 0; k < W; ++k) {
        uint64_t b = src[k];
        int bit = (W - 1 - k) * 8;
        if (bit < 64)
          lo = (lo & ~(0xFFull << bit)) | (b << bit);
        else
          hi = (hi & ~(0xFFull << (bit - 64))) | (b << (bit - 64));
      }
      out[row * 2] = (int64_t)lo;
      out[row * 2 + 1] = (int64_t)hi;
    }
  }
}

}  // namespace srj

using namespace srj;

// ---------------------------------------------------------------------------
// DELTA encodings (parquet-mr v2 defaults; reference decodes via libcudf):
// DELTA_BINARY_PACKED ints, DELTA_LENGTH_BYTE_ARRAY and DELTA_BYTE_ARRAY
// strings, BYTE_STREAM_SPLIT floats. One wave per page; miniblocks unpack
// lane-parallel with a wave-wide prefix sum carrying the running value.
// ---------------------------------------------------------------------------
struct DeltaDesc {
  const uint8_t* src;   // DBP section start (page body)
  int64_t src_len;
  uint8_t* out;         // dense values, `width` bytes each, valid-order
  int64_t expected;     // capacity of out (upper bound on value count)
  int64_t* consumed;    // out: bytes consumed by the DBP section (or null)
  int32_t width;        // 4 or 8
  int32_t pad;
};

__device__ inline uint64_t dbp_varint(const uint8_t* p, int64_t len,
                                      int64_t& pos) {
  uint64_t v = 0;
  int sh = 0;
  while (pos < len && sh < 64) {
    uint8_t b = p[pos++];
    v |= (uint64_t)(b & 0x7F) << sh;
    if (!(b & 0x80)) break;
    sh += 7;
  }
  return v;
}

__device__ inline int64_t dbp_zigzag(uint64_t u) {
  return (int64_t)((u >> 1) ^ (~(u & 1) + 1));
}

// w bits starting at bit `bitoff` of p[0..plen), little-endian packing
__device__ inline uint64_t load_bits(const uint8_t* p, int64_t plen,
                                     int64_t bitoff, int w) {
  if (w <= 0) return 0;
  if (w > 64) w = 64;  // malformed miniblock width: clamp (no UB shifts)
  int64_t b0 = bitoff >> 3;
  int shift = (int)(bitoff & 7);
  unsigned __int128 acc = 0;
  int need = (shift + w + 7) >> 3;
  for (int b = 0; b < need && b0 + b < plen; ++b)
    acc |= (unsigned __int128)p[b0 + b] << (8 * b);
  uint64_t v = (uint64_t)(acc >> shift);
  if (w < 64) v &= (((uint64_t)1 << w) - 1);
  return v;
}

__device__ inline uint64_t shfl_u64(uint64_t v, int lane) {
  return (uint64_t)__shfl((long long)v, lane, 64);
}

__global__ void delta_binpack_kernel(const DeltaDesc* __restrict__ descs,
                                     int32_t npages) {
  int lane = threadIdx.x & 63;
  int wave = threadIdx.x >> 6;
  int wpb = blockDim.x >> 6;
  for (int32_t page = (int32_t)blockIdx.x * wpb + wave; page < npages;
       page += gridDim.x * wpb) {
    DeltaDesc d = descs[page];
    uint64_t block_size = 0, nmini = 0, total = 0;
    int64_t first = 0, pos = 0;
    if (lane == 0) {
      block_size = dbp_varint(d.src, d.src_len, pos);
      nmini = dbp_varint(d.src, d.src_len, pos);
      total = dbp_varint(d.src, d.src_len, pos);
      first = dbp_zigzag(dbp_varint(d.src, d.src_len, pos));
    }
    block_size = shfl_u64(block_size, 0);
    nmini = shfl_u64(nmini, 0);
    total = shfl_u64(total, 0);
    first = (int64_t)shfl_u64((uint64_t)first, 0);
    pos = (int64_t)shfl_u64((uint64_t)pos, 0);
    int64_t n = (int64_t)total < d.expected ? (int64_t)total : d.expected;
    int64_t vpm = nmini ? (int64_t)(block_size / nmini) : 0;
    if (n > 0 && lane == 0) {
      if (d.width == 8) reinterpret_cast<uint64_t*>(d.out)[0] = (uint64_t)first;
      else reinterpret_cast<uint32_t*>(d.out)[0] = (uint32_t)first;
    }
    uint64_t prev = (uint64_t)first;
    int64_t produced = 1;
    int64_t remaining = n > 0 ? n - 1 : 0;
    while (remaining > 0 && vpm > 0 && pos < d.src_len) {
      uint64_t md = 0;
      int64_t p2 = pos;
      if (lane == 0) md = dbp_varint(d.src, d.src_len, p2);
      md = shfl_u64(md, 0);
      p2 = (int64_t)shfl_u64((uint64_t)p2, 0);
      int64_t min_delta = dbp_zigzag(md);
      int64_t bw_pos = p2;
      int64_t data_pos = bw_pos + (int64_t)nmini;
      for (uint32_t mb = 0; mb < (uint32_t)nmini && remaining > 0; ++mb) {
        int w = (bw_pos + mb < d.src_len) ? d.src[bw_pos + mb] : 0;
        int64_t take = vpm < remaining ? vpm : remaining;
        for (int64_t base = 0; base < take; base += 64) {
          int64_t k = base + lane;
          uint64_t delta = 0;
          if (k < take)
            delta = load_bits(d.src + data_pos, d.src_len - data_pos,
                              k * (int64_t)w, w) + (uint64_t)min_delta;
          uint64_t scan = delta;
          for (int off = 1; off < 64; off <<= 1) {
            uint64_t nb = (uint64_t)__shfl_up((long long)scan, off, 64);
            if (lane >= off) scan += nb;
          }
          uint64_t v = prev + scan;
          if (k < take) {
            if (d.width == 8)
              reinterpret_cast<uint64_t*>(d.out)[produced + k] = v;
            else
              reinterpret_cast<uint32_t*>(d.out)[produced + k] = (uint32_t)v;
          }
          int64_t lastl = take - base - 1;
          prev = shfl_u64(v, (int)(lastl < 63 ? lastl : 63));
        }
        produced += take;
        remaining -= take;
        data_pos += (vpm * (int64_t)w + 7) >> 3;  // miniblocks are padded full
      }
      pos = data_pos;
    }
    // zero-fill any tail so downstream prefix sums see 0-length entries
    for (int64_t k = produced + lane; k < d.expected; k += 64) {
      if (d.width == 8) reinterpret_cast<uint64_t*>(d.out)[k] = 0;
      else reinterpret_cast<uint32_t*>(d.out)[k] = 0;
    }
    if (n == 0 && lane == 0 && d.expected > 0) {
      if (d.width == 8) reinterpret_cast<uint64_t*>(d.out)[0] = 0;
      else reinterpret_cast<uint32_t*>(d.out)[0] = 0;
    }
    if (lane == 0 && d.consumed) *d.consumed = pos;
  }
}

// generic per-page exclusive prefix over int32 lengths -> int64 offsets:
// val_len[v] = lens[v] (+ lens2[v]); val_off[v] = extra + *base_off + prefix
struct StrOffDesc {
  const int32_t* lens;
  const int32_t* lens2;     // null unless DELTA_BYTE_ARRAY (prefix+suffix)
  const int64_t* base_off;  // null or ptr to a device int64 (e.g. consumed)
  int64_t extra;
  int64_t num_values;
  int64_t* val_off;
  int32_t* val_len;
  int64_t* total_out;       // null or out: total bytes
};

__global__ void str_off_kernel(const StrOffDesc* __restrict__ descs,
                               int32_t npages) {
  int lane = threadIdx.x & 63;
  int wave = threadIdx.x >> 6;
  int wpb = blockDim.x >> 6;
  for (int32_t page = (int32_t)blockIdx.x * wpb + wave; page < npages;
       page += gridDim.x * wpb) {
    StrOffDesc d = descs[page];
    int64_t base = d.extra + (d.base_off ? *d.base_off : 0);
    int64_t run = 0;
    for (int64_t b = 0; b < d.num_values; b += 64) {
      int64_t v = b + lane;
      int64_t len = 0;
      if (v < d.num_values) {
        len = d.lens[v];
        if (d.lens2) len += d.lens2[v];
      }
      uint64_t scan = (uint64_t)len;  // inclusive
      for (int off = 1; off < 64; off <<= 1) {
        uint64_t nb = (uint64_t)__shfl_up((long long)scan, off, 64);
        if (lane >= off) scan += nb;
      }
      if (v < d.num_values) {
        d.val_off[v] = base + run + (int64_t)scan - len;  // exclusive
        d.val_len[v] = (int32_t)len;
      }
      int64_t lastl = d.num_values - b - 1;
      run += (int64_t)shfl_u64(scan, (int)(lastl < 63 ? lastl : 63));
    }
    if (lane == 0 && d.total_out) *d.total_out = run;
  }
}

// DELTA_BYTE_ARRAY reconstruction: value v = value[v-1][0:plen[v]] + suffix
struct DbaDesc {
  const uint8_t* src;        // page body (suffix bytes addressed absolutely)
  const int32_t* plen;
  const int32_t* slen;
  const int64_t* suf_off;    // per-value absolute offset of suffix in src
  const int64_t* val_off;    // per-value absolute offset in scratch
  int64_t num_values;
  uint8_t* scratch;
};

__global__ void dba_reconstruct_kernel(const DbaDesc* __restrict__ descs,
                                       int32_t npages) {
  int lane = threadIdx.x & 63;
  int wave = threadIdx.x >> 6;
  int wpb = blockDim.x >> 6;
  for (int32_t page = (int32_t)blockIdx.x * wpb + wave; page < npages;
       page += gridDim.x * wpb) {
    DbaDesc d = descs[page];
    for (int64_t v = 0; v < d.num_values; ++v) {
      uint8_t* dst = d.scratch + d.val_off[v];
      int32_t pl = d.plen[v], sl = d.slen[v];
      if (v > 0) {
        const uint8_t* prevp = d.scratch + d.val_off[v - 1];
        for (int32_t k = lane; k < pl; k += 64) dst[k] = prevp[k];
      }
      const uint8_t* suf = d.src + d.suf_off[v];
      for (int32_t k = lane; k < sl; k += 64) dst[pl + k] = suf[k];
      __builtin_amdgcn_s_waitcnt(0);  // writes visible before next iteration
    }
  }
}

// BYTE_STREAM_SPLIT: byte plane b of value v at src[b*n + v]
struct BssDesc {
  const uint8_t* src;
  int64_t n;           // value count (body_len / width)
  uint8_t* out;
  int32_t width;
  int32_t pad;
};

__global__ void bss_kernel(const BssDesc* __restrict__ descs,
                           int32_t npages) {
  for (int32_t page = blockIdx.x; page < npages; page += gridDim.x) {
    BssDesc d = descs[page];
    for (int64_t v = threadIdx.x; v < d.n; v += blockDim.x)
      for (int b = 0; b < d.width; ++b)
        d.out[v * d.width + b] = d.src[(int64_t)b * d.n + v];
  }
}

extern "C" {

void srj_pq_delta_binpack(const void* descs, int32_t npages,
                          hipStream_t stream) {
  if (npages == 0) return;
  int wpb = DEFAULT_BLOCK / 64;
  int blocks = (npages + wpb - 1) / wpb;
  delta_binpack_kernel<<<blocks < MAX_GRID ? blocks : MAX_GRID, DEFAULT_BLOCK,
                         0, stream>>>(
      reinterpret_cast<const DeltaDesc*>(descs), npages);
}

void srj_pq_str_off(const void* descs, int32_t npages, hipStream_t stream) {
  if (npages == 0) return;
  int wpb = DEFAULT_BLOCK / 64;
  int blocks = (npages + wpb - 1) / wpb;
  str_off_kernel<<<blocks < MAX_GRID ? blocks : MAX_GRID, DEFAULT_BLOCK, 0,
                   stream>>>(reinterpret_cast<const StrOffDesc*>(descs),
                             npages);
}

void srj_pq_dba_reconstruct(const void* descs, int32_t npages,
                            hipStream_t stream) {
  if (npages == 0) return;
  int wpb = DEFAULT_BLOCK / 64;
  int blocks = (npages + wpb - 1) / wpb;
  dba_reconstruct_kernel<<<blocks < MAX_GRID ? blocks : MAX_GRID,
                           DEFAULT_BLOCK, 0, stream>>>(
      reinterpret_cast<const DbaDesc*>(descs), npages);
}

void srj_pq_bss(const void* descs, int32_t npages, hipStream_t stream) {
  if (npages == 0) return;
  bss_kernel<<<npages < MAX_GRID ? npages : MAX_GRID, DEFAULT_BLOCK, 0,
               stream>>>(reinterpret_cast<const BssDesc*>(descs), npages);
}

void srj_rle_decode(const void* descs, int32_t npages, hipStream_t stream) {
  if (npages == 0) return;
  int64_t g = npages < MAX_GRID ? npages : MAX_GRID;
  rle_decode_kernel<<<g, DEFAULT_BLOCK, 0, stream>>>(
      reinterpret_cast<const RleDesc*>(descs), npages);
}

void srj_scatter_fixed(const void* descs, int32_t npages, uint8_t* out,
                       hipStream_t stream) {
  if (npages == 0) return;
  int64_t g = npages < MAX_GRID ? npages : MAX_GRID;
  scatter_fixed_kernel<<<g, DEFAULT_BLOCK, 0, stream>>>(
      reinterpret_cast<const ScatterDesc*>(descs), npages, out);
}

void srj_string_plain_index(const void* descs, int32_t npages,
                            hipStream_t stream) {
  if (npages == 0) return;
  int64_t g = npages < MAX_GRID ? npages : MAX_GRID;
  string_plain_index_kernel<<<g, 64, 0, stream>>>(
      reinterpret_cast<const StrIndexDesc*>(descs), npages);
}

void srj_string_copy(const void* descs, int32_t npages, int32_t phase,
                     int32_t* lens, const int32_t* offsets, uint8_t* chars,
                     hipStream_t stream) {
  if (npages == 0) return;
  int64_t g = npages < MAX_GRID ? npages : MAX_GRID;
  if (phase == 0)
    string_copy_kernel<0><<<g, DEFAULT_BLOCK, 0, stream>>>(
        reinterpret_cast<const StrCopyDesc*>(descs), npages, lens, nullptr,
        nullptr);
  else
    string_copy_kernel<1><<<g, DEFAULT_BLOCK, 0, stream>>>(
        reinterpret_cast<const StrCopyDesc*>(descs), npages, nullptr, offsets,
        chars);
}

void srj_def_to_validity(const uint8_t* def, int64_t nrows, uint8_t* validity,
                         hipStream_t stream) {
  def_to_validity_kernel<<<grid_1d(nrows), DEFAULT_BLOCK, 0, stream>>>(
      def, nrows, validity);
}

void srj_pq_snappy_decomp(const void* descs, int32_t n, hipStream_t stream) {
  int waves_per_block = DEFAULT_BLOCK / 64;
  int blocks = (n + waves_per_block - 1) / waves_per_block;
  snappy_decomp_kernel<<<blocks, DEFAULT_BLOCK, 0, stream>>>(
      reinterpret_cast<const SnapDesc*>(descs), n);
}

void srj_gather_u8_at(const uint64_t* addrs, int64_t n, uint8_t* out,
                      hipStream_t stream) {
  gather_u8_at_kernel<<<grid_1d(n), DEFAULT_BLOCK, 0, stream>>>(addrs, n, out);
}

void srj_pq_flba_dec128(const void* descs, int32_t npages, int64_t* out,
                        hipStream_t stream) {
  flba_dec128_kernel<<<npages < MAX_GRID ? (npages ? npages : 1) : MAX_GRID,
                       DEFAULT_BLOCK, 0, stream>>>(
      reinterpret_cast<const ScatterDesc*>(descs), npages, out);
}

}  // extern "C"
