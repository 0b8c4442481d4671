// Device-side shuffle serialization kernels (Kudo GPU serializer).
//
// Reference parity: KudoGpuSerializer.splitAndSerializeToDevice /
// assembleFromDeviceRaw and the shuffle_split/shuffle_assemble native pair
// (shuffle_split.hpp:136,183). The MI355X design replaces the reference's
// src/dst buf-info grids with three generic kernels driven by host-built
// descriptor arrays:
//   * segmented_copy   — batched byte copies (headers, validity slices,
//                        unadjusted offsets, data slices) with a 4B-aligned
//                        fast path; one block per 16 KiB chunk, chunk->segment
//                        lookup by binary search over a host prefix array.
//   * validity_merge   — bit-shifted merge of unaligned validity slices into
//                        the assembled column (64-bit words, atomicOr on the
//                        piece-boundary words only) — the wave64 re-design of
//                        the reference's copy_validity (shuffle_assemble.cu:1355).
//   * offsets_rebase   — dst[i] = src[i] - src[0] + base per piece
//                        (reference copy_offsets, shuffle_assemble.cu:1537).
#include "srj_common.hpp"

namespace srj {

struct CopySeg {
  const uint8_t* src;
  uint8_t* dst;
  int64_t nbytes;
};

constexpr int64_t COPY_CHUNK = 16384;  // bytes per block per chunk

__global__ void segmented_copy_kernel(const CopySeg* __restrict__ segs,
                                      const int64_t* __restrict__ chunk_prefix,
                                      int32_t nsegs, int64_t total_chunks) {
  for (int64_t chunk = blockIdx.x; chunk < total_chunks; chunk += gridDim.x) {
    // binary search: greatest s with chunk_prefix[s] <= chunk
    int32_t lo = 0, hi = nsegs - 1;
    while (lo < hi) {
      int32_t mid = (lo + hi + 1) >> 1;
      if (chunk_prefix[mid] <= chunk) lo = mid;
      else hi = mid - 1;
    }
    const CopySeg seg = segs[lo];
    int64_t off = (chunk - chunk_prefix[lo]) * COPY_CHUNK;
    int64_t n = seg.nbytes - off < COPY_CHUNK ? seg.nbytes - off : COPY_CHUNK;
    const uint8_t* s = seg.src + off;
    uint8_t* d = seg.dst + off;
    // 4B fast path when relative alignment matches
    if ((((uintptr_t)s) & 3) == (((uintptr_t)d) & 3)) {
      int64_t head = (4 - (((uintptr_t)d) & 3)) & 3;
      if (head > n) head = n;
      for (int64_t i = threadIdx.x; i < head; i += blockDim.x) d[i] = s[i];
      int64_t body = (n - head) & ~(int64_t)3;
      const uint32_t* s4 = reinterpret_cast<const uint32_t*>(s + head);
      uint32_t* d4 = reinterpret_cast<uint32_t*>(d + head);
      for (int64_t i = threadIdx.x; i * 4 < body; i += blockDim.x)
        d4[i] = s4[i];
      for (int64_t i = head + body + threadIdx.x; i < n; i += blockDim.x)
        d[i] = s[i];
    } else {
      for (int64_t i = threadIdx.x; i < n; i += blockDim.x) d[i] = s[i];
    }
  }
}

// ---------------------------------------------------------------------------
// validity merge: piece -> dst bit range
// ---------------------------------------------------------------------------
struct ValiditySeg {
  const uint8_t* src;  // validity slice bytes (bit 0 = src_start_bit's byte)
  uint8_t* dst;        // output bitmask base
  int64_t src_start_bit;  // within src (0..7)
  int64_t dst_start_bit;  // absolute bit position in dst
  int64_t nbits;
};

__device__ inline uint64_t load_src_word(const uint8_t* src, int64_t bit_off,
                                         int64_t nbits_avail) {
  // load 64 bits starting at bit_off (byte-granular reads, bounds-checked)
  int64_t byte0 = bit_off >> 3;
  int shift = (int)(bit_off & 7);
  int64_t total_bytes = (bit_off + nbits_avail + 7) >> 3;
  uint64_t lo = 0, hi = 0;
  for (int i = 0; i < 8; ++i) {
    int64_t b = byte0 + i;
    if (b < total_bytes) lo |= (uint64_t)src[b] << (8 * i);
  }
  if (shift) {
    int64_t b = byte0 + 8;
    if (b < total_bytes) hi = (uint64_t)src[b];
    return (lo >> shift) | (hi << (64 - shift));
  }
  return lo;
}

__global__ void validity_merge_kernel(const ValiditySeg* __restrict__ segs,
                                      const int64_t* __restrict__ word_prefix,
                                      int32_t nsegs, int64_t total_words) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t w = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       w < total_words; w += stride) {
    int32_t lo = 0, hi = nsegs - 1;
    while (lo < hi) {
      int32_t mid = (lo + hi + 1) >> 1;
      if (word_prefix[mid] <= w) lo = mid;
      else hi = mid - 1;
    }
    const ValiditySeg seg = segs[lo];
    int64_t wi = w - word_prefix[lo];  // word index within this piece's span
    // dst words this piece touches: [dst_start_bit>>6 .. (dst_start_bit+nbits-1)>>6]
    int64_t dw = (seg.dst_start_bit >> 6) + wi;
    int64_t dst_word_bit0 = dw << 6;
    // bits of this piece that fall into dst word dw:
    int64_t piece_bit0 = dst_word_bit0 > seg.dst_start_bit
                             ? dst_word_bit0 - seg.dst_start_bit : 0;
    int64_t dst_bit_in_word = seg.dst_start_bit + piece_bit0 - dst_word_bit0;
    int64_t navail = seg.nbits - piece_bit0;
    if (navail <= 0) continue;
    int64_t ntake = 64 - dst_bit_in_word;
    if (ntake > navail) ntake = navail;
    // src == null: piece had no validity buffer -> all rows valid
    uint64_t bits = seg.src == nullptr
                        ? ~0ull
                        : load_src_word(seg.src, seg.src_start_bit + piece_bit0,
                                        navail);
    if (ntake < 64) bits &= (1ull << ntake) - 1ull;
    uint64_t word = bits << dst_bit_in_word;
    uint64_t* dst_words = reinterpret_cast<uint64_t*>(seg.dst);
    bool partial = dst_bit_in_word != 0 || ntake < 64;
    if (partial) {
      atomicOr(reinterpret_cast<unsigned long long*>(dst_words + dw),
               (unsigned long long)word);
    } else {
      dst_words[dw] = word;
    }
  }
}

// ---------------------------------------------------------------------------
// offsets rebase
// ---------------------------------------------------------------------------
struct OffsetSeg {
  const int32_t* src;
  int32_t* dst;
  int64_t n;        // number of offsets to write (nrows, excluding final)
  int32_t base;     // dst char/elem base for this piece
  int32_t write_last;  // 1 for the final piece of a column: also write dst[n]
};

__global__ void offsets_rebase_kernel(const OffsetSeg* __restrict__ segs,
                                      const int64_t* __restrict__ prefix,
                                      int32_t nsegs, int64_t total) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += stride) {
    int32_t lo = 0, hi = nsegs - 1;
    while (lo < hi) {
      int32_t mid = (lo + hi + 1) >> 1;
      if (prefix[mid] <= i) lo = mid;
      else hi = mid - 1;
    }
    const OffsetSeg seg = segs[lo];
    int64_t j = i - prefix[lo];
    int32_t first = seg.src[0];
    int64_t limit = seg.n + (seg.write_last ? 1 : 0);
    if (j < limit) seg.dst[j] = seg.src[j] - first + seg.base;
  }
}

// gather int32 values at arbitrary device addresses (for reading piece
// offset[first]/offset[last] during assemble planning)
__global__ void gather_i32_at_kernel(const uint64_t* __restrict__ addrs,
                                     int32_t n, int32_t* __restrict__ out) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) out[i] = *reinterpret_cast<const int32_t*>(addrs[i]);
}

}  // namespace srj

using namespace srj;

extern "C" {

void srj_segmented_copy(const void* segs, const int64_t* chunk_prefix,
                        int32_t nsegs, int64_t total_chunks, hipStream_t stream) {
  if (total_chunks == 0 || nsegs == 0) return;
  int64_t blocks = total_chunks < MAX_GRID ? total_chunks : MAX_GRID;
  segmented_copy_kernel<<<blocks, DEFAULT_BLOCK, 0, stream>>>(
      reinterpret_cast<const CopySeg*>(segs), chunk_prefix, nsegs, total_chunks);
}

void srj_validity_merge(const void* segs, const int64_t* word_prefix,
                        int32_t nsegs, int64_t total_words, hipStream_t stream) {
  if (total_words == 0 || nsegs == 0) return;
  validity_merge_kernel<<<grid_1d(total_words), DEFAULT_BLOCK, 0, stream>>>(
      reinterpret_cast<const ValiditySeg*>(segs), word_prefix, nsegs, total_words);
}

void srj_offsets_rebase(const void* segs, const int64_t* prefix, int32_t nsegs,
                        int64_t total, hipStream_t stream) {
  if (total == 0 || nsegs == 0) return;
  offsets_rebase_kernel<<<grid_1d(total), DEFAULT_BLOCK, 0, stream>>>(
      reinterpret_cast<const OffsetSeg*>(segs), prefix, nsegs, total);
}

void srj_gather_i32_at(const uint64_t* addrs, int32_t n, int32_t* out,
                       hipStream_t stream) {
  if (n == 0) return;
  gather_i32_at_kernel<<<(n + 255) / 256, 256, 0, stream>>>(addrs, n, out);
}

}  // extern "C"
