// JCUDF row <-> columnar conversion.
//
// Reference parity: RowConversion.java:35-158 + row_conversion.cu (2,626 LoC,
// the reference's largest kernel file). Row layout (RowConversion.java:59-101):
// each row is a C-struct-like packing of the columns in order (each column
// aligned to its width), validity bytes (1 per 8 columns, bit i%8 of byte
// i/8, set = valid) immediately after the last column, row padded to 8 bytes.
//
// MI355X design: one thread per row; column loads are coalesced (consecutive
// lanes read consecutive rows); row-major writes go through L2 (64B lines
// amortize across a wave's 64 adjacent rows when row_size is small, the
// common Spark UDF case). A future LDS-tiled variant can stage 64x row_size
// tiles for fully-coalesced stores (reference copy_to_rows tile design);
// correctness and API shape come first.
#include "srj_common.hpp"

namespace srj {

struct RowColDesc {
  const void* data;       // column data (write target for from_rows)
  const uint8_t* valid;   // column validity (or write target)
  int32_t width;          // element bytes (1/2/4/8/16)
  int32_t row_off;        // byte offset within row
};

// load unconditionally, store value-or-zero: keeps the column loads
// branch-free so consecutive elements pipeline (counted vmcnt) instead of
// stalling on a per-element validity branch
__device__ inline void copy_elem_z(uint8_t* dst, const uint8_t* src, int w,
                                   bool valid) {
  switch (w) {
    case 1: *dst = valid ? *src : 0; break;
    case 2: *reinterpret_cast<uint16_t*>(dst) =
                valid ? *reinterpret_cast<const uint16_t*>(src) : 0;
            break;
    case 4: *reinterpret_cast<uint32_t*>(dst) =
                valid ? *reinterpret_cast<const uint32_t*>(src) : 0;
            break;
    case 8: *reinterpret_cast<uint64_t*>(dst) =
                valid ? *reinterpret_cast<const uint64_t*>(src) : 0;
            break;
    case 16: {
      uint64_t lo = *reinterpret_cast<const uint64_t*>(src);
      uint64_t hi = *reinterpret_cast<const uint64_t*>(src + 8);
      *reinterpret_cast<uint64_t*>(dst) = valid ? lo : 0;
      *reinterpret_cast<uint64_t*>(dst + 8) = valid ? hi : 0;
      break;
    }
  }
}

__device__ inline void copy_elem(uint8_t* dst, const uint8_t* src, int w) {
  switch (w) {
    case 1: *dst = *src; break;
    case 2: *reinterpret_cast<uint16_t*>(dst) = *reinterpret_cast<const uint16_t*>(src); break;
    case 4: *reinterpret_cast<uint32_t*>(dst) = *reinterpret_cast<const uint32_t*>(src); break;
    case 8: *reinterpret_cast<uint64_t*>(dst) = *reinterpret_cast<const uint64_t*>(src); break;
    case 16: {
      *reinterpret_cast<uint64_t*>(dst) = *reinterpret_cast<const uint64_t*>(src);
      *reinterpret_cast<uint64_t*>(dst + 8) =
          *reinterpret_cast<const uint64_t*>(src + 8);
      break;
    }
  }
}


// 4B-granular variants for the PITCHED LDS tile: the tile row pitch is
// padded to an odd dword count to spread strided row accesses across all
// 32 LDS banks (SQ_LDS_BANK_CONFLICT measured ~40x per instruction with
// the natural multiple-of-8 pitch), which caps tile alignment at 4B.
__device__ inline void copy_elem_z4(uint8_t* dst, const uint8_t* src, int w,
                                    bool valid) {
  switch (w) {
    case 1: *dst = valid ? *src : 0; break;
    case 2: *reinterpret_cast<uint16_t*>(dst) =
                valid ? *reinterpret_cast<const uint16_t*>(src) : 0;
            break;
    case 4: *reinterpret_cast<uint32_t*>(dst) =
                valid ? *reinterpret_cast<const uint32_t*>(src) : 0;
            break;
    case 8: {
      uint64_t v = valid ? *reinterpret_cast<const uint64_t*>(src) : 0;
      reinterpret_cast<uint32_t*>(dst)[0] = (uint32_t)v;
      reinterpret_cast<uint32_t*>(dst)[1] = (uint32_t)(v >> 32);
      break;
    }
    case 16: {
      uint64_t lo = *reinterpret_cast<const uint64_t*>(src);
      uint64_t hi = *reinterpret_cast<const uint64_t*>(src + 8);
      if (!valid) { lo = 0; hi = 0; }
      reinterpret_cast<uint32_t*>(dst)[0] = (uint32_t)lo;
      reinterpret_cast<uint32_t*>(dst)[1] = (uint32_t)(lo >> 32);
      reinterpret_cast<uint32_t*>(dst)[2] = (uint32_t)hi;
      reinterpret_cast<uint32_t*>(dst)[3] = (uint32_t)(hi >> 32);
      break;
    }
  }
}

__device__ inline void copy_elem4(uint8_t* dst, const uint8_t* src, int w) {
  switch (w) {
    case 1: *dst = *src; break;
    case 2: *reinterpret_cast<uint16_t*>(dst) =
                *reinterpret_cast<const uint16_t*>(src);
            break;
    case 4: *reinterpret_cast<uint32_t*>(dst) =
                *reinterpret_cast<const uint32_t*>(src);
            break;
    case 8: {
      uint64_t v = ((uint64_t)reinterpret_cast<const uint32_t*>(src)[1]
                    << 32) | reinterpret_cast<const uint32_t*>(src)[0];
      *reinterpret_cast<uint64_t*>(dst) = v;
      break;
    }
    case 16: {
      const uint32_t* s = reinterpret_cast<const uint32_t*>(src);
      uint64_t lo = ((uint64_t)s[1] << 32) | s[0];
      uint64_t hi = ((uint64_t)s[3] << 32) | s[2];
      *reinterpret_cast<uint64_t*>(dst) = lo;
      *reinterpret_cast<uint64_t*>(dst + 8) = hi;
      break;
    }
  }
}

__device__ inline int32_t lds_pitch(int32_t row_size) {
  // odd dword count -> stride coprime with the 32 LDS banks
  return ((row_size >> 2) & 1) ? row_size : row_size + 4;
}

__global__ void to_rows_kernel(const RowColDesc* __restrict__ cols, int32_t ncols,
                               int64_t nrows, int32_t row_size,
                               int32_t validity_off, uint8_t* __restrict__ out) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t row = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; row < nrows;
       row += stride) {
    uint8_t* r = out + row * row_size;
    uint8_t vbyte = 0;
    int32_t vcount = 0;
    for (int32_t c = 0; c < ncols; ++c) {
      const RowColDesc& d = cols[c];
      bool valid = is_valid(d.valid, row);
      if (valid) {
        copy_elem(r + d.row_off,
                  reinterpret_cast<const uint8_t*>(d.data) + row * d.width,
                  d.width);
      } else {
        for (int b = 0; b < d.width; ++b) r[d.row_off + b] = 0;
      }
      vbyte |= (uint8_t)valid << (c & 7);
      if ((c & 7) == 7) {
        r[validity_off + (c >> 3)] = vbyte;
        vbyte = 0;
      }
      ++vcount;
    }
    if (ncols & 7) r[validity_off + (ncols >> 3)] = vbyte;
  }
}

__global__ void from_rows_kernel(const RowColDesc* __restrict__ cols,
                                 int32_t ncols, int64_t nrows, int32_t row_size,
                                 int32_t validity_off,
                                 const uint8_t* __restrict__ in) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  int64_t nrows_pad = (nrows + WAVE - 1) & ~(int64_t)(WAVE - 1);
  for (int64_t row = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       row < nrows_pad; row += stride) {
    bool in_range = row < nrows;
    const uint8_t* r = in + (in_range ? row : 0) * row_size;
    for (int32_t c = 0; c < ncols; ++c) {
      const RowColDesc& d = cols[c];
      bool valid =
          in_range && ((r[validity_off + (c >> 3)] >> (c & 7)) & 1);
      if (in_range) {
        copy_elem(const_cast<uint8_t*>(
                      reinterpret_cast<const uint8_t*>(d.data)) + row * d.width,
                  r + d.row_off, d.width);
      }
      if (d.valid != nullptr) {
        ballot_write_validity(const_cast<uint8_t*>(d.valid), row, valid);
      }
    }
  }
}


// ---------------------------------------------------------------------------
// LDS-tiled fixed-width transpose (reference copy_to_rows/copy_from_rows
// tile design, row_conversion.cu:591/912): a workgroup stages TILE_ROWS rows
// in LDS, so the global side of both directions is fully coalesced — column
// reads are 64-lane contiguous AND row writes leave as wide dword4 copies
// instead of per-element scattered stores. Used when 64 rows of the layout
// fit in LDS; the one-thread-per-row kernels above remain the wide-row
// fallback.
// ---------------------------------------------------------------------------
constexpr int TILE_ROWS = 64;

__global__ void to_rows_tiled_kernel(const RowColDesc* __restrict__ cols,
                                     int32_t ncols, int64_t nrows,
                                     int32_t row_size, int32_t validity_off,
                                     uint8_t* __restrict__ out) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  uint8_t* tile = reinterpret_cast<uint8_t*>(smem);
  int32_t const pitch = lds_pitch(row_size);
  int32_t const rs8 = row_size >> 3;
  int64_t ntiles = (nrows + TILE_ROWS - 1) / TILE_ROWS;
  for (int64_t t = blockIdx.x; t < ntiles; t += gridDim.x) {
    int64_t row0 = t * TILE_ROWS;
    int32_t m = (int32_t)(nrows - row0 < TILE_ROWS ? nrows - row0 : TILE_ROWS);
    // zero the tile (padding + null slots are zero by contract)
    for (int32_t i = threadIdx.x * 4; i < TILE_ROWS * pitch;
         i += blockDim.x * 4)
      *reinterpret_cast<uint32_t*>(tile + i) = 0;
    __syncthreads();
    // data: flat (column, tile-row) work — each wave covers one column's 64
    // consecutive rows, so the global column loads are fully coalesced and
    // the whole block stays busy
    for (int32_t w = threadIdx.x; w < ncols * TILE_ROWS; w += blockDim.x) {
      int32_t c = w >> 6;
      int32_t r = w & (TILE_ROWS - 1);
      if (r >= m) continue;
      int64_t row = row0 + r;
      const RowColDesc& d = cols[c];
      copy_elem_z4(tile + (int64_t)r * pitch + d.row_off,
                   reinterpret_cast<const uint8_t*>(d.data) + row * d.width,
                   d.width, is_valid(d.valid, row));
    }
    // validity bytes: one thread per tile row (no cross-thread byte RMW)
    for (int32_t r = threadIdx.x; r < TILE_ROWS; r += blockDim.x) {
      if (r >= m) continue;
      int64_t row = row0 + r;
      uint8_t* dst = tile + (int64_t)r * pitch;
      uint8_t vbyte = 0;
      for (int32_t c = 0; c < ncols; ++c) {
        vbyte |= (uint8_t)is_valid(cols[c].valid, row) << (c & 7);
        if ((c & 7) == 7) {
          dst[validity_off + (c >> 3)] = vbyte;
          vbyte = 0;
        }
      }
      if (ncols & 7) dst[validity_off + (ncols >> 3)] = vbyte;
    }
    __syncthreads();
    // coalesced copy LDS -> global in 8B grains (global u64 stores stay
    // wide; the pitched LDS side reads as 2x u32)
    uint8_t* gdst = out + row0 * row_size;
    int32_t ngrains = m * rs8;
    for (int32_t g = threadIdx.x; g < ngrains; g += blockDim.x) {
      int32_t r = g / rs8;
      int32_t o = (g - r * rs8) << 3;
      const uint32_t* lp =
          reinterpret_cast<const uint32_t*>(tile + (int64_t)r * pitch + o);
      uint64_t v = ((uint64_t)lp[1] << 32) | lp[0];
      *reinterpret_cast<uint64_t*>(gdst + (int64_t)r * row_size + o) = v;
    }
    __syncthreads();
  }
}

// TR (tile rows) is a template parameter: bigger tiles turn the column
// write side into longer contiguous bursts (24 column streams x 512B at
// TR=64 round-robin poorly through HBM write combining; TR=256 makes
// them 2KB) at the cost of LDS per block.
template <int TR>
__global__ void from_rows_tiled_kernel(const RowColDesc* __restrict__ cols,
                                       int32_t ncols, int64_t nrows,
                                       int32_t row_size, int32_t validity_off,
                                       const uint8_t* __restrict__ in) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  uint8_t* tile = reinterpret_cast<uint8_t*>(smem);
  int64_t ntiles = (nrows + TR - 1) / TR;
  for (int64_t t = blockIdx.x; t < ntiles; t += gridDim.x) {
    int64_t row0 = t * TR;
    int32_t m = (int32_t)(nrows - row0 < TR ? nrows - row0 : TR);
    const uint8_t* gsrc = in + row0 * row_size;
    int32_t const pitch = lds_pitch(row_size);
    int32_t const rs8 = row_size >> 3;
    // coalesced copy global -> LDS in 8B grains (pitched LDS writes as
    // 2x u32)
    int32_t ngrains = m * rs8;
    for (int32_t g = threadIdx.x; g < ngrains; g += blockDim.x) {
      int32_t r = g / rs8;
      int32_t o = (g - r * rs8) << 3;
      uint64_t v =
          *reinterpret_cast<const uint64_t*>(gsrc + (int64_t)r * row_size + o);
      uint32_t* lp =
          reinterpret_cast<uint32_t*>(tile + (int64_t)r * pitch + o);
      lp[0] = (uint32_t)v;
      lp[1] = (uint32_t)(v >> 32);
    }
    __syncthreads();
    // data: flat (column, tile-row) work — coalesced global column writes
    for (int32_t w = threadIdx.x; w < ncols * TR; w += blockDim.x) {
      int32_t c = w / TR;
      int32_t r = w % TR;
      if (r >= m) continue;
      const RowColDesc& d = cols[c];
      copy_elem4(const_cast<uint8_t*>(
                     reinterpret_cast<const uint8_t*>(d.data)) +
                     (row0 + r) * d.width,
                 tile + (int64_t)r * pitch + d.row_off, d.width);
    }
    // validity: (column, 64-row subtile) pairs round-robin over all waves
    // (each wave's 64 lanes map 1:1 onto the subtile's rows, so one
    // ballot covers one validity word)
    {
      int32_t wave = threadIdx.x / WAVE;
      int32_t nwaves = blockDim.x / WAVE;
      int32_t lane = threadIdx.x & (WAVE - 1);
      int32_t nsub = TR / WAVE;
      for (int32_t cs = wave; cs < ncols * nsub; cs += nwaves) {
        int32_t c = cs / nsub;
        int32_t sub = cs % nsub;
        int32_t r = sub * WAVE + lane;
        bool in_range = r < m;
        const uint8_t* src = tile + (int64_t)(in_range ? r : 0) * pitch;
        int64_t row = row0 + r;
        const RowColDesc& d = cols[c];
        bool valid =
            in_range && ((src[validity_off + (c >> 3)] >> (c & 7)) & 1);
        if (d.valid != nullptr) {
          ballot_write_validity(const_cast<uint8_t*>(d.valid), row, valid);
        }
      }
    }
    __syncthreads();
  }
}

// ---------------------------------------------------------------------------
// variable-width (strings) JCUDF path (reference copy_strings_to_rows
// row_conversion.cu:839 / copy_strings_from_rows:1145): string columns hold
// an (offset-in-row, length) int32 pair in the fixed section; string bytes
// are appended after the validity bytes; rows are 8-byte aligned with a
// row-offsets array (build_string_row_offsets analog runs as a size kernel +
// host cumsum).
// ---------------------------------------------------------------------------
__global__ void var_row_sizes_kernel(const RowColDesc* __restrict__ cols,
                                     int32_t ncols, int64_t nrows,
                                     int32_t fixed_size,
                                     int32_t* __restrict__ sizes) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t row = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; row < nrows;
       row += stride) {
    int32_t sz = fixed_size;
    for (int32_t c = 0; c < ncols; ++c) {
      const RowColDesc& d = cols[c];
      if (d.width != 0) continue;  // fixed col
      const int32_t* offs = reinterpret_cast<const int32_t*>(d.data);
      if (is_valid(d.valid, row)) sz += offs[row + 1] - offs[row];
    }
    sizes[row] = (sz + 7) & ~7;
  }
}

// var col RowColDesc: width == 0, data = offsets ptr, row_off = fixed-section
// position of the (offset,len) pair; chars pointer passed separately.
__global__ void to_rows_var_kernel(const RowColDesc* __restrict__ cols,
                                   const uint64_t* __restrict__ char_ptrs,
                                   int32_t ncols, int64_t nrows,
                                   int32_t fixed_size, int32_t validity_off,
                                   const int32_t* __restrict__ row_offsets,
                                   uint8_t* __restrict__ out) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t row = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; row < nrows;
       row += stride) {
    uint8_t* r = out + row_offsets[row];
    int32_t var_pos = fixed_size;
    uint8_t vbyte = 0;
    for (int32_t c = 0; c < ncols; ++c) {
      const RowColDesc& d = cols[c];
      bool valid = is_valid(d.valid, row);
      if (d.width == 0) {
        const int32_t* offs = reinterpret_cast<const int32_t*>(d.data);
        int32_t len = valid ? offs[row + 1] - offs[row] : 0;
        reinterpret_cast<int32_t*>(r + d.row_off)[0] = var_pos;
        reinterpret_cast<int32_t*>(r + d.row_off)[1] = len;
        if (valid) {
          const char* src = reinterpret_cast<const char*>(char_ptrs[c]) +
                            offs[row];
          for (int32_t k = 0; k < len; ++k) r[var_pos + k] = src[k];
        }
        var_pos += len;
      } else if (valid) {
        copy_elem(r + d.row_off,
                  reinterpret_cast<const uint8_t*>(d.data) + row * d.width,
                  d.width);
      } else {
        for (int b = 0; b < d.width; ++b) r[d.row_off + b] = 0;
      }
      vbyte |= (uint8_t)valid << (c & 7);
      if ((c & 7) == 7) {
        r[validity_off + (c >> 3)] = vbyte;
        vbyte = 0;
      }
    }
    if (ncols & 7) r[validity_off + (ncols >> 3)] = vbyte;
  }
}

// from rows: phase 0 emits per-row string lengths per var column; phase 1
// copies fixed cols + chars using the per-column char offsets (cumsum'd).
template <int PHASE>
__global__ void from_rows_var_kernel(const RowColDesc* __restrict__ cols,
                                     const uint64_t* __restrict__ char_ptrs,
                                     const uint64_t* __restrict__ len_ptrs,
                                     int32_t ncols, int64_t nrows,
                                     int32_t validity_off,
                                     const int32_t* __restrict__ row_offsets,
                                     const uint8_t* __restrict__ in) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  int64_t npad = (nrows + WAVE - 1) & ~(int64_t)(WAVE - 1);
  for (int64_t row = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       row < npad; row += stride) {
    bool in_range = row < nrows;
    const uint8_t* r = in + (in_range ? row_offsets[row] : 0);
    for (int32_t c = 0; c < ncols; ++c) {
      const RowColDesc& d = cols[c];
      bool valid = in_range && ((r[validity_off + (c >> 3)] >> (c & 7)) & 1);
      if (d.width == 0) {
        int32_t pos = in_range ? reinterpret_cast<const int32_t*>(r + d.row_off)[0] : 0;
        int32_t len = (in_range && valid)
                          ? reinterpret_cast<const int32_t*>(r + d.row_off)[1]
                          : 0;
        if (PHASE == 0) {
          if (in_range)
            reinterpret_cast<int32_t*>(len_ptrs[c])[row] = len;
        } else if (in_range && valid) {
          // out offsets (cumsum of lengths) live where len_ptrs points now
          const int32_t* ooffs = reinterpret_cast<const int32_t*>(len_ptrs[c]);
          char* dst = reinterpret_cast<char*>(char_ptrs[c]) + ooffs[row];
          for (int32_t k = 0; k < len; ++k) dst[k] = (char)r[pos + k];
        }
      } else if (PHASE == 1 && in_range) {
        copy_elem(const_cast<uint8_t*>(
                      reinterpret_cast<const uint8_t*>(d.data)) + row * d.width,
                  r + d.row_off, d.width);
      }
      if (PHASE == 1 && d.valid != nullptr) {
        ballot_write_validity(const_cast<uint8_t*>(d.valid), row, valid);
      }
    }
  }
}

template <int TR>
void launch_from_rows_tiled(const void* cols, int32_t ncols,
                                   int64_t nrows, int32_t row_size,
                                   int32_t validity_off, const uint8_t* in,
                                   hipStream_t stream) {
  int32_t pitch = ((row_size >> 2) & 1) ? row_size : row_size + 4;
  size_t lds = (size_t)TR * pitch;
  int64_t ntiles = (nrows + TR - 1) / TR;
  int64_t nblk = ntiles < MAX_GRID ? ntiles : MAX_GRID;
  from_rows_tiled_kernel<TR><<<nblk, DEFAULT_BLOCK, lds, stream>>>(
      reinterpret_cast<const RowColDesc*>(cols), ncols, nrows, row_size,
      validity_off, in);
}

}  // namespace srj

using namespace srj;

extern "C" {

void srj_to_rows(const void* cols, int32_t ncols, int64_t nrows, int32_t row_size,
                 int32_t validity_off, uint8_t* out, hipStream_t stream) {
  int32_t pitch = ((row_size >> 2) & 1) ? row_size : row_size + 4;
  size_t lds = (size_t)TILE_ROWS * pitch;
  if (lds <= 64 * 1024) {
    int64_t ntiles = (nrows + TILE_ROWS - 1) / TILE_ROWS;
    int64_t nblk = ntiles < MAX_GRID ? ntiles : MAX_GRID;
    to_rows_tiled_kernel<<<nblk, DEFAULT_BLOCK, lds, stream>>>(
        reinterpret_cast<const RowColDesc*>(cols), ncols, nrows, row_size,
        validity_off, out);
    return;
  }
  to_rows_kernel<<<grid_1d(nrows), DEFAULT_BLOCK, 0, stream>>>(
      reinterpret_cast<const RowColDesc*>(cols), ncols, nrows, row_size,
      validity_off, out);
}


void srj_from_rows(const void* cols, int32_t ncols, int64_t nrows,
                   int32_t row_size, int32_t validity_off, const uint8_t* in,
                   hipStream_t stream) {
  // biggest tile whose LDS fits 64 KB: larger tiles give longer
  // per-column write bursts (SRJ_FROM_ROWS_TILE overrides for tuning)
  int tr = 0;
  if (const char* env = getenv("SRJ_FROM_ROWS_TILE")) tr = atoi(env);
  size_t l64 = 64 * 1024;
  size_t pitch = ((row_size >> 2) & 1) ? row_size : row_size + 4;
  if (tr == 0)
    tr = (256 * pitch <= l64) ? 256
         : (128 * pitch <= l64) ? 128
         : (64 * pitch <= l64) ? 64 : 0;
  if (tr == 256 && 256 * pitch <= l64) {
    srj::launch_from_rows_tiled<256>(cols, ncols, nrows, row_size, validity_off,
                                in, stream);
    return;
  }
  if (tr == 128 && 128 * pitch <= l64) {
    srj::launch_from_rows_tiled<128>(cols, ncols, nrows, row_size, validity_off,
                                in, stream);
    return;
  }
  if (tr == 64 && 64 * pitch <= l64) {
    srj::launch_from_rows_tiled<64>(cols, ncols, nrows, row_size, validity_off,
                               in, stream);
    return;
  }
  from_rows_kernel<<<grid_1d(nrows), DEFAULT_BLOCK, 0, stream>>>(
      reinterpret_cast<const RowColDesc*>(cols), ncols, nrows, row_size,
      validity_off, in);
}

void srj_var_row_sizes(const void* cols, int32_t ncols, int64_t nrows,
                       int32_t fixed_size, int32_t* sizes, hipStream_t stream) {
  var_row_sizes_kernel<<<grid_1d(nrows), DEFAULT_BLOCK, 0, stream>>>(
      reinterpret_cast<const RowColDesc*>(cols), ncols, nrows, fixed_size,
      sizes);
}

void srj_to_rows_var(const void* cols, const uint64_t* char_ptrs, int32_t ncols,
                     int64_t nrows, int32_t fixed_size, int32_t validity_off,
                     const int32_t* row_offsets, uint8_t* out,
                     hipStream_t stream) {
  to_rows_var_kernel<<<grid_1d(nrows), DEFAULT_BLOCK, 0, stream>>>(
      reinterpret_cast<const RowColDesc*>(cols), char_ptrs, ncols, nrows,
      fixed_size, validity_off, row_offsets, out);
}

void srj_from_rows_var(const void* cols, const uint64_t* char_ptrs,
                       const uint64_t* len_ptrs, int32_t ncols, int64_t nrows,
                       int32_t validity_off, const int32_t* row_offsets,
                       const uint8_t* in, int32_t phase, hipStream_t stream) {
  if (phase == 0)
    from_rows_var_kernel<0><<<grid_1d(nrows), DEFAULT_BLOCK, 0, stream>>>(
        reinterpret_cast<const RowColDesc*>(cols), char_ptrs, len_ptrs, ncols,
        nrows, validity_off, row_offsets, in);
  else
    from_rows_var_kernel<1><<<grid_1d(nrows), DEFAULT_BLOCK, 0, stream>>>(
        reinterpret_cast<const RowColDesc*>(cols), char_ptrs, len_ptrs, ncols,
        nrows, validity_off, row_offsets, in);
}

}  // extern "C"
