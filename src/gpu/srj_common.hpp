// Common device-side infrastructure for the MI355X-native Spark columnar engine.
//
// MI355X/CDNA4 ground rules baked in here (see /opt/skills/guides):
//  * wave = 64 lanes; __ballot() returns a 64-bit mask. Validity bitmasks are
//    written one aligned 64-bit word per wave via ballot (the reference's
//    32-bit warp-ballot idiom, re-done for wave64).
//  * memory-bound kernels use grid-stride loops capped at ~2048 blocks
//    (256 CUs x 8 blocks) so the launch fills all 8 XCDs without oversubscribing.
//  * row indices are int64 (10B-row configs are first-class).
//
// Columnar layout (Arrow-compatible, matches the reference's cudf layout so the
// Kudo wire format stays byte-identical):
//  * validity: 1 bit per row, LSB-first within a byte, buffer padded to a
//    multiple of 8 bytes so kernels may write whole uint64 words.
//  * strings: int32 offsets[n+1] + uint8 char data (2GB per column; batching
//    above this layer, as in the reference's row_conversion 2GB batches).
#pragma once
#include <hip/hip_runtime.h>
#include <cstdint>

namespace srj {

constexpr int WAVE = 64;
constexpr int DEFAULT_BLOCK = 256;
// 256 CUs x 8 workgroups of 256 threads per CU
constexpr int64_t MAX_GRID = 2048;

inline int64_t grid_1d(int64_t n, int block = DEFAULT_BLOCK, int64_t cap = MAX_GRID) {
  int64_t g = (n + block - 1) / block;
  return g < cap ? (g > 0 ? g : 1) : cap;
}

// ---------------------------------------------------------------------------
// dtype ids shared with Python (spark_rapids_jni_amd/columnar.py keeps the
// authoritative mirror; keep the numeric values in sync).
// ---------------------------------------------------------------------------
enum DType : int32_t {
  BOOL8 = 0,
  INT8 = 1,
  INT16 = 2,
  INT32 = 3,
  INT64 = 4,
  FLOAT32 = 5,
  FLOAT64 = 6,
  DATE32 = 7,        // days since epoch
  TIMESTAMP_US = 8,  // micros since epoch
  STRING = 9,
  DECIMAL32 = 10,
  DECIMAL64 = 11,
  DECIMAL128 = 12,
  LIST = 13,
  STRUCT = 14,
};

__host__ __device__ inline int dtype_size(int32_t t) {
  switch (t) {
    case BOOL8: case INT8: return 1;
    case INT16: return 2;
    case INT32: case DATE32: case DECIMAL32: return 4;
    case INT64: case TIMESTAMP_US: case DECIMAL64: case FLOAT64: return 8;
    case FLOAT32: return 4;
    case DECIMAL128: return 16;
    default: return 0;  // STRING/LIST/STRUCT have no fixed width
  }
}

// ---------------------------------------------------------------------------
// Column descriptor passed across the C ABI. Flat (nested columns are passed
// as a flattened pre-order array; `child0` indexes into that array).
// ---------------------------------------------------------------------------
struct ColDesc {
  int32_t dtype;
  int32_t scale;            // decimal scale (Spark: positive = digits right of dot)
  const void* data;         // fixed-width values or string chars
  const uint8_t* valid;     // validity bitmask, may be null (= all valid)
  const int32_t* offsets;   // strings/lists: n+1 entries
  int32_t num_children;
  int32_t child0;           // index of first child in flattened array
  int64_t size;             // number of rows
};

// ---------------------------------------------------------------------------
// validity helpers
// ---------------------------------------------------------------------------
__device__ inline bool is_valid(const uint8_t* mask, int64_t i) {
  return mask == nullptr || ((mask[i >> 3] >> (i & 7)) & 1);
}

// Each wave handles 64 consecutive rows (thread i -> row base+lane). All 64
// lanes must call this; writes one aligned uint64 word per wave.
__device__ inline void ballot_write_validity(uint8_t* mask, int64_t row, bool valid) {
  uint64_t word = __ballot(valid);
  if ((threadIdx.x & (WAVE - 1)) == 0 && mask != nullptr) {
    reinterpret_cast<uint64_t*>(mask)[row >> 6] = word;
  }
}

// Count of set bits in a bitmask over [0, n) — for null_count computation.
__device__ inline uint64_t word_with_tail_masked(const uint64_t* words, int64_t w,
                                                 int64_t n) {
  uint64_t v = words[w];
  int64_t bits_left = n - (w << 6);
  if (bits_left < 64) v &= (1ull << bits_left) - 1ull;
  return v;
}

// ---------------------------------------------------------------------------
// string view
// ---------------------------------------------------------------------------
struct StrView {
  const char* ptr;
  int32_t len;
};

__device__ inline StrView get_string(const ColDesc& c, int64_t row) {
  int32_t s = c.offsets[row], e = c.offsets[row + 1];
  return {reinterpret_cast<const char*>(c.data) + s, e - s};
}

// ---------------------------------------------------------------------------
// wave-level reductions (64-wide)
// ---------------------------------------------------------------------------
template <typename T>
__device__ inline T wave_sum(T v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_down(v, off, WAVE);
  return v;
}

// inclusive prefix sum across the wave; all 64 lanes must participate
template <typename T>
__device__ inline T wave_prefix_incl(T v) {
  int lane = threadIdx.x & (WAVE - 1);
#pragma unroll
  for (int off = 1; off < WAVE; off <<= 1) {
    T n = __shfl_up(v, off, WAVE);
    if (lane >= off) v += n;
  }
  return v;
}

template <typename T>
__device__ inline T wave_max(T v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    T o = __shfl_down(v, off, WAVE);
    v = o > v ? o : v;
  }
  return v;
}

// ---------------------------------------------------------------------------
// hashing primitives (Spark-exact; shared by murmur/xxhash/hive + hash table)
// Semantics mirrored from the reference's murmur_hash.cu / xxhash64.cu /
// hive_hash.cu (spark-rapids-jni) but implemented fresh.
// ---------------------------------------------------------------------------
__device__ __host__ inline uint32_t rotl32(uint32_t x, int8_t r) {
  return (x << r) | (x >> (32 - r));
}
__device__ __host__ inline uint64_t rotl64(uint64_t x, int8_t r) {
  return (x << r) | (x >> (64 - r));
}

// --- Spark Murmur3_x86_32 ---
__device__ __host__ inline uint32_t mm3_mix_k1(uint32_t k1) {
  k1 *= 0xcc9e2d51u;
  k1 = rotl32(k1, 15);
  k1 *= 0x1b873593u;
  return k1;
}
__device__ __host__ inline uint32_t mm3_mix_h1(uint32_t h1, uint32_t k1) {
  h1 ^= k1;
  h1 = rotl32(h1, 13);
  h1 = h1 * 5u + 0xe6546b64u;
  return h1;
}
__device__ __host__ inline uint32_t mm3_fmix(uint32_t h1, uint32_t len) {
  h1 ^= len;
  h1 ^= h1 >> 16;
  h1 *= 0x85ebca6bu;
  h1 ^= h1 >> 13;
  h1 *= 0xc2b2ae35u;
  h1 ^= h1 >> 16;
  return h1;
}
__device__ __host__ inline uint32_t mm3_hash_int(int32_t v, uint32_t seed) {
  return mm3_fmix(mm3_mix_h1(seed, mm3_mix_k1((uint32_t)v)), 4);
}
__device__ __host__ inline uint32_t mm3_hash_long(int64_t v, uint32_t seed) {
  uint32_t low = (uint32_t)v, high = (uint32_t)(((uint64_t)v) >> 32);
  uint32_t h1 = mm3_mix_h1(seed, mm3_mix_k1(low));
  h1 = mm3_mix_h1(h1, mm3_mix_k1(high));
  return mm3_fmix(h1, 8);
}
// Spark's hashUnsafeBytes: 4-byte LE blocks, then each tail byte processed as
// a full (sign-extended) int block.
__device__ __host__ inline uint32_t mm3_hash_bytes(const char* p, int32_t len,
                                                   uint32_t seed) {
  uint32_t h1 = seed;
  int32_t aligned = len & ~3;
  for (int32_t i = 0; i < aligned; i += 4) {
    uint32_t b;
#if defined(__HIP_DEVICE_COMPILE__)
    b = (uint8_t)p[i] | ((uint8_t)p[i + 1] << 8) | ((uint8_t)p[i + 2] << 16) |
        ((uint8_t)p[i + 3] << 24);
#else
    __builtin_memcpy(&b, p + i, 4);
#endif
    h1 = mm3_mix_h1(h1, mm3_mix_k1(b));
  }
  for (int32_t i = aligned; i < len; ++i) {
    h1 = mm3_mix_h1(h1, mm3_mix_k1((uint32_t)(int32_t)(int8_t)p[i]));
  }
  return mm3_fmix(h1, (uint32_t)len);
}

// STANDARD MurmurHash3_x86_32 (proper 1-3 byte tail) — Iceberg's bucket
// transform requires this variant (the Spark mm3_hash_bytes above mixes
// tail bytes as full int blocks; reference iceberg_bucket.cu delegates
// to cuco::detail::MurmurHash3_32 which is the standard algorithm).
__device__ __host__ inline uint32_t mm3_hash_bytes_std(const char* p,
                                                       int32_t len,
                                                       uint32_t seed) {
  uint32_t h1 = seed;
  int32_t aligned = len & ~3;
  for (int32_t i = 0; i < aligned; i += 4) {
    uint32_t b = (uint8_t)p[i] | ((uint8_t)p[i + 1] << 8) |
                 ((uint8_t)p[i + 2] << 16) | ((uint8_t)p[i + 3] << 24);
    h1 = mm3_mix_h1(h1, mm3_mix_k1(b));
  }
  uint32_t k1 = 0;
  switch (len & 3) {
    case 3: k1 ^= (uint32_t)(uint8_t)p[aligned + 2] << 16; [[fallthrough]];
    case 2: k1 ^= (uint32_t)(uint8_t)p[aligned + 1] << 8; [[fallthrough]];
    case 1:
      k1 ^= (uint32_t)(uint8_t)p[aligned];
      h1 ^= mm3_mix_k1(k1);
      break;
    default: break;
  }
  return mm3_fmix(h1, (uint32_t)len);
}

// normalization Spark applies before hashing floats (NaN -> canonical NaN,
// -0.0 -> +0.0). Bit-based so it is immune to fast-math flags.
__device__ __host__ inline int32_t norm_float_bits(float f) {
  int32_t b;
  __builtin_memcpy(&b, &f, 4);
  if ((b & 0x7f800000) == 0x7f800000 && (b & 0x007fffff)) return 0x7fc00000;
  if ((b & 0x7fffffff) == 0) return 0;
  return b;
}
__device__ __host__ inline int64_t norm_double_bits(double d) {
  int64_t b;
  __builtin_memcpy(&b, &d, 8);
  if ((b & 0x7ff0000000000000LL) == 0x7ff0000000000000LL &&
      (b & 0x000fffffffffffffLL))
    return 0x7ff8000000000000LL;
  if ((b & 0x7fffffffffffffffLL) == 0) return 0;
  return b;
}

// --- Spark XXHash64 (seed chained per column; Spark hashes every fixed-width
// integral value as an 8-byte long, floats as 4 bytes, doubles as 8) ---
constexpr uint64_t XXH_PRIME1 = 0x9E3779B185EBCA87ull;
constexpr uint64_t XXH_PRIME2 = 0xC2B2AE3D27D4EB4Full;
constexpr uint64_t XXH_PRIME3 = 0x165667B19E3779F9ull;
constexpr uint64_t XXH_PRIME4 = 0x85EBCA77C2B2AE63ull;
constexpr uint64_t XXH_PRIME5 = 0x27D4EB2F165667C5ull;

__device__ __host__ inline uint64_t xxh64_avalanche(uint64_t h) {
  h ^= h >> 33;
  h *= XXH_PRIME2;
  h ^= h >> 29;
  h *= XXH_PRIME3;
  h ^= h >> 32;
  return h;
}

__device__ __host__ inline uint64_t xxh64_load64(const char* p) {
  uint64_t v;
  __builtin_memcpy(&v, p, 8);
  return v;
}
__device__ __host__ inline uint32_t xxh64_load32(const char* p) {
  uint32_t v;
  __builtin_memcpy(&v, p, 4);
  return v;
}

__device__ __host__ inline uint64_t xxh64_round(uint64_t acc, uint64_t input) {
  acc += input * XXH_PRIME2;
  acc = rotl64(acc, 31);
  acc *= XXH_PRIME1;
  return acc;
}
__device__ __host__ inline uint64_t xxh64_merge_round(uint64_t acc, uint64_t val) {
  val = xxh64_round(0, val);
  acc ^= val;
  acc = acc * XXH_PRIME1 + XXH_PRIME4;
  return acc;
}

__device__ __host__ inline uint64_t xxhash64_bytes(const char* data, int64_t len,
                                                   uint64_t seed) {
  uint64_t h;
  const char* p = data;
  const char* end = data + len;
  if (len >= 32) {
    uint64_t v1 = seed + XXH_PRIME1 + XXH_PRIME2;
    uint64_t v2 = seed + XXH_PRIME2;
    uint64_t v3 = seed;
    uint64_t v4 = seed - XXH_PRIME1;
    do {
      v1 = xxh64_round(v1, xxh64_load64(p)); p += 8;
      v2 = xxh64_round(v2, xxh64_load64(p)); p += 8;
      v3 = xxh64_round(v3, xxh64_load64(p)); p += 8;
      v4 = xxh64_round(v4, xxh64_load64(p)); p += 8;
    } while (p <= end - 32);
    h = rotl64(v1, 1) + rotl64(v2, 7) + rotl64(v3, 12) + rotl64(v4, 18);
    h = xxh64_merge_round(h, v1);
    h = xxh64_merge_round(h, v2);
    h = xxh64_merge_round(h, v3);
    h = xxh64_merge_round(h, v4);
  } else {
    h = seed + XXH_PRIME5;
  }
  h += (uint64_t)len;
  while (p + 8 <= end) {
    h ^= xxh64_round(0, xxh64_load64(p));
    h = rotl64(h, 27) * XXH_PRIME1 + XXH_PRIME4;
    p += 8;
  }
  if (p + 4 <= end) {
    h ^= (uint64_t)xxh64_load32(p) * XXH_PRIME1;
    h = rotl64(h, 23) * XXH_PRIME2 + XXH_PRIME3;
    p += 4;
  }
  while (p < end) {
    h ^= (uint8_t)(*p) * XXH_PRIME5;
    h = rotl64(h, 11) * XXH_PRIME1;
    ++p;
  }
  return xxh64_avalanche(h);
}

// fixed-width helper: hash `width` bytes stored in a local buffer
__device__ __host__ inline uint64_t xxhash64_fixed(uint64_t bits, int width,
                                                   uint64_t seed) {
  char buf[8];
  __builtin_memcpy(buf, &bits, 8);
  return xxhash64_bytes(buf, width, seed);
}

// --- proleptic Gregorian civil-date helpers (Howard Hinnant's algorithms) ---
__device__ __host__ inline int64_t days_from_civil(int y, int m, int d) {
  y -= m <= 2;
  int64_t era = (y >= 0 ? y : y - 399) / 400;
  unsigned yoe = (unsigned)(y - era * 400);
  unsigned doy = (153u * (m + (m > 2 ? -3 : 9)) + 2) / 5 + d - 1;
  unsigned doe = yoe * 365 + yoe / 4 - yoe / 100 + doy;
  return (int64_t)era * 146097 + (int64_t)doe - 719468;
}

// --- 64-bit mix for internal hash tables (not Spark-visible) ---
__device__ __host__ inline uint64_t mix64(uint64_t x) {
  x ^= x >> 33;
  x *= 0xff51afd7ed558ccdull;
  x ^= x >> 33;
  x *= 0xc4ceb9fe1a85ec53ull;
  x ^= x >> 33;
  return x;
}

#define SRJ_CHECK_HIP(expr)                                              \
  do {                                                                   \
    hipError_t _e = (expr);                                              \
    if (_e != hipSuccess) return _e;                                     \
  } while (0)

// Minimal big-endian two's-complement byte form of a 128-bit little-endian
// value, matching java.math.BigDecimal.unscaledValue().toByteArray(): strip
// redundant sign-extension bytes, keep one extra byte when needed to preserve
// the sign bit, reverse to big-endian. Spark hashes DECIMAL128 (precision>18)
// over exactly these bytes (ref hash/hash.cuh:64 to_java_bigdecimal).
__device__ int dec128_java_bytes(const uint8_t* p, uint8_t out[16]) {
  bool neg = (p[15] & 0x80) != 0;
  uint8_t ext = neg ? 0xff : 0x00;
  int len = 16;
  while (len > 1 && p[len - 1] == ext) --len;
  if (len < 16 && (neg != ((p[len - 1] & 0x80) != 0))) ++len;
  for (int i = 0; i < len; ++i) out[i] = p[len - 1 - i];
  return len;
}

}  // namespace srj
