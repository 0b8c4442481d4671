"""Pure-Python Spark-semantics reference implementations (the test oracle).

These mirror Apache Spark's own algorithms (Murmur3_x86_32 from
org.apache.spark.unsafe.hash.Murmur3_x86_32, XxHash64 from
org.apache.spark.sql.catalyst.expressions.XXH64, Hive's ObjectInspectorUtils
hashCode) and are used to verify the HIP kernels bit-for-bit in tests.
"""
import math
import struct

M32 = 0xFFFFFFFF
M64 = 0xFFFFFFFFFFFFFFFF


def _rotl32(x, r):
    return ((x << r) | (x >> (32 - r))) & M32


def _rotl64(x, r):
    return ((x << r) | (x >> (64 - r))) & M64


# --- Murmur3_x86_32 (Spark) -------------------------------------------------

def _mm3_mix_k1(k1):
    k1 = (k1 * 0xCC9E2D51) & M32
    k1 = _rotl32(k1, 15)
    return (k1 * 0x1B873593) & M32


def _mm3_mix_h1(h1, k1):
    h1 ^= k1
    h1 = _rotl32(h1, 13)
    return (h1 * 5 + 0xE6546B64) & M32


def _mm3_fmix(h1, length):
    h1 ^= length
    h1 ^= h1 >> 16
    h1 = (h1 * 0x85EBCA6B) & M32
    h1 ^= h1 >> 13
    h1 = (h1 * 0xC2B2AE35) & M32
    h1 ^= h1 >> 16
    return h1


def murmur3_int(v, seed):
    return _mm3_fmix(_mm3_mix_h1(seed & M32, _mm3_mix_k1(v & M32)), 4)


def murmur3_long(v, seed):
    v &= M64
    h1 = _mm3_mix_h1(seed & M32, _mm3_mix_k1(v & M32))
    h1 = _mm3_mix_h1(h1, _mm3_mix_k1((v >> 32) & M32))
    return _mm3_fmix(h1, 8)


def murmur3_bytes(b: bytes, seed):
    h1 = seed & M32
    aligned = len(b) & ~3
    for i in range(0, aligned, 4):
        (blk,) = struct.unpack_from("<I", b, i)
        h1 = _mm3_mix_h1(h1, _mm3_mix_k1(blk))
    for i in range(aligned, len(b)):
        byte = b[i]
        if byte >= 128:
            byte -= 256  # sign-extended
        h1 = _mm3_mix_h1(h1, _mm3_mix_k1(byte & M32))
    return _mm3_fmix(h1, len(b))


def norm_float_bits(f):
    if math.isnan(f):
        return 0x7FC00000
    if f == 0.0:
        f = 0.0
    (b,) = struct.unpack("<i", struct.pack("<f", f))
    return b


def norm_double_bits(d):
    if math.isnan(d):
        return 0x7FF8000000000000
    if d == 0.0:
        d = 0.0
    (b,) = struct.unpack("<q", struct.pack("<d", d))
    return b


def _to_signed32(v):
    v &= M32
    return v - (1 << 32) if v >= (1 << 31) else v


def _to_signed64(v):
    v &= M64
    return v - (1 << 64) if v >= (1 << 63) else v


def murmur3_bytes_std(b: bytes, seed):
    """STANDARD MurmurHash3_x86_32 (proper 1-3 byte tail mixing) — what
    Iceberg's bucket transform uses (the Spark variant above processes
    tail bytes as full int blocks instead)."""
    h = seed & M32
    n = len(b)
    aligned = n & ~3
    for i in range(0, aligned, 4):
        k = struct.unpack_from("<I", b, i)[0]
        k = (k * 0xCC9E2D51) & M32
        k = _rotl32(k, 15)
        k = (k * 0x1B873593) & M32
        h ^= k
        h = _rotl32(h, 13)
        h = (h * 5 + 0xE6546B64) & M32
    k = 0
    rem = n & 3
    if rem >= 3:
        k ^= b[aligned + 2] << 16
    if rem >= 2:
        k ^= b[aligned + 1] << 8
    if rem >= 1:
        k ^= b[aligned]
        k = (k * 0xCC9E2D51) & M32
        k = _rotl32(k, 15)
        k = (k * 0x1B873593) & M32
        h ^= k
    h ^= n
    h ^= h >> 16
    h = (h * 0x85EBCA6B) & M32
    h ^= h >> 13
    h = (h * 0xC2B2AE35) & M32
    h ^= h >> 16
    return h


def java_bigint_bytes(v: int) -> bytes:
    """java.math.BigInteger.toByteArray(): minimal big-endian two's
    complement, at least one byte. Spark hashes DECIMAL128 over these bytes
    (ref hash/hash.cuh:64 to_java_bigdecimal)."""
    n = 1
    while True:
        try:
            return int(v).to_bytes(n, "big", signed=True)
        except OverflowError:
            n += 1


def murmur3_row(values, dtypes, seed=42):
    """Chained murmur3 over one row. values: python values (None = null)."""
    from ..columnar import DType
    h = seed
    for v, dt in zip(values, dtypes):
        if v is None:
            continue
        if dt in (DType.BOOL8,):
            h = murmur3_int(1 if v else 0, h)
        elif dt in (DType.INT8, DType.INT16, DType.INT32, DType.DATE32):
            h = murmur3_int(v, h)
        # Spark hashes decimals with precision<=18 as hashLong of the
        # unscaled value, so DECIMAL32 promotes to 64-bit too
        # (ref murmur_hash.cuh:185-197).
        elif dt in (DType.INT64, DType.TIMESTAMP_US, DType.DECIMAL64,
                    DType.DECIMAL32):
            h = murmur3_long(v, h)
        elif dt == DType.DECIMAL128:
            h = murmur3_bytes(java_bigint_bytes(v), h)
        elif dt == DType.FLOAT32:
            h = murmur3_int(norm_float_bits(v), h)
        elif dt == DType.FLOAT64:
            h = murmur3_long(norm_double_bits(v), h)
        elif dt == DType.STRING:
            h = murmur3_bytes(v.encode() if isinstance(v, str) else v, h)
        else:
            raise NotImplementedError(dt)
    return _to_signed32(h)


# --- XXH64 (Spark) ----------------------------------------------------------

P1 = 0x9E3779B185EBCA87
P2 = 0xC2B2AE3D27D4EB4F
P3 = 0x165667B19E3779F9
P4 = 0x85EBCA77C2B2AE63
P5 = 0x27D4EB2F165667C5


def _xxh_avalanche(h):
    h ^= h >> 33
    h = (h * P2) & M64
    h ^= h >> 29
    h = (h * P3) & M64
    h ^= h >> 32
    return h


def _xxh_round(acc, inp):
    acc = (acc + inp * P2) & M64
    acc = _rotl64(acc, 31)
    return (acc * P1) & M64


def _xxh_merge(acc, val):
    val = _xxh_round(0, val)
    acc ^= val
    return (acc * P1 + P4) & M64


def xxhash64_bytes(b: bytes, seed):
    seed &= M64
    n = len(b)
    i = 0
    if n >= 32:
        v1 = (seed + P1 + P2) & M64
        v2 = (seed + P2) & M64
        v3 = seed
        v4 = (seed - P1) & M64
        while i <= n - 32:
            v1 = _xxh_round(v1, struct.unpack_from("<Q", b, i)[0]); i += 8
            v2 = _xxh_round(v2, struct.unpack_from("<Q", b, i)[0]); i += 8
            v3 = _xxh_round(v3, struct.unpack_from("<Q", b, i)[0]); i += 8
            v4 = _xxh_round(v4, struct.unpack_from("<Q", b, i)[0]); i += 8
        h = (_rotl64(v1, 1) + _rotl64(v2, 7) + _rotl64(v3, 12) +
             _rotl64(v4, 18)) & M64
        h = _xxh_merge(h, v1)
        h = _xxh_merge(h, v2)
        h = _xxh_merge(h, v3)
        h = _xxh_merge(h, v4)
    else:
        h = (seed + P5) & M64
    h = (h + n) & M64
    while i + 8 <= n:
        h ^= _xxh_round(0, struct.unpack_from("<Q", b, i)[0])
        h = (_rotl64(h, 27) * P1 + P4) & M64
        i += 8
    if i + 4 <= n:
        h ^= (struct.unpack_from("<I", b, i)[0] * P1) & M64
        h = (_rotl64(h, 23) * P2 + P3) & M64
        i += 4
    while i < n:
        h ^= (b[i] * P5) & M64
        h = (_rotl64(h, 11) * P1) & M64
        i += 1
    return _xxh_avalanche(h)


def xxhash64_row(values, dtypes, seed=42):
    from ..columnar import DType
    h = seed & M64
    for v, dt in zip(values, dtypes):
        if v is None:
            continue
        if dt == DType.BOOL8:
            h = xxhash64_bytes(struct.pack("<i", 1 if v else 0), h)
        elif dt in (DType.INT8, DType.INT16, DType.INT32, DType.DATE32):
            h = xxhash64_bytes(struct.pack("<i", v), h)
        elif dt in (DType.INT64, DType.TIMESTAMP_US, DType.DECIMAL64):
            h = xxhash64_bytes(struct.pack("<q", v), h)
        elif dt == DType.DECIMAL32:
            h = xxhash64_bytes(struct.pack("<q", v), h)
        elif dt == DType.DECIMAL128:
            h = xxhash64_bytes(java_bigint_bytes(v), h)
        elif dt == DType.FLOAT32:
            h = xxhash64_bytes(struct.pack("<i", norm_float_bits(v)), h)
        elif dt == DType.FLOAT64:
            h = xxhash64_bytes(struct.pack("<q", norm_double_bits(v)), h)
        elif dt == DType.STRING:
            h = xxhash64_bytes(v.encode() if isinstance(v, str) else v, h)
        else:
            raise NotImplementedError(dt)
    return _to_signed64(h)


# --- Hive hash --------------------------------------------------------------

def hive_hash_row(values, dtypes):
    from ..columnar import DType
    h = 0
    for v, dt in zip(values, dtypes):
        h = _to_signed32(h * 31 + _hive_one(v, dt))
    return h


def _hive_one(v, dt):
    from ..columnar import DType
    if v is None:
        return 0
    if dt == DType.BOOL8:
        return 1 if v else 0
    if dt in (DType.INT8, DType.INT16, DType.INT32, DType.DATE32):
        return v
    if dt == DType.INT64:
        return _to_signed32(_to_signed64(v) ^ ((v & M64) >> 32))
    if dt == DType.TIMESTAMP_US:
        sec, sub = divmod(v, 1000000)
        nanos = sub * 1000
        t = sec * 1000000000 + nanos
        return _to_signed32(t ^ ((t & M64) >> 32))
    if dt == DType.FLOAT32:
        return _to_signed32(norm_float_bits(v))
    if dt == DType.FLOAT64:
        b = norm_double_bits(v)
        return _to_signed32(b ^ ((b & M64) >> 32))
    if dt == DType.STRING:
        h = 0
        for byte in (v.encode() if isinstance(v, str) else v):
            if byte >= 128:
                byte -= 256
            h = _to_signed32(h * 31 + byte)
        return h
    raise NotImplementedError(dt)
