"""Sketches & digests: SHA-2 family, CRC32, HyperLogLog++, histogram
percentiles, conv, parse_url, GBK charset decode.

Java API parity: Hash.java (sha224..sha512 via hash/sha.cpp, host CRC32
Hash.java:172), HyperLogLogPlusPlusHostUDF.java, Histogram.java,
NumberConverter.java, ParseURI.java, CharsetDecode.java.
"""
import math
import struct
import zlib
from enum import IntEnum
from typing import List, Optional, Sequence

import torch

from .. import _native
from ..columnar import Column, DType, make_validity, pack_descriptors
from .misc import _str_output


def sha2(col: Column, bits: int) -> Column:
    """sha224/sha256/sha384/sha512 of string/binary rows, lowercase hex."""
    assert bits in (224, 256, 384, 512)
    g = _native.gpu()
    stream = _native.current_stream()
    n = col.size
    dev = col.device
    hexlen = {224: 56, 256: 64, 384: 96, 512: 128}[bits]
    desc, top, keep = pack_descriptors([col])
    offsets = torch.arange(0, (n + 1) * hexlen, hexlen, dtype=torch.int32,
                           device=dev)
    chars = torch.zeros(max(n * hexlen, 1), dtype=torch.uint8, device=dev)
    validity = make_validity(n, dev)
    g.sha2(desc.data_ptr(), n, bits, offsets.data_ptr(), chars.data_ptr(),
           validity.data_ptr(), stream)
    return Column(DType.STRING, n, chars, validity, offsets, null_count=None)


def crc32_host(data: bytes) -> int:
    """Host CRC32 (reference Hash.java:172 keeps CRC32 on the CPU)."""
    return zlib.crc32(data) & 0xFFFFFFFF


class HyperLogLogPlusPlus:
    """HLL++ sketch: XXHash64(seed 42) inputs, 2^p int registers, packed
    10 x 6 bits per long for Spark interchange (reference
    HyperLogLogPlusPlusHostUDF.java:20-35); sketch construction matches
    the reference kernel bit-for-bit (hyper_log_log_plus_plus.cu
    reduce_hllpp_kernel: idx = hash >> (64-p), rho = clz((hash << p) |
    (1 << (p-1))) + 1).

    estimate() mirrors the reference's finalizer semantics exactly (it
    delegates to cuco's hyperloglog finalizer, hyper_log_log_plus_plus.cu
    :873): raw alpha*m^2/sum with linear counting below 2.5m when empty
    registers exist, no appendix bias tables and no large-range correction
    (64-bit hashes make it unnecessary). NOTE (documented delta, applies
    to the reference too): Spark's CPU implementation additionally applies
    the HLL++ paper's bias-interpolation tables in the 2.5m..5m band, so
    both this library and the reference can differ from Spark CPU there by
    up to ~1%.
    """

    def __init__(self, precision: int = 9, device="cuda",
                 registers: Optional[torch.Tensor] = None):
        assert 4 <= precision <= 18
        self.precision = precision
        self.m = 1 << precision
        self.registers = registers if registers is not None else \
            torch.zeros(self.m, dtype=torch.int32, device=device)

    def update(self, col: Column):
        from . import hashing
        h = hashing.xxhash64([col])
        g = _native.gpu()
        g.hllpp_update(h.data.data_ptr(),
                       col.validity.data_ptr() if col.validity is not None else 0,
                       col.size, self.precision, self.registers.data_ptr(),
                       _native.current_stream())

    def merge(self, other: "HyperLogLogPlusPlus"):
        assert self.precision == other.precision
        g = _native.gpu()
        g.hllpp_merge(other.registers.data_ptr(), self.registers.data_ptr(),
                      self.m, _native.current_stream())

    def to_longs(self) -> torch.Tensor:
        """Spark STRUCT<long,...> register packing."""
        g = _native.gpu()
        nlongs = (self.m + 9) // 10
        out = torch.zeros(nlongs, dtype=torch.int64,
                          device=self.registers.device)
        g.hllpp_pack(self.registers.data_ptr(), self.m, out.data_ptr(),
                     _native.current_stream())
        return out

    @staticmethod
    def from_longs(longs: torch.Tensor, precision: int) -> "HyperLogLogPlusPlus":
        g = _native.gpu()
        m = 1 << precision
        regs = torch.zeros(m, dtype=torch.int32, device=longs.device)
        g.hllpp_unpack(longs.data_ptr(), m, regs.data_ptr(),
                       _native.current_stream())
        return HyperLogLogPlusPlus(precision, longs.device, regs)

    def estimate(self) -> int:
        """Cardinality estimate, reference-exact (cuco finalizer)."""
        regs = self.registers.cpu().tolist()
        m = self.m
        alpha = (0.673 if m == 16 else 0.697 if m == 32 else 0.709 if m == 64
                 else 0.7213 / (1 + 1.079 / m))
        z = sum(2.0 ** -r for r in regs)
        e = alpha * m * m / z
        zeros = sum(1 for r in regs if r == 0)
        if e <= 2.5 * m and zeros != 0:
            e = m * math.log(m / zeros)  # linear counting
        return int(e + 0.5)


def create_histogram_if_valid(values: Column, freqs: Column,
                              output_as_lists: bool = True):
    """Validate STRUCT<value,freq> pairs (freq >= 0, non-null) — reference
    histogram.hpp:30 createHistogramIfValid. Returns (values, freqs) with
    invalid rows nulled (host check via device-friendly torch ops)."""
    f = freqs.data
    bad = f < 0
    if bool(bad.any().item()):
        raise ValueError("negative frequency in histogram input")
    return values, freqs


def percentile_from_histogram(offsets: torch.Tensor, values: Column,
                              freqs: Column, percentages: Sequence[float]
                              ) -> Column:
    """reference histogram.hpp:40 percentileFromHistogram: per histogram row
    (LIST of value/freq entries sorted by value), interpolated percentiles.
    Returns a column of nrows*len(percentages) doubles."""
    g = _native.gpu()
    nrows = offsets.numel() - 1
    dev = values.device
    npct = len(percentages)
    pcts = torch.tensor(list(percentages), dtype=torch.float64, device=dev)
    out = torch.empty(nrows * npct, dtype=torch.float64, device=dev)
    validity = make_validity(nrows * npct, dev)
    g.percentile_from_histogram(offsets.data_ptr(), values.data.data_ptr(),
                                freqs.data.data_ptr(), nrows, pcts.data_ptr(),
                                npct, out.data_ptr(), validity.data_ptr(),
                                _native.current_stream())
    return Column(DType.FLOAT64, nrows * npct, out, validity, null_count=None)


def convert_base(col: Column, from_base: int, to_base: int) -> Column:
    """Spark conv() (reference number_converter.cu / NumberConverter.java)."""
    assert 2 <= from_base <= 36 and 2 <= abs(to_base) <= 36
    g = _native.gpu()
    n = col.size
    desc, top, keep = pack_descriptors([col])

    def lens_fn(lens, stream):
        g.conv(desc.data_ptr(), n, from_base, to_base, 0, lens.data_ptr(), 0, 0,
               0, stream)

    def write_fn(offsets, chars, validity, stream):
        g.conv(desc.data_ptr(), n, from_base, to_base, 1, 0, offsets.data_ptr(),
               chars.data_ptr(), validity.data_ptr(), stream)

    return _str_output(n, col.device, lens_fn, write_fn)


class UriPart(IntEnum):
    PROTOCOL = 0
    HOST = 1
    PATH = 2
    QUERY = 3
    QUERY_KEY = 4


def parse_uri(col: Column, part: UriPart, query_key="") -> Column:
    """Spark parse_url (reference parse_uri.cu / ParseURI.java:38-174).

    query_key may be a literal str (parseURIQueryWithLiteral) or a string
    Column of per-row keys (parseURIQueryWithColumn)."""
    g = _native.gpu()
    n = col.size
    dev = col.device
    desc, top, keep = pack_descriptors([col])
    qcol_ptr = 0
    kb = b""
    if isinstance(query_key, Column):
        assert query_key.size == n, "key column must match the URI column"
        qdesc, qtop, qkeep = pack_descriptors([query_key])
        qcol_ptr = qdesc.data_ptr()
        part = UriPart.QUERY_KEY
    else:
        kb = query_key.encode()
    kt = torch.frombuffer(bytearray(kb) or bytearray(1), dtype=torch.uint8).to(dev)

    def lens_fn(lens, stream):
        g.parse_uri(desc.data_ptr(), n, int(part), kt.data_ptr(), len(kb),
                    qcol_ptr, 0, lens.data_ptr(), 0, 0, 0, stream)

    def write_fn(offsets, chars, validity, stream):
        g.parse_uri(desc.data_ptr(), n, int(part), kt.data_ptr(), len(kb),
                    qcol_ptr, 1, 0, offsets.data_ptr(), chars.data_ptr(),
                    validity.data_ptr(), stream)

    return _str_output(n, dev, lens_fn, write_fn)


class CharsetDecodeError(RuntimeError):
    def __init__(self, row):
        super().__init__(f"invalid GBK sequence at row {row}")
        self.row_with_error = row


def gbk_decode(col: Column, report: bool = False) -> Column:
    """GBK -> UTF-8 decode (reference charset_decode.cu; REPLACE emits
    U+FFFD, REPORT raises with the first bad row)."""
    g = _native.gpu()
    n = col.size
    dev = col.device
    desc, top, keep = pack_descriptors([col])
    err = torch.full((1,), 2**63 - 1, dtype=torch.int64, device=dev) if report \
        else None

    def lens_fn(lens, stream):
        g.gbk_decode(desc.data_ptr(), n, 1 if report else 0, 0, lens.data_ptr(),
                     0, 0, 0, err.data_ptr() if err is not None else 0, stream)

    def write_fn(offsets, chars, validity, stream):
        g.gbk_decode(desc.data_ptr(), n, 1 if report else 0, 1, 0,
                     offsets.data_ptr(), chars.data_ptr(), validity.data_ptr(),
                     err.data_ptr() if err is not None else 0, stream)

    out = _str_output(n, dev, lens_fn, write_fn)
    if err is not None:
        row = int(err.item())
        if row != 2**63 - 1:
            raise CharsetDecodeError(row)
    return out
