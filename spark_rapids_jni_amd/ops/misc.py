"""Misc op wrappers: case_when, BloomFilter, ZOrder, hex, uuid,
substring_index, regex-rewrite literal_range, Aggregation64Utils, ANSI
multiply, datetime rebase/truncate.

Java API parity: CaseWhen.java, BloomFilter.java, ZOrder.java,
StringUtils (bytesToHex/randomUUIDs), SubstringIndexUtils.java,
RegexRewriteUtils.java, Aggregation64Utils.java, ANSI Arithmetic
(multiply.cu), DateTimeUtils.java — see SURVEY.md §2.4/§2.6.
"""
import struct
from enum import IntEnum
from typing import List, Optional, Sequence

import torch

from .. import _native
from ..columnar import Column, DType, Table, make_validity, pack_descriptors


def _str_output(n, dev, lens_fn, write_fn):
    g = _native.gpu()
    stream = _native.current_stream()
    lens = torch.empty(n, dtype=torch.int32, device=dev)
    lens_fn(lens, stream)
    offsets = torch.zeros(n + 1, dtype=torch.int32, device=dev)
    torch.cumsum(lens, 0, out=offsets[1:].view(n))
    nchars = int(offsets[-1].item())
    chars = torch.empty(max(nchars, 1), dtype=torch.uint8, device=dev)
    validity = make_validity(n, dev)
    write_fn(offsets, chars, validity, stream)
    return Column(DType.STRING, n, chars[:nchars], validity, offsets,
                  null_count=None)


def select_first_true_index(bool_cols: Sequence[Column]) -> Column:
    """reference case_when.hpp:25 select_first_true_index."""
    g = _native.gpu()
    n = bool_cols[0].size
    dev = bool_cols[0].device
    out = torch.empty(n, dtype=torch.int32, device=dev)
    desc, top, keep = pack_descriptors(list(bool_cols))
    g.select_first_true(desc.data_ptr(), top.data_ptr(), len(bool_cols), n,
                        out.data_ptr(), _native.current_stream())
    return Column(DType.INT32, n, out)


class BloomFilter:
    """Spark-compatible bloom filter in one device byte buffer
    (reference bloom_filter.hpp:37-80; V1 12B / V2 16B big-endian headers)."""

    def __init__(self, version: int, num_hashes: int, num_longs: int,
                 seed: int = 0, device="cuda", buf: Optional[torch.Tensor] = None):
        self.version = version
        self.num_hashes = num_hashes
        self.num_longs = num_longs
        self.seed = seed
        hdr = (struct.pack(">iii", version, num_hashes, num_longs)
               if version == 1 else
               struct.pack(">iiii", version, num_hashes, seed, num_longs))
        self.header_size = len(hdr)
        if buf is None:
            buf = torch.zeros(self.header_size + num_longs * 8,
                              dtype=torch.uint8, device=device)
            buf[:self.header_size] = torch.frombuffer(bytearray(hdr),
                                                      dtype=torch.uint8)
        self.buf = buf

    @staticmethod
    def from_buffer(buf: torch.Tensor) -> "BloomFilter":
        hdr = buf[:16].cpu().numpy().tobytes()
        version = struct.unpack(">i", hdr[:4])[0]
        if version == 1:
            _, nh, nl = struct.unpack(">iii", hdr[:12])
            return BloomFilter(1, nh, nl, 0, buf.device, buf)
        _, nh, seed, nl = struct.unpack(">iiii", hdr[:16])
        return BloomFilter(2, nh, nl, seed, buf.device, buf)

    def _bits_ptr(self):
        return self.buf.data_ptr() + self.header_size

    def put(self, col: Column):
        g = _native.gpu()
        g.bloom_filter(self._bits_ptr(), self.num_longs * 64,
                       col.data.data_ptr(),
                       col.validity.data_ptr() if col.validity is not None else 0,
                       col.size, self.num_hashes, self.seed, self.version, 0, 0, 0,
                       _native.current_stream())

    def might_contain(self, col: Column) -> Column:
        g = _native.gpu()
        n = col.size
        out = torch.empty(n, dtype=torch.int8, device=col.device)
        validity = make_validity(n, col.device)
        g.bloom_filter(self._bits_ptr(), self.num_longs * 64,
                       col.data.data_ptr(),
                       col.validity.data_ptr() if col.validity is not None else 0,
                       n, self.num_hashes, self.seed, self.version, 1,
                       out.data_ptr(), validity.data_ptr(),
                       _native.current_stream())
        return Column(DType.BOOL8, n, out, validity, null_count=None)

    def merge(self, other: "BloomFilter"):
        assert (self.version, self.num_hashes, self.num_longs) == \
               (other.version, other.num_hashes, other.num_longs)
        g = _native.gpu()
        g.bitmask_or(other._bits_ptr(), self._bits_ptr(), self.num_longs * 2,
                     _native.current_stream())


def interleave_bits(cols: Sequence[Column]) -> Column:
    """Z-order interleave (reference zorder.hpp:34). Returns LIST<UINT8>-style
    fixed-width byte rows as a STRING column of ncols*width bytes per row."""
    from ..columnar import FIXED_WIDTH
    g = _native.gpu()
    n = cols[0].size
    dev = cols[0].device
    width = FIXED_WIDTH[cols[0].dtype]
    assert all(FIXED_WIDTH[c.dtype] == width for c in cols)
    out = torch.empty(n * len(cols) * width, dtype=torch.uint8, device=dev)
    desc, top, keep = pack_descriptors(list(cols))
    g.interleave_bits(desc.data_ptr(), top.data_ptr(), len(cols), width, n,
                      out.data_ptr(), _native.current_stream())
    offsets = torch.arange(0, (n + 1) * len(cols) * width, len(cols) * width,
                           dtype=torch.int32, device=dev)
    return Column(DType.STRING, n, out, None, offsets)


def hilbert_index(nbits: int, cols: Sequence[Column]) -> Column:
    """reference zorder.hpp hilbert_index (Moten/Skilling)."""
    g = _native.gpu()
    n = cols[0].size
    out = torch.empty(n, dtype=torch.int64, device=cols[0].device)
    desc, top, keep = pack_descriptors(list(cols))
    g.hilbert_index(desc.data_ptr(), top.data_ptr(), len(cols), nbits, n,
                    out.data_ptr(), _native.current_stream())
    return Column(DType.INT64, n, out)


def bytes_to_hex(col: Column) -> Column:
    g = _native.gpu()
    desc, top, keep = pack_descriptors([col])
    n = col.size

    def lens_fn(lens, stream):
        g.bytes_to_hex(desc.data_ptr(), n, 0, lens.data_ptr(), 0, 0, 0, stream)

    def write_fn(offsets, chars, validity, stream):
        g.bytes_to_hex(desc.data_ptr(), n, 1, 0, offsets.data_ptr(),
                       chars.data_ptr(), validity.data_ptr(), stream)

    return _str_output(n, col.device, lens_fn, write_fn)


def random_uuids(n: int, seed: int = 0, device="cuda") -> Column:
    g = _native.gpu()
    chars = torch.empty(max(n * 36, 1), dtype=torch.uint8, device=device)
    g.uuid(n, seed, chars.data_ptr(), _native.current_stream())
    offsets = torch.arange(0, (n + 1) * 36, 36, dtype=torch.int32, device=device)
    return Column(DType.STRING, n, chars, None, offsets)


def substring_index(col: Column, delim: str, count: int) -> Column:
    g = _native.gpu()
    n = col.size
    dev = col.device
    desc, top, keep = pack_descriptors([col])
    dbytes = delim.encode()
    dstage = torch.frombuffer(bytearray(dbytes) or bytearray(1),
                              dtype=torch.uint8).to(dev)

    def lens_fn(lens, stream):
        g.substring_index(desc.data_ptr(), dstage.data_ptr(), len(dbytes), count,
                          n, 0, lens.data_ptr(), 0, 0, 0, stream)

    def write_fn(offsets, chars, validity, stream):
        g.substring_index(desc.data_ptr(), dstage.data_ptr(), len(dbytes), count,
                          n, 1, 0, offsets.data_ptr(), chars.data_ptr(),
                          validity.data_ptr(), stream)

    return _str_output(n, dev, lens_fn, write_fn)


def literal_range_pattern(col: Column, literal: str, length: int,
                          start: str, end: str) -> Column:
    """reference regex_rewrite_utils.hpp:28 literal_range_pattern."""
    g = _native.gpu()
    n = col.size
    dev = col.device
    desc, top, keep = pack_descriptors([col])
    lit = literal.encode()
    lstage = torch.frombuffer(bytearray(lit) or bytearray(1),
                              dtype=torch.uint8).to(dev)
    out = torch.empty(n, dtype=torch.int8, device=dev)
    validity = make_validity(n, dev)
    g.literal_range(desc.data_ptr(), lstage.data_ptr(), len(lit), length,
                    ord(start), ord(end), n, out.data_ptr(),
                    validity.data_ptr(), _native.current_stream())
    return Column(DType.BOOL8, n, out, validity, null_count=None)


class Aggregation64Utils:
    """Overflow-safe 64-bit SUM decomposition (reference
    Aggregation64Utils.java:42-61)."""

    @staticmethod
    def extract_int32_chunk(col: Column, chunk: int) -> Column:
        g = _native.gpu()
        n = col.size
        out = torch.empty(n, dtype=torch.int64, device=col.device)
        g.extract_chunk32(col.data.data_ptr(),
                          col.validity.data_ptr() if col.validity is not None else 0,
                          n, chunk, out.data_ptr(), _native.current_stream())
        return Column(DType.INT64, n, out, col.validity, null_count=None)

    @staticmethod
    def combine_int64_sum_chunks(lo: Column, hi: Column):
        """Returns (sum Column, overflow flags Column)."""
        g = _native.gpu()
        n = lo.size
        out = torch.empty(n, dtype=torch.int64, device=lo.device)
        overflow = torch.empty(n, dtype=torch.int8, device=lo.device)
        g.combine_chunks(lo.data.data_ptr(), hi.data.data_ptr(), n,
                         out.data_ptr(), overflow.data_ptr(),
                         _native.current_stream())
        return (Column(DType.INT64, n, out),
                Column(DType.BOOL8, n, overflow))


class OverflowError64(RuntimeError):
    def __init__(self, row):
        super().__init__(f"int64 overflow at row {row}")
        self.row_with_error = row


def multiply_int64(a: Column, b: Column, ansi: bool = False) -> Column:
    """ANSI overflow-checked multiply reporting the first bad row
    (reference multiply.cu + ExceptionWithRowIndex)."""
    g = _native.gpu()
    n = a.size
    dev = a.device
    out = torch.empty(n, dtype=torch.int64, device=dev)
    validity = make_validity(n, dev)
    err = torch.full((1,), 2**63 - 1, dtype=torch.int64, device=dev) if ansi \
        else None
    g.multiply_i64(a.data.data_ptr(),
                   a.validity.data_ptr() if a.validity is not None else 0,
                   b.data.data_ptr(),
                   b.validity.data_ptr() if b.validity is not None else 0,
                   n, out.data_ptr(), validity.data_ptr(),
                   err.data_ptr() if err is not None else 0,
                   _native.current_stream())
    if err is not None:
        row = int(err.item())
        if row != 2**63 - 1:
            raise OverflowError64(row)
    return Column(DType.INT64, n, out, validity, null_count=None)


class TruncUnit(IntEnum):
    YEAR = 0
    QUARTER = 1
    MONTH = 2
    WEEK = 3
    DAY = 4
    HOUR = 5
    MINUTE = 6
    SECOND = 7
    MILLISECOND = 8
    MICROSECOND = 9


_TRUNC_ALIASES = {
    "YEAR": TruncUnit.YEAR, "YYYY": TruncUnit.YEAR, "YY": TruncUnit.YEAR,
    "QUARTER": TruncUnit.QUARTER, "MONTH": TruncUnit.MONTH,
    "MON": TruncUnit.MONTH, "MM": TruncUnit.MONTH, "WEEK": TruncUnit.WEEK,
    "DAY": TruncUnit.DAY, "DD": TruncUnit.DAY, "HOUR": TruncUnit.HOUR,
    "MINUTE": TruncUnit.MINUTE, "SECOND": TruncUnit.SECOND,
    "MILLISECOND": TruncUnit.MILLISECOND, "MICROSECOND": TruncUnit.MICROSECOND,
}


def truncate_timestamp(col: Column, fmt: str) -> Column:
    """Spark date_trunc (reference datetime_truncate.cu)."""
    g = _native.gpu()
    unit = _TRUNC_ALIASES[fmt.upper()]
    n = col.size
    out = torch.empty(n, dtype=torch.int64, device=col.device)
    g.trunc_timestamp(col.data.data_ptr(),
                      col.validity.data_ptr() if col.validity is not None else 0,
                      n, int(unit), out.data_ptr(), _native.current_stream())
    return Column(DType.TIMESTAMP_US, n, out, col.validity, null_count=None)


def rebase_gregorian_to_julian(col: Column) -> Column:
    """reference datetime_rebase.cu (DATE32 path)."""
    g = _native.gpu()
    n = col.size
    out = torch.empty(n, dtype=torch.int32, device=col.device)
    g.rebase_days(col.data.data_ptr(),
                  col.validity.data_ptr() if col.validity is not None else 0,
                  n, 1, out.data_ptr(), _native.current_stream())
    return Column(DType.DATE32, n, out, col.validity, null_count=None)


def rebase_julian_to_gregorian(col: Column) -> Column:
    g = _native.gpu()
    n = col.size
    out = torch.empty(n, dtype=torch.int32, device=col.device)
    g.rebase_days(col.data.data_ptr(),
                  col.validity.data_ptr() if col.validity is not None else 0,
                  n, 0, out.data_ptr(), _native.current_stream())
    return Column(DType.DATE32, n, out, col.validity, null_count=None)
