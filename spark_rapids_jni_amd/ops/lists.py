"""List/map ops, iceberg transforms, round_float, JoinPrimitives tail.

Java API parity: GpuListSliceUtils.java (list_slice 4 scalar/column combos),
Map.java/MapUtils.java (sortMapColumn, isValidMap, mapFromEntries),
MapZipWithUtils.java (mapZip full-outer key union), iceberg/* (bucket,
truncate, year/month/day/hour), RoundFloat (round_float.cu),
JoinPrimitives.filterGatherMapsByAst / getMatchedRows.
"""
import struct
from typing import List, Optional, Tuple

import torch

from .. import _native
from ..columnar import Column, DType, Table, make_validity, pack_descriptors
from .copying import gather_column


def list_slice(col: Column, start, length, ansi: bool = False) -> Column:
    """Spark slice(list, start, length): 1-based start, negative from end.
    start/length may be Python ints or INT32 Columns (4 combos)."""
    g = _native.gpu()
    stream = _native.current_stream()
    n = col.size
    dev = col.device
    start_ptr = start.data.data_ptr() if isinstance(start, Column) else 0
    len_ptr = length.data.data_ptr() if isinstance(length, Column) else 0
    out_lens = torch.empty(n, dtype=torch.int32, device=dev)
    child_start = torch.empty(n, dtype=torch.int32, device=dev)
    validity = make_validity(n, dev)
    err = torch.full((1,), 2**63 - 1, dtype=torch.int64, device=dev) if ansi \
        else None
    g.list_slice(col.offsets.data_ptr(),
                 col.validity.data_ptr() if col.validity is not None else 0, n,
                 start_ptr, start if isinstance(start, int) else 0,
                 len_ptr, length if isinstance(length, int) else 0,
                 out_lens.data_ptr(), child_start.data_ptr(),
                 validity.data_ptr(),
                 err.data_ptr() if err is not None else 0, stream)
    if err is not None:
        row = int(err.item())
        if row != 2**63 - 1:
            raise ValueError(f"slice: invalid start/length at row {row}")
    out_offsets = torch.zeros(n + 1, dtype=torch.int32, device=dev)
    torch.cumsum(out_lens, 0, out=out_offsets[1:].view(n))
    nchild = int(out_offsets[-1].item())
    gmap = torch.empty(max(nchild, 1), dtype=torch.int64, device=dev)
    g.list_slice_gather(child_start.data_ptr(), out_offsets.data_ptr(), n,
                        gmap.data_ptr(), stream)
    child = gather_column(col.children[0], gmap[:nchild])
    return Column(DType.LIST, n, None, validity, out_offsets, [child],
                  null_count=None)


def _map_parts(m: Column):
    entry = m.children[0]
    return m.offsets, entry.children[0], entry.children[1]


def is_valid_map(m: Column) -> Column:
    """Non-null, unique keys per row (reference map_utils.hpp:27)."""
    g = _native.gpu()
    n = m.size
    offs, keys, vals = _map_parts(m)
    desc, top, keep = pack_descriptors([keys])
    out = torch.empty(n, dtype=torch.int8, device=m.device)
    g.validate_map(offs.data_ptr(),
                   m.validity.data_ptr() if m.validity is not None else 0, n,
                   desc.data_ptr(), 0, out.data_ptr(), _native.current_stream())
    return Column(DType.BOOL8, n, out)


def map_from_entries(entries: Column, check: bool = True) -> Column:
    """LIST<STRUCT<k,v>> -> MAP, zero-copy after validation
    (reference map_utils.hpp map_from_entries)."""
    if check:
        valid = is_valid_map(entries)
        if not bool(torch.all(valid.data != 0).item()):
            raise ValueError("duplicate or null keys in map_from_entries")
    return entries


def sort_map_column(m: Column) -> Column:
    """Sort each row's entries by key (reference map.hpp:25)."""
    g = _native.gpu()
    n = m.size
    offs, keys, vals = _map_parts(m)
    desc, top, keep = pack_descriptors([keys])
    nent = keys.size
    perm = torch.empty(max(nent, 1), dtype=torch.int64, device=m.device)
    g.sort_map(offs.data_ptr(), n, desc.data_ptr(), 0, perm.data_ptr(),
               _native.current_stream())
    new_keys = gather_column(keys, perm[:nent])
    new_vals = gather_column(vals, perm[:nent])
    entry = Column(DType.STRUCT, nent, None, None, None, [new_keys, new_vals])
    return Column(DType.LIST, n, None, m.validity, m.offsets, [entry],
                  null_count=None)


def map_zip(m1: Column, m2: Column):
    """Full-outer key union of two maps (reference map_zip_with_utils.cu).
    Inputs must be key-sorted (sort_map_column). Returns a MAP column of
    STRUCT<key, STRUCT<v1, v2>> with nulls for missing sides."""
    g = _native.gpu()
    stream = _native.current_stream()
    n = m1.size
    dev = m1.device
    o1, k1, v1 = _map_parts(m1)
    o2, k2, v2 = _map_parts(m2)
    desc, top, keep = pack_descriptors([k1, k2])
    top_h = top.cpu().tolist()
    counts = torch.zeros(n, dtype=torch.int32, device=dev)
    g.map_zip(o1.data_ptr(), o2.data_ptr(), n, desc.data_ptr(), top_h[0],
              top_h[1], 0, counts.data_ptr(), 0, 0, 0, 0, stream)
    out_offsets = torch.zeros(n + 1, dtype=torch.int32, device=dev)
    torch.cumsum(counts, 0, out=out_offsets[1:].view(n))
    nent = int(out_offsets[-1].item())
    kmap = torch.empty(max(nent, 1), dtype=torch.int64, device=dev)
    v1map = torch.empty(max(nent, 1), dtype=torch.int64, device=dev)
    v2map = torch.empty(max(nent, 1), dtype=torch.int64, device=dev)
    g.map_zip(o1.data_ptr(), o2.data_ptr(), n, desc.data_ptr(), top_h[0],
              top_h[1], 1, 0, out_offsets.data_ptr(), kmap.data_ptr(),
              v1map.data_ptr(), v2map.data_ptr(), stream)
    kmap = kmap[:nent]
    # keys: from m1 where kmap>=0 else from m2 at -(kmap+2)
    from_m2 = kmap < 0
    k1_idx = torch.where(from_m2, torch.zeros_like(kmap), kmap)
    keys_a = gather_column(k1, k1_idx)
    k2_idx = torch.where(from_m2, -(kmap + 2), torch.zeros_like(kmap))
    keys_b = gather_column(k2, k2_idx)
    # merge: pick b where from_m2 (host-free merge via torch.where on data)
    if keys_a.dtype == DType.STRING:
        # rebuild via python-level select (string merge): gather with map
        sel_map = torch.where(from_m2, -(kmap + 2) + k1.size, kmap)
        both = _concat_strings(k1, k2)
        keys = gather_column(both, sel_map)
    else:
        data = torch.where(from_m2, keys_b.data, keys_a.data)
        keys = Column(keys_a.dtype, nent, data)
    va = gather_column(v1, v1map[:nent], has_nulls=True)
    vb = gather_column(v2, v2map[:nent], has_nulls=True)
    entry = Column(DType.STRUCT, nent, None, None, None,
                   [keys, Column(DType.STRUCT, nent, None, None, None, [va, vb])])
    return Column(DType.LIST, n, None, m1.validity, out_offsets, [entry],
                  null_count=None)


def _concat_strings(a: Column, b: Column) -> Column:
    dev = a.device
    na, nb = a.size, b.size
    chars = torch.cat([a.data if a.data is not None else
                       torch.zeros(0, dtype=torch.uint8, device=dev),
                       b.data if b.data is not None else
                       torch.zeros(0, dtype=torch.uint8, device=dev)])
    base = int(a.offsets[-1].item())
    offsets = torch.cat([a.offsets[:-1], b.offsets + base])
    return Column(DType.STRING, na + nb, chars, None, offsets)


# --- iceberg ----------------------------------------------------------------

def iceberg_bucket(col: Column, nbuckets: int) -> Column:
    """Iceberg bucket transform: (murmur3(bytes) & MAX_INT) % N
    (reference iceberg/iceberg_bucket.hpp:26; int/long/date/ts hash as
    8-byte LE long, strings as UTF-8 bytes)."""
    g = _native.gpu()
    n = col.size
    dev = col.device
    out = torch.empty(n, dtype=torch.int32, device=dev)
    validity = make_validity(n, dev)
    if col.dtype == DType.STRING:
        # strings AND binary: raw UTF-8/byte payload
        desc, top, keep = pack_descriptors([col])
        g.iceberg_bucket_string(desc.data_ptr(), n, nbuckets, out.data_ptr(),
                                validity.data_ptr(), _native.current_stream())
    elif col.dtype in (DType.DECIMAL32, DType.DECIMAL64, DType.DECIMAL128):
        # minimal big-endian two's-complement unscaled bytes (iceberg spec)
        width = {DType.DECIMAL32: 4, DType.DECIMAL64: 8,
                 DType.DECIMAL128: 16}[col.dtype]
        g.iceberg_bucket_decimal(
            col.data.data_ptr(),
            col.validity.data_ptr() if col.validity is not None else 0,
            n, width, nbuckets, out.data_ptr(), validity.data_ptr(),
            _native.current_stream())
    else:
        data = col.data
        if col.dtype in (DType.INT32, DType.DATE32):
            data = col.data.to(torch.int64)
        g.iceberg_bucket_long(data.data_ptr(),
                              col.validity.data_ptr() if col.validity is not None
                              else 0, n, nbuckets, out.data_ptr(),
                              validity.data_ptr(), _native.current_stream())
    return Column(DType.INT32, n, out, validity, null_count=None)


def iceberg_truncate(col: Column, width: int) -> Column:
    """Iceberg truncate transform for integral columns."""
    g = _native.gpu()
    n = col.size
    data = col.data.to(torch.int64) if col.dtype == DType.INT32 else col.data
    out = torch.empty(n, dtype=torch.int64, device=col.device)
    g.iceberg_truncate_long(data.data_ptr(),
                            col.validity.data_ptr() if col.validity is not None
                            else 0, n, width, out.data_ptr(),
                            _native.current_stream())
    out_t = out.to(torch.int32) if col.dtype == DType.INT32 else out
    return Column(col.dtype, n, out_t, col.validity, null_count=None)


def iceberg_datetime_transform(col: Column, part: str) -> Column:
    """year/month/day/hour transforms (reference iceberg_datetime_util.cu)."""
    g = _native.gpu()
    parts = {"year": 0, "month": 1, "day": 2, "hour": 3}
    n = col.size
    out = torch.empty(n, dtype=torch.int32, device=col.device)
    from_micros = 1 if col.dtype == DType.TIMESTAMP_US else 0
    g.iceberg_datetime(col.data.data_ptr(),
                       col.validity.data_ptr() if col.validity is not None else 0,
                       n, from_micros, parts[part], out.data_ptr(),
                       _native.current_stream())
    return Column(DType.INT32, n, out, col.validity, null_count=None)


# --- round_float -------------------------------------------------------------

def round_double(col: Column, scale: int, half_even: bool = False) -> Column:
    """Spark round/bround on doubles — exact decimal rounding via the Ryu /
    Eisel-Lemire digit space (reference round_float.cu's decimal-string
    trick)."""
    g = _native.gpu()
    n = col.size
    out = torch.empty(n, dtype=torch.float64, device=col.device)
    g.round_double(col.data.data_ptr(),
                   col.validity.data_ptr() if col.validity is not None else 0, n,
                   scale, 1 if half_even else 0, out.data_ptr(),
                   _native.current_stream())
    return Column(DType.FLOAT64, n, out, col.validity, null_count=None)


# --- JoinPrimitives tail -----------------------------------------------------

class Ast:
    """Postfix AST program builder for filter_gather_maps_by_ast
    (reference join_primitives.hpp:115). Columns are referenced by index into
    the combined [left..., right...] column list."""
    OPS = {"<": 4, "<=": 5, ">": 6, ">=": 7, "==": 8, "!=": 9,
           "&&": 10, "||": 11, "!": 12, "+": 13, "-": 14}

    def __init__(self):
        self.prog = []

    def left_col(self, idx):
        self.prog.append((0, idx, 0))
        return self

    def right_col(self, idx):
        self.prog.append((1, idx, 0))
        return self

    def lit(self, v):
        if isinstance(v, int):
            self.prog.append((2, 0, v))
        else:
            self.prog.append((3, 0, struct.unpack("<q", struct.pack("<d", v))[0]))
        return self

    def op(self, name):
        self.prog.append((self.OPS[name], 0, 0))
        return self

    def pack(self, dev):
        raw = bytearray()
        for op, arg, lit in self.prog:
            raw += struct.pack("<iiq", op, arg, lit)
        return torch.frombuffer(raw or bytearray(1), dtype=torch.uint8).to(dev)


def filter_gather_maps_by_ast(cols: List[Column], ast: Ast,
                              lmap: torch.Tensor, rmap: torch.Tensor):
    """Keep join pairs where the AST predicate evaluates true
    (mixed-join post-filter)."""
    g = _native.gpu()
    stream = _native.current_stream()
    dev = lmap.device
    n = lmap.numel()
    desc, top, keep = pack_descriptors(cols)
    # AST col indices are into `cols`; remap through top indices on the host
    prog = Ast()
    top_h = top.cpu().tolist()
    for op, arg, lit in ast.prog:
        prog.prog.append((op, top_h[arg] if op in (0, 1) else arg, lit))
    pt = prog.pack(dev)
    counter = torch.zeros(1, dtype=torch.int64, device=dev)
    g.ast_filter_pairs(desc.data_ptr(), pt.data_ptr(), len(ast.prog),
                       lmap.data_ptr(), rmap.data_ptr(), n, counter.data_ptr(),
                       0, 0, 0, 0, stream)
    total = int(counter.item())
    counter.zero_()
    out_l = torch.empty(max(total, 1), dtype=torch.int32, device=dev)
    out_r = torch.empty(max(total, 1), dtype=torch.int64, device=dev)
    g.ast_filter_pairs(desc.data_ptr(), pt.data_ptr(), len(ast.prog),
                       lmap.data_ptr(), rmap.data_ptr(), n, counter.data_ptr(),
                       out_l.data_ptr(), out_r.data_ptr(), total, 1, stream)
    return out_l[:total], out_r[:total]


def get_matched_rows(gmap: torch.Tensor, table_size: int) -> Column:
    """Bitmap of build rows present in a gather map
    (reference join_primitives.hpp:237)."""
    g = _native.gpu()
    flags = torch.zeros(table_size, dtype=torch.uint8, device=gmap.device)
    g.matched_rows(gmap.data_ptr(), gmap.numel(), flags.data_ptr(),
                   _native.current_stream())
    return Column(DType.BOOL8, table_size, flags.view(torch.int8))
