"""Gather and hash-partition primitives.

Gather follows the cudf convention the reference relies on: a negative
gather-map entry produces a null output row (make_left_outer/full_outer use
this, reference join_primitives.hpp:130-170).
"""
from typing import Optional

import torch

from .. import _native
from ..columnar import (Column, DType, FIXED_WIDTH, Table, make_validity,
                        pack_descriptors)


def gather_column(col: Column, gmap: torch.Tensor,
                  has_nulls: Optional[bool] = None) -> Column:
    g = _native.gpu()
    stream = _native.current_stream()
    n = gmap.numel()
    dev = gmap.device
    assert gmap.dtype == torch.int64
    out_nullable = (has_nulls if has_nulls is not None
                    else (col.validity is not None))
    if col.dtype == DType.STRING:
        lens = torch.empty(n, dtype=torch.int32, device=dev)
        g.gather_str_lengths(col.offsets.data_ptr(), gmap.data_ptr(), n,
                             lens.data_ptr(), stream)
        offsets = torch.zeros(n + 1, dtype=torch.int32, device=dev)
        torch.cumsum(lens, 0, out=offsets[1:].view(n))
        nchars = int(offsets[-1].item())
        chars = torch.empty(max(nchars, 1), dtype=torch.uint8, device=dev)
        validity = make_validity(n, dev) if out_nullable else None
        g.gather_str_chars(col.data.data_ptr() if col.data is not None else 0,
                           col.offsets.data_ptr(),
                           col.validity.data_ptr() if col.validity is not None else 0,
                           gmap.data_ptr(), offsets.data_ptr(), n,
                           chars.data_ptr(),
                           validity.data_ptr() if validity is not None else 0,
                           stream)
        return Column(DType.STRING, n, chars[:nchars], validity, offsets,
                      null_count=None)
    if col.dtype == DType.STRUCT:
        validity = None
        if out_nullable:
            validity = make_validity(n, dev)
            g.gather_validity(
                col.validity.data_ptr() if col.validity is not None else 0,
                gmap.data_ptr(), n, validity.data_ptr(), stream)
        children = [gather_column(ch, gmap, has_nulls) for ch in col.children]
        return Column(DType.STRUCT, n, None, validity, children=children,
                      null_count=None)
    if col.dtype == DType.LIST:
        lens = torch.empty(max(n, 1), dtype=torch.int32, device=dev)
        g.gather_str_lengths(col.offsets.data_ptr(), gmap.data_ptr(), n,
                             lens.data_ptr(), stream)
        offsets = torch.zeros(n + 1, dtype=torch.int32, device=dev)
        if n:
            torch.cumsum(lens[:n], 0, out=offsets[1:].view(n))
        validity = None
        if out_nullable:
            validity = make_validity(n, dev)
            g.gather_validity(
                col.validity.data_ptr() if col.validity is not None else 0,
                gmap.data_ptr(), n, validity.data_ptr(), stream)
        # child gather map: src child rows of each gathered list, in order
        lens64 = lens[:n].to(torch.int64)
        total_child = int(offsets[-1].item())
        if total_child > 0:
            starts = col.offsets.to(torch.int64)[gmap.clamp(min=0)]
            base = torch.repeat_interleave(starts, lens64)
            first = torch.repeat_interleave(offsets[:-1].to(torch.int64),
                                            lens64)
            child_map = base + (
                torch.arange(total_child, dtype=torch.int64, device=dev) -
                first)
        else:
            child_map = torch.empty(0, dtype=torch.int64, device=dev)
        child = gather_column(col.children[0], child_map, has_nulls)
        return Column(DType.LIST, n, None, validity, offsets,
                      children=[child], null_count=None)
    width = FIXED_WIDTH[col.dtype]
    numel = n * 2 if col.dtype == DType.DECIMAL128 else n
    out = torch.empty(numel, dtype=col.data.dtype, device=dev)
    validity = make_validity(n, dev) if out_nullable else None
    g.gather_fixed(col.data.data_ptr(),
                   col.validity.data_ptr() if col.validity is not None else 0,
                   gmap.data_ptr(), n, out.data_ptr(),
                   validity.data_ptr() if validity is not None else 0,
                   width, stream)
    return Column(col.dtype, n, out, validity, scale=col.scale, null_count=None)


def gather(table: Table, gmap: torch.Tensor, has_nulls=None) -> Table:
    return Table([gather_column(c, gmap, has_nulls) for c in table.columns])


def partition_map(parts: torch.Tensor, nparts: int):
    """Group rows by partition id.

    Returns (offsets, perm): partition p occupies perm[offsets[p]:offsets[p+1]]
    as a gather map into the original rows (Spark hash-partition shuffle write).
    """
    g = _native.gpu()
    stream = _native.current_stream()
    n = parts.numel()
    dev = parts.device
    hist = torch.zeros(nparts, dtype=torch.int64, device=dev)
    g.partition_hist(parts.data_ptr(), n, nparts, hist.data_ptr(), stream)
    offsets = torch.zeros(nparts + 1, dtype=torch.int64, device=dev)
    torch.cumsum(hist, 0, out=offsets[1:].view(nparts))
    # clone: the scatter kernel advances the cursors in place and a slice's
    # .contiguous() would alias the offsets storage
    cursors = offsets[:nparts].clone()
    perm = torch.empty(n, dtype=torch.int64, device=dev)
    g.partition_scatter(parts.data_ptr(), n, nparts, cursors.data_ptr(),
                        perm.data_ptr(), stream)
    return offsets, perm


def spark_partition_ids(hash_col: Column, nparts: int) -> torch.Tensor:
    """pmod(murmur3, nparts) — Spark HashPartitioning semantics."""
    g = _native.gpu()
    n = hash_col.size
    out = torch.empty(n, dtype=torch.int32, device=hash_col.device)
    g.pmod(hash_col.data.data_ptr(), n, nparts, out.data_ptr(),
           _native.current_stream())
    return out
