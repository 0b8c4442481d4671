"""JCUDF row <-> columnar conversion (Java API parity: RowConversion.java).

Layout per RowConversion.java:59-101: columns packed in order, each aligned
to its width; validity bytes (bit c%8 of byte c/8, set = valid) right after
the last column; row size padded to 8 bytes. Output batched so each batch
stays under 2GB (reference row_conversion.cu batching).
"""
import struct
from typing import List, Tuple

import torch

from .. import _native
from ..columnar import (Column, DType, FIXED_WIDTH, Table, make_validity)

_ROWCOL_FMT = "<QQii"  # data, valid, width, row_off
MAX_BATCH_BYTES = 2**31 - 8


def row_layout(dtypes: List[DType]) -> Tuple[List[int], int, int]:
    """Returns (per-column row offsets, validity_off, row_size)."""
    off = 0
    offs = []
    for dt in dtypes:
        w = FIXED_WIDTH[dt]
        a = min(w, 8)
        off = (off + a - 1) // a * a
        offs.append(off)
        off += w
    validity_off = off
    off += (len(dtypes) + 7) // 8
    row_size = (off + 7) // 8 * 8
    return offs, validity_off, row_size


def _pack_desc(cols: List[Column], offs, dev):
    raw = bytearray(len(cols) * struct.calcsize(_ROWCOL_FMT))
    for i, c in enumerate(cols):
        struct.pack_into(_ROWCOL_FMT, raw, i * struct.calcsize(_ROWCOL_FMT),
                         c.data.data_ptr(),
                         c.validity.data_ptr() if c.validity is not None else 0,
                         FIXED_WIDTH[c.dtype], offs[i])
    return torch.frombuffer(raw, dtype=torch.uint8).to(dev)


def convert_to_rows(table: Table) -> List[Tuple[torch.Tensor, int]]:
    """Returns a list of (bytes tensor, num_rows) batches; each batch is
    rows laid out back to back at fixed row_size (a LIST<INT8> column in the
    reference API)."""
    cols = table.columns
    for c in cols:
        assert c.dtype not in (DType.STRING, DType.LIST, DType.STRUCT), \
            "fixed-width only (reference convertToRows contract)"
    g = _native.gpu()
    stream = _native.current_stream()
    dev = table.device
    offs, validity_off, row_size = row_layout([c.dtype for c in cols])
    n = table.num_rows
    # 64-row-aligned batches keep validity slicing byte-aligned
    rows_per_batch = max(MAX_BATCH_BYTES // row_size // 64 * 64, 64)
    out = []
    start = 0
    while start < n or n == 0:
        m = min(rows_per_batch, n - start)
        views = []
        for c in cols:
            es = c.data.element_size()
            elems_per_row = FIXED_WIDTH[c.dtype] // es
            views.append(Column(
                c.dtype, m, c.data[start * elems_per_row:],
                c.validity[start // 8:] if c.validity is not None else None,
                scale=c.scale))
        # zeros: padding bytes are part of the format contract
        buf = torch.zeros(max(m * row_size, 1), dtype=torch.uint8, device=dev)
        desc = _pack_desc(views, offs, dev)
        g.to_rows(desc.data_ptr(), len(cols), m, row_size, validity_off,
                  buf.data_ptr(), stream)
        out.append((buf, m))
        start += m
        if n == 0:
            break
    return out


def convert_from_rows(batches: List[Tuple[torch.Tensor, int]],
                      dtypes: List[DType]) -> Table:
    """Inverse: rebuild columns from JCUDF row batches."""
    g = _native.gpu()
    stream = _native.current_stream()
    dev = batches[0][0].device
    offs, validity_off, row_size = row_layout(dtypes)
    total = sum(m for _, m in batches)
    from ..columnar import TORCH_DTYPE
    out_cols = []
    for dt in dtypes:
        numel = total * (2 if dt == DType.DECIMAL128 else 1)
        data = torch.empty(numel, dtype=TORCH_DTYPE[dt], device=dev)
        validity = make_validity(total, dev, fill_valid=False)
        out_cols.append(Column(dt, total, data, validity, null_count=None))
    start = 0
    for buf, m in batches:
        assert start % 64 == 0 or start == 0, "batches must be 64-row aligned"
        views = []
        for c in out_cols:
            w = FIXED_WIDTH[c.dtype]
            es = c.data.element_size()
            views.append(Column(c.dtype, m, c.data[start * (w // es):],
                                c.validity[start // 8:]))
        desc = _pack_desc(views, offs, dev)
        g.from_rows(desc.data_ptr(), len(dtypes), m, row_size, validity_off,
                    buf.data_ptr(), stream)
        start += m
    return Table(out_cols)
