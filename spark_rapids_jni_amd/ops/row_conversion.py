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


# ---------------------------------------------------------------------------
# variable-width (strings) path — reference row_conversion.cu
# copy_strings_to_rows/copy_strings_from_rows + build_string_row_offsets:
# string columns hold an (offset-in-row, length) int32 pair in the fixed
# section; chars follow the validity bytes; rows vary in size and are tracked
# by a row-offsets array (the reference's LIST<INT8> offsets).
# ---------------------------------------------------------------------------

def var_row_layout(dtypes: List[DType]) -> Tuple[List[int], int, int]:
    """(per-column fixed-section offsets, validity_off, fixed_size).
    STRING columns take an 8-byte (offset,len) slot aligned to 4."""
    off = 0
    offs = []
    for dt in dtypes:
        if dt == DType.STRING:
            w, a = 8, 4
        else:
            w = FIXED_WIDTH[dt]
            a = min(w, 8)
        off = (off + a - 1) // a * a
        offs.append(off)
        off += w
    validity_off = off
    off += (len(dtypes) + 7) // 8
    return offs, validity_off, (off + 7) // 8 * 8


def _pack_desc_var(cols: List[Column], offs, dev):
    sz = struct.calcsize(_ROWCOL_FMT)
    raw = bytearray(len(cols) * sz)
    chars = []
    for i, c in enumerate(cols):
        is_str = c.dtype == DType.STRING
        struct.pack_into(
            _ROWCOL_FMT, raw, i * sz,
            c.offsets.data_ptr() if is_str else c.data.data_ptr(),
            c.validity.data_ptr() if c.validity is not None else 0,
            0 if is_str else FIXED_WIDTH[c.dtype], offs[i])
        chars.append(c.data.data_ptr() if is_str else 0)
    desc = torch.frombuffer(raw, dtype=torch.uint8).to(dev)
    cptr = torch.tensor(chars, dtype=torch.int64).to(dev)
    return desc, cptr


def convert_to_rows_varwidth(table: Table):
    """Returns (bytes tensor, int32 row_offsets [n+1]) — one LIST<INT8>
    column with per-row offsets (rows are 8-byte padded, variable size)."""
    cols = table.columns
    g = _native.gpu()
    stream = _native.current_stream()
    dev = table.device
    dtypes = [c.dtype for c in cols]
    offs, validity_off, fixed_size = var_row_layout(dtypes)
    n = table.num_rows
    sizes = torch.empty(max(n, 1), dtype=torch.int32, device=dev)
    desc, cptr = _pack_desc_var(cols, offs, dev)
    g.var_row_sizes(desc.data_ptr(), len(cols), n, fixed_size,
                    sizes.data_ptr(), stream)
    row_offsets = torch.zeros(n + 1, dtype=torch.int64, device=dev)
    if n:
        torch.cumsum(sizes[:n], 0, out=row_offsets[1:])
    total = int(row_offsets[-1].item())
    assert total < MAX_BATCH_BYTES, "var-width batch exceeds 2GB"
    row_offsets = row_offsets.to(torch.int32)
    buf = torch.zeros(max(total, 1), dtype=torch.uint8, device=dev)
    if n:
        g.to_rows_var(desc.data_ptr(), cptr.data_ptr(), len(cols), n,
                      fixed_size, validity_off, row_offsets.data_ptr(),
                      buf.data_ptr(), stream)
    return buf, row_offsets


def convert_from_rows_varwidth(buf: torch.Tensor, row_offsets: torch.Tensor,
                               dtypes: List[DType]) -> Table:
    """Inverse of convert_to_rows_varwidth."""
    g = _native.gpu()
    stream = _native.current_stream()
    dev = buf.device
    offs, validity_off, _fixed = var_row_layout(dtypes)
    n = row_offsets.numel() - 1
    from ..columnar import TORCH_DTYPE
    out_cols = []
    lens = {}
    for i, dt in enumerate(dtypes):
        validity = make_validity(n, dev, fill_valid=False)
        if dt == DType.STRING:
            lens[i] = torch.empty(max(n, 1), dtype=torch.int32, device=dev)
            out_cols.append(Column(dt, n, torch.empty(0, dtype=torch.uint8,
                                                      device=dev), validity,
                                   offsets=torch.zeros(n + 1, dtype=torch.int32,
                                                       device=dev),
                                   null_count=None))
        else:
            numel = n * (2 if dt == DType.DECIMAL128 else 1)
            data = torch.empty(max(numel, 1), dtype=TORCH_DTYPE[dt], device=dev)
            out_cols.append(Column(dt, n, data, validity, null_count=None))
    if n == 0:
        return Table(out_cols)
    sz = struct.calcsize(_ROWCOL_FMT)
    raw = bytearray(len(dtypes) * sz)
    for i, c in enumerate(out_cols):
        struct.pack_into(_ROWCOL_FMT, raw, i * sz,
                         0 if c.dtype == DType.STRING else c.data.data_ptr(),
                         c.validity.data_ptr(),
                         0 if c.dtype == DType.STRING else FIXED_WIDTH[c.dtype],
                         offs[i])
    desc = torch.frombuffer(raw, dtype=torch.uint8).to(dev)
    lptr = torch.tensor(
        [lens[i].data_ptr() if i in lens else 0 for i in range(len(dtypes))],
        dtype=torch.int64).to(dev)
    zero = torch.zeros(len(dtypes), dtype=torch.int64).to(dev)
    # phase 0: per-row string lengths
    g.from_rows_var(desc.data_ptr(), zero.data_ptr(), lptr.data_ptr(),
                    len(dtypes), n, validity_off, row_offsets.data_ptr(),
                    buf.data_ptr(), 0, stream)
    # cumsum lengths -> char offsets, allocate chars, phase 1
    chars = []
    ooffs = {}
    for i, dt in enumerate(dtypes):
        if dt == DType.STRING:
            o = torch.zeros(n + 1, dtype=torch.int64, device=dev)
            torch.cumsum(lens[i][:n], 0, out=o[1:])
            o32 = o.to(torch.int32)
            ooffs[i] = o32
            nb = int(o[-1].item())
            data = torch.empty(max(nb, 1), dtype=torch.uint8, device=dev)
            out_cols[i] = Column(dt, n, data, out_cols[i].validity,
                                 offsets=o32, null_count=None)
            chars.append(data.data_ptr())
        else:
            chars.append(0)
    cptr = torch.tensor(chars, dtype=torch.int64).to(dev)
    optr = torch.tensor(
        [ooffs[i].data_ptr() if i in ooffs else 0 for i in range(len(dtypes))],
        dtype=torch.int64).to(dev)
    g.from_rows_var(desc.data_ptr(), cptr.data_ptr(), optr.data_ptr(),
                    len(dtypes), n, validity_off, row_offsets.data_ptr(),
                    buf.data_ptr(), 1, stream)
    return Table(out_cols)
