"""Columnar core: Column / Table over torch tensors.

Arrow/cudf-compatible layout (keeps the Kudo wire format byte-identical with
the reference plugin):
  * fixed-width data: one typed torch tensor
  * validity: uint8 bitmask, 1 bit/row LSB-first, padded to a multiple of
    8 bytes so wave64 kernels can write whole 64-bit ballot words
  * strings: int32 offsets[n+1] + uint8 char data
  * lists: int32 offsets[n+1] + one child column; structs: N children

Device memory is owned by torch's caching allocator (the RMM-pool analog;
288 GB HBM3E per MI355X). Kernels receive raw pointers via the packed ColDesc
ABI below (struct layout must match src/gpu/srj_common.hpp).
"""
from __future__ import annotations

import struct
from enum import IntEnum
from typing import List, Optional, Sequence

import torch

from . import _native


class DType(IntEnum):
    BOOL8 = 0
    INT8 = 1
    INT16 = 2
    INT32 = 3
    INT64 = 4
    FLOAT32 = 5
    FLOAT64 = 6
    DATE32 = 7
    TIMESTAMP_US = 8
    STRING = 9
    DECIMAL32 = 10
    DECIMAL64 = 11
    DECIMAL128 = 12
    LIST = 13
    STRUCT = 14


TORCH_DTYPE = {
    DType.BOOL8: torch.int8,
    DType.INT8: torch.int8,
    DType.INT16: torch.int16,
    DType.INT32: torch.int32,
    DType.INT64: torch.int64,
    DType.FLOAT32: torch.float32,
    DType.FLOAT64: torch.float64,
    DType.DATE32: torch.int32,
    DType.TIMESTAMP_US: torch.int64,
    DType.DECIMAL32: torch.int32,
    DType.DECIMAL64: torch.int64,
    DType.DECIMAL128: torch.int64,  # 2 words per row
}

FIXED_WIDTH = {
    DType.BOOL8: 1, DType.INT8: 1, DType.INT16: 2, DType.INT32: 4,
    DType.INT64: 8, DType.FLOAT32: 4, DType.FLOAT64: 8, DType.DATE32: 4,
    DType.TIMESTAMP_US: 8, DType.DECIMAL32: 4, DType.DECIMAL64: 8,
    DType.DECIMAL128: 16,
}

_FROM_TORCH = {
    torch.int8: DType.INT8,
    torch.int16: DType.INT16,
    torch.int32: DType.INT32,
    torch.int64: DType.INT64,
    torch.float32: DType.FLOAT32,
    torch.float64: DType.FLOAT64,
    torch.bool: DType.BOOL8,
}

# must match srj::ColDesc in src/gpu/srj_common.hpp (48 bytes)
_COLDESC_FMT = "<iiQQQiiq"
COLDESC_BYTES = struct.calcsize(_COLDESC_FMT)
assert COLDESC_BYTES == 48


def validity_nbytes(nrows: int) -> int:
    return ((nrows + 63) // 64) * 8


def make_validity(nrows: int, device, fill_valid: bool = True) -> torch.Tensor:
    buf = torch.full((validity_nbytes(nrows),), 0xFF if fill_valid else 0,
                     dtype=torch.uint8, device=device)
    return buf


def validity_from_bools(valid: Sequence[bool], device) -> torch.Tensor:
    n = len(valid)
    buf = bytearray(validity_nbytes(n))
    for i, v in enumerate(valid):
        if v:
            buf[i >> 3] |= 1 << (i & 7)
    return torch.tensor(list(buf), dtype=torch.uint8, device=device)


class Column:
    """One cudf-style column. `data`/`validity`/`offsets` are torch tensors."""

    def __init__(self, dtype: DType, size: int, data: Optional[torch.Tensor],
                 validity: Optional[torch.Tensor] = None,
                 offsets: Optional[torch.Tensor] = None,
                 children: Optional[List["Column"]] = None,
                 scale: int = 0,
                 null_count: Optional[int] = None):
        self.dtype = DType(dtype)
        self.size = int(size)
        self.data = data
        self.validity = validity
        self.offsets = offsets
        self.children = children or []
        self.scale = scale
        self._null_count = null_count if validity is not None else 0

    # -- construction ------------------------------------------------------
    @staticmethod
    def from_torch(t: torch.Tensor, valid: Optional[Sequence[bool]] = None,
                   dtype: Optional[DType] = None) -> "Column":
        dt = dtype if dtype is not None else _FROM_TORCH[t.dtype]
        if t.dtype == torch.bool:
            t = t.to(torch.int8)
        v = validity_from_bools(valid, t.device) if valid is not None else None
        nc = (len(valid) - sum(valid)) if valid is not None else 0
        return Column(dt, t.numel(), t.contiguous(), v, null_count=nc)

    @staticmethod
    def from_pylist(values, dtype: DType, device="cpu", scale: int = 0) -> "Column":
        n = len(values)
        valid = [v is not None for v in values]
        has_null = not all(valid)
        if dtype == DType.STRING:
            chunks = [(v.encode() if isinstance(v, str) else (v or b""))
                      for v in values]
            offs = [0]
            for c in chunks:
                offs.append(offs[-1] + len(c))
            data = torch.frombuffer(bytearray(b"".join(chunks)) or bytearray(1),
                                    dtype=torch.uint8)[:offs[-1]].to(device)
            offsets = torch.tensor(offs, dtype=torch.int32, device=device)
            v = validity_from_bools(valid, device) if has_null else None
            return Column(dtype, n, data, v, offsets,
                          null_count=n - sum(valid))
        tdt = TORCH_DTYPE[dtype]
        fill = [v if v is not None else 0 for v in values]
        if dtype == DType.BOOL8:
            fill = [1 if v else 0 for v in fill]
        if dtype == DType.DECIMAL128:
            import struct as _struct
            words = []
            for v in fill:
                words.extend(_struct.unpack(
                    "<2q", int(v).to_bytes(16, "little", signed=True)))
            data = torch.tensor(words, dtype=torch.int64, device=device)
            v = validity_from_bools(valid, device) if has_null else None
            return Column(dtype, n, data, v, scale=scale,
                          null_count=n - sum(valid))
        data = torch.tensor(fill, dtype=tdt, device=device)
        v = validity_from_bools(valid, device) if has_null else None
        return Column(dtype, n, data, v, scale=scale, null_count=n - sum(valid))

    # -- properties --------------------------------------------------------
    @property
    def device(self):
        for t in (self.data, self.offsets, self.validity):
            if t is not None:
                return t.device
        if self.children:
            return self.children[0].device
        return torch.device("cpu")

    @property
    def null_count(self) -> int:
        if self.validity is None:
            return 0
        if self._null_count is None:
            if self.validity.is_cuda:
                g = _native.gpu()
                out = torch.zeros(1, dtype=torch.int64, device=self.validity.device)
                g.count_set_bits(self.validity.data_ptr(), self.size,
                                 out.data_ptr(), _native.current_stream())
                self._null_count = self.size - int(out.item())
            else:
                bits = 0
                mv = self.validity.numpy()
                for i in range(self.size):
                    bits += (mv[i >> 3] >> (i & 7)) & 1
                self._null_count = self.size - bits
        return self._null_count

    def to(self, device) -> "Column":
        return Column(
            self.dtype, self.size,
            None if self.data is None else self.data.to(device),
            None if self.validity is None else self.validity.to(device),
            None if self.offsets is None else self.offsets.to(device),
            [c.to(device) for c in self.children], self.scale, self._null_count)

    def is_valid_host(self, i: int) -> bool:
        if self.validity is None:
            return True
        b = int(self.validity[i >> 3].item())
        return bool((b >> (i & 7)) & 1)

    def to_pylist(self):
        """Host materialization for tests/debug (never on the hot path)."""
        c = self.to("cpu")
        out = []
        if self.dtype == DType.STRING:
            data = bytes(c.data.numpy().tobytes()) if c.data is not None else b""
            offs = c.offsets.tolist()
            for i in range(c.size):
                out.append(None if not c.is_valid_host(i)
                           else data[offs[i]:offs[i + 1]].decode("utf-8", "replace"))
            return out
        if self.dtype == DType.LIST:
            child = c.children[0].to_pylist()
            offs = c.offsets.tolist()
            return [None if not c.is_valid_host(i) else child[offs[i]:offs[i + 1]]
                    for i in range(c.size)]
        if self.dtype == DType.STRUCT:
            kids = [ch.to_pylist() for ch in c.children]
            return [None if not c.is_valid_host(i) else tuple(k[i] for k in kids)
                    for i in range(c.size)]
        if self.dtype == DType.DECIMAL128:
            raw = c.data.numpy().tobytes()
            vals = [int.from_bytes(raw[i * 16:(i + 1) * 16], "little",
                                   signed=True) for i in range(c.size)]
            return [v if c.is_valid_host(i) else None
                    for i, v in enumerate(vals)]
        vals = c.data.tolist()
        if self.dtype == DType.BOOL8:
            vals = [bool(v) for v in vals]
        return [v if c.is_valid_host(i) else None for i, v in enumerate(vals)]

    def __repr__(self):
        return (f"Column({self.dtype.name}, size={self.size}, "
                f"nulls={self.null_count}, device={self.device})")


class Table:
    def __init__(self, columns: List[Column]):
        assert columns, "empty table"
        n = columns[0].size
        assert all(c.size == n for c in columns)
        self.columns = list(columns)

    @property
    def num_rows(self) -> int:
        return self.columns[0].size

    @property
    def num_columns(self) -> int:
        return len(self.columns)

    @property
    def device(self):
        return self.columns[0].device

    def to(self, device) -> "Table":
        return Table([c.to(device) for c in self.columns])


# ---------------------------------------------------------------------------
# ColDesc packing (flattened pre-order, matching src/gpu/srj_common.hpp)
# ---------------------------------------------------------------------------

def _flatten(cols: Sequence[Column]):
    flat: List[Column] = []
    top: List[int] = []

    def add(c: Column) -> int:
        idx = len(flat)
        flat.append(c)
        return idx

    def walk(c: Column) -> int:
        idx = add(c)
        child_idxs = [walk(ch) for ch in c.children]
        # record first-child index; children are contiguous in pre-order only
        # if each child subtree is flattened consecutively, which walk() does.
        c._child0 = child_idxs[0] if child_idxs else 0
        return idx

    for c in cols:
        top.append(walk(c))
    return flat, top


# pinned staging for descriptor uploads. During hipGraph capture no pinned
# allocation (hipHostRegister) is permitted, so capture-time descriptors are
# bump-allocated from a pre-created pinned arena; regions referenced by a
# captured graph are never recycled (the graph re-reads them at replay).
_PIN_ARENA = None
_PIN_POS = 0


def ensure_pinned_arena() -> None:
    """Create the capture-time pinned staging arena. MUST be called before
    hipGraph capture (pinned allocation is illegal mid-capture);
    graphs.CapturedPipeline does this automatically."""
    global _PIN_ARENA
    if _PIN_ARENA is None:
        _PIN_ARENA = torch.empty(1 << 22, dtype=torch.uint8, pin_memory=True)


def _stage_pinned(raw) -> torch.Tensor:
    global _PIN_ARENA, _PIN_POS
    host = torch.frombuffer(bytearray(raw) if not isinstance(raw, bytearray)
                            else raw, dtype=torch.uint8)
    if not torch.cuda.is_current_stream_capturing():
        ensure_pinned_arena()
        return host.pin_memory()
    n = (len(host) + 255) & ~255
    assert _PIN_ARENA is not None and _PIN_POS + n <= _PIN_ARENA.numel(), \
        "pinned arena exhausted during graph capture"
    view = _PIN_ARENA[_PIN_POS:_PIN_POS + len(host)]
    view.copy_(host)  # host->pinned memcpy, no stream work
    _PIN_POS += n
    return view


def pack_descriptors(cols: Sequence[Column], device=None):
    """Pack columns into a device ColDesc array + top-level index array.

    Returns (desc_tensor, top_tensor, keepalive) where keepalive holds tensor
    references that must outlive the kernel launch.
    """
    flat, top = _flatten(cols)
    device = device or flat[0].device
    raw = bytearray(COLDESC_BYTES * len(flat))
    for i, c in enumerate(flat):
        data_ptr = c.data.data_ptr() if c.data is not None and c.data.numel() else 0
        valid_ptr = c.validity.data_ptr() if c.validity is not None else 0
        offs_ptr = c.offsets.data_ptr() if c.offsets is not None else 0
        struct.pack_into(_COLDESC_FMT, raw, i * COLDESC_BYTES,
                         int(c.dtype), c.scale, data_ptr, valid_ptr, offs_ptr,
                         len(c.children), getattr(c, "_child0", 0), c.size)
    if (torch.cuda.is_available() and str(device).startswith("cuda")
            and torch.cuda.is_current_stream_capturing()):
        # hipGraph capture: pageable H2D is not capturable — stage through
        # the pre-created pinned arena (regions live as long as the graph)
        pinned = _stage_pinned(raw)
        desc = pinned.to(device, non_blocking=True)
        traw = bytearray(max(len(top), 1) * 4)
        struct.pack_into(f"<{len(top)}i", traw, 0, *top)
        tpin = _stage_pinned(traw)
        top_t = tpin.view(torch.int32)[:len(top)].to(device,
                                                     non_blocking=True)
        return desc, top_t, (flat, pinned, tpin)
    desc_host = torch.frombuffer(raw, dtype=torch.uint8)
    desc = desc_host.to(device)
    top_t = torch.tensor(top, dtype=torch.int32, device=device)
    return desc, top_t, (flat, desc_host)


def validity_to_bool(mask: torch.Tensor, n: int) -> torch.Tensor:
    """LSB-first validity bitmask -> bool tensor of length n (device op)."""
    shifts = torch.arange(8, dtype=torch.uint8, device=mask.device)
    bits = (mask.unsqueeze(1) >> shifts) & 1
    return bits.reshape(-1)[:n].to(torch.bool)


# ---------------------------------------------------------------------------
# Arrow interop: the Column layout IS Arrow's (validity bitmap LSB-first,
# int32 string offsets + data buffer), so conversion is buffer re-wrapping,
# not element-wise work.
# ---------------------------------------------------------------------------

_ARROW_TYPES = None


def _arrow_map():
    global _ARROW_TYPES
    if _ARROW_TYPES is None:
        import pyarrow as pa
        _ARROW_TYPES = {
            DType.INT8: pa.int8(), DType.INT16: pa.int16(),
            DType.INT32: pa.int32(), DType.INT64: pa.int64(),
            DType.FLOAT32: pa.float32(), DType.FLOAT64: pa.float64(),
            DType.BOOL8: pa.int8(), DType.DATE32: pa.date32(),
            DType.TIMESTAMP_US: pa.timestamp("us"),
            DType.STRING: pa.string(),
        }
    return _ARROW_TYPES


def from_arrow(arr) -> Column:
    """pyarrow Array/ChunkedArray -> host Column (zero-copy where the Arrow
    buffers are already in our layout; bools are unpacked to int8)."""
    import pyarrow as pa
    if isinstance(arr, pa.ChunkedArray):
        arr = arr.combine_chunks()
    if arr.offset != 0 or isinstance(arr, pa.BooleanArray):
        # normalize slices / bit-packed bools through pylist (cold path)
        dt = ({pa.bool_(): DType.BOOL8}.get(arr.type)
              or next(k for k, v in _arrow_map().items() if v == arr.type))
        return Column.from_pylist(arr.to_pylist(), dt)
    import numpy as np
    bufs = arr.buffers()
    n = len(arr)
    validity = None
    if bufs[0] is not None and arr.null_count:
        vb = np.frombuffer(bufs[0], dtype=np.uint8)
        padded = np.zeros(validity_nbytes(n), dtype=np.uint8)
        padded[:len(vb)] = vb
        validity = torch.from_numpy(padded)
    if pa.types.is_string(arr.type) or pa.types.is_binary(arr.type):
        offs = np.frombuffer(bufs[1], dtype=np.int32)[:n + 1].copy()
        chars = (np.frombuffer(bufs[2], dtype=np.uint8).copy()
                 if bufs[2] is not None else np.zeros(0, dtype=np.uint8))
        return Column(DType.STRING, n, torch.from_numpy(chars), validity,
                      torch.from_numpy(offs), null_count=arr.null_count)
    dt = next(k for k, v in _arrow_map().items() if v == arr.type)
    np_dt = {DType.INT8: np.int8, DType.INT16: np.int16, DType.INT32: np.int32,
             DType.INT64: np.int64, DType.FLOAT32: np.float32,
             DType.FLOAT64: np.float64, DType.DATE32: np.int32,
             DType.TIMESTAMP_US: np.int64}[dt]
    data = np.frombuffer(bufs[1], dtype=np_dt)[:n].copy()
    return Column(dt, n, torch.from_numpy(data), validity,
                  null_count=arr.null_count)


def to_arrow(col: Column):
    """host/device Column -> pyarrow Array (host copy if on GPU)."""
    import numpy as np
    import pyarrow as pa
    c = col.to("cpu")
    n = c.size
    at = _arrow_map()[c.dtype]
    vbuf = None
    if c.validity is not None:
        vbuf = pa.py_buffer(c.validity.numpy().tobytes())
    if c.dtype == DType.STRING:
        return pa.StringArray.from_buffers(
            n, pa.py_buffer(c.offsets.numpy().tobytes()),
            pa.py_buffer((c.data.numpy().tobytes()
                          if c.data is not None else b"")),
            vbuf, c.null_count if c.validity is not None else 0)
    if c.dtype == DType.BOOL8:
        return pa.array(
            [None if not c.is_valid_host(i) else bool(int(c.data[i]))
             for i in range(n)], type=pa.bool_())
    buf = pa.py_buffer(c.data.numpy().tobytes())
    return pa.Array.from_buffers(
        at, n, [vbuf, buf],
        null_count=c.null_count if c.validity is not None else 0)
