"""Device-side Kudo shuffle: split a GPU table into per-partition Kudo
records and assemble received records back into one GPU table.

Java API parity: kudo/KudoGpuSerializer.java:50,71
(splitAndSerializeToDevice / assembleFromDeviceRaw) and the native
shuffle_split / shuffle_assemble pair.

The per-partition output is a byte-exact Kudo record (same format as
spark_rapids_jni_amd/kudo.py writes on the host — cross-validated in
tests/test_shuffle_gpu.py), so records can be merged by either path. The
buffers are laid out contiguously per destination rank and feed RCCL
all_to_all_single directly — no host bounce.

Nested schemas (STRUCT/LIST, arbitrarily deep) are handled natively:
child row ranges derive from parent offset values (fetched at partition
boundaries on split; staged offset sections on assemble), matching the
host-path slicing semantics exactly.
"""
import struct
from typing import List, Optional, Sequence, Tuple

import numpy as np
import torch

from . import _native
from .columnar import Column, DType, FIXED_WIDTH, Table, validity_nbytes
from .kudo import KudoTableHeader, MAGIC

_COPYSEG = "<QQq"          # src, dst, nbytes
_VALIDSEG = "<QQqqq"       # src, dst, src_start_bit, dst_start_bit, nbits
_OFFSEG = "<QQqii"         # src, dst, n, base, write_last
COPY_CHUNK = 16384


class _SegBatch:
    """Accumulates copy segments + chunk prefix, then launches one kernel."""

    def __init__(self):
        self.segs = []
        self.chunks = []

    def add(self, src_ptr: int, dst_ptr: int, nbytes: int):
        if nbytes <= 0:
            return
        self.segs.append(struct.pack(_COPYSEG, src_ptr, dst_ptr, nbytes))
        self.chunks.append((nbytes + COPY_CHUNK - 1) // COPY_CHUNK)

    def run(self, device, g, stream):
        if not self.segs:
            return
        raw = b"".join(self.segs)
        prefix = np.zeros(len(self.chunks), dtype=np.int64)
        np.cumsum(self.chunks[:-1], out=prefix[1:])
        total = int(prefix[-1] + self.chunks[-1])
        segs_t = torch.frombuffer(bytearray(raw), dtype=torch.uint8).to(device)
        pref_t = torch.from_numpy(prefix).to(device)
        g.segmented_copy(segs_t.data_ptr(), pref_t.data_ptr(), len(self.chunks),
                         total, stream)
        return segs_t, pref_t  # keepalive


def _flatten_plan(cols, bounds, dev):
    """Depth-first flatten with per-partition absolute row bounds.

    bounds: list of nparts+1 absolute row indices for this nesting level.
    Returns [(col, bounds, char_bounds)] — char_bounds only for STRING
    (absolute char positions at partition boundaries, one small D2H per
    offset-bearing column, mirroring the host writer kudo.py:152).
    """
    import torch as _t
    out = []
    for c in cols:
        cb = None
        child_bounds = None
        if c.dtype in (DType.STRING, DType.LIST):
            idx = _t.tensor(bounds, dtype=_t.int64, device=dev)
            vals = c.offsets[idx].cpu().tolist()
            if c.dtype == DType.STRING:
                cb = vals
            else:
                child_bounds = vals
        out.append((c, bounds, cb))
        if c.dtype == DType.STRUCT:
            out.extend(_flatten_plan(c.children, bounds, dev))
        elif c.dtype == DType.LIST:
            out.extend(_flatten_plan([c.children[0]], child_bounds, dev))
    return out


def split_and_serialize_to_device(table: Table, offsets: torch.Tensor,
                                  perm: torch.Tensor
                                  ) -> Tuple[torch.Tensor, List[int]]:
    """Split rows (already partition-grouped by `perm`) into P contiguous
    device Kudo records. Returns (one uint8 buffer, per-partition byte sizes).
    Supports nested schemas (STRUCT/LIST): child row ranges derive from the
    parent's offset values at partition boundaries, as in the host writer.
    """
    from .ops.copying import gather
    g = _native.gpu()
    stream = _native.current_stream()
    dev = table.device
    G = gather(table, perm)
    offs = offsets.cpu().tolist()
    nparts = len(offs) - 1

    flat = _flatten_plan(G.columns, offs, dev)
    ncols = len(flat)
    hlen = 28 + (ncols + 7) // 8

    # plan per-partition record layout
    headers = bytearray()
    sizes = []
    seg_plan = []  # (src_ptr, rec_off, nbytes) rec-relative; headers separate
    positions = []
    pos = 0
    for p in range(nparts):
        top_start, top_n = offs[p], offs[p + 1] - offs[p]
        bitset = bytearray((ncols + 7) // 8)
        validity_parts = []  # (src_ptr, nbytes)
        offset_parts = []
        data_parts = []
        for ci, (c, bnds, cb) in enumerate(flat):
            start, end = bnds[p], bnds[p + 1]
            n = end - start
            if c.validity is not None and n > 0:
                # n == 0: prefer "no validity" (kudo spec) over a filler byte
                bitset[ci // 8] |= 1 << (ci % 8)
                b0 = start // 8
                b1 = max((start + n + 7) // 8, b0 + 1)
                validity_parts.append((c.validity.data_ptr() + b0, b1 - b0))
            if c.dtype in (DType.STRING, DType.LIST):
                if n > 0:
                    offset_parts.append((c.offsets.data_ptr() + start * 4,
                                         (n + 1) * 4))
            if c.dtype == DType.STRING:
                if n > 0:
                    nchars = cb[p + 1] - cb[p]
                    if nchars > 0:
                        data_parts.append((c.data.data_ptr() + cb[p], nchars))
            elif c.dtype not in (DType.STRUCT, DType.LIST):
                w = FIXED_WIDTH[c.dtype]
                if n > 0:
                    data_parts.append((c.data.data_ptr() + start * w, n * w))
        vlen = sum(x[1] for x in validity_parts)
        vpad = (4 - (hlen + vlen) % 4) % 4
        olen = sum(x[1] for x in offset_parts)
        dlen = sum(x[1] for x in data_parts)
        dpad = (4 - dlen % 4) % 4
        h = KudoTableHeader(top_start, top_n, vlen + vpad, olen, 0, ncols,
                            bytes(bitset))
        h.total_len = h.validity_len + olen + dlen + dpad
        import io
        b = io.BytesIO()
        h.write(b)
        headers.extend(b.getvalue())
        rec_len = hlen + h.total_len
        # record-relative positions of body parts
        rp = hlen
        for src, nb in validity_parts:
            seg_plan.append((src, pos + rp, nb))
            rp += nb
        rp += vpad
        for src, nb in offset_parts:
            seg_plan.append((src, pos + rp, nb))
            rp += nb
        for src, nb in data_parts:
            seg_plan.append((src, pos + rp, nb))
            rp += nb
        positions.append(pos)
        sizes.append(rec_len)
        pos += rec_len

    out = torch.zeros(max(pos, 1), dtype=torch.uint8, device=dev)
    base = out.data_ptr()
    batch = _SegBatch()
    # headers: staged once, copied into place
    hdr_stage = torch.frombuffer(headers or bytearray(1),
                                 dtype=torch.uint8).to(dev)
    for p in range(nparts):
        batch.add(hdr_stage.data_ptr() + p * hlen, base + positions[p], hlen)
    for src, dst_off, nb in seg_plan:
        batch.add(src, base + dst_off, nb)
    keep = batch.run(dev, g, stream)
    return out, sizes


def _flat_count(cols) -> int:
    n = 0
    for c in cols:
        n += 1
        if c.dtype == DType.STRUCT:
            n += _flat_count(c.children)
        elif c.dtype == DType.LIST:
            n += _flat_count([c.children[0]])
    return n


def assemble_from_device(buffers: Sequence[torch.Tensor],
                         schema: Sequence[Column]) -> Table:
    """Merge device Kudo records into one device table (inverse of split).

    Nested schemas supported: each piece's offsets section is staged to host
    once for layout planning (counts/positions); validity merge, offset
    rebase and data copies all run in batched device kernels.
    """
    g = _native.gpu()
    stream = _native.current_stream()
    dev = buffers[0].device
    ncols = _flat_count(schema)
    hlen = 28 + (ncols + 7) // 8

    import io

    # parse headers + stage offset sections (small D2H per piece)
    headers: List[KudoTableHeader] = []
    off_sections: List[np.ndarray] = []
    for b in buffers:
        raw = b[:hlen].cpu().numpy().tobytes()
        h = KudoTableHeader.read(io.BytesIO(raw))
        assert h.num_columns == ncols, \
            f"schema mismatch: record has {h.num_columns} flat columns"
        headers.append(h)
        o0 = hlen + h.validity_len
        sec = (b[o0:o0 + h.offset_len].cpu().numpy().view(np.int32)
               if h.offset_len else np.empty(0, dtype=np.int32))
        off_sections.append(sec)

    # walk each piece's flattened schema computing per-column layout
    # rec: dict(n, start_bit, valid_ptr?, off_ptr?, first_off?, data_ptr?, ...)
    piece_cols: List[List[dict]] = []
    for b, h, osec in zip(buffers, headers, off_sections):
        base = b.data_ptr()
        vpos = hlen
        opos = hlen + h.validity_len
        dpos = opos + h.offset_len
        oidx = 0  # index into osec (int32 words)
        cols: List[dict] = []

        def walk(cs, start_bit, n):
            nonlocal vpos, opos, dpos, oidx
            for c in cs:
                ci = len(cols)
                rec = {"n": n, "start_bit": start_bit}
                cols.append(rec)
                if h.has_validity_buffer(ci):
                    if n > 0:
                        rec["valid_ptr"] = base + vpos
                        vpos += (start_bit + n + 7) // 8
                    else:
                        vpos += 1  # host-writer filler byte
                first = last = 0
                if c.dtype in (DType.STRING, DType.LIST) and n > 0:
                    rec["off_ptr"] = base + opos
                    first = int(osec[oidx])
                    last = int(osec[oidx + n])
                    rec["first_off"] = first
                    opos += (n + 1) * 4
                    oidx += n + 1
                if c.dtype == DType.STRING:
                    nch = last - first
                    rec["nchars"] = nch
                    if nch > 0:
                        rec["data_ptr"] = base + dpos
                        dpos += nch
                elif c.dtype == DType.STRUCT:
                    walk(c.children, start_bit, n)
                elif c.dtype == DType.LIST:
                    walk([c.children[0]], first % 8, last - first)
                elif n > 0:
                    w = FIXED_WIDTH[c.dtype]
                    rec["data_ptr"] = base + dpos
                    dpos += n * w
        walk(list(schema), h.offset % 8, h.num_rows)
        piece_cols.append(cols)

    copy_batch = _SegBatch()
    vsegs, vwords = [], []
    osegs, ocounts = [], []
    keepalive = []
    flat_idx = [0]

    def build(cs) -> List[Column]:
        out_cols = []
        for c in cs:
            ci = flat_idx[0]
            flat_idx[0] += 1
            recs = [pc[ci] for pc in piece_cols]
            total_rows = sum(r["n"] for r in recs)
            any_valid = any("valid_ptr" in r for r in recs)
            validity = None
            if any_valid:
                validity = torch.zeros(validity_nbytes(total_rows),
                                       dtype=torch.uint8, device=dev)
                row_pos = 0
                for r in recs:
                    n = r["n"]
                    if n > 0:
                        src = r.get("valid_ptr", 0)
                        sbit = r["start_bit"] if src else 0
                        vsegs.append(struct.pack(
                            _VALIDSEG, src, validity.data_ptr(), sbit,
                            row_pos, n))
                        w0 = row_pos // 64
                        w1 = (row_pos + n - 1) // 64
                        vwords.append(w1 - w0 + 1)
                    row_pos += n
            offsets_t = None
            if c.dtype in (DType.STRING, DType.LIST):
                offsets_t = torch.zeros(total_rows + 1, dtype=torch.int32,
                                        device=dev)
                row_pos = 0
                base_val = 0
                for r in recs:
                    n = r["n"]
                    if n > 0:
                        child_n = (r["nchars"] if c.dtype == DType.STRING
                                   else pcount(r, c))
                        is_last = row_pos + n == total_rows
                        osegs.append(struct.pack(
                            _OFFSEG, r["off_ptr"],
                            offsets_t.data_ptr() + row_pos * 4, n, base_val,
                            1 if is_last else 0))
                        ocounts.append(n + (1 if is_last else 0))
                        base_val += child_n
                        row_pos += n
            if c.dtype == DType.STRUCT:
                children = build(c.children)
                out_cols.append(Column(DType.STRUCT, total_rows, None,
                                       validity, children=children,
                                       null_count=None))
                continue
            if c.dtype == DType.LIST:
                children = build([c.children[0]])
                out_cols.append(Column(DType.LIST, total_rows, None, validity,
                                       offsets_t, children, c.scale,
                                       null_count=None))
                continue
            if c.dtype == DType.STRING:
                nchars_total = sum(r.get("nchars", 0) for r in recs)
                out_chars = torch.empty(max(nchars_total, 1),
                                        dtype=torch.uint8, device=dev)
                char_pos = 0
                for r in recs:
                    nch = r.get("nchars", 0)
                    if nch > 0:
                        copy_batch.add(r["data_ptr"],
                                       out_chars.data_ptr() + char_pos, nch)
                    char_pos += nch
                out_cols.append(Column(DType.STRING, total_rows, out_chars,
                                       validity, offsets_t, null_count=None))
                continue
            from .columnar import TORCH_DTYPE
            w = FIXED_WIDTH[c.dtype]
            numel = total_rows * (2 if c.dtype == DType.DECIMAL128 else 1)
            out_data = torch.empty(max(numel, 1), dtype=TORCH_DTYPE[c.dtype],
                                   device=dev)
            row_pos = 0
            for r in recs:
                n = r["n"]
                if n > 0:
                    copy_batch.add(r["data_ptr"],
                                   out_data.data_ptr() + row_pos * w, n * w)
                row_pos += n
            out_cols.append(Column(c.dtype, total_rows, out_data, validity,
                                   scale=c.scale, null_count=None))
        return out_cols

    def pcount(rec, c):
        # child rows contributed by this piece for a LIST column
        return rec["_child_n"] if rec["n"] > 0 else 0

    # fill _child_n from the already-walked per-piece layouts
    def fill_child_counts(cs, idx):
        for c in cs:
            ci = idx[0]
            idx[0] += 1
            if c.dtype == DType.LIST:
                child_idx = idx[0]
                for pc in piece_cols:
                    r = pc[ci]
                    r["_child_n"] = pc[child_idx]["n"] if r["n"] > 0 else 0
                fill_child_counts([c.children[0]], idx)
            elif c.dtype == DType.STRUCT:
                fill_child_counts(c.children, idx)
    fill_child_counts(list(schema), [0])

    out_cols = build(list(schema))

    keepalive.append(copy_batch.run(dev, g, stream))
    if vsegs:
        raw = b"".join(vsegs)
        prefix = np.zeros(len(vwords), dtype=np.int64)
        np.cumsum(vwords[:-1], out=prefix[1:])
        total_words = int(prefix[-1] + vwords[-1])
        segs_t = torch.frombuffer(bytearray(raw), dtype=torch.uint8).to(dev)
        pref_t = torch.from_numpy(prefix).to(dev)
        g.validity_merge(segs_t.data_ptr(), pref_t.data_ptr(), len(vwords),
                         total_words, stream)
        keepalive.append((segs_t, pref_t))
    if osegs:
        raw = b"".join(osegs)
        prefix = np.zeros(len(ocounts), dtype=np.int64)
        np.cumsum(ocounts[:-1], out=prefix[1:])
        total = int(prefix[-1] + ocounts[-1])
        segs_t = torch.frombuffer(bytearray(raw), dtype=torch.uint8).to(dev)
        pref_t = torch.from_numpy(prefix).to(dev)
        g.offsets_rebase(segs_t.data_ptr(), pref_t.data_ptr(), len(ocounts),
                         total, stream)
        keepalive.append((segs_t, pref_t))
    torch.cuda.synchronize()  # keepalive tensors may go out of scope
    return Table(out_cols)


def assemble_from_device_raw(buffer: torch.Tensor, sizes: Sequence[int],
                             schema: Sequence[Column]) -> Table:
    """KudoGpuSerializer.assembleFromDeviceRaw: one concatenated buffer."""
    views = []
    pos = 0
    for s in sizes:
        views.append(buffer[pos:pos + s])
        pos += s
    return assemble_from_device(views, schema)
