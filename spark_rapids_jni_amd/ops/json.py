"""JSON ops (Java API parity: JSONUtils.java — getJsonObject,
extractRawMapFromJsonString, fromJsonToStructs).

JSONPath support: $.key, ['key'], [n], [*], .* — Spark/Hive get_json_object
semantics (single string match unquoted+unescaped, container matches raw,
multiple wildcard matches wrapped as a JSON array, missing -> null).
"""
import re
import struct
from typing import List, Optional, Tuple

import torch

from .. import _native
from ..columnar import Column, DType, Table, make_validity, pack_descriptors

MAX_PATH_DEPTH = 16

_PATH_TOKEN = re.compile(
    r"\.(\*)|\.([A-Za-z_][A-Za-z0-9_\- ]*)|\['([^']*)'\]|\[(\*)\]|\[(\d+)\]")


class JsonPathError(ValueError):
    pass


def compile_path(path: str) -> List[Tuple[int, Optional[str], int]]:
    """'$.a[2].b[*]' -> [(0,'a',0),(1,None,2),(0,'b',0),(2,None,0)]"""
    if not path.startswith("$"):
        raise JsonPathError(f"path must start with $: {path!r}")
    pos = 1
    out = []
    while pos < len(path):
        m = _PATH_TOKEN.match(path, pos)
        if not m:
            raise JsonPathError(f"bad path at {pos}: {path!r}")
        if m.group(1) or m.group(4):
            out.append((2, None, 0))
        elif m.group(2) is not None:
            out.append((0, m.group(2), 0))
        elif m.group(3) is not None:
            out.append((0, m.group(3), 0))
        else:
            out.append((1, None, int(m.group(5))))
        pos = m.end()
    if len(out) > MAX_PATH_DEPTH:
        raise JsonPathError("path too deep")
    return out


def _pack_instrs(instrs, dev):
    keychars = bytearray()
    raw = bytearray()
    for kind, key, idx in instrs:
        off = len(keychars)
        klen = 0
        if key is not None:
            kb = key.encode()
            keychars.extend(kb)
            klen = len(kb)
        raw += struct.pack("<iiii", kind, off, klen, idx)
    it = torch.frombuffer(raw or bytearray(1), dtype=torch.uint8).to(dev)
    kt = torch.frombuffer(keychars or bytearray(1), dtype=torch.uint8).to(dev)
    return it, kt


def get_json_object(col: Column, path: str) -> Column:
    """reference get_json_object.hpp:44 / JSONUtils.java:28."""
    g = _native.gpu()
    stream = _native.current_stream()
    n = col.size
    dev = col.device
    instrs = compile_path(path)
    it, kt = _pack_instrs(instrs, dev)
    desc, top, keep = pack_descriptors([col])
    lens = torch.empty(n, dtype=torch.int32, device=dev)
    g.get_json_object(desc.data_ptr(), n, it.data_ptr(), kt.data_ptr(),
                      len(instrs), 0, lens.data_ptr(), 0, 0, 0, stream)
    offsets = torch.zeros(n + 1, dtype=torch.int32, device=dev)
    torch.cumsum(lens, 0, out=offsets[1:].view(n))
    nchars = int(offsets[-1].item())
    chars = torch.empty(max(nchars, 1), dtype=torch.uint8, device=dev)
    validity = make_validity(n, dev)
    g.get_json_object(desc.data_ptr(), n, it.data_ptr(), kt.data_ptr(),
                      len(instrs), 1, 0, offsets.data_ptr(), chars.data_ptr(),
                      validity.data_ptr(), stream)
    return Column(DType.STRING, n, chars[:nchars], validity, offsets,
                  null_count=None)


_MULTIOUT_FMT = "<QQQQQ"  # lens, offsets, chars, out_valid, overflow


def get_json_object_multiple_paths(col: Column, paths: List[str]) -> List[Column]:
    """reference JSONUtils.getJsonObjectMultiplePaths (get_json_object.cu
    multi-path kernel): the document is tokenized ONCE per row and a bitmask
    of still-viable paths rides the single walk — N paths cost roughly one
    scan instead of N. Falls back per path when a row exceeds the multi-path
    match cap (>4 wildcard matches)."""
    if len(paths) == 0:
        return []
    if len(paths) == 1:
        return [get_json_object(col, paths[0])]
    out: List[Optional[Column]] = [None] * len(paths)
    for c0 in range(0, len(paths), 8):
        chunk = paths[c0:c0 + 8]
        cols = _multi_path_chunk(col, chunk)
        for j, oc in enumerate(cols):
            out[c0 + j] = oc
    return out  # type: ignore[return-value]


def _multi_path_chunk(col: Column, paths: List[str]) -> List[Optional[Column]]:
    g = _native.gpu()
    stream = _native.current_stream()
    n = col.size
    dev = col.device
    npaths = len(paths)
    all_instrs = []
    path_off = []
    path_len = []
    for p in paths:
        ins = compile_path(p)
        path_off.append(len(all_instrs))
        path_len.append(len(ins))
        all_instrs.extend(ins)
    it, kt = _pack_instrs(all_instrs, dev)
    poff = torch.tensor(path_off, dtype=torch.int32).to(dev)
    plen = torch.tensor(path_len, dtype=torch.int32).to(dev)
    desc, top, keep = pack_descriptors([col])

    lens = [torch.empty(n, dtype=torch.int32, device=dev) for _ in paths]
    overflow = torch.zeros(npaths, dtype=torch.int32, device=dev)

    def pack_outs(offsets=None, chars=None, valid=None):
        raw = bytearray(npaths * struct.calcsize(_MULTIOUT_FMT))
        for i in range(npaths):
            struct.pack_into(
                _MULTIOUT_FMT, raw, i * struct.calcsize(_MULTIOUT_FMT),
                lens[i].data_ptr(),
                offsets[i].data_ptr() if offsets else 0,
                chars[i].data_ptr() if chars else 0,
                valid[i].data_ptr() if valid else 0,
                overflow.data_ptr() + i * 4)
        return torch.frombuffer(raw, dtype=torch.uint8).to(dev)

    outs0 = pack_outs()
    g.get_json_multi(desc.data_ptr(), n, it.data_ptr(), kt.data_ptr(),
                     poff.data_ptr(), plen.data_ptr(), npaths, 0,
                     outs0.data_ptr(), stream)
    offsets = []
    for i in range(npaths):
        o = torch.zeros(n + 1, dtype=torch.int32, device=dev)
        torch.cumsum(lens[i], 0, out=o[1:].view(n))
        offsets.append(o)
    totals = torch.stack([o[-1] for o in offsets]).cpu().tolist()  # one D2H
    chars = [torch.empty(max(t, 1), dtype=torch.uint8, device=dev)
             for t in totals]
    valid = [make_validity(n, dev) for _ in range(npaths)]
    outs1 = pack_outs(offsets, chars, valid)
    g.get_json_multi(desc.data_ptr(), n, it.data_ptr(), kt.data_ptr(),
                     poff.data_ptr(), plen.data_ptr(), npaths, 1,
                     outs1.data_ptr(), stream)
    ovf = overflow.cpu().tolist()
    res: List[Optional[Column]] = []
    for i in range(npaths):
        if ovf[i]:
            # some row had >4 matches: redo this path on the exact kernel
            res.append(get_json_object(col, paths[i]))
        else:
            res.append(Column(DType.STRING, n, chars[i][:totals[i]], valid[i],
                              offsets[i], null_count=None))
    return res


def from_json_to_raw_map(col: Column) -> Column:
    """JSON object per row -> LIST<STRUCT<key STRING, value STRING>>
    (reference from_json_to_raw_map.cu)."""
    g = _native.gpu()
    stream = _native.current_stream()
    n = col.size
    dev = col.device
    desc, top, keep = pack_descriptors([col])
    counts = torch.zeros(n, dtype=torch.int32, device=dev)
    g.json_map_count(desc.data_ptr(), n, counts.data_ptr(), stream)
    entry_offsets = torch.zeros(n + 1, dtype=torch.int32, device=dev)
    torch.cumsum(counts, 0, out=entry_offsets[1:].view(n))
    nent = int(entry_offsets[-1].item())
    key_lens = torch.zeros(max(nent, 1), dtype=torch.int32, device=dev)
    val_lens = torch.zeros(max(nent, 1), dtype=torch.int32, device=dev)
    g.json_map_entry_lens(desc.data_ptr(), n, entry_offsets.data_ptr(),
                          key_lens.data_ptr(), val_lens.data_ptr(), stream)
    key_offsets = torch.zeros(nent + 1, dtype=torch.int32, device=dev)
    val_offsets = torch.zeros(nent + 1, dtype=torch.int32, device=dev)
    if nent:
        torch.cumsum(key_lens[:nent], 0, out=key_offsets[1:].view(nent))
        torch.cumsum(val_lens[:nent], 0, out=val_offsets[1:].view(nent))
    nk = int(key_offsets[-1].item())
    nv = int(val_offsets[-1].item())
    key_chars = torch.empty(max(nk, 1), dtype=torch.uint8, device=dev)
    val_chars = torch.empty(max(nv, 1), dtype=torch.uint8, device=dev)
    validity = make_validity(n, dev)
    g.json_map_write(desc.data_ptr(), n, entry_offsets.data_ptr(),
                     key_offsets.data_ptr(), val_offsets.data_ptr(),
                     key_chars.data_ptr(), val_chars.data_ptr(),
                     validity.data_ptr(), stream)
    keys = Column(DType.STRING, nent, key_chars[:nk], None, key_offsets)
    vals = Column(DType.STRING, nent, val_chars[:nv], None, val_offsets)
    entry = Column(DType.STRUCT, nent, None, None, None, [keys, vals])
    return Column(DType.LIST, n, None, validity, entry_offsets, [entry],
                  null_count=None)


def from_json_to_structs(col: Column, field_names: List[str],
                         field_types: Optional[List] = None) -> Table:
    """Spark from_json to STRUCT (reference from_json_to_structs.cu).
    All fields are extracted in ONE shared-scan pass (multi-path kernel),
    then coerced to the requested types with the Spark-exact cast kernels.
    A field type may be ("struct", child_names, child_types) — nested
    struct schemas recurse on the extracted sub-object text."""
    from . import cast as cast_ops
    extracted = get_json_object_multiple_paths(
        col, [f"$.{name}" for name in field_names])
    cols = []
    for i, name in enumerate(field_names):
        s = extracted[i]
        dt = field_types[i] if field_types else DType.STRING
        if isinstance(dt, tuple) and dt[0] == "struct":
            sub = from_json_to_structs(s, dt[1], dt[2])
            cols.append(Column(DType.STRUCT, s.size, None, s.validity,
                               children=list(sub.columns), null_count=None))
            continue
        if dt == DType.STRING:
            cols.append(s)
        elif dt in (DType.INT8, DType.INT16, DType.INT32, DType.INT64):
            cols.append(cast_ops.to_integer(s, dtype=dt))
        elif dt in (DType.FLOAT32, DType.FLOAT64):
            cols.append(cast_ops.to_float(s, dtype=dt))
        elif dt == DType.BOOL8:
            cols.append(cast_ops.to_bool(s))
        elif dt == DType.DATE32:
            cols.append(cast_ops.to_date(s))
        elif dt == DType.TIMESTAMP_US:
            cols.append(cast_ops.to_timestamp(s))
        else:
            raise NotImplementedError(dt)
    return Table(cols)
