"""GPU protobuf decode (Java API parity: Protobuf.java 4-pass design +
ProtobufSchemaDescriptor.java flattened field tables).

Schema: list of (field_number, kind) where kind in {"int64","int32","bool",
"sint64","double","float","string","bytes"} or "repeated_<k>" for k in
{int64,int32,bool,sint64,double,float,string,bytes} (packed or unpacked
encodings -> LIST columns; reference protobuf_kernels.cuh:150-361 batched
variants) or ("message", child_schema) — nested messages decode
recursively to STRUCT columns.
"""
import struct
from typing import List, Tuple

import torch

from .. import _native
from ..columnar import Column, DType, Table, make_validity, pack_descriptors

_KINDS = {"int64": (0, DType.INT64, torch.int64),
          "int32": (1, DType.INT32, torch.int32),
          "bool": (2, DType.BOOL8, torch.int8),
          "sint64": (3, DType.INT64, torch.int64),
          "double": (4, DType.FLOAT64, torch.float64),
          "float": (5, DType.FLOAT32, torch.float32),
          "string": (6, DType.STRING, None),
          "bytes": (6, DType.STRING, None)}

# fnum, kind, rep_slot, pad, data, valid, lens, offsets, chars,
# lens2, elem_offsets, char_base
_FIELD_FMT = "<iiiiQQQQQQQQ"

_REP_KINDS = {"repeated_int64": (7, DType.INT64, torch.int64),
              "repeated_int32": (8, DType.INT32, torch.int32),
              "repeated_bool": (9, DType.BOOL8, torch.int8),
              "repeated_sint64": (10, DType.INT64, torch.int64),
              "repeated_double": (11, DType.FLOAT64, torch.float64),
              "repeated_float": (12, DType.FLOAT32, torch.float32),
              "repeated_string": (13, DType.STRING, None),
              "repeated_bytes": (13, DType.STRING, None)}


def decode(col: Column, schema) -> Table:
    """Decode one serialized protobuf message per row into columns.

    schema: list of (field_number, kind); kind is a scalar kind name from
    _KINDS or ("message", child_schema) — nested messages decode
    recursively into STRUCT columns (reference Protobuf.java nested-message
    passes; here the length-delimited blob pass feeds a recursive decode).
    """
    def _is_msg(k):
        return isinstance(k, tuple) and k[0] == "message"

    # reference ProtobufSchemaDescriptor rejects duplicate field numbers
    # under the same parent message (they would double-decode); different
    # parents may reuse numbers — validated recursively BEFORE any kernel
    # work so invalid schemas fail fast on any host
    def _check(sch):
        fnums = [f for f, _k in sch]
        if len(set(fnums)) != len(fnums):
            raise ValueError(
                "duplicate protobuf field numbers in one message")
        for _f, k in sch:
            if _is_msg(k):
                _check(k[1])
    _check(schema)

    if any(_is_msg(k) for _, k in schema):
        flat = [(f, "bytes" if _is_msg(k) else k) for f, k in schema]
        tbl = decode(col, flat)
        out = []
        for (f, k), c in zip(schema, tbl.columns):
            if _is_msg(k):
                sub = decode(c, k[1])
                out.append(Column(DType.STRUCT, c.size, None, c.validity,
                                  children=list(sub.columns),
                                  null_count=None))
            else:
                out.append(c)
        return Table(out)
    assert len(schema) <= 64
    g = _native.gpu()
    stream = _native.current_stream()
    n = col.size
    dev = col.device
    desc, top, keep = pack_descriptors([col])

    outs = []
    nrep = 0
    for fnum, kindname in schema:
        if kindname in _REP_KINDS:
            assert nrep < 8, "at most 8 repeated fields per message"
            kind, edt, etdt = _REP_KINDS[kindname]
            o = {"kind": kind, "dtype": DType.LIST, "fnum": fnum,
                 "rep_slot": nrep, "elem_dtype": edt, "elem_tdt": etdt,
                 "lens": torch.zeros(n, dtype=torch.int32, device=dev),
                 "valid": make_validity(n, dev, fill_valid=False)}
            if kind == 13:
                o["lens2"] = torch.zeros(n, dtype=torch.int32, device=dev)
            outs.append(o)
            nrep += 1
            continue
        kind, dt, tdt = _KINDS[kindname]
        if dt == DType.STRING:
            outs.append({"kind": kind, "dtype": dt, "fnum": fnum,
                         "lens": torch.zeros(n, dtype=torch.int32, device=dev),
                         "valid": make_validity(n, dev, fill_valid=False)})
        else:
            outs.append({"kind": kind, "dtype": dt, "fnum": fnum,
                         "data": torch.zeros(n, dtype=tdt, device=dev),
                         "valid": make_validity(n, dev, fill_valid=False)})

    def pack(phase):
        raw = bytearray()
        for o in outs:
            raw += struct.pack(
                _FIELD_FMT, o["fnum"], o["kind"], o.get("rep_slot", 0), 0,
                o["data"].data_ptr() if "data" in o else 0,
                o["valid"].data_ptr(),
                o["lens"].data_ptr() if "lens" in o and phase == 0 else 0,
                o["offsets"].data_ptr() if "offsets" in o else 0,
                o["chars"].data_ptr() if "chars" in o else 0,
                o["lens2"].data_ptr() if "lens2" in o and phase == 0 else 0,
                o["elem_offsets"].data_ptr() if "elem_offsets" in o else 0,
                o["char_base"].data_ptr() if "char_base" in o else 0)
        return torch.frombuffer(raw, dtype=torch.uint8).to(dev)

    row_ok = torch.zeros(n, dtype=torch.uint8, device=dev)
    ft = pack(0)
    g.pb_decode(desc.data_ptr(), n, ft.data_ptr(), len(outs),
                row_ok.data_ptr(), 0, stream)
    # phase 2: allocate string/list outputs, re-run writing bytes/values
    any_bytes = False
    for o in outs:
        if "lens" in o:
            any_bytes = True
            offsets = torch.zeros(n + 1, dtype=torch.int32, device=dev)
            torch.cumsum(o["lens"], 0, out=offsets[1:].view(n))
            o["offsets"] = offsets
            nch = int(offsets[-1].item())
            if o["kind"] == 13:
                # LIST<STRING>: list offsets + per-element char offsets
                char_base = torch.zeros(n + 1, dtype=torch.int32, device=dev)
                torch.cumsum(o["lens2"], 0, out=char_base[1:].view(n))
                o["char_base"] = char_base
                tot_chars = int(char_base[-1].item())
                o["tot_chars"] = tot_chars
                o["elem_offsets"] = torch.zeros(max(nch, 1) + 1,
                                                dtype=torch.int32,
                                                device=dev)
                o["chars"] = torch.empty(max(tot_chars, 1),
                                         dtype=torch.uint8, device=dev)
            elif 7 <= o["kind"] <= 12:
                o["data"] = torch.empty(max(nch, 1), dtype=o["elem_tdt"],
                                        device=dev)
            else:
                o["chars"] = torch.empty(max(nch, 1), dtype=torch.uint8,
                                         device=dev)
    if any_bytes:
        ft2 = pack(1)
        g.pb_decode(desc.data_ptr(), n, ft2.data_ptr(), len(outs),
                    0, 1, stream)
    cols = []
    for o in outs:
        if o["kind"] == 13:
            nel = int(o["offsets"][-1].item())
            o["elem_offsets"][nel] = o["tot_chars"]
            child = Column(DType.STRING, nel, o["chars"][:max(
                o["tot_chars"], 1)], None, o["elem_offsets"][:nel + 1],
                null_count=0)
            cols.append(Column(DType.LIST, n, None, o["valid"],
                               o["offsets"], [child], null_count=None))
        elif 7 <= o["kind"] <= 12:
            nch = int(o["offsets"][-1].item())
            child = Column(o["elem_dtype"], nch, o["data"][:max(nch, 1)])
            cols.append(Column(DType.LIST, n, None, o["valid"],
                               o["offsets"], [child], null_count=None))
        elif "offsets" in o:
            nch = int(o["offsets"][-1].item())
            cols.append(Column(DType.STRING, n, o["chars"][:nch], o["valid"],
                               o["offsets"], null_count=None))
        else:
            cols.append(Column(o["dtype"], n, o["data"], o["valid"],
                               null_count=None))
    return Table(cols)
