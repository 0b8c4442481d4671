"""HostTable: whole-table host copy in one pinned buffer, async both ways.

Java API parity: HostTable.java:30-96 (fromTableAsync / toTableAsync) +
host_table_view.hpp. Used for spill-to-host and host-shuffle staging; copies
ride a side stream so spill overlaps compute.
"""
from typing import List, Optional

import torch

from .columnar import Column, DType, FIXED_WIDTH, Table, validity_nbytes
from .schema import flatten_columns, has_data, has_offsets


def _align8(x):
    return (x + 7) & ~7


class HostTable:
    """One pinned host buffer holding every buffer of a table, plus layout
    metadata to reconstruct the device table."""

    def __init__(self, buf: torch.Tensor, layout, num_rows: int):
        self.buf = buf
        self.layout = layout  # list of per-flat-column dicts
        self.num_rows = num_rows

    @staticmethod
    def from_table_async(table: Table,
                         stream: Optional[torch.cuda.Stream] = None
                         ) -> "HostTable":
        flat = flatten_columns(table.columns)
        layout = []
        pos = 0
        for c in flat:
            ent = {"dtype": int(c.dtype), "scale": c.scale, "size": c.size,
                   "nchildren": len(c.children)}
            if c.validity is not None:
                ent["valid"] = (pos, c.validity.numel())
                pos = _align8(pos + c.validity.numel())
            if has_offsets(c):
                nb = c.offsets.numel() * 4
                ent["offsets"] = (pos, nb)
                pos = _align8(pos + nb)
            if has_data(c) and c.data is not None:
                nb = c.data.numel() * c.data.element_size()
                ent["data"] = (pos, nb, str(c.data.dtype))
                pos = _align8(pos + nb)
            layout.append(ent)
        buf = torch.empty(max(pos, 1), dtype=torch.uint8,
                          pin_memory=torch.cuda.is_available())
        ctx = torch.cuda.stream(stream) if stream else _null_ctx()
        with ctx:
            for c, ent in zip(flat, layout):
                if "valid" in ent:
                    o, nb = ent["valid"]
                    buf[o:o + nb].copy_(c.validity, non_blocking=True)
                if "offsets" in ent:
                    o, nb = ent["offsets"]
                    buf[o:o + nb].view(torch.int32).copy_(c.offsets,
                                                          non_blocking=True)
                if "data" in ent:
                    o, nb, _ = ent["data"]
                    buf[o:o + nb].copy_(
                        c.data.contiguous().view(torch.uint8), non_blocking=True)
        return HostTable(buf, layout, table.num_rows)

    def to_table_async(self, device="cuda",
                       stream: Optional[torch.cuda.Stream] = None) -> Table:
        ctx = torch.cuda.stream(stream) if stream else _null_ctx()
        cols: List[Column] = []
        with ctx:
            it = iter(self.layout)
            cols = [self._rebuild(next(it), it, device) for _ in
                    range(self._num_top())]
        return Table(cols)

    def _num_top(self):
        # count top-level columns by walking children counts
        i = 0
        tops = 0
        n = len(self.layout)

        def skip(idx):
            nchildren = self.layout[idx]["nchildren"]
            idx += 1
            for _ in range(nchildren):
                idx = skip(idx)
            return idx

        while i < n:
            i = skip(i)
            tops += 1
        return tops

    def _rebuild(self, ent, it, device):
        validity = offsets = data = None
        if "valid" in ent:
            o, nb = ent["valid"]
            validity = self.buf[o:o + nb].to(device, non_blocking=True)
        if "offsets" in ent:
            o, nb = ent["offsets"]
            offsets = self.buf[o:o + nb].view(torch.int32).to(device,
                                                              non_blocking=True)
        if "data" in ent:
            o, nb, dt = ent["data"]
            tdt = getattr(torch, dt.split(".")[-1])
            data = self.buf[o:o + nb].view(tdt).to(device, non_blocking=True)
        children = [self._rebuild(next(it), it, device)
                    for _ in range(ent["nchildren"])]
        return Column(DType(ent["dtype"]), ent["size"], data, validity, offsets,
                      children, ent["scale"], null_count=None)


class _null_ctx:
    def __enter__(self):
        return self

    def __exit__(self, *a):
        return False
