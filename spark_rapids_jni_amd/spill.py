"""Spill manager: LRU device->host spilling over HostTable on a side stream.

Reference analog: the plugin's spillable-batch store driving
RmmSpark.spillRangeStart/Done (SURVEY.md §2.1 spill-range bookkeeping; the
MI355X design note: pinned hipMemcpyAsync spill on a side HIP stream so
spills overlap compute on 288 GB HBM3E).
"""
import threading
from collections import OrderedDict
from typing import Optional

import torch

from .columnar import Table
from .hosttable import HostTable
from .memory import RmmSpark
from .schema import flatten_columns


def _table_bytes(table: Table) -> int:
    total = 0
    for c in flatten_columns(table.columns):
        if c.data is not None:
            total += c.data.numel() * c.data.element_size()
        if c.validity is not None:
            total += c.validity.numel()
        if c.offsets is not None:
            total += c.offsets.numel() * 4
    return total


class SpillableTable:
    """A device table that can round-trip to pinned host memory."""

    def __init__(self, table: Table, manager: "SpillManager" = None):
        self._device: Optional[Table] = table
        self._host: Optional[HostTable] = None
        self.device_bytes = _table_bytes(table)
        self._mgr = manager

    @property
    def spilled(self) -> bool:
        return self._device is None

    def spill(self, stream: Optional[torch.cuda.Stream] = None) -> int:
        """Move to host; returns device bytes released."""
        if self._device is None:
            return 0
        try:
            RmmSpark.spill_range_start()
        except Exception:
            pass
        try:
            self._host = HostTable.from_table_async(self._device, stream)
            if stream is not None:
                stream.synchronize()
            elif torch.cuda.is_available():
                torch.cuda.synchronize()
            self._device = None
        finally:
            try:
                RmmSpark.spill_range_done()
            except Exception:
                pass
        return self.device_bytes

    def get(self, device="cuda") -> Table:
        """Device table, unspilling (and re-registering as MRU) if needed."""
        if self._device is None:
            self._device = self._host.to_table_async(device)
            self._host = None
        if self._mgr is not None:
            self._mgr._touch(self)
        return self._device


class SpillManager:
    """LRU registry of spillable tables; `spill_until(bytes)` frees device
    memory oldest-first (called from the OOM-retry path or proactively)."""

    def __init__(self):
        self._lock = threading.Lock()
        self._lru: "OrderedDict[int, SpillableTable]" = OrderedDict()
        self._stream: Optional[torch.cuda.Stream] = None

    def _side_stream(self):
        if self._stream is None and torch.cuda.is_available():
            self._stream = torch.cuda.Stream()
        return self._stream

    def register(self, table: Table) -> SpillableTable:
        st = SpillableTable(table, self)
        with self._lock:
            self._lru[id(st)] = st
        return st

    def _touch(self, st: SpillableTable):
        with self._lock:
            if id(st) in self._lru:
                self._lru.move_to_end(id(st))

    def unregister(self, st: SpillableTable):
        with self._lock:
            self._lru.pop(id(st), None)

    @property
    def spillable_bytes(self) -> int:
        with self._lock:
            return sum(s.device_bytes for s in self._lru.values()
                       if not s.spilled)

    def spill_until(self, nbytes: int) -> int:
        """Spill least-recently-used tables until `nbytes` freed (or out of
        candidates). Returns bytes actually released."""
        freed = 0
        with self._lock:
            victims = [s for s in self._lru.values() if not s.spilled]
        for v in victims:
            if freed >= nbytes:
                break
            freed += v.spill(self._side_stream())
        return freed
