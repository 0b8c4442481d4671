"""Composable join building blocks.

Java API parity: com.nvidia.spark.rapids.jni.JoinPrimitives
(reference join_primitives.hpp:64-237 / JoinPrimitives.java:80-302):
hash_inner_join producing gather maps, plus gather-map algebra
(make_left_outer / make_full_outer / semi / anti).

MI355X design: a multimap of packed (fingerprint32|row+1) u64 slots at 50%
load, linear probing (src/gpu/hashtable.hip). Build side is capped at 2^31-1
rows (the reference has the same per-column row cap); probe side is int64.
"""
from typing import List, Optional, Sequence, Tuple, Union

import torch

from .. import _native
from ..columnar import Column, Table, pack_descriptors


def _keys(x) -> List[Column]:
    if isinstance(x, Table):
        return x.columns
    if isinstance(x, Column):
        return [x]
    return list(x)


def _next_pow2(n: int) -> int:
    p = 1
    while p < n:
        p <<= 1
    return p


_FAST_KEY_DTYPES = None


def _is_i64_fast(cols: List[Column]) -> bool:
    """Single integer-typed key -> the specialized 16B-slot table (narrower
    ints are upcast to int64 once; sign extension preserves equality, and
    real NDS surrogate keys are INT32)."""
    global _FAST_KEY_DTYPES
    from ..columnar import DType
    if _FAST_KEY_DTYPES is None:
        _FAST_KEY_DTYPES = {DType.INT64, DType.INT32, DType.INT16, DType.INT8,
                            DType.DATE32, DType.TIMESTAMP_US}
    return len(cols) == 1 and cols[0].dtype in _FAST_KEY_DTYPES


def _as_i64_keys(c: Column) -> Column:
    """int64 view/upcast of an integer key column (validity shared)."""
    if c.data.dtype == torch.int64:
        return c
    return Column(c.dtype, c.size, c.data.to(torch.int64), c.validity,
                  null_count=None)


def _int_packable(cols: List[Column]) -> bool:
    from ..columnar import DType
    return (len(cols) > 1 and all(
        c.data is not None
        and c.data.dtype in (torch.int64, torch.int32, torch.int16,
                             torch.int8)
        and c.dtype not in (DType.FLOAT64, DType.FLOAT32) for c in cols))


def _pack_ranges(cols: List[Column]):
    """(mins, widths) for folding an int-key tuple into one int64, or None
    if the range product overflows."""
    if cols[0].size == 0:
        return None  # nothing to scan; generic path handles empties
    pairs = [torch.aminmax(c.data) for c in cols]
    flat = torch.stack([t for lo_hi in pairs for t in lo_hi]).cpu().tolist()
    mins, widths = [], []
    total = 1
    for k in range(len(cols)):  # ONE device sync for all columns
        lo, hi = int(flat[2 * k]), int(flat[2 * k + 1])
        span = hi - lo + 1
        if span <= 0:
            return None
        mins.append(lo)
        widths.append(span)
        total *= span
        if total >= 2**62:
            return None
    return mins, widths


def _pack_keys(cols: List[Column], mins, widths):
    """Fold key columns into (packed int64, validity bitmask|None) using the
    given ranges. Rows with a null key or an out-of-range value (probe side
    of a join: can never match) get their validity bit cleared — join
    semantics, where nulls never match."""
    from ..columnar import validity_to_bool
    n = cols[0].size
    dev = cols[0].device
    packed = torch.zeros(n, dtype=torch.int64, device=dev)
    ok = torch.ones(n, dtype=torch.bool, device=dev)
    for c, lo, w in zip(cols, mins, widths):
        v = c.data.to(torch.int64) - lo
        in_range = (v >= 0) & (v < w)
        if c.validity is not None:
            in_range &= validity_to_bool(c.validity, n)
        ok &= in_range
        packed = packed * w + torch.where(in_range, v, torch.zeros_like(v))
    if bool(ok.all().item()):
        return packed, None
    from ..ops.aggregate import _validity_from_bool
    return packed, _validity_from_bool(ok)


class HashJoinTable:
    """Build-side hash table reusable across probes (reference: the build/probe
    split of hash_inner_join).

    Two layouts: a generic (fingerprint32|row+1) single-word table for any key
    schema, and a specialized 16-byte {key,row} table for single-int64 keys
    (the NDS surrogate-key hot path) with software-pipelined probing — see
    src/gpu/hashtable_i64.hip.
    """

    def __init__(self, build_keys: List[Column], slots: torch.Tensor,
                 capacity: int, keep, i64_fast: bool):
        self.build_keys = build_keys
        self.slots = slots
        self.capacity = capacity
        self._keep = keep
        self.i64_fast = i64_fast
        self.num_build_rows = build_keys[0].size

    @staticmethod
    def build(keys: Union[Column, Table, Sequence[Column]],
              force_generic: bool = False) -> "HashJoinTable":
        cols = _keys(keys)
        n = cols[0].size
        assert n < 2**31, "build side capped at 2^31-1 rows (chunk the build)"
        g = _native.gpu()
        dev = cols[0].device
        # int64 fast path runs at 25% max load: random-probe scans touch
        # ~1.3 slots instead of ~2.5 at 50% load (measured faster than
        # double-slot prefetching); generic path stays at 50%.
        capacity = max(_next_pow2(n * 2), 64)
        pack = None
        if not force_generic and _int_packable(cols):
            pack = _pack_ranges(cols)
        if pack is not None:
            packed, pvalid = _pack_keys(cols, *pack)
            capacity = max(_next_pow2(n * 4), 64)
            slots = torch.zeros(capacity * 2, dtype=torch.int64, device=dev)
            g.join_build_i64(packed.data_ptr(),
                             pvalid.data_ptr() if pvalid is not None else 0,
                             n, slots.data_ptr(), capacity,
                             _native.current_stream())
            tbl = HashJoinTable(cols, slots, capacity, (packed, pvalid), True)
            tbl._pack = pack
            return tbl
        if _is_i64_fast(cols) and not force_generic:
            capacity = max(_next_pow2(n * 4), 64)
            c = _as_i64_keys(cols[0])
            slots = torch.zeros(capacity * 2, dtype=torch.int64, device=dev)
            g.join_build_i64(c.data.data_ptr(),
                             c.validity.data_ptr() if c.validity is not None else 0,
                             n, slots.data_ptr(), capacity, _native.current_stream())
            return HashJoinTable(cols, slots, capacity, None, True)
        slots = torch.zeros(capacity, dtype=torch.int64, device=dev)
        desc, top, keep = pack_descriptors(cols)
        g.join_build(desc.data_ptr(), top.data_ptr(), len(cols), n,
                     slots.data_ptr(), capacity, _native.current_stream())
        return HashJoinTable(cols, slots, capacity, (desc, top, keep), False)

    def _descs(self, probe_cols):
        bdesc, btop, bkeep = pack_descriptors(self.build_keys)
        pdesc, ptop, pkeep = pack_descriptors(probe_cols)
        return bdesc, btop, pdesc, ptop, (bkeep, pkeep, bdesc, btop, pdesc, ptop)

    def inner_join(self, probe: Union[Column, Table, Sequence[Column]],
                   out_hint: Optional[int] = None,
                   track_build_matches: bool = False):
        """Returns (build_idx int32, probe_idx int64[, build_matched uint8])."""
        pcols = _keys(probe)
        nprobe = pcols[0].size
        g = _native.gpu()
        stream = _native.current_stream()
        dev = pcols[0].device
        pack = getattr(self, "_pack", None)
        fast = self.i64_fast and (
            (pack is not None and len(pcols) == len(self.build_keys))
            or (pack is None and _is_i64_fast(pcols)))
        if fast and pack is not None:
            packed, pvalid = _pack_keys(pcols, *pack)
            pcols = [Column(pcols[0].dtype, nprobe, packed, pvalid,
                            null_count=None)]
        elif fast:
            pcols = [_as_i64_keys(pcols[0])]
        counter = torch.zeros(1, dtype=torch.int64, device=dev)
        if not fast:
            bdesc, btop, pdesc, ptop, keep = self._descs(pcols)
        # NOTE measured negative: radix-partitioning the probe keys by hash
        # prefix (part_hist/part_scatter + idxmap probe) did NOT speed up the
        # 10B x 1B config — each 64B table line is touched only ~1.4x per 1B
        # chunk, so LLC locality saves almost no DRAM traffic while the
        # scatter adds 29 ms/chunk (profiles/r01_bench_join_22.1B.json).
        # The kernels stay exposed (g.part_hist/part_scatter, idxmap probe)
        # for schema-partitioned shuffle use; the probe here stays direct.
        probe_keys_ptr = pcols[0].data.data_ptr() if fast else 0
        idxmap_ptr = 0
        if out_hint is None:
            if fast:
                p = pcols[0]
                g.join_probe_i64(probe_keys_ptr,
                                 p.validity.data_ptr() if p.validity is not None else 0,
                                 nprobe, self.slots.data_ptr(), self.capacity,
                                 counter.data_ptr(), 0, 0, 0, 0, 0, 0, stream)
            else:
                g.join_probe_count(bdesc.data_ptr(), btop.data_ptr(), pdesc.data_ptr(),
                                   ptop.data_ptr(), len(pcols), nprobe,
                                   self.slots.data_ptr(), self.capacity,
                                   counter.data_ptr(), stream)
            total = int(counter.item())
            counter.zero_()
        else:
            total = out_hint
        out_build = torch.empty(max(total, 1), dtype=torch.int32, device=dev)
        out_probe = torch.empty(max(total, 1), dtype=torch.int64, device=dev)
        matched = (torch.zeros(self.num_build_rows, dtype=torch.uint8, device=dev)
                   if track_build_matches else None)
        if fast:
            p = pcols[0]
            g.join_probe_i64(probe_keys_ptr,
                             p.validity.data_ptr() if p.validity is not None else 0,
                             nprobe, self.slots.data_ptr(), self.capacity,
                             counter.data_ptr(), out_build.data_ptr(),
                             out_probe.data_ptr(), total,
                             matched.data_ptr() if matched is not None else 0,
                             1, idxmap_ptr, stream)
        else:
            g.join_probe_fill(bdesc.data_ptr(), btop.data_ptr(), pdesc.data_ptr(),
                              ptop.data_ptr(), len(pcols), nprobe,
                              self.slots.data_ptr(), self.capacity, counter.data_ptr(),
                              out_build.data_ptr(), out_probe.data_ptr(), total,
                              matched.data_ptr() if matched is not None else 0, stream)
        actual = int(counter.item())
        if actual > total:
            # hint was too small: rerun with the exact size
            return self.inner_join(probe, out_hint=actual,
                                   track_build_matches=track_build_matches)
        out_build = out_build[:actual]
        out_probe = out_probe[:actual]
        if track_build_matches:
            return out_build, out_probe, matched
        return out_build, out_probe

    def mark_matches(self, probe) -> torch.Tensor:
        """Stream the probe rows and return per-BUILD-row matched flags
        (uint8) without materializing join pairs. This is the right-semi
        building block: to semi/anti-filter a small table against a huge
        one, build on the small side and mark while probing the huge side
        (reference get_matched_rows, join_primitives.hpp:237)."""
        pcols = _keys(probe)
        nprobe = pcols[0].size
        g = _native.gpu()
        stream = _native.current_stream()
        dev = pcols[0].device
        matched = torch.zeros(self.num_build_rows, dtype=torch.uint8,
                              device=dev)
        counter = torch.zeros(1, dtype=torch.int64, device=dev)
        pack = getattr(self, "_pack", None)
        fast = self.i64_fast and (
            (pack is not None and len(pcols) == len(self.build_keys))
            or (pack is None and _is_i64_fast(pcols)))
        if fast and pack is not None:
            packed, pvalid = _pack_keys(pcols, *pack)
            pcols = [Column(pcols[0].dtype, nprobe, packed, pvalid,
                            null_count=None)]
        elif fast:
            pcols = [_as_i64_keys(pcols[0])]
        if fast:
            p = pcols[0]
            g.join_probe_i64(p.data.data_ptr(),
                             p.validity.data_ptr() if p.validity is not None
                             else 0,
                             nprobe, self.slots.data_ptr(), self.capacity,
                             counter.data_ptr(), 0, 0, 0,
                             matched.data_ptr(), 1, 0, stream)
        else:
            bdesc, btop, pdesc, ptop, keep = self._descs(pcols)
            g.join_probe_fill(bdesc.data_ptr(), btop.data_ptr(),
                              pdesc.data_ptr(), ptop.data_ptr(),
                              len(pcols), nprobe, self.slots.data_ptr(),
                              self.capacity, counter.data_ptr(), 0, 0, 0,
                              matched.data_ptr(), stream)
        return matched

    def semi_join(self, probe, anti: bool = False) -> torch.Tensor:
        """Left semi/anti join: probe-side row indices with (no) match."""
        if self.i64_fast:
            # semi/anti runs on the generic slot layout
            if not hasattr(self, "_generic"):
                g = _native.gpu()
                n = self.build_keys[0].size
                slots = torch.zeros(self.capacity, dtype=torch.int64,
                                    device=self.build_keys[0].device)
                desc, top, keep = pack_descriptors(self.build_keys)
                # hash over ALL key columns — a single-column hash here
                # silently misses every multi-key match (r2 fix)
                g.join_build(desc.data_ptr(), top.data_ptr(),
                             len(self.build_keys), n,
                             slots.data_ptr(), self.capacity,
                             _native.current_stream())
                self._generic = HashJoinTable(self.build_keys, slots,
                                              self.capacity, (desc, top, keep),
                                              False)
            return self._generic.semi_join(probe, anti)
        pcols = _keys(probe)
        nprobe = pcols[0].size
        g = _native.gpu()
        stream = _native.current_stream()
        dev = pcols[0].device
        bdesc, btop, pdesc, ptop, keep = self._descs(pcols)
        counter = torch.zeros(1, dtype=torch.int64, device=dev)
        out = torch.empty(nprobe, dtype=torch.int64, device=dev)
        g.join_semi(bdesc.data_ptr(), btop.data_ptr(), pdesc.data_ptr(),
                    ptop.data_ptr(), len(pcols), nprobe, self.slots.data_ptr(),
                    self.capacity, counter.data_ptr(), out.data_ptr(), nprobe,
                    1 if anti else 0, stream)
        return out[:int(counter.item())]


def hash_inner_join(build, probe) -> Tuple[torch.Tensor, torch.Tensor]:
    """reference join_primitives.hpp:87 hash_inner_join"""
    return HashJoinTable.build(build).inner_join(probe)


def make_left_outer(probe_size: int, build_idx: torch.Tensor,
                    probe_idx: torch.Tensor):
    """Extend an inner-join gather map pair to LEFT OUTER (probe side = left):
    unmatched probe rows appear with build index -1 (null gather).
    reference join_primitives.hpp:130 make_left_outer."""
    dev = probe_idx.device
    matched = torch.zeros(probe_size, dtype=torch.bool, device=dev)
    matched[probe_idx] = True
    unmatched = torch.nonzero(~matched, as_tuple=False).view(-1)
    lo_build = torch.cat([build_idx.long(),
                          torch.full((unmatched.numel(),), -1, dtype=torch.int64,
                                     device=dev)])
    lo_probe = torch.cat([probe_idx, unmatched])
    return lo_build, lo_probe


def make_full_outer(build_matched: torch.Tensor, build_idx: torch.Tensor,
                    probe_idx: torch.Tensor, probe_size: int):
    """FULL OUTER: left-outer plus unmatched build rows with probe index -1.
    reference join_primitives.hpp:150 make_full_outer."""
    lo_build, lo_probe = make_left_outer(probe_size, build_idx, probe_idx)
    unmatched_b = torch.nonzero(build_matched == 0, as_tuple=False).view(-1)
    fo_build = torch.cat([lo_build, unmatched_b])
    fo_probe = torch.cat([lo_probe,
                          torch.full((unmatched_b.numel(),), -1,
                                     dtype=torch.int64, device=lo_probe.device)])
    return fo_build, fo_probe
