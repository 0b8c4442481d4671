"""Hash group-by aggregation (COUNT/SUM/MIN/MAX) on GPU.

Spark semantics: nulls form their own group (null-safe key equality);
SUM/MIN/MAX skip nulls (all-null group -> null result); COUNT(col) counts
non-null, COUNT(*) counts rows. Integer SUM accumulates in int64, float in
float64 (Spark's widened accumulators). The reference delegates this to cudf
groupby + Aggregation64Utils overflow helpers; here it
is a first-class MI355X kernel (src/gpu/hashtable.hip groupby_kernel).
"""
import struct
from enum import IntEnum
from typing import List, Optional, Sequence, Tuple

import torch

from .. import _native
from ..columnar import Column, DType, Table, pack_descriptors
from .copying import gather_column


class Agg(IntEnum):
    COUNT_ALL = 0
    COUNT_VALID = 1
    SUM = 2
    MIN = 4
    MAX = 5


_AGGDESC_FMT = "<iiQQQ"  # op, in_dtype, data, valid, state
_AGGDESC_SZ = struct.calcsize(_AGGDESC_FMT)


def _next_pow2(n):
    p = 1
    while p < n:
        p <<= 1
    return p


def groupby(keys, aggs: Sequence[Tuple[Agg, Optional[Column]]],
            num_groups_hint: Optional[int] = None):
    """Group rows of `keys` and aggregate.

    aggs: list of (Agg, value_column_or_None). Returns
    (key_table, [result Column, ...]) with one row per group.
    """
    kcols = keys.columns if isinstance(keys, Table) else (
        [keys] if isinstance(keys, Column) else list(keys))
    n = kcols[0].size
    assert n < 2**31, "group-by input capped at 2^31-1 rows per batch"
    g = _native.gpu()
    stream = _native.current_stream()
    dev = kcols[0].device

    # specialized path: single non-null int64-valued key — 16B {key,row}
    # slots claimed by CAS-on-key, software-pipelined probing
    # (src/gpu/hashtable_i64.hip groupby_i64_kernel)
    def _int_keyable(c):
        return (c.data is not None
                and c.data.dtype in (torch.int64, torch.int32, torch.int16,
                                     torch.int8)
                and c.dtype not in (DType.FLOAT64, DType.FLOAT32))

    i64_fast = (len(kcols) == 1 and kcols[0].validity is None
                and _int_keyable(kcols[0]))
    i64_keys = None
    if i64_fast:
        i64_keys = (kcols[0].data if kcols[0].data.dtype == torch.int64
                    else kcols[0].data.to(torch.int64))
    elif n > 0 and len(kcols) > 1 and all(_int_keyable(c) for c in kcols):
        # multi-key packing (spark-rapids-style): if the per-column value
        # ranges multiply into < 2^63, fold the key tuple into ONE int64 and
        # take the specialized path. Nulls are null-safe-equal group keys:
        # each column packs as 0 = null, value - min + 1 otherwise. The
        # output keys are gathered from the ORIGINAL columns by the
        # representative rows, so packing never leaks into results.
        widths = []
        mins = []
        ok = True
        pairs = [torch.aminmax(c.data) for c in kcols]
        flat = torch.stack(
            [t for lo_hi in pairs for t in lo_hi]).cpu().tolist()
        for k in range(len(kcols)):  # ONE device sync for all columns
            lo, hi = int(flat[2 * k]), int(flat[2 * k + 1])
            span = hi - lo + 2  # +1 for the null code
            if span <= 0:
                ok = False
                break
            widths.append(span)
            mins.append(lo)
        if ok:
            total = 1
            for w in widths:
                total *= w
                if total >= 2**62:
                    ok = False
                    break
        if ok:
            packed = torch.zeros(n, dtype=torch.int64, device=dev)
            for c, lo, w in zip(kcols, mins, widths):
                code = (c.data.to(torch.int64) - lo) + 1
                if c.validity is not None:
                    from ..columnar import validity_to_bool
                    code = torch.where(validity_to_bool(c.validity, n), code,
                                       torch.zeros_like(code))
                packed = packed * w + code
            i64_fast = True
            i64_keys = packed
    # native agg list = user aggs + a hidden COUNT_VALID per nullable SUM col
    # (to derive the all-null-group -> null result in the SAME compaction pass,
    # since compaction order is nondeterministic). States are (re)allocated
    # inside the capacity loop below.
    native: List[Tuple[int, Optional[Column]]] = []
    metas = []  # (user op, is_float, native_index, hidden_count_index|None)

    def add_native(native_op, col):
        native.append((native_op, col))
        return len(native) - 1

    def _hidden_count(col):
        # all-null-group detection needs a per-group valid count — but a
        # column with no validity buffer can never produce an all-null
        # group, so skip the extra atomic stream entirely (the bench shape:
        # one atomicAdd per row saved)
        if col.validity is None:
            return None
        return add_native(1, col)

    for op, col in aggs:
        if op in (Agg.COUNT_ALL, Agg.COUNT_VALID):
            idx = add_native(int(op), col)
            metas.append((op, False, idx, None))
            continue
        assert col is not None
        is_float = col.dtype in (DType.FLOAT32, DType.FLOAT64)
        if op == Agg.SUM:
            idx = add_native(3 if is_float else 2, col)
        elif op == Agg.MIN:
            idx = add_native(6 if is_float else 4, col)
        else:  # MAX
            idx = add_native(7 if is_float else 5, col)
        metas.append((op, is_float, idx, _hidden_count(col)))

    naggs = len(native)

    def _state_for(native_op, is_float_dtype, nstates):
        if native_op in (0, 1):
            return torch.zeros(nstates, dtype=torch.int64, device=dev)
        if native_op in (2, 3):
            return torch.zeros(nstates, dtype=torch.float64 if native_op == 3
                               else torch.int64, device=dev)
        if native_op == 4:
            return torch.full((nstates,), 2**63 - 1, dtype=torch.int64,
                              device=dev)
        if native_op == 5:
            return torch.full((nstates,), -2**63, dtype=torch.int64,
                              device=dev)
        if native_op == 6:
            return torch.full((nstates,), float("inf"), dtype=torch.float64,
                              device=dev)
        return torch.full((nstates,), float("-inf"), dtype=torch.float64,
                          device=dev)

    # capacity schedule: start from the hint (or a 4M-slot guess) and grow
    # 16x on overflow up to the exact bound. A 288M-row aggregation with
    # thousands of groups must not allocate 2n-slot tables (that is ~17 GB
    # of slots plus an 8.6 GB state per aggregate).
    full_cap = max(_next_pow2(n * 2), 64)
    if num_groups_hint:
        capacity = max(_next_pow2(min(num_groups_hint, n) * 2), 64)
    else:
        capacity = min(full_cap, 1 << 22)
    lds_ok = (num_groups_hint is not None and num_groups_hint <= 1024
              and naggs <= 3)
    if not i64_fast:
        kdesc, ktop, keep = pack_descriptors(kcols)
    while True:
        nstates = capacity + 1 if i64_fast else capacity
        if i64_fast:
            # interleaved {key, row1} pairs; keys init to the EMPTY sentinel
            # (INT64_MIN), +1 reserved slot for rows whose key equals it
            slots = torch.zeros(2 * (capacity + 1), dtype=torch.int64,
                                device=dev)
            slots.view(-1, 2)[:, 0] = -2**63
        else:
            slots = torch.zeros(capacity, dtype=torch.int64, device=dev)
        states = [_state_for(op, None, nstates) for op, _c in native]
        raw = bytearray(max(naggs, 1) * _AGGDESC_SZ)
        for i, ((native_op, col), st) in enumerate(zip(native, states)):
            struct.pack_into(
                _AGGDESC_FMT, raw, i * _AGGDESC_SZ, native_op,
                int(col.dtype) if col is not None else 0,
                col.data.data_ptr() if col is not None else 0,
                (col.validity.data_ptr()
                 if col is not None and col.validity is not None else 0),
                st.data_ptr())
        agg_desc = torch.frombuffer(raw, dtype=torch.uint8).to(dev)
        overflow = torch.zeros(1, dtype=torch.int32, device=dev)
        if i64_fast:
            if lds_ok:
                # low-cardinality: per-workgroup LDS pre-aggregation
                # collapses per-row global atomics into per-(block x
                # group) merges
                _IDENT = {0: 0, 1: 0, 2: 0, 3: 0,
                          4: 2**63 - 1, 5: -2**63,
                          6: 0x7FF0000000000000,           # +inf bits
                          7: 0xFFF0000000000000 - 2**64}   # -inf bits
                idents = torch.tensor([_IDENT[op] for op, _c in native],
                                      dtype=torch.int64).to(dev)
                g.groupby_i64_lds(i64_keys.data_ptr(), n, slots.data_ptr(),
                                  capacity, agg_desc.data_ptr(), naggs,
                                  idents.data_ptr(), overflow.data_ptr(),
                                  stream)
            else:
                g.groupby_i64(i64_keys.data_ptr(), n, slots.data_ptr(),
                              capacity, agg_desc.data_ptr(), naggs,
                              overflow.data_ptr(), stream)
        else:
            g.groupby(kdesc.data_ptr(), ktop.data_ptr(), len(kcols), n,
                      slots.data_ptr(), capacity, agg_desc.data_ptr(),
                      naggs, overflow.data_ptr(), stream)
        if i64_fast:
            # speculative compact-count: enqueue the two-pass compaction's
            # counting half WITH the groupby so the overflow flag and the
            # group count come back in ONE device sync (host gaps, not
            # kernel time, dominate the NDS power run at this point)
            cap1 = capacity + 1
            nblk = max(1, min((cap1 + 255) // 256, 2048))  # mirrors grid_1d
            blk = torch.empty(nblk, dtype=torch.int64, device=dev)
            g.groupby_compact_i64_count(slots.data_ptr(), cap1,
                                        blk.data_ptr(), stream)
            csum = torch.cumsum(blk, 0)
            probe = torch.cat([overflow.to(torch.int64),
                               csum[-1:]]).cpu().tolist()
            ovf, ngroups = int(probe[0]), int(probe[1])
        else:
            ovf = int(overflow.item())
        if not ovf:
            break
        assert capacity < full_cap, "groupby overflow at full capacity"
        # a second guess would fill to ~100% load before overflowing again
        # (long probe chains make that pass expensive) — go straight to
        # the exact bound (NDS SF100: groupby share 45.8% -> see profiles)
        capacity = full_cap
        lds_ok = False
        del slots, states, agg_desc

    out_repr = torch.empty(nstates, dtype=torch.int64, device=dev)
    out_agg = torch.empty(max(naggs, 1) * nstates, dtype=torch.int64,
                          device=dev)
    if i64_fast:
        # two-pass compaction fill (counting half ran inside the loop): the
        # single global counter a wave-leader atomicAdd hammered was the
        # bottleneck on big tables (101 ms on a 512M-slot scan); per-block
        # bases + an LDS cursor run at slot-scan bandwidth, deterministic.
        bases = csum - blk
        g.groupby_compact_i64_fill(slots.data_ptr(), cap1,
                                   agg_desc.data_ptr(), naggs,
                                   bases.data_ptr(), out_repr.data_ptr(),
                                   out_agg.data_ptr(), nstates, stream)
    else:
        counter = torch.zeros(1, dtype=torch.int64, device=dev)
        g.groupby_compact(slots.data_ptr(), capacity, agg_desc.data_ptr(),
                          naggs, counter.data_ptr(), out_repr.data_ptr(),
                          out_agg.data_ptr(), nstates, stream)
        ngroups = int(counter.item())
    out_repr = out_repr[:ngroups]

    def agg_vals(idx):
        return out_agg[idx * nstates:idx * nstates + ngroups]

    key_out = Table([gather_column(c, out_repr) for c in kcols])
    results: List[Column] = []
    for (op, is_float, idx, hidden) in metas:
        vals = agg_vals(idx).clone()
        if op in (Agg.COUNT_ALL, Agg.COUNT_VALID):
            results.append(Column(DType.INT64, ngroups, vals))
        else:  # SUM / MIN / MAX: null result for all-null groups
            if hidden is None:  # value column had no nulls
                if is_float:
                    results.append(Column(DType.FLOAT64, ngroups,
                                          vals.view(torch.float64).clone()))
                else:
                    results.append(Column(DType.INT64, ngroups, vals))
                continue
            cnt = agg_vals(hidden)
            validity = _validity_from_bool(cnt > 0)
            if is_float:
                out_vals = vals.view(torch.float64).clone()
                out_vals[cnt == 0] = 0.0
                results.append(Column(DType.FLOAT64, ngroups, out_vals, validity,
                                      null_count=None))
            else:
                out_vals = vals.clone()
                out_vals[cnt == 0] = 0
                results.append(Column(DType.INT64, ngroups, out_vals, validity,
                                      null_count=None))
    return key_out, results


def _validity_from_bool(flags: torch.Tensor):
    n = flags.numel()
    dev = flags.device
    nbytes = ((n + 63) // 64) * 8
    bits = torch.zeros(nbytes * 8, dtype=torch.bool, device=dev)
    bits[:n] = flags
    weights = torch.tensor([1, 2, 4, 8, 16, 32, 64, 128], dtype=torch.uint8,
                           device=dev)
    packed = (bits.view(-1, 8).to(torch.uint8) * weights).sum(1, dtype=torch.uint8)
    return packed
