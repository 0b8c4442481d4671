"""Relational plan IR + the two executors (GPU kernels / CPU torch oracle).

Plan nodes: Scan, Filter, Project, Join, Agg, Window, Sort, Limit, Union,
Distinct. The Engine interprets a plan bottom-up over Frames (dict of named
torch tensors + validity masks). The hot relational ops route to the
hand-written HIP kernels on GPU (ops/join.py HashJoinTable,
ops/aggregate.py groupby — src/gpu/hashtable*.hip); the CPU backend is an
intentionally simple torch/Python reference used as the correctness oracle
(SURVEY.md §6 — NDS is the reference's headline metric).

Distribution (one rank per GPU, RCCL over xGMI): fact scans are sharded
per-rank, dimension scans replicated. A join with a sharded input
co-partitions both sides by key hash (all-to-all exchange for sharded
sides, local partition filter for replicated sides); an aggregation over
sharded rows does local partial aggregation, allgathers the (small)
partials, and re-aggregates — the classic map-side combine.
"""
from dataclasses import dataclass, field
from typing import Dict, List, Optional, Sequence, Tuple, Union as PyUnion

import torch

from .expr import Expr, Col, Lit, Val, eval_expr, _wrap


# ---------------------------------------------------------------------------
# plan nodes
# ---------------------------------------------------------------------------

class Plan:
    pass


@dataclass
class Scan(Plan):
    table: str
    columns: Optional[List[str]] = None


@dataclass
class Filter(Plan):
    child: Plan
    cond: Expr


@dataclass
class Project(Plan):
    child: Plan
    outs: List[Tuple[str, Expr]]   # (name, expr); replaces the schema
    extend: bool = False           # True: keep existing columns, add outs


@dataclass
class Join(Plan):
    left: Plan
    right: Plan
    on: List[Tuple[str, str]]      # (left_col, right_col)
    how: str = "inner"             # inner | left | full | semi | anti
    # build side is always `right`; put the smaller input there


@dataclass
class Agg(Plan):
    child: Plan
    keys: List[str]
    aggs: List[Tuple[str, str, Optional[Expr]]]  # (out_name, fn, expr)
    rollup: bool = False
    # fn: sum count min max avg countd stddev_samp


@dataclass
class Window(Plan):
    child: Plan
    partition: List[str]
    funcs: List[tuple]  # (out_name, fn, arg_col|None, order|None)
    # fn: rank dense_rank row_number sum avg min max cumsum
    # order: list of (col, asc) — required for rank-like and cumsum


@dataclass
class Sort(Plan):
    child: Plan
    by: List[Tuple[str, bool]]     # (col, ascending)


@dataclass
class Limit(Plan):
    child: Plan
    n: int


@dataclass
class Union(Plan):
    children: List[Plan]           # union all, by column name


@dataclass
class Distinct(Plan):
    child: Plan


# ---------------------------------------------------------------------------
# runtime frame
# ---------------------------------------------------------------------------

class Frame:
    __slots__ = ("cols", "nrows", "sharded")

    def __init__(self, cols: Dict[str, Val], nrows: int, sharded=False):
        self.cols = cols
        self.nrows = nrows
        self.sharded = sharded

    def names(self):
        return list(self.cols.keys())

    def gather(self, idx: torch.Tensor, null_for_neg=False) -> "Frame":
        out = {}
        n = idx.numel()
        if null_for_neg:
            neg = idx < 0
            safe = idx.clamp(min=0)
        for name, v in self.cols.items():
            if null_for_neg:
                if v.data.numel() == 0:
                    # empty side of an outer join: all rows are null
                    data = torch.zeros(n, dtype=v.data.dtype,
                                       device=idx.device)
                    out[name] = Val(data, torch.zeros(
                        n, dtype=torch.bool, device=idx.device), v.dict)
                    continue
                data = v.data[safe]
                valid = v.valid[safe] if v.valid is not None else \
                    torch.ones(n, dtype=torch.bool, device=data.device)
                valid = valid & ~neg
                out[name] = Val(data, valid, v.dict)
            else:
                data = v.data[idx]
                valid = v.valid[idx] if v.valid is not None else None
                out[name] = Val(data, valid, v.dict)
        return Frame(out, n, self.sharded)

    def mask(self, m: torch.Tensor) -> "Frame":
        idx = torch.nonzero(m, as_tuple=False).view(-1)
        return self.gather(idx)

    def to_rows(self, decode=True):
        """Host list-of-tuples with None for nulls (tests/oracle compare)."""
        cols = []
        for v in self.cols.values():
            data = v.data.cpu()
            valid = v.valid.cpu() if v.valid is not None else None
            vals = data.tolist()
            if decode and v.dict is not None:
                vals = [v.dict[int(x)] if (valid is None or valid[i])
                        else None for i, x in enumerate(vals)]
            else:
                vals = [x if (valid is None or valid[i]) else None
                        for i, x in enumerate(vals)]
            cols.append(vals)
        return list(zip(*cols)) if cols else []


def _bool_valid(v: Val, n, device):
    return v.valid if v.valid is not None else None


# ---------------------------------------------------------------------------
# backends: join + groupby primitives
# ---------------------------------------------------------------------------

def _pack_validity(valid: Optional[torch.Tensor]):
    if valid is None:
        return None
    from ..ops.aggregate import _validity_from_bool
    return _validity_from_bool(valid)


_TORCH2DT = None


def _val_to_column(v: Val):
    from ..columnar import Column, DType
    global _TORCH2DT
    if _TORCH2DT is None:
        _TORCH2DT = {torch.int64: DType.INT64, torch.int32: DType.INT32,
                     torch.int16: DType.INT16, torch.int8: DType.INT8,
                     torch.float64: DType.FLOAT64,
                     torch.float32: DType.FLOAT32}
    data = v.data
    if data.dtype == torch.bool:
        data = data.to(torch.int8)
    dt = _TORCH2DT[data.dtype]
    valid = v.valid
    # expressions like case_when(..., otherwise=lit(0)) produce an all-true
    # mask: drop it, or every SUM downstream pays a hidden per-group
    # count_valid atomic stream AND global aggregates lose the LDS
    # pre-aggregation path (naggs<=3 limit) — q97's final agg was 2x255 ms
    # on a single contended CAS slot because of exactly this
    if valid is not None and data.numel():
        at = getattr(v, "_all_true", None)
        if at is None:
            at = bool(valid.all().item())
            try:
                v._all_true = at
            except AttributeError:
                pass
        if at:
            valid = None
    return Column(dt, data.numel(), data, _pack_validity(valid),
                  null_count=None)


def _precheck_all_true(vals):
    """Resolve the all-true memo for many Vals in ONE device sync."""
    cand = [v for v in vals
            if v is not None and v.valid is not None
            and getattr(v, "_all_true", False) is None and v.data.numel()]
    if not cand:
        return
    flags = torch.stack([v.valid.all() for v in cand]).cpu().tolist()
    for v, fl in zip(cand, flags):
        try:
            v._all_true = bool(fl)
        except AttributeError:
            pass


class GpuBackend:
    """Joins and aggregations through the HIP kernels."""

    def join(self, left_keys: List[Val], right_keys: List[Val], how: str,
             nleft: int):
        from ..ops.join import HashJoinTable
        _precheck_all_true(list(left_keys) + list(right_keys))
        bcols = [_val_to_column(v) for v in right_keys]
        pcols = [_val_to_column(v) for v in left_keys]
        if how in ("semi", "anti"):
            nright = right_keys[0].data.numel() if right_keys else 0
            if nright > 8 * max(nleft, 1):
                # right-semi: build on the (small) left side and mark
                # matches while streaming the huge right side — avoids a
                # hash table over the big input for a handful of probes
                ltbl = HashJoinTable.build(pcols)
                matched = ltbl.mark_matches(bcols).bool()
                sel = torch.nonzero(matched if how == "semi" else ~matched,
                                    as_tuple=False).view(-1)
                return sel, None
            tbl = HashJoinTable.build(bcols)
            sel = tbl.semi_join(pcols, anti=(how == "anti"))
            return sel, None
        tbl = HashJoinTable.build(bcols)
        if how == "full":
            from ..ops.join import make_full_outer, make_left_outer
            bi, pi, matched = tbl.inner_join(pcols, track_build_matches=True)
            fb, fp = make_full_outer(matched, bi, pi, nleft)
            return fp, fb  # -1 entries on both sides
        bi, pi = tbl.inner_join(pcols)
        if how == "left":
            from ..ops.join import make_left_outer
            lb, lp = make_left_outer(nleft, bi, pi)
            return lp, lb  # (left_idx, right_idx with -1 for unmatched)
        return pi, bi.long()

    def groupby(self, keys: List[Val], aggs, hint=None):
        """aggs: list of (fn∈{count,count_valid,sum,min,max}, Val|None).
        Returns (key Vals, result Vals)."""
        from ..columnar import validity_to_bool
        from ..ops.aggregate import Agg as A, groupby as gb
        _precheck_all_true(list(keys) + [v for _fn, v in aggs])
        kcols = [_val_to_column(v) for v in keys]
        fnmap = {"count": A.COUNT_ALL, "count_valid": A.COUNT_VALID,
                 "sum": A.SUM, "min": A.MIN, "max": A.MAX}
        gaggs = []
        for fn, v in aggs:
            gaggs.append((fnmap[fn], _val_to_column(v) if v is not None
                          else None))
        kt, res = gb(kcols, gaggs, num_groups_hint=hint)
        ngroups = kt.num_rows
        out_keys = []
        for c, orig in zip(kt.columns, keys):
            valid = (validity_to_bool(c.validity, ngroups)
                     if c.validity is not None else None)
            out_keys.append(Val(c.data, valid, orig.dict))
        out_res = []
        for c in res:
            valid = (validity_to_bool(c.validity, ngroups)
                     if c.validity is not None else None)
            out_res.append(Val(c.data, valid))
        return out_keys, out_res, ngroups


class CpuBackend:
    """Vectorized numpy sort-merge reference (the oracle) — deliberately a
    different algorithm family than the GPU hash kernels so agreement is
    meaningful."""

    @staticmethod
    def _codes(keys: List[Val], null_is_group: bool):
        """Fold key columns into one int64 code per row.

        null_is_group=True: nulls form their own key value (group-by).
        null_is_group=False: rows with any null key get code -1 (join keys
        never match on null)."""
        import numpy as np
        n = keys[0].data.numel()
        packed = np.zeros(n, dtype=np.int64)
        any_null = np.zeros(n, dtype=bool)
        for v in keys:
            d = v.data.cpu().numpy()
            if d.dtype == np.bool_:
                d = d.astype(np.int64)
            uniq, inv = np.unique(d, return_inverse=True)
            inv = inv.astype(np.int64) + 1  # 0 reserved for null
            if v.valid is not None:
                m = v.valid.cpu().numpy()
                inv = np.where(m, inv, 0)
                any_null |= ~m
            packed = packed * (len(uniq) + 1) + inv
            # re-code to keep the product small
            u2, packed = np.unique(packed, return_inverse=True)
            packed = packed.astype(np.int64)
        if not null_is_group:
            packed = np.where(any_null, -1, packed)
        return packed

    def join(self, left_keys: List[Val], right_keys: List[Val], how: str,
             nleft: int):
        import numpy as np
        dev = left_keys[0].data.device if left_keys else "cpu"
        nl = left_keys[0].data.numel()
        nr = right_keys[0].data.numel()
        # joint coding: codes must agree across sides
        joint = [Val(torch.cat([l.data.cpu(), r.data.cpu()]),
                     (None if l.valid is None and r.valid is None else
                      torch.cat([
                          l.valid.cpu() if l.valid is not None else
                          torch.ones(nl, dtype=torch.bool),
                          r.valid.cpu() if r.valid is not None else
                          torch.ones(nr, dtype=torch.bool)])))
                 for l, r in zip(left_keys, right_keys)]
        codes = self._codes(joint, null_is_group=False)
        lc, rc = codes[:nl], codes[nl:]
        rvalid = rc >= 0
        rs = np.sort(rc[rvalid])
        rs_idx = np.arange(nr)[rvalid][np.argsort(rc[rvalid], kind="stable")]
        lo = np.searchsorted(rs, lc, side="left")
        hi = np.searchsorted(rs, lc, side="right")
        cnt = np.where(lc >= 0, hi - lo, 0)
        if how == "semi":
            sel = np.nonzero(cnt > 0)[0]
            return torch.from_numpy(sel).to(dev), None
        if how == "anti":
            sel = np.nonzero(cnt == 0)[0]
            return torch.from_numpy(sel).to(dev), None
        li = np.repeat(np.arange(nl), cnt)
        # right positions: for each left row, rs_idx[lo[i] : lo[i]+cnt[i]]
        offs = np.repeat(lo, cnt) + (np.arange(cnt.sum()) -
                                     np.repeat(np.cumsum(cnt) - cnt, cnt))
        ri = rs_idx[offs]
        if how in ("left", "full"):
            miss = np.nonzero(cnt == 0)[0]
            li = np.concatenate([li, miss])
            ri = np.concatenate([ri, np.full(len(miss), -1, dtype=ri.dtype)])
        if how == "full":
            hit = np.zeros(nr, dtype=bool)
            hit[ri[ri >= 0]] = True
            miss_r = np.nonzero(~hit)[0]
            li = np.concatenate([li, np.full(len(miss_r), -1,
                                             dtype=li.dtype)])
            ri = np.concatenate([ri, miss_r])
        return (torch.from_numpy(li.astype(np.int64)).to(dev),
                torch.from_numpy(ri.astype(np.int64)).to(dev))

    def groupby(self, keys: List[Val], aggs, hint=None):
        import numpy as np
        n = keys[0].data.numel() if keys else (
            aggs[0][1].data.numel() if aggs and aggs[0][1] is not None else 0)
        dev = keys[0].data.device if keys else "cpu"
        if keys:
            codes = self._codes(keys, null_is_group=True)
        else:
            codes = np.zeros(n, dtype=np.int64)
        uniq, gid = np.unique(codes, return_inverse=True)
        ng = len(uniq) if n > 0 else 0
        # representative (first) row per group
        order = np.argsort(gid, kind="stable")
        sorted_gid = gid[order]
        seg_start = np.searchsorted(sorted_gid, np.arange(ng), side="left")
        rep = order[seg_start] if ng else np.zeros(0, dtype=np.int64)
        rep_t = torch.from_numpy(rep.astype(np.int64))
        out_keys = []
        for v in keys:
            data = v.data[rep_t]
            valid = v.valid[rep_t] if v.valid is not None else None
            out_keys.append(Val(data, valid, v.dict))
        out_res = []
        gid_t = gid
        for fn, v in aggs:
            if fn == "count":
                acc = np.bincount(gid_t, minlength=ng)
                out_res.append(Val(torch.from_numpy(
                    acc.astype(np.int64)).to(dev)))
                continue
            vals = v.data.cpu().numpy()
            if vals.dtype == np.bool_:
                vals = vals.astype(np.int64)
            vm = (v.valid.cpu().numpy() if v.valid is not None
                  else np.ones(n, dtype=bool))
            if fn == "count_valid":
                acc = np.bincount(gid_t[vm], minlength=ng)
                out_res.append(Val(torch.from_numpy(
                    acc.astype(np.int64)).to(dev)))
                continue
            is_float = vals.dtype.kind == "f"
            vcnt = np.bincount(gid_t[vm], minlength=ng)
            if fn == "sum":
                acc = np.bincount(gid_t[vm], weights=vals[vm].astype(
                    np.float64), minlength=ng)
                if not is_float:
                    acc = np.round(acc).astype(np.int64)
            else:  # min / max on sorted segments
                sv = vals[order]
                sm = vm[order]
                big = np.inf if is_float else np.iinfo(np.int64).max
                if fn == "max":
                    big = -np.inf if is_float else np.iinfo(np.int64).min
                sv = np.where(sm, sv.astype(np.float64), big)
                if len(sv):
                    red = (np.minimum.reduceat(sv, seg_start) if fn == "min"
                           else np.maximum.reduceat(sv, seg_start))
                else:
                    red = np.zeros(0)
                acc = red if is_float else \
                    np.where(vcnt > 0, red, 0).astype(np.int64)
            has_null = (vcnt == 0).any()
            dt = torch.float64 if is_float else torch.int64
            data = torch.from_numpy(np.ascontiguousarray(
                np.where(vcnt > 0, acc, 0)).astype(
                np.float64 if is_float else np.int64)).to(dev)
            valid = (torch.from_numpy(vcnt > 0).to(dev)
                     if has_null else None)
            out_res.append(Val(data, valid))
        return out_keys, out_res, ng


# ---------------------------------------------------------------------------
# engine
# ---------------------------------------------------------------------------

class Engine:
    def __init__(self, catalog, device="cuda", world: int = 1, rank: int = 0,
                 backend=None):
        """catalog: dict name -> Dataset (see schema.py) or Frame."""
        self.catalog = catalog
        self.device = torch.device(device)
        self.world = world
        self.rank = rank
        self.backend = backend or (GpuBackend() if self.device.type == "cuda"
                                   else CpuBackend())
        self.temps: Dict[str, Frame] = {}
        self._scan_cache: Dict[str, Frame] = {}

    # -- public ------------------------------------------------------------
    def run(self, plan: Plan) -> Frame:
        f = self._exec(plan)
        if f.sharded and self.world > 1:
            f = self._allgather(f)
            f.sharded = False
        return f

    def register(self, name: str, frame: Frame):
        self.temps[name] = frame

    def const_frame(self, **cols) -> Frame:
        """One-row frame from python scalars (scalar-subquery composition
        queries like q9/q28/q88)."""
        out = {}
        for name, v in cols.items():
            if v is None:
                out[name] = Val(torch.zeros(1, dtype=torch.float64,
                                            device=self.device),
                                torch.zeros(1, dtype=torch.bool,
                                            device=self.device))
            elif isinstance(v, str):
                out[name] = Val(torch.zeros(1, dtype=torch.int64,
                                            device=self.device), None, [v])
            elif isinstance(v, int):
                out[name] = Val(torch.full((1,), v, dtype=torch.int64,
                                           device=self.device))
            else:
                out[name] = Val(torch.full((1,), float(v),
                                           dtype=torch.float64,
                                           device=self.device))
        return Frame(out, 1)

    def scalar(self, plan: Plan, col: Optional[str] = None):
        f = self.run(plan)
        name = col or f.names()[0]
        v = f.cols[name]
        if f.nrows == 0:
            return None
        if v.valid is not None and not bool(v.valid[0]):
            return None
        x = v.data[0].item()
        return v.dict[int(x)] if v.dict is not None else x

    # -- execution ---------------------------------------------------------
    def _exec(self, p: Plan) -> Frame:
        m = getattr(self, "_exec_" + type(p).__name__.lower())
        return m(p)

    def _exec_scan(self, p: Scan) -> Frame:
        if p.table in self.temps:
            f = self.temps[p.table]
            if p.columns:
                return Frame({c: f.cols[c] for c in p.columns}, f.nrows,
                             f.sharded)
            return f
        return self._load(p.table, p.columns)

    def _load(self, name: str, columns=None) -> Frame:
        """Columns load lazily (only what queries reference) and stay
        cached on the device across queries."""
        ds = self.catalog[name]
        cache = self._scan_cache.setdefault(name, {})
        want = list(columns) if columns else list(ds.columns.keys())
        for cname in want:
            if cname in cache:
                continue
            t = torch.as_tensor(ds.columns[cname]).to(self.device)
            valid = None
            if ds.valid.get(cname) is not None:
                valid = torch.as_tensor(ds.valid[cname]).to(self.device)
            cache[cname] = Val(t, valid, ds.dicts.get(cname))
        cols = {c: cache[c] for c in want}
        return Frame(cols, ds.nrows,
                     sharded=(ds.sharded and self.world > 1))

    def _exec_filter(self, p: Filter) -> Frame:
        f = self._exec(p.child)
        v = eval_expr(p.cond, f.cols, f.nrows, self.device)
        m = v.data.bool()
        if v.valid is not None:
            m = m & v.valid
        return f.mask(m)

    def _exec_project(self, p: Project) -> Frame:
        f = self._exec(p.child)
        out = dict(f.cols) if p.extend else {}
        for name, e in p.outs:
            out[name] = eval_expr(_wrap(e), f.cols, f.nrows, self.device)
        return Frame(out, f.nrows, f.sharded)

    def _exec_join(self, p: Join) -> Frame:
        lf = self._exec(p.left)
        rf = self._exec(p.right)
        if self.world > 1 and (lf.sharded or rf.sharded):
            lf, rf = self._copartition(lf, rf, p.on)
        lkeys = [lf.cols[a] for a, _ in p.on]
        rkeys = [rf.cols[b] for _, b in p.on]
        li, ri = self.backend.join(lkeys, rkeys, p.how, lf.nrows)
        sharded = lf.sharded or rf.sharded
        if p.how in ("semi", "anti"):
            out = lf.gather(li)
            out.sharded = sharded
            return out
        left_out = lf.gather(li, null_for_neg=(p.how == "full"))
        # right join-key columns equal the left keys for inner/left; drop
        # them (full outer keeps both so queries can coalesce)
        rnames = {b for _, b in p.on} if p.how != "full" else set()
        out = dict(left_out.cols)
        right_g = rf.gather(ri, null_for_neg=(p.how in ("left", "full")))
        for name, v in right_g.cols.items():
            if name in rnames:
                continue
            if name in out:
                raise ValueError(f"join output column clash: {name} "
                                 "(Project/rename one side first)")
            out[name] = v
        return Frame(out, left_out.nrows, sharded)

    def _exec_agg(self, p: Agg) -> Frame:
        f = self._exec(p.child)
        if p.rollup:
            return self._rollup(f, p)
        return self._agg_frame(f, p.keys, p.aggs)

    def _agg_frame(self, f: Frame, keys: List[str],
                   aggs: List[Tuple[str, str, Optional[Expr]]],
                   extra_keys: Optional[Dict[str, Val]] = None) -> Frame:
        # two-phase when distributed: partial -> allgather -> final
        distributed = f.sharded and self.world > 1
        plain, countd = [], []
        for name, fn, e in aggs:
            (countd if fn == "countd" else plain).append((name, fn, e))
        kv = [f.cols[k] for k in keys]
        if extra_keys:
            kv = kv + list(extra_keys.values())
            keynames = keys + list(extra_keys.keys())
        else:
            keynames = list(keys)
        dummy_key = not kv
        if dummy_key:  # global aggregate: constant key, dropped at the end
            kv = [Val(torch.zeros(f.nrows, dtype=torch.int64,
                                  device=self.device))]
            keynames = ["__k0"]

        native = []          # (fn, Val|None)
        slots = []           # per plain agg: dict of native indices
        for name, fn, e in plain:
            v = (eval_expr(_wrap(e), f.cols, f.nrows, self.device)
                 if e is not None else None)
            if fn == "count":
                if v is None:
                    slots.append({"fn": fn, "n": self._push(native, ("count",
                                                                    None))})
                else:
                    slots.append({"fn": "count_valid",
                                  "n": self._push(native, ("count_valid", v))})
            elif fn in ("sum", "min", "max"):
                fv = self._float(v)
                slots.append({"fn": fn, "n": self._push(native, (fn, fv)),
                              "cv": self._push(native, ("count_valid", fv))})
            elif fn == "avg":
                fv = self._float(v)
                slots.append({"fn": fn, "s": self._push(native, ("sum", fv)),
                              "c": self._push(native, ("count_valid", fv))})
            elif fn == "stddev_samp":
                fv = self._float(v)
                sq = Val(fv.data * fv.data, fv.valid)
                slots.append({"fn": fn, "s": self._push(native, ("sum", fv)),
                              "q": self._push(native, ("sum", sq)),
                              "c": self._push(native, ("count_valid", fv))})
            else:
                raise ValueError(fn)
        if not native:
            native.append(("count", None))  # keys-only: need group discovery
        # a global aggregate has exactly one group: the hint engages the
        # LDS pre-aggregation kernel instead of a per-row contended CAS
        # table; chunk to <=3 aggs per pass (the LDS kernel's limit)
        if dummy_key:
            # chunk by EFFECTIVE agg count: a nullable SUM/MIN/MAX carries a
            # hidden count_valid stream inside the kernel, so it costs 2 of
            # the LDS kernel's <=3 agg slots (q97 postmortem: 3 sums over
            # masked values -> naggs 6 -> no LDS -> 255 ms on one CAS slot)
            _precheck_all_true([v for _f, v in native])

            def _cost(fn_v):
                fn, v = fn_v
                if fn in ("count", "count_valid") or v is None:
                    return 1
                if v.valid is None or getattr(v, "_all_true", False):
                    return 1
                return 2
            chunks, cur, budget = [], [], 3
            for item in native:
                c = _cost(item)
                if cur and budget < c:
                    chunks.append(cur)
                    cur, budget = [], 3
                cur.append(item)
                budget -= c
            if cur:
                chunks.append(cur)
            kvals = rvals = None
            ng = 0
            out_r = []
            for chunk in chunks:
                kvals, rv, ng = self.backend.groupby(kv, chunk, hint=1)
                out_r.extend(rv)
            rvals = out_r
        else:
            kvals, rvals, ng = self.backend.groupby(kv, native)

        if distributed:
            # partials -> replicated frame -> re-reduce
            part = Frame(self._kv_dict(keynames, kvals, rvals), ng,
                         sharded=True)
            gathered = self._allgather(part)
            kvals2 = [gathered.cols[k] for k in keynames]
            native2 = []
            for fn, _v in native:
                i = len(native2)
                merged_fn = {"count": "sum", "count_valid": "sum"}.get(fn, fn)
                native2.append((merged_fn, gathered.cols[f"__a{i}"]))
            kvals, rvals, ng = self.backend.groupby(kvals2, native2)

        out: Dict[str, Val] = {}
        for k, v in zip(keynames, kvals):
            out[k] = v
        for (name, fn, _e), s in zip(plain, slots):
            if s["fn"] in ("count", "count_valid"):
                out[name] = rvals[s["n"]]
            elif s["fn"] in ("sum", "min", "max"):
                out[name] = rvals[s["n"]]
            elif s["fn"] == "avg":
                out[name] = self._div(rvals[s["s"]], rvals[s["c"]])
            elif s["fn"] == "stddev_samp":
                ssum, ssq = rvals[s["s"]], rvals[s["q"]]
                cnt = rvals[s["c"]].data.to(torch.float64)
                ok = cnt >= 2
                mean_sq = ssum.data * ssum.data / cnt.clamp(min=1)
                var = (ssq.data - mean_sq) / (cnt - 1).clamp(min=1)
                out[name] = Val(var.clamp(min=0).sqrt(), ok)
        result = Frame(out, ng, sharded=False)
        if dummy_key and ng == 0:
            # SQL global aggregate over empty input still yields one row:
            # counts are 0, everything else null
            cols = {"__k0": Val(torch.zeros(1, dtype=torch.int64,
                                            device=self.device))}
            for n_, f_, _x in plain:
                if f_ == "count":
                    cols[n_] = Val(torch.zeros(1, dtype=torch.int64,
                                               device=self.device))
                else:
                    cols[n_] = Val(torch.zeros(1, dtype=torch.float64,
                                               device=self.device),
                                   torch.zeros(1, dtype=torch.bool,
                                               device=self.device))
            result = Frame(cols, 1)
            ng = 1

        for name, fn, e in countd:
            d = self._count_distinct(f, keys, e, extra_keys)
            result = self._merge_on_keys(result, d,
                                         keynames if not dummy_key else [],
                                         name)
        if dummy_key:
            result.cols.pop("__k0", None)
        return result

    @staticmethod
    def _push(native, item):
        native.append(item)
        return len(native) - 1

    @staticmethod
    def _float(v: Val) -> Val:
        if v.data.dtype in (torch.float32, torch.float64):
            return Val(v.data.to(torch.float64), v.valid)
        return v

    @staticmethod
    def _div(s: Val, c: Val) -> Val:
        cnt = c.data.to(torch.float64)
        ok = cnt > 0
        data = s.data.to(torch.float64) / cnt.clamp(min=1)
        valid = ok if s.valid is None else (ok & s.valid)
        return Val(data, valid)

    def _kv_dict(self, keynames, kvals, rvals):
        d = {}
        for k, v in zip(keynames, kvals):
            d[k] = v
        for i, v in enumerate(rvals):
            d[f"__a{i}"] = v
        return d

    def _count_distinct(self, f: Frame, keys, e, extra_keys) -> Frame:
        v = eval_expr(_wrap(e), f.cols, f.nrows, self.device)
        kv = [f.cols[k] for k in keys]
        keynames = list(keys)
        if extra_keys:
            kv = kv + list(extra_keys.values())
            keynames = keys + list(extra_keys.keys())
        # distinct (keys, value): drop null values first (countd skips nulls)
        if v.valid is not None:
            keep = torch.nonzero(v.valid, as_tuple=False).view(-1)
            kv = [Val(x.data[keep],
                      x.valid[keep] if x.valid is not None else None,
                      x.dict) for x in kv]
            v = Val(v.data[keep], None, v.dict)
        kvals, _r, ng = self.backend.groupby(kv + [v], [("count", None)])
        dist = Frame(self._kv_dict(keynames + ["__v"], kvals, []), ng,
                     sharded=f.sharded)
        if dist.sharded and self.world > 1:
            dist = self._allgather(dist)
            kv2 = [dist.cols[k] for k in keynames + ["__v"]]
            kvals, _r, ng = self.backend.groupby(kv2, [("count", None)])
            dist = Frame(self._kv_dict(keynames + ["__v"], kvals, []), ng)
        # count per key
        if not keynames:  # global count distinct
            return Frame({"__cd": Val(torch.full(
                (1,), dist.nrows, dtype=torch.int64,
                device=self.device))}, 1)
        kv3 = [dist.cols[k] for k in keynames]
        kvals, rvals, ng = self.backend.groupby(kv3, [("count", None)])
        out = self._kv_dict(keynames, kvals, [])
        out["__cd"] = rvals[0]
        return Frame(out, ng)

    def _merge_on_keys(self, main: Frame, d: Frame, keynames, out_name):
        if not keynames:
            # global aggregate: single row each
            out = dict(main.cols)
            out[out_name] = d.cols["__cd"] if d.nrows else \
                Val(torch.zeros(main.nrows, dtype=torch.int64,
                                device=self.device))
            if main.nrows == 0 and d.nrows:
                # main had no plain aggs -> take d's row
                return Frame({out_name: d.cols["__cd"]}, d.nrows)
            return Frame(out, main.nrows)
        li, ri = self.backend.join([main.cols[k] for k in keynames],
                                   [d.cols[k] for k in keynames], "left",
                                   main.nrows)
        out_f = main.gather(li)
        cd = d.cols["__cd"]
        neg = ri < 0
        data = cd.data[ri.clamp(min=0)]
        data = torch.where(neg, torch.zeros_like(data), data)
        out_f.cols[out_name] = Val(data)
        return out_f

    def _rollup(self, f: Frame, p: Agg) -> Frame:
        # cascade re-aggregation when every agg is decomposable: level L-1
        # re-aggregates level L's RESULT (sum of sums, min of mins, count ->
        # sum) instead of rescanning the fact input once per level — q67's
        # 9-level rollup over 58M rows becomes one 58M pass + 8 passes over
        # group-count-sized inputs. Both backends run the same cascade, so
        # GPU-vs-oracle verification is unaffected.
        from .expr import col as _col
        cascade = all(fn in ("sum", "min", "max", "count")
                      for _n, fn, _e in p.aggs)
        frames = []
        src = None
        for lvl in range(len(p.keys), -1, -1):
            keys = p.keys[:lvl]
            if src is None or not cascade:
                g = self._agg_frame(f, keys, p.aggs)
            else:
                reaggs = [(n, "sum" if fn == "count" else fn, _col(n))
                          for n, fn, _e in p.aggs]
                g = self._agg_frame(src, keys, reaggs)
            src = g
            cols = dict(g.cols)
            for miss in p.keys[lvl:]:
                proto = f.cols[miss]
                data = torch.zeros(g.nrows, dtype=proto.data.dtype,
                                   device=self.device)
                cols[miss] = Val(data, torch.zeros(g.nrows, dtype=torch.bool,
                                                   device=self.device),
                                 proto.dict)
            cols["__lvl"] = Val(torch.full((g.nrows,), lvl,
                                           dtype=torch.int64,
                                           device=self.device))
            ordered = {k: cols[k] for k in
                       p.keys + [n for n, _f, _e in p.aggs] + ["__lvl"]}
            frames.append(Frame(ordered, g.nrows))
        return self._union(frames)

    def _exec_window(self, p: Window) -> Frame:
        f = self._exec(p.child)
        if f.sharded and self.world > 1:
            f = self._allgather(f)
            f.sharded = False
        n = f.nrows
        dev = self.device
        if n == 0:
            out = dict(f.cols)
            for spec in p.funcs:
                out[spec[0]] = Val(torch.zeros(0, dtype=torch.int64,
                                               device=dev))
            return Frame(out, 0)
        out = dict(f.cols)
        for spec in p.funcs:
            name, fn, arg = spec[0], spec[1], spec[2]
            order = spec[3] if len(spec) > 3 else None
            # sort rows: partition keys asc, then order keys
            perm = torch.arange(n, dtype=torch.int64, device=dev)
            sort_keys = []
            if order:
                sort_keys.extend([(c, asc) for c, asc in order])
            for k in reversed(p.partition):
                sort_keys.insert(0, (k, True))
            perm = self._argsort(f, sort_keys) if sort_keys else perm
            inv = torch.empty_like(perm)
            inv[perm] = torch.arange(n, dtype=torch.int64, device=dev)
            # segment boundaries on partition keys (nulls group together:
            # a data difference only counts when both rows are valid)
            bnd = torch.zeros(n, dtype=torch.bool, device=dev)
            bnd[0] = True
            for k in p.partition:
                v = f.cols[k]
                d = v.data[perm]
                diff = d[1:] != d[:-1]
                if v.valid is not None:
                    vv = v.valid[perm]
                    diff = (diff & vv[1:] & vv[:-1]) | (vv[1:] != vv[:-1])
                bnd[1:] |= diff
            seg = torch.cumsum(bnd.to(torch.int64), 0) - 1
            nseg = int(seg[-1].item()) + 1
            seg_start = torch.zeros(nseg, dtype=torch.int64, device=dev)
            seg_start.scatter_(0, seg[bnd], torch.nonzero(
                bnd, as_tuple=False).view(-1))
            pos = torch.arange(n, dtype=torch.int64, device=dev) - \
                seg_start[seg]
            if fn in ("sum", "avg", "min", "max"):
                v = f.cols[arg]
                data = v.data[perm].to(torch.float64)
                vm = (v.valid[perm] if v.valid is not None
                      else torch.ones(n, dtype=torch.bool, device=dev))
                acc = torch.zeros(nseg, dtype=torch.float64, device=dev)
                if fn in ("sum", "avg"):
                    acc.index_add_(0, seg, torch.where(vm, data,
                                                       torch.zeros_like(data)))
                    if fn == "avg":
                        cnt = torch.zeros(nseg, dtype=torch.float64,
                                          device=dev)
                        cnt.index_add_(0, seg, vm.to(torch.float64))
                        acc = acc / cnt.clamp(min=1)
                elif fn == "min":
                    acc = torch.full((nseg,), float("inf"),
                                     dtype=torch.float64, device=dev)
                    acc.scatter_reduce_(0, seg, torch.where(
                        vm, data, torch.full_like(data, float("inf"))),
                        reduce="amin")
                else:
                    acc = torch.full((nseg,), float("-inf"),
                                     dtype=torch.float64, device=dev)
                    acc.scatter_reduce_(0, seg, torch.where(
                        vm, data, torch.full_like(data, float("-inf"))),
                        reduce="amax")
                res_sorted = acc[seg]
                out[name] = Val(res_sorted[inv])
            elif fn == "cumsum":
                v = f.cols[arg]
                data = v.data[perm].to(torch.float64)
                vm = (v.valid[perm] if v.valid is not None
                      else torch.ones(n, dtype=torch.bool, device=dev))
                data = torch.where(vm, data, torch.zeros_like(data))
                cs = torch.cumsum(data, 0)
                base = torch.zeros(nseg, dtype=torch.float64, device=dev)
                starts = torch.nonzero(bnd, as_tuple=False).view(-1)
                base[1:] = cs[starts[1:] - 1]
                res_sorted = cs - base[seg]
                out[name] = Val(res_sorted[inv])
            elif fn in ("rank", "dense_rank", "row_number"):
                assert order, f"{fn} needs an order"
                if fn == "row_number":
                    res_sorted = pos + 1
                else:
                    change = bnd.clone()
                    for c, _asc in order:
                        v = f.cols[c]
                        d = v.data[perm]
                        diff = d[1:] != d[:-1]
                        if v.valid is not None:
                            vv = v.valid[perm]
                            diff = (diff & vv[1:] & vv[:-1]) | \
                                (vv[1:] != vv[:-1])
                        change[1:] |= diff
                    if fn == "rank":
                        idx = torch.arange(n, dtype=torch.int64, device=dev)
                        last_change = torch.cummax(
                            torch.where(change, idx,
                                        torch.full_like(idx, -1)), 0)[0]
                        res_sorted = last_change - seg_start[seg] + 1
                    else:
                        dr = torch.cumsum(change.to(torch.int64), 0)
                        seg_base = dr[torch.nonzero(bnd,
                                                    as_tuple=False).view(-1)]
                        res_sorted = dr - seg_base[seg] + 1
                out[name] = Val(res_sorted[inv])
            else:
                raise ValueError(fn)
        return Frame(out, n, f.sharded)

    def _argsort(self, f: Frame, by: List[Tuple[str, bool]]) -> torch.Tensor:
        """Stable multi-key argsort. Spark default null order: NULLS FIRST
        when ascending, NULLS LAST when descending."""
        n = f.nrows
        perm = torch.arange(n, dtype=torch.int64, device=self.device)
        for name, asc in reversed(by):
            v = f.cols[name]
            d = v.data[perm]
            if d.dtype == torch.bool:
                d = d.to(torch.int8)
            if v.valid is not None:
                vm = v.valid[perm]
                if d.dtype.is_floating_point:
                    fill = float("-inf") if asc else float("inf")
                else:
                    ii = torch.iinfo(d.dtype)
                    fill = ii.min if asc else ii.max
                d = torch.where(vm, d, torch.full_like(d, fill))
            o = torch.argsort(d, stable=True, descending=not asc)
            perm = perm[o]
        return perm

    def _exec_sort(self, p: Sort) -> Frame:
        f = self._exec(p.child)
        if f.sharded and self.world > 1:
            f = self._allgather(f)
            f.sharded = False
        # append the remaining columns as tie-breakers so a following Limit
        # cuts deterministically regardless of execution order (both
        # backends and any rank count agree on the kept rows)
        named = {n for n, _a in p.by}
        by = list(p.by) + [(n, True) for n in f.names() if n not in named]
        perm = self._argsort(f, by)
        return f.gather(perm)

    def _exec_limit(self, p: Limit) -> Frame:
        f = self._exec(p.child)
        if f.sharded and self.world > 1:
            f = self._allgather(f)
            f.sharded = False
        if f.nrows <= p.n:
            return f
        idx = torch.arange(p.n, dtype=torch.int64, device=self.device)
        return f.gather(idx)

    def _exec_union(self, p: Union) -> Frame:
        frames = [self._exec(c) for c in p.children]
        frames = [self._allgather(f) if f.sharded and self.world > 1 else f
                  for f in frames]
        return self._union(frames)

    def _union(self, frames: List[Frame]) -> Frame:
        names = frames[0].names()
        out = {}
        total = sum(f.nrows for f in frames)
        for name in names:
            protos = [f.cols[name] for f in frames]
            dicts = [v.dict for v in protos]
            d = None
            if any(x is not None for x in dicts):
                assert all(x is not None for x in dicts), \
                    f"union mixes dict and plain column {name}"
                if all(x is dicts[0] for x in dicts):
                    d = dicts[0]
                else:  # merge dictionaries, remap codes per branch
                    d = sorted(set().union(*[set(x) for x in dicts]))
                    code = {s: i for i, s in enumerate(d)}
                    protos = [Val(torch.tensor(
                        [code[s] for s in v.dict], dtype=torch.int64,
                        device=self.device)[v.data.long()], v.valid, d)
                        for v in protos]
            dt = protos[0].data.dtype
            for v in protos[1:]:
                dt = torch.promote_types(dt, v.data.dtype)
            data = torch.cat([v.data.to(dt) for v in protos])
            if any(v.valid is not None for v in protos):
                valid = torch.cat([
                    v.valid if v.valid is not None else
                    torch.ones(f.nrows, dtype=torch.bool, device=self.device)
                    for v, f in zip(protos, frames)])
            else:
                valid = None
            out[name] = Val(data, valid, d)
        return Frame(out, total)

    def _exec_distinct(self, p: Distinct) -> Frame:
        f = self._exec(p.child)
        names = f.names()
        kvals, _r, ng = self.backend.groupby([f.cols[k] for k in names],
                                             [("count", None)])
        out = {k: v for k, v in zip(names, kvals)}
        g = Frame(out, ng, sharded=f.sharded)
        if g.sharded and self.world > 1:
            g = self._allgather(g)
            kvals, _r, ng = self.backend.groupby(
                [g.cols[k] for k in names], [("count", None)])
            g = Frame({k: v for k, v in zip(names, kvals)}, ng)
        return g

    # -- distribution ------------------------------------------------------
    def _allgather(self, f: Frame) -> Frame:
        import torch.distributed as dist
        if self.world == 1 or not dist.is_initialized():
            return Frame(dict(f.cols), f.nrows, sharded=False)
        sizes = [torch.zeros(1, dtype=torch.int64) for _ in range(self.world)]
        mine = torch.tensor([f.nrows], dtype=torch.int64)
        if dist.get_backend() == "nccl":
            sizes = [s.to(self.device) for s in sizes]
            mine = mine.to(self.device)
        dist.all_gather(sizes, mine)
        sizes = [int(s.item()) for s in sizes]
        mx = max(sizes + [1])

        def gather_tensor(t):
            as_bool = t.dtype == torch.bool
            if as_bool:
                t = t.to(torch.uint8)
            pad = torch.zeros(mx, dtype=t.dtype, device=t.device)
            pad[:f.nrows] = t
            recv = [torch.empty_like(pad) for _ in range(self.world)]
            dist.all_gather(recv, pad)
            cat = torch.cat([r[:s] for r, s in zip(recv, sizes)])
            return cat.bool() if as_bool else cat

        out_cols = {}
        for name, v in f.cols.items():
            data = gather_tensor(v.data)
            valid = gather_tensor(v.valid) if v.valid is not None else None
            out_cols[name] = Val(data, valid, v.dict)
        return Frame(out_cols, sum(sizes), sharded=False)

    def _copartition(self, lf: Frame, rf: Frame, on) -> Tuple[Frame, Frame]:
        """Hash-partition both join inputs by key so matching rows land on
        the same rank; sharded sides move via all-to-all, replicated sides
        just keep their local partition."""
        lout = self._partition_side(lf, [a for a, _ in on])
        rout = self._partition_side(rf, [b for _, b in on])
        return lout, rout

    def _partition_side(self, f: Frame, keys: List[str]) -> Frame:
        h = self._key_hash(f, keys)
        part = h % self.world
        if not f.sharded:
            keep = torch.nonzero(part == self.rank, as_tuple=False).view(-1)
            g = f.gather(keep)
            g.sharded = True
            return g
        return self._exchange(f, part)

    def _key_hash(self, f: Frame, keys: List[str]) -> torch.Tensor:
        h = torch.zeros(f.nrows, dtype=torch.int64, device=self.device)
        for k in keys:
            v = f.cols[k]
            d = v.data.to(torch.int64) if not v.data.dtype.is_floating_point \
                else v.data.to(torch.float64).view(torch.int64)
            h = h * 0x9E3779B97F4A7C15 + d
        return (h & (2**63 - 1))

    def _exchange(self, f: Frame, part: torch.Tensor) -> Frame:
        """All-to-all repartition of a sharded frame by partition id."""
        from ..parallel.exchange import _exchange_bytes
        order = torch.argsort(part, stable=True)
        counts = torch.bincount(part, minlength=self.world)
        g = f.gather(order)
        splits = [int(c) for c in counts.cpu()]
        out_cols = {}
        out_n = None
        for name, v in g.cols.items():
            data8 = v.data.contiguous().view(torch.uint8).view(-1)
            esz = v.data.element_size()
            b, out_splits = _exchange_bytes(
                data8, [s * esz for s in splits])
            data = b.view(v.data.dtype)
            valid = None
            if v.valid is not None:
                v8 = v.valid.to(torch.uint8)
                bb, _os = _exchange_bytes(v8, splits)
                valid = bb.bool()
            out_cols[name] = Val(data, valid, v.dict)
            out_n = data.numel()
        return Frame(out_cols, out_n if out_n is not None else 0,
                     sharded=True)
