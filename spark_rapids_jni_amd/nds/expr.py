"""Expression IR + torch evaluator for the NDS plan engine.

Spark SQL null semantics (the ones the reference's kernels implement):
arithmetic/comparison propagate null; AND/OR are three-valued Kleene;
Filter keeps rows whose predicate is TRUE (null -> dropped).

String dimension attributes are dictionary-encoded at generation time
(codes in the column, the dictionary in the table metadata), so string
predicates (=, IN, LIKE, substr) evaluate against the dictionary on host
and become integer-code predicates on device — a deliberate MI355X-first
design: the scan stays numeric and HBM-bandwidth-bound.
"""
import datetime
from typing import List, Optional, Sequence

import torch


class Val:
    """An evaluated expression: data tensor + optional validity (bool, True =
    valid) + optional string dictionary (data holds codes)."""
    __slots__ = ("data", "valid", "dict", "_all_true")

    def __init__(self, data, valid=None, dict=None):
        self.data = data
        self.valid = valid
        self.dict = dict
        self._all_true = None  # memo: valid mask known all-true (1 sync max)

    def valid_mask(self):
        if self.valid is None:
            return torch.ones_like(self.data, dtype=torch.bool)
        return self.valid


class Expr:
    def __add__(self, o): return BinOp("+", self, _wrap(o))
    def __radd__(self, o): return BinOp("+", _wrap(o), self)
    def __sub__(self, o): return BinOp("-", self, _wrap(o))
    def __rsub__(self, o): return BinOp("-", _wrap(o), self)
    def __mul__(self, o): return BinOp("*", self, _wrap(o))
    def __rmul__(self, o): return BinOp("*", _wrap(o), self)
    def __truediv__(self, o): return BinOp("/", self, _wrap(o))
    def __rtruediv__(self, o): return BinOp("/", _wrap(o), self)
    def __gt__(self, o): return BinOp(">", self, _wrap(o))
    def __ge__(self, o): return BinOp(">=", self, _wrap(o))
    def __lt__(self, o): return BinOp("<", self, _wrap(o))
    def __le__(self, o): return BinOp("<=", self, _wrap(o))
    def __eq__(self, o): return BinOp("==", self, _wrap(o))
    def __ne__(self, o): return BinOp("!=", self, _wrap(o))
    def __and__(self, o): return BinOp("and", self, _wrap(o))
    def __or__(self, o): return BinOp("or", self, _wrap(o))
    def __invert__(self): return Not(self)
    def __neg__(self): return BinOp("-", Lit(0), self)
    def __hash__(self): return id(self)

    def isin(self, values): return InList(self, list(values))
    def between(self, lo, hi):
        return BinOp("and", BinOp(">=", self, _wrap(lo)),
                     BinOp("<=", self, _wrap(hi)))
    def like(self, pattern): return Like(self, pattern)
    def substr(self, start, length): return Substr(self, start, length)
    def cast_float(self): return Cast(self, "float")
    def cast_int(self): return Cast(self, "int")


def _wrap(v):
    return v if isinstance(v, Expr) else Lit(v)


class Col(Expr):
    def __init__(self, name: str):
        self.name = name

    def __repr__(self):
        return f"col({self.name})"


class Lit(Expr):
    def __init__(self, value):
        self.value = value

    def __repr__(self):
        return f"lit({self.value!r})"


class BinOp(Expr):
    def __init__(self, op, l, r):
        self.op, self.l, self.r = op, l, r


class Not(Expr):
    def __init__(self, e):
        self.e = e


class IsNull(Expr):
    def __init__(self, e, negate=False):
        self.e, self.negate = e, negate


class InList(Expr):
    def __init__(self, e, values):
        self.e, self.values = e, values


class Like(Expr):
    def __init__(self, e, pattern):
        self.e, self.pattern = e, pattern


class Substr(Expr):
    def __init__(self, e, start, length):
        self.e, self.start, self.length = e, start, length


class Cast(Expr):
    def __init__(self, e, to):
        self.e, self.to = e, to


class CaseWhen(Expr):
    def __init__(self, branches, otherwise):
        self.branches = [(c, _wrap(v)) for c, v in branches]
        self.otherwise = _wrap(otherwise) if otherwise is not None else None


class Coalesce(Expr):
    def __init__(self, exprs):
        self.exprs = [_wrap(e) for e in exprs]


class Abs(Expr):
    def __init__(self, e):
        self.e = _wrap(e)


def abs_(e) -> Abs:
    return Abs(e)


def col(name: str) -> Col:
    return Col(name)


def lit(v) -> Lit:
    return Lit(v)


_EPOCH = datetime.date(1970, 1, 1)


def date_lit(s: str) -> Lit:
    """Date literal as days since epoch (d_date columns use this encoding)."""
    y, m, d = map(int, s.split("-"))
    return Lit((datetime.date(y, m, d) - _EPOCH).days)


def case_when(*branches, otherwise=None) -> CaseWhen:
    """case_when((cond, value), ..., otherwise=value)"""
    return CaseWhen(list(branches), otherwise)


def coalesce(*exprs) -> Coalesce:
    return Coalesce(list(exprs))


def is_null(e) -> IsNull:
    return IsNull(_wrap(e))


def is_not_null(e) -> IsNull:
    return IsNull(_wrap(e), negate=True)


# ---------------------------------------------------------------------------
# evaluation
# ---------------------------------------------------------------------------

def _like_match(s: str, pattern: str) -> bool:
    """SQL LIKE with % and _ (no escapes — TPC-DS uses only these)."""
    import re
    rx = "^" + re.escape(pattern).replace("%", ".*").replace("_", ".") + "$"
    return re.match(rx, s) is not None


def eval_expr(e: Expr, env: dict, nrows: int, device) -> Val:
    """env: name -> Val. Returns Val with data on `device`."""
    if isinstance(e, Col):
        v = env[e.name]
        if v is None:
            raise KeyError(e.name)
        return v
    if isinstance(e, Lit):
        v = e.value
        if v is None:
            data = torch.zeros(nrows, dtype=torch.int64, device=device)
            return Val(data, torch.zeros(nrows, dtype=torch.bool, device=device))
        if isinstance(v, str):
            # standalone string literal: a single-entry-dictionary column
            return Val(torch.zeros(nrows, dtype=torch.int64, device=device),
                       None, [v])
        if isinstance(v, bool):
            dt = torch.bool
        elif isinstance(v, int):
            dt = torch.int64
        elif isinstance(v, float):
            dt = torch.float64
        else:
            raise TypeError(f"unsupported literal {v!r}")
        return Val(torch.full((nrows,), v, dtype=dt, device=device))
    if isinstance(e, Abs):
        v = eval_expr(e.e, env, nrows, device)
        return Val(v.data.abs(), v.valid)
    if isinstance(e, BinOp):
        return _eval_binop(e, env, nrows, device)
    if isinstance(e, Not):
        v = eval_expr(e.e, env, nrows, device)
        return Val(~v.data.bool(), v.valid)
    if isinstance(e, IsNull):
        v = eval_expr(e.e, env, nrows, device)
        m = v.valid_mask()
        return Val(m if e.negate else ~m)
    if isinstance(e, InList):
        v = eval_expr(e.e, env, nrows, device)
        vals = e.values
        if v.dict is not None:
            codes = [i for i, s in enumerate(v.dict) if s in set(vals)]
            vals = codes
        if not vals:
            return Val(torch.zeros(nrows, dtype=torch.bool, device=device),
                       v.valid)
        t = torch.tensor(vals, dtype=v.data.dtype, device=device)
        return Val(torch.isin(v.data, t), v.valid)
    if isinstance(e, Like):
        v = eval_expr(e.e, env, nrows, device)
        assert v.dict is not None, "LIKE requires a dict-encoded column"
        codes = [i for i, s in enumerate(v.dict) if _like_match(s, e.pattern)]
        if not codes:
            return Val(torch.zeros(nrows, dtype=torch.bool, device=device),
                       v.valid)
        t = torch.tensor(codes, dtype=v.data.dtype, device=device)
        return Val(torch.isin(v.data, t), v.valid)
    if isinstance(e, Substr):
        v = eval_expr(e.e, env, nrows, device)
        assert v.dict is not None, "substr requires a dict-encoded column"
        # remap codes through a substring'd dictionary
        sub = [s[e.start - 1:e.start - 1 + e.length] for s in v.dict]
        new_dict = sorted(set(sub))
        code_of = {s: i for i, s in enumerate(new_dict)}
        remap = torch.tensor([code_of[s] for s in sub], dtype=v.data.dtype,
                             device=device)
        return Val(remap[v.data.long()], v.valid, new_dict)
    if isinstance(e, Cast):
        v = eval_expr(e.e, env, nrows, device)
        dt = torch.float64 if e.to == "float" else torch.int64
        return Val(v.data.to(dt), v.valid)
    if isinstance(e, CaseWhen):
        out_data = None
        out_valid = None
        decided = torch.zeros(nrows, dtype=torch.bool, device=device)
        for cond, value in e.branches:
            c = eval_expr(cond, env, nrows, device)
            hit = c.data.bool() & c.valid_mask() & ~decided
            val = eval_expr(value, env, nrows, device)
            if out_data is None:
                out_data = torch.zeros(nrows, dtype=val.data.dtype,
                                       device=device)
                out_valid = torch.zeros(nrows, dtype=torch.bool, device=device)
            if val.data.dtype != out_data.dtype:
                if out_data.dtype == torch.float64 or val.data.dtype == torch.float64:
                    out_data = out_data.to(torch.float64)
                    val = Val(val.data.to(torch.float64), val.valid, val.dict)
            out_data = torch.where(hit, val.data.to(out_data.dtype), out_data)
            out_valid = torch.where(hit, val.valid_mask(), out_valid)
            decided |= hit
        if e.otherwise is not None:
            val = eval_expr(e.otherwise, env, nrows, device)
            if out_data is None:
                return val
            if val.data.dtype != out_data.dtype:
                out_data = out_data.to(torch.promote_types(out_data.dtype,
                                                           val.data.dtype))
            out_data = torch.where(decided, out_data,
                                   val.data.to(out_data.dtype))
            out_valid = torch.where(decided, out_valid, val.valid_mask())
        if out_data is None:
            out_data = torch.zeros(nrows, dtype=torch.int64, device=device)
            out_valid = torch.zeros(nrows, dtype=torch.bool, device=device)
        return Val(out_data, out_valid)
    if isinstance(e, Coalesce):
        out = eval_expr(e.exprs[0], env, nrows, device)
        data, valid = out.data, out.valid_mask()
        for nxt in e.exprs[1:]:
            v = eval_expr(nxt, env, nrows, device)
            need = ~valid
            if v.data.dtype != data.dtype:
                data = data.to(torch.promote_types(data.dtype, v.data.dtype))
            data = torch.where(need, v.data.to(data.dtype), data)
            valid = valid | (need & v.valid_mask())
        return Val(data, valid)
    raise TypeError(f"cannot evaluate {type(e).__name__}")


_CMP = {">": torch.gt, ">=": torch.ge, "<": torch.lt, "<=": torch.le,
        "==": torch.eq, "!=": torch.ne}


def _eval_binop(e: BinOp, env, nrows, device) -> Val:
    # string literal against a dict-encoded column: map to code space
    if e.op in ("==", "!="):
        lit_side = None
        if isinstance(e.r, Lit) and isinstance(e.r.value, str):
            col_v = eval_expr(e.l, env, nrows, device)
            lit_side = e.r.value
        elif isinstance(e.l, Lit) and isinstance(e.l.value, str):
            col_v = eval_expr(e.r, env, nrows, device)
            lit_side = e.l.value
        if lit_side is not None:
            assert col_v.dict is not None, "string literal vs non-dict column"
            try:
                code = col_v.dict.index(lit_side)
            except ValueError:
                code = -(2**40)  # never matches
            res = (col_v.data == code) if e.op == "==" else (col_v.data != code)
            return Val(res, col_v.valid)
    l = eval_expr(e.l, env, nrows, device)
    r = eval_expr(e.r, env, nrows, device)
    if (l.dict is not None and r.dict is not None
            and l.dict is not r.dict and e.op in ("==", "!=")):
        # two dict-encoded columns with different dictionaries: remap both
        # into a merged dictionary before comparing codes
        merged = {s: i for i, s in enumerate(sorted(set(l.dict) |
                                                    set(r.dict)))}
        lmap = torch.tensor([merged[s] for s in l.dict], dtype=torch.int64,
                            device=device)
        rmap = torch.tensor([merged[s] for s in r.dict], dtype=torch.int64,
                            device=device)
        l = Val(lmap[l.data.long()], l.valid)
        r = Val(rmap[r.data.long()], r.valid)
    if e.op == "and":
        ld, rd = l.data.bool(), r.data.bool()
        data = ld & rd
        if l.valid is None and r.valid is None:
            return Val(data)
        lv, rv = l.valid_mask(), r.valid_mask()
        # Kleene: FALSE and X = FALSE is definite
        valid = (lv & rv) | (lv & ~ld) | (rv & ~rd)
        return Val(data & valid, valid)
    if e.op == "or":
        ld, rd = l.data.bool(), r.data.bool()
        data = ld | rd
        if l.valid is None and r.valid is None:
            return Val(data)
        lv, rv = l.valid_mask(), r.valid_mask()
        valid = (lv & rv) | (lv & ld) | (rv & rd)
        return Val(data, valid)
    ld, rd = l.data, r.data
    if ld.dtype != rd.dtype:
        t = torch.promote_types(ld.dtype, rd.dtype)
        ld, rd = ld.to(t), rd.to(t)
    if e.op in _CMP:
        data = _CMP[e.op](ld, rd)
    elif e.op == "+":
        data = ld + rd
    elif e.op == "-":
        data = ld - rd
    elif e.op == "*":
        data = ld * rd
    elif e.op == "/":
        # SQL division: null on divide-by-zero, float result
        ld = ld.to(torch.float64)
        rd = rd.to(torch.float64)
        zero = rd == 0
        data = ld / torch.where(zero, torch.ones_like(rd), rd)
        valid = ~zero
        if l.valid is not None:
            valid &= l.valid
        if r.valid is not None:
            valid &= r.valid
        return Val(data, valid)
    else:
        raise ValueError(e.op)
    if l.valid is None and r.valid is None:
        return Val(data)
    return Val(data, l.valid_mask() & r.valid_mask())
