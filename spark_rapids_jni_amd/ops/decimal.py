"""DECIMAL128 arithmetic (Java API parity: DecimalUtils.java:48-207 —
multiply128 / divide128 / integerDivide128 / remainder128 / add128 / sub128
with explicit target precision/scale, Spark overflow -> null or ANSI error).

Helpers compute Spark's default result types (DecimalType arithmetic with
the 38-digit cap and minimum-scale-6 adjustment).
"""
import torch

from .. import _native
from ..columnar import Column, DType, make_validity

MAX_PRECISION = 38
MIN_ADJUSTED_SCALE = 6


def _adjust(precision, scale):
    if precision <= MAX_PRECISION:
        return precision, scale
    digits = precision - scale
    min_scale = min(scale, MIN_ADJUSTED_SCALE)
    return MAX_PRECISION, max(MAX_PRECISION - digits, min_scale)


def multiply_result_type(p1, s1, p2, s2):
    return _adjust(p1 + p2 + 1, s1 + s2)


def divide_result_type(p1, s1, p2, s2):
    scale = max(MIN_ADJUSTED_SCALE, s1 + p2 + 1)
    return _adjust(p1 - s1 + s2 + scale, scale)


def add_result_type(p1, s1, p2, s2):
    scale = max(s1, s2)
    return _adjust(max(p1 - s1, p2 - s2) + scale + 1, scale)


class DecimalOverflowError(RuntimeError):
    def __init__(self, row):
        super().__init__(f"decimal overflow at row {row}")
        self.row_with_error = row


def _prep(a: Column, b: Column, ansi):
    assert a.dtype == DType.DECIMAL128 and b.dtype == DType.DECIMAL128
    n = a.size
    dev = a.device
    out = torch.zeros(n * 2, dtype=torch.int64, device=dev)
    validity = make_validity(n, dev)
    err = torch.full((1,), 2**63 - 1, dtype=torch.int64, device=dev) if ansi \
        else None
    return n, dev, out, validity, err


def _check(err, ansi):
    if ansi and err is not None:
        row = int(err.item())
        if row != 2**63 - 1:
            raise DecimalOverflowError(row)


def _vp(c):
    return c.validity.data_ptr() if c.validity is not None else 0


def multiply_128(a: Column, b: Column, out_scale: int,
                 out_precision: int = MAX_PRECISION,
                 ansi: bool = False) -> Column:
    g = _native.gpu()
    n, dev, out, validity, err = _prep(a, b, ansi)
    g.dec128_mul(a.data.data_ptr(), _vp(a), b.data.data_ptr(), _vp(b), n,
                 a.scale + b.scale, out_scale, out_precision, out.data_ptr(),
                 validity.data_ptr(), err.data_ptr() if err is not None else 0,
                 _native.current_stream())
    _check(err, ansi)
    return Column(DType.DECIMAL128, n, out, validity, scale=out_scale,
                  null_count=None)


def divide_128(a: Column, b: Column, out_scale: int,
               out_precision: int = MAX_PRECISION, ansi: bool = False,
               integer_divide: bool = False) -> Column:
    g = _native.gpu()
    n, dev, out, validity, err = _prep(a, b, ansi)
    g.dec128_div(a.data.data_ptr(), _vp(a), b.data.data_ptr(), _vp(b), n,
                 a.scale, b.scale, out_scale, out_precision,
                 1 if integer_divide else 0, 0, out.data_ptr(),
                 validity.data_ptr(), err.data_ptr() if err is not None else 0,
                 _native.current_stream())
    _check(err, ansi)
    out_dt = DType.DECIMAL128
    return Column(out_dt, n, out, validity,
                  scale=0 if integer_divide else out_scale, null_count=None)


def integer_divide_128(a: Column, b: Column, ansi: bool = False) -> Column:
    """reference DecimalUtils.integerDivide128 (DIV -> LONG-ranged result)."""
    return divide_128(a, b, 0, MAX_PRECISION, ansi, integer_divide=True)


def remainder_128(a: Column, b: Column, ansi: bool = False) -> Column:
    g = _native.gpu()
    n, dev, out, validity, err = _prep(a, b, ansi)
    out_scale = max(a.scale, b.scale)
    g.dec128_div(a.data.data_ptr(), _vp(a), b.data.data_ptr(), _vp(b), n,
                 out_scale - a.scale, out_scale - b.scale, out_scale,
                 MAX_PRECISION, 0, 1, out.data_ptr(), validity.data_ptr(),
                 err.data_ptr() if err is not None else 0,
                 _native.current_stream())
    _check(err, ansi)
    return Column(DType.DECIMAL128, n, out, validity, scale=out_scale,
                  null_count=None)


def _addsub(a, b, sub, out_precision, ansi):
    g = _native.gpu()
    n, dev, out, validity, err = _prep(a, b, ansi)
    out_scale = max(a.scale, b.scale)
    g.dec128_addsub(a.data.data_ptr(), _vp(a), b.data.data_ptr(), _vp(b), n,
                    out_scale - a.scale, out_scale - b.scale, 1 if sub else 0,
                    out_precision, out.data_ptr(), validity.data_ptr(),
                    err.data_ptr() if err is not None else 0,
                    _native.current_stream())
    _check(err, ansi)
    return Column(DType.DECIMAL128, n, out, validity, scale=out_scale,
                  null_count=None)


def add_128(a: Column, b: Column, out_precision: int = MAX_PRECISION,
            ansi: bool = False) -> Column:
    return _addsub(a, b, False, out_precision, ansi)


def subtract_128(a: Column, b: Column, out_precision: int = MAX_PRECISION,
                 ansi: bool = False) -> Column:
    return _addsub(a, b, True, out_precision, ansi)
