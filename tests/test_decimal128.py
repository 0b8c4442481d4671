"""DECIMAL128 arithmetic vs Python Decimal oracle."""
import random
from decimal import ROUND_HALF_UP, Decimal, localcontext

import pytest
import torch

from spark_rapids_jni_amd.columnar import Column, DType

random.seed(53)


def _mk(vals, scale, device="cuda"):
    """vals: python Decimals-as-unscaled-ints (None = null)."""
    n = len(vals)
    words = []
    valid = [v is not None for v in vals]
    for v in vals:
        u = (v if v is not None else 0) & (2**128 - 1)
        words.append(u & (2**64 - 1))
        words.append(u >> 64)
    # int64 two's complement words
    words = [w - 2**64 if w >= 2**63 else w for w in words]
    data = torch.tensor(words, dtype=torch.int64, device=device)
    from spark_rapids_jni_amd.columnar import validity_from_bools
    vmask = validity_from_bools(valid, device) if not all(valid) else None
    return Column(DType.DECIMAL128, n, data, vmask, scale=scale,
                  null_count=None)


def _vals(col):
    words = col.data.cpu().tolist()
    out = []
    for i in range(col.size):
        lo = words[2 * i] & (2**64 - 1)
        hi = words[2 * i + 1] & (2**64 - 1)
        u = (hi << 64) | lo
        if u >= 2**127:
            u -= 2**128
        out.append(u if col.is_valid_host(i) else None)
    return out


def _oracle_mul(x, s1, y, s2, out_scale, out_precision):
    if x is None or y is None:
        return None
    with localcontext() as ctx:
        ctx.prec = 80
        d = (Decimal(x).scaleb(-s1) * Decimal(y).scaleb(-s2)).quantize(
            Decimal(1).scaleb(-out_scale), rounding=ROUND_HALF_UP)
        u = int(d.scaleb(out_scale))
    if abs(u) >= 10**out_precision or abs(u) >= 2**127:
        return None
    return u


@pytest.mark.gpu
def test_dec128_multiply():
    from spark_rapids_jni_amd.ops.decimal import multiply_128
    s1, s2, out_scale, prec = 3, 2, 4, 38
    xs = [0, 1, -1, 12345, -99999, 10**30, -(10**30), None,
          123456789012345678901234567, 5]
    ys = [7, -7, 123, 10**9, 10**9, 10**8, 10**8, 3, 100, None]
    a, b = _mk(xs, s1), _mk(ys, s2)
    got = _vals(multiply_128(a, b, out_scale, prec))
    for i, (x, y) in enumerate(zip(xs, ys)):
        exp = _oracle_mul(x, s1, y, s2, out_scale, prec)
        assert got[i] == exp, f"row {i}: {x}*{y}: {got[i]} != {exp}"


@pytest.mark.gpu
def test_dec128_multiply_fuzz():
    from spark_rapids_jni_amd.ops.decimal import multiply_128
    s1, s2, out_scale = 5, 4, 6
    xs = [random.randint(-10**25, 10**25) for _ in range(500)]
    ys = [random.randint(-10**10, 10**10) for _ in range(500)]
    got = _vals(multiply_128(_mk(xs, s1), _mk(ys, s2), out_scale))
    for i, (x, y) in enumerate(zip(xs, ys)):
        exp = _oracle_mul(x, s1, y, s2, out_scale, 38)
        assert got[i] == exp, f"row {i}"


@pytest.mark.gpu
def test_dec128_divide():
    from spark_rapids_jni_amd.ops.decimal import divide_128
    s1, s2, out_scale = 2, 3, 6
    xs = [100, 1, -1000, 10**20, 7, None, 5]
    ys = [3000, 7000, 9000, 11, 2000, 1000, 0]
    got = _vals(divide_128(_mk(xs, s1), _mk(ys, s2), out_scale))
    for i, (x, y) in enumerate(zip(xs, ys)):
        if x is None or y is None or y == 0:
            assert got[i] is None, i
            continue
        with localcontext() as ctx:
            ctx.prec = 80
            d = (Decimal(x).scaleb(-s1) / Decimal(y).scaleb(-s2)).quantize(
                Decimal(1).scaleb(-out_scale), rounding=ROUND_HALF_UP)
            exp = int(d.scaleb(out_scale))
        assert got[i] == exp, f"row {i}: {got[i]} != {exp}"


@pytest.mark.gpu
def test_dec128_divide_fuzz():
    from spark_rapids_jni_amd.ops.decimal import divide_128
    s1, s2, out_scale = 4, 2, 8
    xs = [random.randint(-10**20, 10**20) for _ in range(300)]
    ys = [random.choice([random.randint(1, 10**12),
                         -random.randint(1, 10**12)]) for _ in range(300)]
    got = _vals(divide_128(_mk(xs, s1), _mk(ys, s2), out_scale))
    for i, (x, y) in enumerate(zip(xs, ys)):
        with localcontext() as ctx:
            ctx.prec = 80
            d = (Decimal(x).scaleb(-s1) / Decimal(y).scaleb(-s2)).quantize(
                Decimal(1).scaleb(-out_scale), rounding=ROUND_HALF_UP)
            exp = int(d.scaleb(out_scale))
        if abs(exp) >= 2**127:
            exp = None
        assert got[i] == exp, f"row {i}: {xs[i]}/{ys[i]}"


@pytest.mark.gpu
def test_dec128_add_sub_intdiv_rem():
    from spark_rapids_jni_amd.ops.decimal import (add_128, integer_divide_128,
                                                  remainder_128, subtract_128)
    s1, s2 = 2, 4
    xs = [100, -100, 10**30, None, 5]
    ys = [12345, 999, 10**32, 1, 3]
    a, b = _mk(xs, s1), _mk(ys, s2)
    add = _vals(add_128(a, b))
    sub = _vals(subtract_128(a, b))
    for i, (x, y) in enumerate(zip(xs, ys)):
        if x is None or y is None:
            assert add[i] is None and sub[i] is None
            continue
        assert add[i] == x * 100 + y, i
        assert sub[i] == x * 100 - y, i
    # integer divide: (1.00 / 1.2345) -> 0; (10^30/100=10^28 / 10^28) etc.
    idiv = _vals(integer_divide_128(a, b))
    for i, (x, y) in enumerate(zip(xs, ys)):
        if x is None or y is None:
            assert idiv[i] is None
            continue
        exp = abs(x * 10**s2) // abs(y * 10**s1)
        if (x < 0) != (y < 0):
            exp = -exp
        assert idiv[i] == exp, f"row {i}: {idiv[i]} != {exp}"
    rem = _vals(remainder_128(a, b))
    for i, (x, y) in enumerate(zip(xs, ys)):
        if x is None or y is None:
            continue
        xx, yy = x * 100, y  # aligned to scale 4
        exp = abs(xx) % abs(yy)
        if xx < 0:
            exp = -exp
        assert rem[i] == exp, f"row {i}"


@pytest.mark.gpu
def test_dec128_overflow_ansi():
    from spark_rapids_jni_amd.ops.decimal import (DecimalOverflowError,
                                                  multiply_128)
    big = 10**37
    a, b = _mk([big, 2], 0), _mk([big, 3], 0)
    got = _vals(multiply_128(a, b, 0))
    assert got[0] is None and got[1] == 6
    with pytest.raises(DecimalOverflowError) as ei:
        multiply_128(a, b, 0, ansi=True)
    assert ei.value.row_with_error == 0
