"""Parquet footer (CPU) + GPU page decode tests, pyarrow as writer/oracle."""
import os
import random

import pytest

pa = pytest.importorskip("pyarrow")
import pyarrow.parquet as pq  # noqa: E402

from spark_rapids_jni_amd import parquet as srj_pq  # noqa: E402

random.seed(83)


def _make_table(n=1000, with_nulls=True):
    ints = [None if with_nulls and i % 7 == 3 else
            random.randint(-10**9, 10**9) for i in range(n)]
    longs = [None if with_nulls and i % 11 == 5 else
             random.randint(-10**17, 10**17) for i in range(n)]
    floats = [None if with_nulls and i % 5 == 1 else
              random.random() * 1e6 for i in range(n)]
    strs = [None if with_nulls and i % 13 == 2 else
            f"val{i % 100}-{random.randint(0, 999)}" for i in range(n)]
    return pa.table({
        "i": pa.array(ints, type=pa.int32()),
        "l": pa.array(longs, type=pa.int64()),
        "d": pa.array(floats, type=pa.float64()),
        "s": pa.array(strs, type=pa.string()),
    }), {"i": ints, "l": longs, "d": floats, "s": strs}


def test_footer_parse_and_prune(tmp_path):
    t, _ = _make_table(500)
    p = str(tmp_path / "a.parquet")
    pq.write_table(t, p, compression="NONE", row_group_size=200)
    f = srj_pq.read_footer(p)
    assert f.num_rows == 500
    assert [s.name for s in f.schema] == ["i", "l", "d", "s"]
    assert len(f.row_groups) == 3
    assert f.row_groups[0].num_rows == 200
    pruned = f.prune(["l", "s"])
    assert [s.name for s in pruned.schema] == ["l", "s"]
    assert all(len(rg.columns) == 2 for rg in pruned.row_groups)
    # row-group split filtering keeps disjoint coverage
    import os as _os
    flen = _os.path.getsize(p)
    half1 = f.filter_row_groups(0, flen // 2)
    half2 = f.filter_row_groups(flen // 2, flen)
    assert half1.num_rows + half2.num_rows == 500


def _check(table_path, oracle, columns=None, device="cuda"):
    got = srj_pq.read_table(table_path, columns=columns, device=device)
    names = columns if columns else list(oracle.keys())
    for c, name in zip(got.columns, names):
        gv = c.to_pylist()
        exp = oracle[name]
        if name == "d":
            for a, b in zip(gv, exp):
                assert (a is None) == (b is None)
                if a is not None:
                    assert abs(a - b) < 1e-9 * max(1, abs(b))
        else:
            assert gv == exp, name


@pytest.mark.gpu
@pytest.mark.parametrize("dict_enc", [False, True], ids=["plain", "dict"])
@pytest.mark.parametrize("pagever", ["1.0", "2.0"])
def test_decode_roundtrip(tmp_path, dict_enc, pagever):
    t, oracle = _make_table(2000)
    p = str(tmp_path / "t.parquet")
    pq.write_table(t, p, compression="NONE", use_dictionary=dict_enc,
                   data_page_version=pagever, row_group_size=700)
    _check(p, oracle)


@pytest.mark.gpu
def test_decode_no_nulls_required(tmp_path):
    t, oracle = _make_table(800, with_nulls=False)
    schema = pa.schema([pa.field(n, t.schema.field(n).type, nullable=False)
                        for n in t.schema.names])
    t2 = t.cast(schema)
    p = str(tmp_path / "req.parquet")
    pq.write_table(t2, p, compression="NONE", use_dictionary=False)
    _check(p, oracle)


@pytest.mark.gpu
def test_decode_snappy(tmp_path):
    t, oracle = _make_table(1500)
    p = str(tmp_path / "sn.parquet")
    pq.write_table(t, p, compression="SNAPPY", use_dictionary=True)
    _check(p, oracle)


@pytest.mark.gpu
def test_decode_column_pruning(tmp_path):
    t, oracle = _make_table(600)
    p = str(tmp_path / "pr.parquet")
    pq.write_table(t, p, compression="NONE")
    _check(p, oracle, columns=["l", "s"])


@pytest.mark.gpu
def test_decode_small_pages_many_groups(tmp_path):
    t, oracle = _make_table(5000)
    p = str(tmp_path / "mp.parquet")
    pq.write_table(t, p, compression="NONE", use_dictionary=True,
                   row_group_size=512, data_page_size=512)
    _check(p, oracle)


def test_thrift_typed_roundtrip(tmp_path):
    """typed parse -> write reproduces the original footer byte-for-byte."""
    from spark_rapids_jni_amd import _native
    t, _ = _make_table(300)
    p = str(tmp_path / "rt.parquet")
    pq.write_table(t, p, compression="NONE", row_group_size=100)
    raw = open(p, "rb").read()
    import struct as st
    flen = st.unpack("<I", raw[-8:-4])[0]
    footer = raw[-8 - flen:-8]
    host = _native.host()
    tree, end = host.thrift_parse_typed(footer, 0)
    assert end == len(footer)
    out = host.thrift_write(tree)
    assert out == footer


def test_footer_rewrite_prune_columns(tmp_path):
    """pyarrow can read a file whose footer we pruned to a column subset."""
    t, vals = _make_table(400)
    src = str(tmp_path / "src.parquet")
    dst = str(tmp_path / "dst.parquet")
    pq.write_table(t, src, compression="NONE", row_group_size=150)
    srj_pq.rewrite_parquet_file(src, dst, keep_columns=["l", "s"])
    t2 = pq.read_table(dst)
    assert t2.column_names == ["l", "s"]
    assert t2.column("l").to_pylist() == vals["l"]
    assert t2.column("s").to_pylist() == vals["s"]
    # our own reader agrees
    f = srj_pq.read_footer(dst)
    assert [s.name for s in f.schema] == ["l", "s"]
    assert all(len(rg.columns) == 2 for rg in f.row_groups)


def test_footer_rewrite_split_filter(tmp_path):
    """row-group filtering by split range keeps only in-range groups."""
    t, _ = _make_table(600, with_nulls=False)
    src = str(tmp_path / "s2.parquet")
    dst = str(tmp_path / "d2.parquet")
    pq.write_table(t, src, compression="NONE", row_group_size=200)
    full = srj_pq.read_footer(src)
    assert len(full.row_groups) == 3
    # split covering only the first row group's midpoint
    rg0_start = min(c.start_offset for c in full.row_groups[0].columns)
    rg0_mid = rg0_start + full.row_groups[0].total_byte_size // 2
    srj_pq.rewrite_parquet_file(src, dst, part_offset=0,
                                part_length=rg0_mid + 1)
    f = srj_pq.read_footer(dst)
    assert len(f.row_groups) == 1
    assert f.num_rows == 200
    t2 = pq.read_table(dst)
    assert t2.num_rows == 200


@pytest.mark.gpu
def test_decode_snappy_large_plain(tmp_path):
    """Large snappy pages without dictionary force long literal/match streams
    through the device decompressor (one wave per page)."""
    t, oracle = _make_table(200_000)
    p = str(tmp_path / "snl.parquet")
    pq.write_table(t, p, compression="SNAPPY", use_dictionary=False,
                   row_group_size=60_000, data_page_size=256 * 1024)
    _check(p, oracle)


@pytest.mark.gpu
def test_decode_snappy_repetitive(tmp_path):
    """Highly repetitive data creates overlapped back-references (offset <
    length) — the serial-lane fallback path of the snappy kernel."""
    n = 50_000
    ints = [7 for _ in range(n)]
    strs = [("ab" * 30) for _ in range(n)]
    t = pa.table({"i": pa.array(ints, type=pa.int64()),
                  "s": pa.array(strs, type=pa.string())})
    p = str(tmp_path / "snr.parquet")
    pq.write_table(t, p, compression="SNAPPY", use_dictionary=False)
    got = srj_pq.read_table(p, device="cuda")
    assert got.columns[0].to_pylist() == ints
    assert got.columns[1].to_pylist() == strs


def _list_oracle(n, with_nulls=True, strings=False):
    rows = []
    for i in range(n):
        if with_nulls and i % 13 == 4:
            rows.append(None)
        elif i % 7 == 2:
            rows.append([])
        else:
            ln = i % 5
            if strings:
                rows.append([None if with_nulls and (i + k) % 11 == 5
                             else f"v{i}-{k}" for k in range(ln)])
            else:
                rows.append([None if with_nulls and (i + k) % 11 == 5
                             else i * 10 + k for k in range(ln)])
    return rows


@pytest.mark.gpu
@pytest.mark.parametrize("pagever", ["1.0", "2.0"])
def test_decode_list_int64(tmp_path, pagever):
    rows = _list_oracle(4000)
    t = pa.table({"l": pa.array(rows, type=pa.list_(pa.int64()))})
    p = str(tmp_path / f"l{pagever}.parquet")
    pq.write_table(t, p, compression="NONE", use_dictionary=False,
                   data_page_version=pagever, row_group_size=1500)
    got = srj_pq.read_table(p, device="cuda")
    assert got.columns[0].dtype.name == "LIST"
    assert got.columns[0].to_pylist() == rows


@pytest.mark.gpu
def test_decode_list_strings_dict(tmp_path):
    rows = _list_oracle(3000, strings=True)
    t = pa.table({"s": pa.array(rows, type=pa.list_(pa.string()))})
    p = str(tmp_path / "ls.parquet")
    pq.write_table(t, p, compression="NONE", use_dictionary=True,
                   row_group_size=1000)
    got = srj_pq.read_table(p, device="cuda")
    assert got.columns[0].to_pylist() == rows


@pytest.mark.gpu
def test_decode_list_mixed_with_flat(tmp_path):
    rows = _list_oracle(2000)
    ints = list(range(2000))
    t = pa.table({"i": pa.array(ints, type=pa.int64()),
                  "l": pa.array(rows, type=pa.list_(pa.int64()))})
    p = str(tmp_path / "lm.parquet")
    pq.write_table(t, p, compression="SNAPPY", row_group_size=700)
    got = srj_pq.read_table(p, device="cuda")
    assert got.columns[0].to_pylist() == ints
    assert got.columns[1].to_pylist() == rows


@pytest.mark.gpu
def test_decode_decimal_int_backed(tmp_path):
    from decimal import Decimal
    rows32 = [None if i % 9 == 1 else Decimal(i * 7 - 500).scaleb(-2)
              for i in range(2000)]
    rows64 = [None if i % 5 == 2 else Decimal(i * 977 - 10**6).scaleb(-4)
              for i in range(2000)]
    t = pa.table({"d32": pa.array(rows32, type=pa.decimal128(7, 2)),
                  "d64": pa.array(rows64, type=pa.decimal128(15, 4))})
    p = str(tmp_path / "dec.parquet")
    pq.write_table(t, p, compression="NONE", store_decimal_as_integer=True)
    got = srj_pq.read_table(p, device="cuda")
    d32, d64 = got.columns
    assert d32.dtype.name == "DECIMAL32" and d32.scale == 2
    assert d64.dtype.name == "DECIMAL64" and d64.scale == 4
    g32 = d32.to_pylist()
    g64 = d64.to_pylist()
    for i in range(2000):
        exp = rows32[i]
        assert g32[i] == (None if exp is None else int(exp.scaleb(2))), i
        exp = rows64[i]
        assert g64[i] == (None if exp is None else int(exp.scaleb(4))), i


def test_thrift_typed_fuzz_roundtrip():
    """Random typed trees -> compact-protocol write -> typed parse -> equal."""
    import random as rnd
    from spark_rapids_jni_amd import _native
    host = _native.host()
    r = rnd.Random(271)

    def rand_value(ty, depth):
        if ty == 1:
            return r.random() < 0.5
        if ty == 3:
            return r.randint(-128, 127)
        if ty in (4, 5, 6):
            bits = {4: 15, 5: 31, 6: 63}[ty]
            return r.randint(-(2**bits), 2**bits - 1)
        if ty == 7:
            return r.uniform(-1e18, 1e18)
        if ty == 8:
            return bytes(r.randrange(256) for _ in range(r.randrange(20)))
        if ty == 9:
            ety = r.choice([3, 5, 6, 8] + ([12] if depth < 2 else []))
            return (ety, [rand_value(ety, depth + 1)
                          for _ in range(r.randrange(18))])
        if ty == 11:
            kt, vt = r.choice([(5, 8), (8, 6)])
            return (kt, vt, [(rand_value(kt, depth + 1),
                              rand_value(vt, depth + 1))
                             for _ in range(r.randrange(5))])
        if ty == 12:
            return rand_struct(depth + 1)
        raise AssertionError(ty)

    def rand_struct(depth=0):
        d = {}
        fid = 0
        for _ in range(r.randrange(1, 8)):
            fid += r.randint(1, 40)
            ty = r.choice([1, 3, 4, 5, 6, 7, 8, 9, 11] +
                          ([12] if depth < 3 else []))
            d[fid] = (ty, rand_value(ty, depth))
        return d

    def norm(d):
        # dict -> sorted comparable structure (floats exact: same bits)
        out = {}
        for k, (ty, v) in d.items():
            if ty == 12:
                v = norm(v)
            elif ty == 9:
                v = (v[0], [norm(x) if v[0] == 12 else x for x in v[1]])
            elif ty == 11:
                # empty maps lose their kv types on the wire (no kv byte)
                v = (0, 0, []) if not v[2] else (v[0], v[1], list(v[2]))
            out[k] = (ty, v)
        return out

    for _ in range(60):
        tree = rand_struct()
        raw = host.thrift_write(tree)
        back, end = host.thrift_parse_typed(raw, 0)
        assert end == len(raw)
        assert norm(back) == norm(tree)


@pytest.mark.gpu
def test_decode_struct_column(tmp_path):
    rows = [None if i % 13 == 4 else
            {"x": None if i % 7 == 2 else i * 3,
             "y": None if i % 5 == 1 else f"s{i}"}
            for i in range(3000)]
    from decimal import Decimal
    for i, r in enumerate(rows):
        if r is not None:
            r["z"] = None if i % 11 == 6 else Decimal(i - 1500).scaleb(-2)
    t = pa.table({"st": pa.array(rows, type=pa.struct(
        [("x", pa.int64()), ("y", pa.string()),
         ("z", pa.decimal128(20, 2))])),
        "i": pa.array(list(range(3000)), type=pa.int64())})
    p = str(tmp_path / "st.parquet")
    pq.write_table(t, p, compression="NONE", row_group_size=1100)
    got = srj_pq.read_table(p, device="cuda")
    st = got.columns[0]
    assert st.dtype.name == "STRUCT"
    xs = st.children[0].to_pylist()
    ys = st.children[1].to_pylist()
    zc = st.children[2]
    zw = zc.data.cpu().tolist()
    assert zc.dtype.name == "DECIMAL128" and zc.scale == 2
    for i, r in enumerate(rows):
        if r is None:
            assert not st.is_valid_host(i), i
        else:
            assert st.is_valid_host(i), i
            assert xs[i] == r["x"] and ys[i] == r["y"], i
            if r["z"] is None:
                assert not zc.is_valid_host(i), i
            else:
                u = (zw[2 * i] & (2**64 - 1)) | \
                    ((zw[2 * i + 1] & (2**64 - 1)) << 64)
                if u >= 2**127:
                    u -= 2**128
                assert u == int(r["z"].scaleb(2)), i
    assert got.columns[1].to_pylist() == list(range(3000))


@pytest.mark.gpu
def test_decode_decimal_flba(tmp_path):
    """FIXED_LEN_BYTE_ARRAY decimals (pyarrow's default layout; Spark uses it
    for precision > 18): big-endian N-byte values -> DECIMAL128."""
    from decimal import Decimal
    rows_small = [None if i % 9 == 1 else Decimal(i * 7 - 5000).scaleb(-2)
                  for i in range(3000)]
    rows_big = [None if i % 5 == 2 else
                Decimal((i - 1500) * 10**20 + i).scaleb(-4)
                for i in range(3000)]
    t = pa.table({"d7": pa.array(rows_small, type=pa.decimal128(7, 2)),
                  "d38": pa.array(rows_big, type=pa.decimal128(38, 4))})
    p = str(tmp_path / "flba.parquet")
    pq.write_table(t, p, compression="NONE", row_group_size=1300)
    got = srj_pq.read_table(p, device="cuda")
    for col, rows, scale in ((got.columns[0], rows_small, 2),
                             (got.columns[1], rows_big, 4)):
        assert col.dtype.name == "DECIMAL128" and col.scale == scale
        words = col.data.cpu().tolist()
        for i, exp in enumerate(rows):
            if exp is None:
                assert not col.is_valid_host(i), i
                continue
            lo = words[2 * i] & (2**64 - 1)
            hi = words[2 * i + 1] & (2**64 - 1)
            u = (hi << 64) | lo
            if u >= 2**127:
                u -= 2**128
            assert u == int(exp.scaleb(scale)), (i, u, exp)


@pytest.mark.parametrize("codec", ["ZSTD", "GZIP", "LZ4"])
@pytest.mark.gpu
def test_decode_host_codecs(tmp_path, codec):
    """ZSTD/GZIP/LZ4 pages decompress on the host thread pool before the
    device decode kernels run."""
    t, oracle = _make_table(3000)
    p = str(tmp_path / f"{codec}.parquet")
    pq.write_table(t, p, compression=codec, use_dictionary=True,
                   row_group_size=900)
    _check(p, oracle)


@pytest.mark.parametrize("codec", ["ZSTD", "GZIP"])
@pytest.mark.parametrize("pagever", ["1.0", "2.0"])
@pytest.mark.gpu
def test_decode_host_codecs_gpu(tmp_path, codec, pagever):
    t, oracle = _make_table(3000)
    p = str(tmp_path / f"{codec}_{pagever}.parquet")
    pq.write_table(t, p, compression=codec, use_dictionary=True,
                   data_page_version=pagever, row_group_size=900)
    _check(p, oracle)


@pytest.mark.gpu
def test_decode_snappy_v2_device(tmp_path):
    """v2 pages keep their uncompressed level prefix; the snappy body now
    decompresses on device (previously host)."""
    t, oracle = _make_table(4000)
    p = str(tmp_path / "snv2.parquet")
    pq.write_table(t, p, compression="SNAPPY", use_dictionary=True,
                   data_page_version="2.0", row_group_size=1100)
    _check(p, oracle)


@pytest.mark.gpu
def test_decode_zstd_list(tmp_path):
    vals = [None if i % 9 == 4 else
            [random.randint(0, 100) for _ in range(i % 5)]
            for i in range(1200)]
    t = pa.table({"lst": pa.array(vals, type=pa.list_(pa.int64()))})
    p = str(tmp_path / "zl.parquet")
    pq.write_table(t, p, compression="ZSTD", row_group_size=500)
    got = srj_pq.read_table(p, device="cuda")
    assert got.columns[0].to_pylist() == vals


@pytest.mark.gpu
def test_decode_struct_of_struct(tmp_path):
    n = 2500
    vals = []
    for i in range(n):
        if i % 11 == 3:
            vals.append(None)
        elif i % 7 == 2:
            vals.append({"a": i, "inner": None,
                         "s": None if i % 5 == 0 else f"x{i}"})
        else:
            vals.append({"a": i,
                         "inner": {"b": i * 2,
                                   "c": None if i % 3 == 0 else i * 1.5},
                         "s": f"x{i}"})
    typ = pa.struct([("a", pa.int64()),
                     ("inner", pa.struct([("b", pa.int64()),
                                          ("c", pa.float64())])),
                     ("s", pa.string())])
    t = pa.table({"st": pa.array(vals, type=typ)})
    p = str(tmp_path / "sos.parquet")
    pq.write_table(t, p, compression="SNAPPY", row_group_size=800)
    got = srj_pq.read_table(p, device="cuda").columns[0].to_pylist()
    for i, (g, x) in enumerate(zip(got, vals)):
        if x is None:
            assert g is None, i
            continue
        assert g is not None, i
        a, inner, s = g
        assert a == x["a"] and s == x["s"], i
        if x["inner"] is None:
            assert inner is None, i
        else:
            b, c = inner
            assert b == x["inner"]["b"], i
            if x["inner"]["c"] is None:
                assert c is None, i
            else:
                assert abs(c - x["inner"]["c"]) < 1e-9, i


@pytest.mark.gpu
@pytest.mark.parametrize("codec", ["NONE", "ZSTD"])
def test_decode_map_column(tmp_path, codec):
    import random as _r
    _r.seed(4)
    n = 2000
    vals = []
    for i in range(n):
        if i % 13 == 5:
            vals.append(None)
        else:
            m = [(f"k{j}", None if (i + j) % 7 == 0 else i * 10 + j)
                 for j in range(i % 4)]
            vals.append(m)
    t = pa.table({"m": pa.array(vals, type=pa.map_(pa.string(),
                                                   pa.int64()))})
    p = str(tmp_path / f"map_{codec}.parquet")
    pq.write_table(t, p, compression=codec, row_group_size=700)
    got = srj_pq.read_table(p, device="cuda").columns[0].to_pylist()
    for i, (g, x) in enumerate(zip(got, vals)):
        if x is None:
            assert g is None, i
        else:
            assert g == [tuple(kv) for kv in x], (i, g, x)


def _lstruct_rows(n, seed):
    rng = random.Random(seed)
    rows = []
    for i in range(n):
        if i % 17 == 3:
            rows.append(None)
            continue
        lst = []
        for j in range(rng.randrange(0, 4)):
            if (i + j) % 11 == 5:
                lst.append(None)  # null struct element
            else:
                lst.append({"a": None if (i * 3 + j) % 7 == 2 else i + j,
                            "b": None if j % 5 == 1 else f"s{i % 40}_{j}"})
        rows.append(lst)
    return rows


def test_footer_nested_list_schema(tmp_path):
    """Schema parse (CPU): LIST<STRUCT> and LIST<LIST> shapes and levels."""
    typ_ls = pa.list_(pa.struct([("a", pa.int64()), ("b", pa.string())]))
    typ_ll = pa.list_(pa.list_(pa.int64()))
    t = pa.table({"ls": pa.array([[{"a": 1, "b": "x"}]], type=typ_ls),
                  "ll": pa.array([[[1, 2], []]], type=typ_ll)})
    p = str(tmp_path / "sch.parquet")
    pq.write_table(t, p, compression="NONE")
    f = srj_pq.read_footer(p)
    ls, ll = f.schema
    assert ls.is_list and ls.element.is_struct and ls.max_rep == 1
    # optional list + repeated + (a,b optional leaves under the elem struct)
    leaves = srj_pq._flatten_struct_leaves(ls.element)
    assert [lf.max_def for lf in leaves] == [ls.max_def] * 2
    assert ll.is_list and ll.element.is_list and ll.max_rep == 2
    assert ll.element.element.physical_type == srj_pq.T_INT64


@pytest.mark.parametrize("pagever", ["1.0", "2.0"])
@pytest.mark.gpu
def test_decode_list_of_struct(tmp_path, pagever):
    rows = _lstruct_rows(2500, 7)
    typ = pa.list_(pa.struct([("a", pa.int64()), ("b", pa.string())]))
    t = pa.table({"ls": pa.array(rows, type=typ)})
    p = str(tmp_path / "ls.parquet")
    pq.write_table(t, p, compression="NONE", data_page_version=pagever,
                   row_group_size=900)
    col = srj_pq.read_table(p, device="cuda").columns[0]
    exp = [None if r is None else
           [None if e is None else (e["a"], e["b"]) for e in r] for r in rows]
    assert col.to_pylist() == exp


@pytest.mark.gpu
def test_decode_list_of_nested_struct(tmp_path):
    """struct-in-struct inside a list element; snappy + dictionary pages."""
    rows = []
    for i, base in enumerate(_lstruct_rows(1500, 13)):
        if base is None or i % 3 == 0:
            rows.append(base)
            continue
        rows.append([None if e is None else
                     {"inner": e, "c": (i * 7) % 23} for e in base])
    typ = pa.list_(pa.struct([
        ("inner", pa.struct([("a", pa.int64()), ("b", pa.string())])),
        ("c", pa.int32())]))
    fixed = [None if r is None else
             [e if e is None or "inner" in e else {"inner": e, "c": None}
              for e in r] for r in rows]
    t = pa.table({"ls": pa.array(fixed, type=typ)})
    p = str(tmp_path / "lns.parquet")
    pq.write_table(t, p, compression="SNAPPY", use_dictionary=True,
                   row_group_size=700)
    col = srj_pq.read_table(p, device="cuda").columns[0]

    def conv(e):
        if e is None:
            return None
        inner = e["inner"]
        return ((None if inner is None else (inner["a"], inner["b"])),
                e["c"])
    exp = [None if r is None else [conv(e) for e in r] for r in fixed]
    assert col.to_pylist() == exp


@pytest.mark.parametrize("pagever", ["1.0", "2.0"])
@pytest.mark.gpu
def test_decode_list_of_list(tmp_path, pagever):
    rng = random.Random(11)
    rows = []
    for i in range(2500):
        if i % 19 == 2:
            rows.append(None)
            continue
        outer = []
        for j in range(rng.randrange(0, 4)):
            if (i + j) % 13 == 6:
                outer.append(None)
                continue
            outer.append([None if (i * 7 + j + k) % 5 == 1 else i + j * 10 + k
                          for k in range(rng.randrange(0, 3))])
        rows.append(outer)
    t = pa.table({"ll": pa.array(rows, type=pa.list_(pa.list_(pa.int64())))})
    p = str(tmp_path / "ll.parquet")
    pq.write_table(t, p, compression="NONE", data_page_version=pagever,
                   row_group_size=800)
    col = srj_pq.read_table(p, device="cuda").columns[0]
    assert col.to_pylist() == rows


@pytest.mark.gpu
def test_decode_list_of_list_strings(tmp_path):
    """3-deep def chain + dictionary-encoded string leaves."""
    rng = random.Random(29)
    vocab = [f"w{v}" for v in range(50)]
    rows = []
    for i in range(2000):
        rows.append(None if i % 23 == 7 else
                    [None if (i + j) % 9 == 4 else
                     [None if (i + j + k) % 6 == 2 else rng.choice(vocab)
                      for k in range(rng.randrange(0, 3))]
                     for j in range(rng.randrange(0, 3))])
    t = pa.table({"ll": pa.array(rows, type=pa.list_(pa.list_(pa.string())))})
    p = str(tmp_path / "lls.parquet")
    pq.write_table(t, p, compression="SNAPPY", use_dictionary=True,
                   row_group_size=650)
    col = srj_pq.read_table(p, device="cuda").columns[0]
    assert col.to_pylist() == rows


@pytest.mark.gpu
def test_decode_struct_with_list(tmp_path):
    """LIST nested inside STRUCT (struct validity from the list leaf too)."""
    rng = random.Random(41)
    rows = []
    for i in range(2200):
        if i % 13 == 4:
            rows.append(None)
            continue
        tags = (None if i % 9 == 2 else
                [None if (i + k) % 7 == 3 else (i * 3 + k) % 1000
                 for k in range(rng.randrange(0, 4))])
        rows.append({"x": None if i % 5 == 1 else f"n{i % 60}",
                     "tags": tags})
    typ = pa.struct([("x", pa.string()), ("tags", pa.list_(pa.int64()))])
    t = pa.table({"st": pa.array(rows, type=typ)})
    p = str(tmp_path / "swl.parquet")
    pq.write_table(t, p, compression="NONE", row_group_size=750)
    col = srj_pq.read_table(p, device="cuda").columns[0]
    exp = [None if r is None else (r["x"], r["tags"]) for r in rows]
    assert col.to_pylist() == exp


@pytest.mark.gpu
def test_decode_struct_with_list_first_leaf(tmp_path):
    """The LIST is the struct's FIRST leaf — struct validity must come from
    the list's row-space def levels; inner struct adds a level."""
    rows = []
    for i in range(1500):
        if i % 11 == 3:
            rows.append(None)
            continue
        inner = (None if i % 17 == 5 else
                 {"ll": None if i % 7 == 1 else
                  [[j, None, i % 50] for j in range(i % 3)]})
        rows.append({"inner": inner, "y": i % 100})
    typ = pa.struct([
        ("inner", pa.struct([("ll", pa.list_(pa.list_(pa.int32())))])),
        ("y", pa.int32())])
    t = pa.table({"st": pa.array(rows, type=typ)})
    p = str(tmp_path / "swl2.parquet")
    pq.write_table(t, p, compression="SNAPPY", use_dictionary=True,
                   row_group_size=600)
    col = srj_pq.read_table(p, device="cuda").columns[0]
    exp = [None if r is None else
           ((None if r["inner"] is None else (r["inner"]["ll"],)), r["y"])
           for r in rows]
    assert col.to_pylist() == exp


@pytest.mark.gpu
def test_decode_map_of_list_value(tmp_path):
    """MAP<string, LIST<int>>: null maps, null/empty value lists."""
    rng = random.Random(53)
    rows = []
    for i in range(1800):
        if i % 15 == 6:
            rows.append(None)
            continue
        ents = []
        for j in range(rng.randrange(0, 4)):
            v = (None if (i + j) % 8 == 3 else
                 [None if (i + j + k) % 6 == 1 else i + 10 * j + k
                  for k in range(rng.randrange(0, 3))])
            ents.append((f"k{j}", v))
        rows.append(ents)
    typ = pa.map_(pa.string(), pa.list_(pa.int64()))
    t = pa.table({"m": pa.array(rows, type=typ)})
    p = str(tmp_path / "mlv.parquet")
    pq.write_table(t, p, compression="NONE", row_group_size=700)
    col = srj_pq.read_table(p, device="cuda").columns[0]
    assert col.to_pylist() == rows


@pytest.mark.gpu
def test_decode_map_of_struct_value(tmp_path):
    """MAP<int, STRUCT<a,b>>: null struct values and null struct fields."""
    rng = random.Random(59)
    rows = []
    for i in range(1800):
        if i % 21 == 8:
            rows.append(None)
            continue
        ents = []
        for j in range(rng.randrange(0, 3)):
            v = (None if (i + j) % 7 == 2 else
                 {"a": None if (i * 5 + j) % 9 == 4 else i + j,
                  "b": None if j % 4 == 1 else f"v{(i + j) % 30}"})
            ents.append((i * 10 + j, v))
        rows.append(ents)
    typ = pa.map_(pa.int64(), pa.struct([("a", pa.int64()),
                                         ("b", pa.string())]))
    t = pa.table({"m": pa.array(rows, type=typ)})
    p = str(tmp_path / "msv.parquet")
    pq.write_table(t, p, compression="SNAPPY", use_dictionary=True,
                   row_group_size=650)
    col = srj_pq.read_table(p, device="cuda").columns[0]
    exp = [None if r is None else
           [(k, None if v is None else (v["a"], v["b"])) for k, v in r]
           for r in rows]
    assert col.to_pylist() == exp


@pytest.mark.parametrize("pagever", ["1.0", "2.0"])
@pytest.mark.gpu
def test_decode_delta_binary_packed(tmp_path, pagever):
    """DELTA_BINARY_PACKED int32/int64 (parquet-mr v2 default): multiple
    blocks/miniblocks, negative deltas, wide jumps, nulls, small pages."""
    rng = random.Random(71)
    n = 4000
    longs, ints = [], []
    cur = 0
    for i in range(n):
        cur += rng.randint(-10**12, 10**12) if i % 97 == 5 else rng.randint(-3, 7)
        longs.append(None if i % 11 == 4 else cur)
        ints.append(None if i % 13 == 6 else (cur % 2**31) - 2**30)
    t = pa.table({"l": pa.array(longs, type=pa.int64()),
                  "i": pa.array(ints, type=pa.int32())})
    p = str(tmp_path / "dbp.parquet")
    pq.write_table(t, p, compression="NONE", use_dictionary=False,
                   data_page_version=pagever, data_page_size=2048,
                   row_group_size=1700,
                   column_encoding={"l": "DELTA_BINARY_PACKED",
                                    "i": "DELTA_BINARY_PACKED"})
    got = srj_pq.read_table(p, device="cuda")
    assert got.columns[0].to_pylist() == longs
    assert got.columns[1].to_pylist() == ints


@pytest.mark.parametrize("enc", ["DELTA_LENGTH_BYTE_ARRAY",
                                 "DELTA_BYTE_ARRAY"])
@pytest.mark.gpu
def test_decode_delta_strings(tmp_path, enc):
    rng = random.Random(73)
    n = 3000
    vals = []
    for i in range(n):
        if i % 9 == 3:
            vals.append(None)
        elif i % 17 == 5:
            vals.append("")
        else:
            # shared prefixes exercise DELTA_BYTE_ARRAY's reconstruction
            vals.append(f"prefix_{i % 7}_" + "x" * rng.randrange(0, 20)
                        + str(rng.randrange(1000)))
    t = pa.table({"s": pa.array(vals, type=pa.string())})
    p = str(tmp_path / "ds.parquet")
    pq.write_table(t, p, compression="NONE", use_dictionary=False,
                   data_page_size=4096, row_group_size=1300,
                   column_encoding={"s": enc})
    got = srj_pq.read_table(p, device="cuda")
    assert got.columns[0].to_pylist() == vals


@pytest.mark.gpu
def test_decode_byte_stream_split(tmp_path):
    rng = random.Random(79)
    f32 = [None if i % 7 == 2 else rng.uniform(-1e6, 1e6) for i in range(2500)]
    f64 = [None if i % 5 == 1 else rng.uniform(-1e12, 1e12)
           for i in range(2500)]
    t = pa.table({"f": pa.array(f32, type=pa.float32()),
                  "d": pa.array(f64, type=pa.float64())})
    p = str(tmp_path / "bss.parquet")
    pq.write_table(t, p, compression="NONE", use_dictionary=False,
                   data_page_size=4096, row_group_size=900,
                   column_encoding={"f": "BYTE_STREAM_SPLIT",
                                    "d": "BYTE_STREAM_SPLIT"})
    got = srj_pq.read_table(p, device="cuda")
    import struct as st
    exp32 = [None if v is None else st.unpack("<f", st.pack("<f", v))[0]
             for v in f32]
    for g_, e_ in zip(got.columns[0].to_pylist(), exp32):
        assert (g_ is None) == (e_ is None) and (g_ is None or g_ == e_)
    for g_, e_ in zip(got.columns[1].to_pylist(), f64):
        assert (g_ is None) == (e_ is None) and (g_ is None or g_ == e_)


@pytest.mark.gpu
def test_decode_delta_nested(tmp_path):
    """DELTA encodings inside LIST and STRUCT columns ride the same
    temp-buffer composition as flat pages."""
    rng = random.Random(83)
    ll, ls, st_rows = [], [], []
    cur = 0
    for i in range(2200):
        if i % 13 == 4:
            ll.append(None)
            ls.append(None)
            st_rows.append(None)
            continue
        row = []
        for k in range(rng.randrange(0, 4)):
            cur += rng.randint(-5, 9)
            row.append(None if (i + k) % 7 == 2 else cur)
        ll.append(row)
        ls.append([None if (i + k) % 8 == 3 else
                   f"pre{(i + k) % 5}_" + str(rng.randrange(500))
                   for k in range(rng.randrange(0, 3))])
        st_rows.append({"a": None if i % 9 == 5 else i * 3 - 1000,
                        "b": None if i % 6 == 1 else f"w{(i * 7) % 90}"})
    t = pa.table({
        "ll": pa.array(ll, type=pa.list_(pa.int64())),
        "ls": pa.array(ls, type=pa.list_(pa.string())),
        "st": pa.array(st_rows, type=pa.struct([("a", pa.int64()),
                                                ("b", pa.string())]))})
    p = str(tmp_path / "nd.parquet")
    pq.write_table(t, p, compression="NONE", use_dictionary=False,
                   data_page_size=4096, row_group_size=800,
                   column_encoding={"ll.list.element": "DELTA_BINARY_PACKED",
                                    "ls.list.element": "DELTA_BYTE_ARRAY",
                                    "st.a": "DELTA_BINARY_PACKED",
                                    "st.b": "DELTA_LENGTH_BYTE_ARRAY"})
    got = srj_pq.read_table(p, device="cuda")
    assert got.columns[0].to_pylist() == ll
    assert got.columns[1].to_pylist() == ls
    exp_st = [None if r is None else (r["a"], r["b"]) for r in st_rows]
    assert got.columns[2].to_pylist() == exp_st


@pytest.mark.gpu
def test_decode_list_of_flba_decimal(tmp_path):
    """LIST<DECIMAL128 (FIXED_LEN_BYTE_ARRAY)>: big-endian N-byte decimals
    inside list elements -> 2x int64 words, same kernel as the flat path."""
    from decimal import Decimal
    rng = random.Random(89)
    rows = []
    for i in range(1800):
        if i % 11 == 3:
            rows.append(None)
            continue
        rows.append([None if (i + k) % 7 == 2 else
                     Decimal((i - 900) * 10**19 + k * 37).scaleb(-4)
                     for k in range(rng.randrange(0, 4))])
    t = pa.table({"ld": pa.array(rows, type=pa.list_(pa.decimal128(38, 4)))})
    p = str(tmp_path / "lflba.parquet")
    pq.write_table(t, p, compression="NONE", row_group_size=700)
    col = srj_pq.read_table(p, device="cuda").columns[0]
    child = col.children[0]
    assert child.dtype.name == "DECIMAL128" and child.scale == 4
    exp = [None if r is None else
           [None if v is None else int(v.scaleb(4)) for v in r]
           for r in rows]
    assert col.to_pylist() == exp


@pytest.mark.gpu
def test_decode_struct_with_map_and_liststruct(tmp_path):
    """MAP and LIST<STRUCT> nested INSIDE a struct: chunk-cursor routing +
    ancestor validity from the nested child's row-space def levels."""
    rng = random.Random(97)
    rows = []
    for i in range(1600):
        if i % 13 == 5:
            rows.append(None)
            continue
        m = (None if i % 9 == 2 else
             [(f"k{j}", None if (i + j) % 6 == 1 else i * 2 + j)
              for j in range(rng.randrange(0, 3))])
        ls = (None if i % 7 == 3 else
              [None if (i + j) % 11 == 4 else
               {"x": None if (i * 3 + j) % 5 == 2 else i + j,
                "y": f"s{(i + j) % 40}"}
               for j in range(rng.randrange(0, 3))])
        rows.append({"m": m, "ls": ls, "z": i % 1000})
    typ = pa.struct([
        ("m", pa.map_(pa.string(), pa.int64())),
        ("ls", pa.list_(pa.struct([("x", pa.int64()), ("y", pa.string())]))),
        ("z", pa.int32())])
    t = pa.table({"st": pa.array(rows, type=typ)})
    p = str(tmp_path / "swml.parquet")
    pq.write_table(t, p, compression="SNAPPY", row_group_size=600)
    col = srj_pq.read_table(p, device="cuda").columns[0]

    def conv(r):
        if r is None:
            return None
        ls = (None if r["ls"] is None else
              [None if e is None else (e["x"], e["y"]) for e in r["ls"]])
        m = None if r["m"] is None else [tuple(kv) for kv in r["m"]]
        return (m, ls, r["z"])
    assert col.to_pylist() == [conv(r) for r in rows]


def test_footer_random_nested_schemas(tmp_path):
    """Randomized nested-schema battery (CPU): the schema parser's chunk
    accounting must match the physical column-chunk count for arbitrary
    compositions of lists/structs/maps, and known-unsupported combos must
    raise NotImplementedError (never mis-parse)."""
    rng = random.Random(2024)
    prims = [pa.int64(), pa.int32(), pa.string(), pa.float64()]

    def rand_type(depth):
        r = rng.random()
        if depth >= 3 or r < 0.35:
            return rng.choice(prims)
        if r < 0.55:
            return pa.list_(rand_type(depth + 1))
        if r < 0.8:
            return pa.struct([(f"f{i}", rand_type(depth + 1))
                              for i in range(rng.randrange(1, 4))])
        return pa.map_(rng.choice([pa.string(), pa.int64()]),
                       rand_type(depth + 1))

    def dummy(t):
        if pa.types.is_list(t):
            return [dummy(t.value_type)]
        if pa.types.is_struct(t):
            return {f.name: dummy(f.type) for f in t}
        if pa.types.is_map(t):
            return [(dummy(t.key_type), dummy(t.item_type))]
        if pa.types.is_string(t):
            return "v"
        return 1

    parsed = skipped = 0
    for trial in range(150):
        typ = rand_type(0)
        t = pa.table({"c": pa.array([dummy(typ), None], type=typ)})
        p = str(tmp_path / f"r{trial}.parquet")
        pq.write_table(t, p, compression="NONE")
        nchunks = t.schema.empty_table().to_batches()  # noqa: F841
        try:
            f = srj_pq.read_footer(p)
        except NotImplementedError:
            skipped += 1
            continue
        got = sum(srj_pq._field_chunk_count(fl) for fl in f.schema)
        assert got == len(f.row_groups[0].columns), (trial, typ)
        # absolute def/rep levels must match pyarrow's column descriptors
        sch = pq.ParquetFile(p).schema
        exp = [(sch.column(i).max_definition_level,
                sch.column(i).max_repetition_level)
               for i in range(len(sch))]
        ours = [lv for fl in f.schema for lv in srj_pq._leaf_levels(fl)]
        assert ours == exp, (trial, typ, ours, exp)
        parsed += 1
    # the generator must exercise both outcomes
    assert parsed >= 60 and skipped >= 1, (parsed, skipped)


@pytest.mark.gpu
def test_decode_random_nested(tmp_path):
    """Randomized end-to-end decode conformance: random nested schemas +
    random null-sprinkled data, GPU decode vs the python values pyarrow
    wrote. Unsupported shapes must raise, never mis-decode."""
    rng = random.Random(515)
    prims = [pa.int64(), pa.int32(), pa.string(), pa.float64()]

    def rand_type(depth):
        r = rng.random()
        if depth >= 3 or r < 0.35:
            return rng.choice(prims)
        if r < 0.55:
            return pa.list_(rand_type(depth + 1))
        if r < 0.8:
            return pa.struct([(f"f{i}", rand_type(depth + 1))
                              for i in range(rng.randrange(1, 4))])
        return pa.map_(rng.choice([pa.string(), pa.int64()]),
                       rand_type(depth + 1))

    def rand_val(t, nullable=True):
        if nullable and rng.random() < 0.18:
            return None
        if pa.types.is_list(t):
            return [rand_val(t.value_type) for _ in range(rng.randrange(3))]
        if pa.types.is_struct(t):
            return {f.name: rand_val(f.type) for f in t}
        if pa.types.is_map(t):
            n = rng.randrange(3)
            keys = (rng.sample(range(100), n)
                    if pa.types.is_integer(t.key_type)
                    else rng.sample([f"k{j}" for j in range(100)], n))
            return [(k, rand_val(t.item_type)) for k in keys]
        if pa.types.is_string(t):
            return f"s{rng.randrange(50)}"
        if pa.types.is_floating(t):
            return rng.randrange(-10**6, 10**6) / 4.0  # exact in f64
        if t == pa.int32():
            return rng.randrange(-2**31, 2**31)
        return rng.randrange(-2**62, 2**62)

    def conv(t, v):
        if v is None:
            return None
        if pa.types.is_struct(t):
            return tuple(conv(f.type, v[f.name]) for f in t)
        if pa.types.is_map(t):
            return [(k, conv(t.item_type, x)) for k, x in v]
        if pa.types.is_list(t):
            return [conv(t.value_type, e) for e in v]
        return v

    parsed = skipped = 0
    for trial in range(25):
        typ = rand_type(0)
        rows = [rand_val(typ) for _ in range(300)]
        t = pa.table({"c": pa.array(rows, type=typ)})
        p = str(tmp_path / f"d{trial}.parquet")
        pq.write_table(t, p,
                       compression=rng.choice(["NONE", "SNAPPY"]),
                       data_page_version=rng.choice(["1.0", "2.0"]),
                       row_group_size=130)
        try:
            col = srj_pq.read_table(p, device="cuda").columns[0]
        except NotImplementedError:
            skipped += 1
            continue
        assert col.to_pylist() == [conv(typ, r) for r in rows], (trial, typ)
        parsed += 1
    assert parsed >= 10, (parsed, skipped)


def test_filter_row_groups_reference_vectors():
    """Reference ParquetFooterTest row-index-offset vectors: midpoint =
    first chunk offset + COMPRESSED size / 2; cumulative row offsets are
    computed over all original groups, so they survive filtering."""
    def rg(rows, dpo, csize):
        col = srj_pq.ColumnChunkMeta(
            path=("c",), physical_type=2, encodings=[0], codec=0,
            num_values=rows, total_compressed_size=csize,
            total_uncompressed_size=csize * 3,  # NOT what midpoints use
            data_page_offset=dpo, dictionary_page_offset=None)
        return srj_pq.RowGroupMeta([col], rows, csize * 3)

    f = srj_pq.ParquetFooter(1, 3500, [], [rg(1000, 100, 200),
                                           rg(2000, 400, 200),
                                           rg(500, 700, 200)])
    # midpoints: 200, 500, 800; cumulative row offsets: 0, 1000, 3000
    def sel(off, length):
        g = f.filter_row_groups(off, length)
        return ([r.num_rows for r in g.row_groups], g.row_index_offsets,
                g.num_rows)

    assert sel(0, 10**9) == ([1000, 2000, 500], [0, 1000, 3000], 3500)
    assert sel(0, 300) == ([1000], [0], 1000)
    assert sel(300, 300) == ([2000], [1000], 2000)
    assert sel(600, 300) == ([500], [3000], 500)
    assert sel(0, 600) == ([1000, 2000], [0, 1000], 3000)
    assert sel(10000, 100) == ([], [], 0)
    # single-group + byte-range-filtered-away
    f1 = srj_pq.ParquetFooter(1, 1000, [], [rg(1000, 100, 200)])
    assert f1.filter_row_groups(0, 300).row_index_offsets == [0]
    assert f1.filter_row_groups(300, 100).row_index_offsets == []
    # prune preserves the offsets
    g = f.filter_row_groups(300, 10**9)
    assert g.prune(["c"]).row_index_offsets == [1000, 3000]
