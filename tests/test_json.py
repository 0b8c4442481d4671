"""JSON op tests (get_json_object / from_json) vs Python oracle."""
import json
import random

import pytest

from spark_rapids_jni_amd.columnar import Column, DType

random.seed(19)


def test_path_compile():
    from spark_rapids_jni_amd.ops.json import JsonPathError, compile_path
    assert compile_path("$.a[2].b[*]") == [(0, "a", 0), (1, None, 2),
                                           (0, "b", 0), (2, None, 0)]
    assert compile_path("$['x y'].z") == [(0, "x y", 0), (0, "z", 0)]
    assert compile_path("$.*") == [(2, None, 0)]
    with pytest.raises(JsonPathError):
        compile_path("a.b")
    with pytest.raises(JsonPathError):
        compile_path("$.a[b]")


DOCS = [
    '{"a":"b"}',
    '{"a":1,"b":{"c":[10,20,30]}}',
    '{"a":[{"x":1},{"x":2},{"y":3}]}',
    '{"a":"line\\nbreak \\u00e9"}',
    '{"a":null}',
    '[1,2,3]',
    'not json at all {',
    "",
    None,
    '{"a": {"deep": {"nest": "v"}}}',
    '{"empty":{},"arr":[]}',
]


@pytest.mark.gpu
@pytest.mark.parametrize("path,expect", [
    ("$.a", ["b", "1", '[{"x":1},{"x":2},{"y":3}]', "line\nbreak é",
             None, None, None, None, None, '{"deep":{"nest":"v"}}', None]),
    ("$.b.c[1]", [None, "20", None, None, None, None, None, None, None, None,
                  None]),
    ("$.a[*].x", [None, None, "[1,2]", None, None, None, None, None, None,
                  None, None]),
    ("$.a[1].x", [None, None, "2", None, None, None, None, None, None, None,
                  None]),
    ("$.a.deep.nest", [None, None, None, None, None, None, None, None, None,
                       "v", None]),
    ("$.empty", [None, None, None, None, None, None, None, None, None, None,
                 "{}"]),
    ("$[1]", [None, None, None, None, None, "2", None, None, None, None,
              None]),
])
def test_get_json_object(path, expect):
    from spark_rapids_jni_amd.ops.json import get_json_object
    col = Column.from_pylist(DOCS, DType.STRING, "cuda")
    got = get_json_object(col, path).to_pylist()
    assert got == expect, f"path {path}"


@pytest.mark.gpu
def test_from_json_raw_map():
    from spark_rapids_jni_amd.ops.json import from_json_to_raw_map
    docs = ['{"k1":"v1","k2":2}', "{}", '{"x":{"nested":1},"y":[1,2]}',
            "bad", None, '{"esc\\t":"tab"}']
    col = Column.from_pylist(docs, DType.STRING, "cuda")
    out = from_json_to_raw_map(col)
    lst = out.to_pylist()
    assert lst[0] == [("k1", "v1"), ("k2", "2")]
    assert lst[1] == []
    assert lst[2] == [("x", '{"nested":1}'), ("y", "[1,2]")]
    assert lst[3] is None
    assert lst[4] is None
    assert lst[5] == [("esc\t", "tab")]


@pytest.mark.gpu
def test_from_json_to_structs():
    from spark_rapids_jni_amd.ops.json import from_json_to_structs
    docs = ['{"i": 42, "s":"hi", "f": 1.5, "b": true}',
            '{"i": "7", "s": 9, "f": "2.25"}',
            '{"s":"only"}', None]
    col = Column.from_pylist(docs, DType.STRING, "cuda")
    t = from_json_to_structs(col, ["i", "s", "f", "b"],
                             [DType.INT64, DType.STRING, DType.FLOAT64,
                              DType.BOOL8])
    assert t.columns[0].to_pylist() == [42, 7, None, None]
    assert t.columns[1].to_pylist() == ["hi", "9", "only", None]
    assert t.columns[2].to_pylist() == [1.5, 2.25, None, None]
    assert t.columns[3].to_pylist() == [True, None, None, None]


@pytest.mark.gpu
def test_get_json_object_fuzz_vs_oracle():
    """random flat docs: $.key extraction must match a json-module oracle."""
    from spark_rapids_jni_amd.ops.json import get_json_object
    docs = []
    for _ in range(300):
        d = {f"k{j}": random.choice([random.randint(0, 99), "s" * (j % 3),
                                     None, True, [1, 2], {"n": j}])
             for j in range(random.randint(0, 5))}
        docs.append(json.dumps(d, separators=(",", ":")))
    col = Column.from_pylist(docs, DType.STRING, "cuda")
    for key in ["k0", "k2", "k4"]:
        got = get_json_object(col, f"$.{key}").to_pylist()
        for doc, gv in zip(docs, got):
            obj = json.loads(doc)
            if key not in obj or obj[key] is None:
                assert gv is None, (doc, key, gv)
            else:
                v = obj[key]
                if isinstance(v, str):
                    exp = v
                else:
                    exp = json.dumps(v, separators=(",", ":"))
                assert gv == exp, (doc, key, gv)

@pytest.mark.gpu
def test_multi_path_shared_scan():
    """Multi-path results must equal per-path get_json_object exactly."""
    from spark_rapids_jni_amd.ops.json import (get_json_object,
                                               get_json_object_multiple_paths)
    docs = ['{"a":%d,"b":{"c":"x%d"},"d":[%d,%d,%d],"e":"v","f":null}' %
            (i, i % 9, i, i + 1, i + 2) for i in range(3000)]
    docs += ['{"a":1', None, '[]', '{"d":[5]}']
    col = Column.from_pylist(docs, DType.STRING, "cuda")
    paths = ["$.a", "$.b.c", "$.d[1]", "$.d[*]", "$.e", "$.f", "$.missing",
             "$.b", "$.d[9]"]
    multi = get_json_object_multiple_paths(col, paths)
    for p, mc in zip(paths, multi):
        sc = get_json_object(col, p)
        assert mc.to_pylist() == sc.to_pylist(), f"path {p}"


@pytest.mark.gpu
def test_multi_path_overflow_fallback():
    """>4 wildcard matches per row triggers the per-path fallback."""
    from spark_rapids_jni_amd.ops.json import (get_json_object,
                                               get_json_object_multiple_paths)
    docs = ['[1,2,3,4,5,6,7]', '[1,2]']
    col = Column.from_pylist(docs, DType.STRING, "cuda")
    paths = ["$[*]", "$[0]"]
    multi = get_json_object_multiple_paths(col, paths)
    for p, mc in zip(paths, multi):
        sc = get_json_object(col, p)
        assert mc.to_pylist() == sc.to_pylist(), f"path {p}"


@pytest.mark.gpu
def test_from_json_nested_struct():
    from spark_rapids_jni_amd.ops.json import from_json_to_structs
    docs = ['{"a":%d,"b":{"x":%d,"y":"s%d"}}' % (i, i * 2, i)
            if i % 7 != 3 else '{"a":%d}' % i for i in range(300)]
    col = Column.from_pylist(docs, DType.STRING, "cuda")
    tbl = from_json_to_structs(
        col, ["a", "b"],
        [DType.INT64, ("struct", ["x", "y"], [DType.INT64, DType.STRING])])
    assert tbl.columns[0].to_pylist() == list(range(300))
    b = tbl.columns[1]
    assert b.dtype == DType.STRUCT
    xs = b.children[0].to_pylist()
    ys = b.children[1].to_pylist()
    for i in range(300):
        if i % 7 == 3:
            assert not b.is_valid_host(i)
        else:
            assert xs[i] == i * 2 and ys[i] == f"s{i}"


@pytest.mark.gpu
def test_get_json_object_normalization():
    """Reference GetJsonObjectTest Escape/Number_Normalization/leading-zeros
    vectors: output is re-serialized through the JSON generator — whitespace
    stripped, single quotes accepted, strings re-escaped canonically,
    numbers through the Java Double.toString formatter."""
    from spark_rapids_jni_amd.ops.json import get_json_object

    bs = "\\"
    doc5 = ("'" + bs + "u4e2d" + bs + "u56FD" + bs + '"' + bs + "'" +
            bs + bs + bs + "/" + bs + "b" + bs + "f" + bs + "n" + bs + "r" +
            bs + "t" + bs + "b'")
    exp5 = '中国"\'' + bs + "/\b\f\n\r\t\b"
    doc6 = "[" + doc5 + "]"
    exp6 = ('["中国' + bs + '"' + "'" + bs + bs + "/" + bs + "b" + bs + "f" +
            bs + "n" + bs + "r" + bs + "t" + bs + 'b"]')

    cases = [
        ('{ "a": "A" }', '{"a":"A"}'),
        ("{'a':'A\"'}", '{"a":"A' + bs + '""}'),
        ("{'a':\"B'\"}", '{"a":"B\'"}'),
        ("['a','b','\"C\"']", '["a","b","' + bs + '"C' + bs + '""]'),
        (doc5, exp5),
        (doc6, exp6),
        # number normalization
        ("[100.0,200.000,351.980]", "[100.0,200.0,351.98]"),
        ("[12345678900000000000.0]", "[1.23456789E19]"),
        ("[0.0]", "[0.0]"),
        ("[-0.0]", "[-0.0]"),
        ("[-0]", "[0]"),
        ("[12345678999999999999999999]", "[12345678999999999999999999]"),
        ("[9.299999257686047e-0005603333574677677]", "[0.0]"),
        ("9.299999257686047e0005603333574677677", '"Infinity"'),
        ("[1E308]", "[1.0E308]"),
        ("[1.0E309,-1E309,1E5000]",
         '["Infinity","-Infinity","Infinity"]'),
        ("0.3", "0.3"),
        ("0.03", "0.03"),
        ("0.003", "0.003"),
        ("0.0003", "3.0E-4"),
        ("0.00003", "3.0E-5"),
        # leading zeros invalid
        ("00", None), ("01", None), ("02", None), ("000", None),
        ("-01", None), ("-00", None), ("-02", None),
    ]
    col = Column.from_pylist([d for d, _ in cases], DType.STRING, "cuda")
    got = get_json_object(col, "$").to_pylist()
    for (doc, exp), g in zip(cases, got):
        assert g == exp, (doc, g, exp)


@pytest.mark.gpu
def test_get_json_object_escaped_keys_and_surrogates():
    """Reference NamesWithEscapedCharacters + testUCS2Surrogates: \\uXXXX
    escapes in KEY names match plain path keys; surrogate pairs in values
    decode to the astral codepoint."""
    from spark_rapids_jni_amd.ops.json import (get_json_object,
                                               get_json_object_multiple_paths)
    col = Column.from_pylist(
        ["{'data': 'TEST1'}", "{'\\u0064\\u0061t\\u0061': 'TEST2'}"],
        DType.STRING, "cuda")
    assert get_json_object(col, "$.data").to_pylist() == ["TEST1", "TEST2"]
    got = get_json_object_multiple_paths(col, ["$.data"])[0].to_pylist()
    assert got == ["TEST1", "TEST2"]
    col2 = Column.from_pylist(
        ["{'package_name': 'TEST1'}", "{'package_name': '\\uD83E\\uDD66'}"],
        DType.STRING, "cuda")
    assert get_json_object(col2, "$.package_name").to_pylist() == \
        ["TEST1", "\U0001F966"]
