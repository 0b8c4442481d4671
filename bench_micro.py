#!/usr/bin/env python3
"""Micro-benchmarks reproducing the reference's six nvbench shapes
(row transpose to/from rows, string->float, long->binary-string,
bloom build/probe, get_json_object 1-N paths, parse_uri) so per-op throughput
is comparable once numbers exist on both sides.

Run on an MI355X: python bench_micro.py [--rows N] [--iters K]
Prints one JSON line per shape: {"bench": ..., "value": ..., "unit": ...}.
"""
import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

import torch


def timeit(fn, iters, warmup=2):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def emit(name, rows, secs, bytes_processed=None):
    rec = {"bench": name, "rows": rows, "ms": round(secs * 1e3, 3),
           "rows_per_sec": round(rows / secs, 1)}
    if bytes_processed:
        rec["GBps"] = round(bytes_processed / secs / 1e9, 2)
    print(json.dumps(rec), flush=True)


def plumbing_bench():
    """BASELINE config[0]: murmur3 + row<->columnar on a 1k-row int64 batch
    via the HOST path (no GPU) — measures binding/plumbing overhead like the
    reference's Java/JNI host path."""
    import io
    import numpy as np
    from spark_rapids_jni_amd import _native, kudo
    from spark_rapids_jni_amd.columnar import Column, DType
    host = _native.host()
    n = 1000
    keys = torch.arange(n, dtype=torch.int64)
    out = torch.empty(n, dtype=torch.int32)
    col = Column.from_torch(keys)
    iters = 2000
    t0 = time.perf_counter()
    for _ in range(iters):
        host.murmur3_long_host(keys.data_ptr(), n, 42, out.data_ptr())
        b = io.BytesIO()
        kudo.write_partition([col], 0, n, b)
        kudo.merge_on_host([b.getvalue()], [col])
    dt = (time.perf_counter() - t0) / iters
    print(json.dumps({"bench": "host_plumbing_murmur3_rowcol_1k",
                      "rows": n, "us_per_batch": round(dt * 1e6, 1),
                      "rows_per_sec": round(n / dt, 1)}), flush=True)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--rows", type=int, default=2**24)
    ap.add_argument("--iters", type=int, default=5)
    args = ap.parse_args()
    plumbing_bench()  # CPU shape (BASELINE config[0]) first
    n = args.rows
    dev = "cuda"

    from spark_rapids_jni_amd.columnar import Column, DType, Table
    from spark_rapids_jni_amd.ops import cast, hashing, misc
    from spark_rapids_jni_amd.ops.row_conversion import (convert_from_rows,
                                                         convert_to_rows)
    from spark_rapids_jni_amd.ops.json import (get_json_object,
                                               get_json_object_multiple_paths)
    from spark_rapids_jni_amd.ops.sketch import parse_uri, UriPart

    g = torch.Generator(device=dev)
    g.manual_seed(7)

    # 1. row transpose: mixed fixed-width table (reference: 212 mixed columns
    #    at 2^10..2^26 rows; here 24 columns x 8 type-pattern, scaled rows)
    cols = []
    dtypes = [DType.INT64, DType.INT32, DType.FLOAT64, DType.FLOAT32,
              DType.INT16, DType.INT8, DType.BOOL8, DType.TIMESTAMP_US] * 3
    for dt in dtypes:
        from spark_rapids_jni_amd.columnar import TORCH_DTYPE
        t = torch.randint(0, 100, (n // 16,), dtype=torch.int64,
                          device=dev, generator=g).to(TORCH_DTYPE[dt])
        cols.append(Column(dt, n // 16, t))
    tbl = Table(cols)
    batches = None

    def to_rows():
        nonlocal batches
        batches = convert_to_rows(tbl)

    secs = timeit(to_rows, args.iters)
    row_size = batches[0][0].numel() // batches[0][1]
    emit("row_conversion_to_rows", n // 16, secs,
         (n // 16) * row_size)
    secs = timeit(lambda: convert_from_rows(batches, dtypes), args.iters)
    emit("row_conversion_from_rows", n // 16, secs, (n // 16) * row_size)

    # 2. string -> float
    fvals = torch.rand(n // 4, dtype=torch.float64, device=dev,
                       generator=g) * 1e6 - 5e5
    fstr = cast.from_floats(Column(DType.FLOAT64, n // 4, fvals))
    secs = timeit(lambda: cast.to_float(fstr), args.iters)
    emit("string_to_float", n // 4, secs,
         int(fstr.offsets[-1].item()))

    # 3. long -> "binary" (base-2) string via conv, plus hex
    lvals = torch.randint(0, 2**62, (n // 4,), dtype=torch.int64, device=dev,
                          generator=g)
    lstr = cast.from_integer(Column(DType.INT64, n // 4, lvals))
    from spark_rapids_jni_amd.ops.sketch import convert_base
    secs = timeit(lambda: convert_base(lstr, 10, 2), args.iters)
    emit("long_to_binary_string", n // 4, secs)

    # 4. bloom filter build + probe
    keys = torch.randint(0, 2**40, (n,), dtype=torch.int64, device=dev,
                         generator=g)
    kc = Column(DType.INT64, n, keys)
    bf = misc.BloomFilter(2, 3, 1 << 20, seed=42)
    secs = timeit(lambda: bf.put(kc), args.iters)
    emit("bloom_filter_build", n, secs)
    secs = timeit(lambda: bf.might_contain(kc), args.iters)
    emit("bloom_filter_probe", n, secs)

    # 5. get_json_object, 1..4 paths over the same docs
    docs = ['{"a":%d,"b":{"c":"x%d"},"d":[%d,%d],"e":"v"}' %
            (i, i % 97, i, i + 1) for i in range(200_000)]
    jc = Column.from_pylist(docs, DType.STRING, dev)
    paths = ["$.a", "$.b.c", "$.d[1]", "$.e"]
    for k in (1, 2, 4):
        def run_paths(k=k):
            if k == 1:
                get_json_object(jc, paths[0])
            else:
                get_json_object_multiple_paths(jc, paths[:k])
        secs = timeit(run_paths, args.iters)
        emit(f"get_json_object_{k}path", jc.size * k, secs,
             int(jc.offsets[-1].item()) * k)

    # 5b. generic (multi-column) hash join probe — the non-int64 join path
    from spark_rapids_jni_amd.ops.join import HashJoinTable
    nb = n // 2
    npr = n
    k1 = torch.randint(0, nb, (nb,), dtype=torch.int32, device=dev, generator=g)
    k2 = torch.randint(0, 8, (nb,), dtype=torch.int32, device=dev, generator=g)
    btbl = Table([Column.from_torch(k1), Column.from_torch(k2)])
    p1 = torch.randint(0, nb, (npr,), dtype=torch.int32, device=dev,
                       generator=g)
    p2 = torch.randint(0, 8, (npr,), dtype=torch.int32, device=dev,
                       generator=g)
    ptbl = Table([Column.from_torch(p1), Column.from_torch(p2)])
    tbl = HashJoinTable.build(btbl)
    secs = timeit(lambda: tbl.inner_join(ptbl, out_hint=npr // 4), args.iters)
    emit("generic_join_probe_2col", npr, secs)

    # 5c. radix sort (north-star hot op; reference delegates to cub/cudf)
    from spark_rapids_jni_amd.ops.sort import sort_pairs_i64
    skeys = torch.randint(-2**62, 2**62, (n,), dtype=torch.int64, device=dev,
                          generator=g)
    secs = timeit(lambda: sort_pairs_i64(skeys), args.iters)
    emit("radix_sort_int64", n, secs, n * 8)

    # 6. parse_uri
    uris = ["https://host%d.example.com:80/p/%d?k=%d&z=9" % (i % 50, i, i)
            for i in range(200_000)]
    uc = Column.from_pylist(uris, DType.STRING, dev)
    secs = timeit(lambda: parse_uri(uc, UriPart.HOST), args.iters)
    emit("parse_uri_host", uc.size, secs, int(uc.offsets[-1].item()))
    secs = timeit(lambda: parse_uri(uc, UriPart.QUERY_KEY, "k"), args.iters)
    emit("parse_uri_query_key", uc.size, secs)


if __name__ == "__main__":
    main()
