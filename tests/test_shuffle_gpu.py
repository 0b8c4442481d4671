"""Device kudo split/assemble tests, cross-validated against the host
serializer (byte-format compatibility both directions)."""
import io
import random

import pytest
import torch

from spark_rapids_jni_amd import kudo
from spark_rapids_jni_amd.columnar import Column, DType, Table

random.seed(17)


def _mk_table(n, device):
    ints = [None if i % 9 == 4 else random.randint(-10**6, 10**6)
            for i in range(n)]
    flts = [random.random() * 100 for i in range(n)]
    strs = [None if i % 7 == 3 else f"row{i}" * (i % 3) for i in range(n)]
    cols = [Column.from_pylist(ints, DType.INT64, device),
            Column.from_pylist(flts, DType.FLOAT64, device),
            Column.from_pylist(strs, DType.STRING, device)]
    return Table(cols), (ints, flts, strs)


def _partition(n, nparts, device):
    import numpy as np
    pids_h = [random.randrange(nparts) for _ in range(n)]
    pids = torch.tensor(pids_h, dtype=torch.int32, device=device)
    from spark_rapids_jni_amd.ops.copying import partition_map
    offsets, perm = partition_map(pids, nparts)
    return pids_h, offsets, perm


@pytest.mark.gpu
def test_device_split_host_merge():
    """Device-written kudo records must merge correctly on the HOST path."""
    from spark_rapids_jni_amd import shuffle_gpu
    n, nparts = 1000, 5
    tbl, (ints, flts, strs) = _mk_table(n, "cuda")
    pids_h, offsets, perm = _partition(n, nparts, "cuda")
    buf, sizes = shuffle_gpu.split_and_serialize_to_device(tbl, offsets, perm)
    raw = buf.cpu().numpy().tobytes()
    pieces = []
    pos = 0
    for s in sizes:
        pieces.append(raw[pos:pos + s])
        pos += s
    merged = kudo.merge_on_host(pieces, [c.to("cpu") for c in tbl.columns])
    permh = perm.cpu().tolist()
    exp_ints = [ints[i] for i in permh]
    exp_flts = [flts[i] for i in permh]
    exp_strs = [strs[i] for i in permh]
    assert merged[0].to_pylist() == exp_ints
    assert merged[1].to_pylist() == exp_flts
    assert merged[2].to_pylist() == exp_strs


@pytest.mark.gpu
def test_host_write_device_assemble():
    """Host-written kudo records must assemble correctly on the DEVICE path."""
    from spark_rapids_jni_amd import shuffle_gpu
    n = 300
    tbl_h, (ints, flts, strs) = _mk_table(n, "cpu")
    slices = [(3, 50), (53, 7), (60, 240)]
    bufs = []
    for off, cnt in slices:
        out = io.BytesIO()
        kudo.write_partition(tbl_h.columns, off, cnt, out)
        bufs.append(torch.frombuffer(bytearray(out.getvalue()),
                                     dtype=torch.uint8).cuda())
    got = shuffle_gpu.assemble_from_device(bufs, tbl_h.columns)
    exp = [[v for off, cnt in slices for v in col[off:off + cnt]]
           for col in (ints, flts, strs)]
    assert got.columns[0].to_pylist() == exp[0]
    assert got.columns[1].to_pylist() == exp[1]
    assert got.columns[2].to_pylist() == exp[2]


@pytest.mark.gpu
def test_device_roundtrip():
    from spark_rapids_jni_amd import shuffle_gpu
    n, nparts = 2000, 8
    tbl, (ints, flts, strs) = _mk_table(n, "cuda")
    pids_h, offsets, perm = _partition(n, nparts, "cuda")
    buf, sizes = shuffle_gpu.split_and_serialize_to_device(tbl, offsets, perm)
    got = shuffle_gpu.assemble_from_device_raw(buf, sizes, tbl.columns)
    permh = perm.cpu().tolist()
    assert got.columns[0].to_pylist() == [ints[i] for i in permh]
    assert got.columns[1].to_pylist() == [flts[i] for i in permh]
    assert got.columns[2].to_pylist() == [strs[i] for i in permh]


@pytest.mark.gpu
def test_device_roundtrip_with_empty_partitions():
    from spark_rapids_jni_amd import shuffle_gpu
    n, nparts = 40, 16  # some partitions will be empty
    tbl, (ints, flts, strs) = _mk_table(n, "cuda")
    pids = torch.tensor([i % 3 for i in range(n)], dtype=torch.int32,
                        device="cuda")
    from spark_rapids_jni_amd.ops.copying import partition_map
    offsets, perm = partition_map(pids, nparts)
    buf, sizes = shuffle_gpu.split_and_serialize_to_device(tbl, offsets, perm)
    got = shuffle_gpu.assemble_from_device_raw(buf, sizes, tbl.columns)
    permh = perm.cpu().tolist()
    assert got.columns[0].to_pylist() == [ints[i] for i in permh]
    assert got.columns[2].to_pylist() == [strs[i] for i in permh]


def _mk_nested(n, device):
    """table: [LIST<INT64> (nullable), STRUCT<INT32, STRING>, LIST<STRING>]"""
    child_vals, offs = [], [0]
    for i in range(n):
        ln = random.randint(0, 4)
        child_vals.extend(random.randint(0, 99) for _ in range(ln))
        offs.append(offs[-1] + ln)
    from spark_rapids_jni_amd.columnar import validity_from_bools
    lvalid = [i % 11 != 7 for i in range(n)]
    lst = Column(DType.LIST, n, None,
                 validity_from_bools(lvalid, device),
                 torch.tensor(offs, dtype=torch.int32, device=device),
                 [Column.from_pylist(child_vals, DType.INT64, device)],
                 null_count=None)
    a = Column.from_pylist([None if i % 6 == 0 else i for i in range(n)],
                           DType.INT32, device)
    b = Column.from_pylist([f"v{i}" if i % 3 else None for i in range(n)],
                           DType.STRING, device)
    st = Column(DType.STRUCT, n, None, None, None, [a, b])
    soffs, svals = [0], []
    for i in range(n):
        ln = random.randint(0, 3)
        svals.extend(f"s{i}-{k}" for k in range(ln))
        soffs.append(soffs[-1] + ln)
    lstr = Column(DType.LIST, n, None, None,
                  torch.tensor(soffs, dtype=torch.int32, device=device),
                  [Column.from_pylist(svals, DType.STRING, device)])
    return Table([lst, st, lstr])


@pytest.mark.gpu
def test_nested_gather():
    from spark_rapids_jni_amd.ops.copying import gather
    n = 200
    tbl = _mk_nested(n, "cuda")
    exp = [c.to_pylist() for c in tbl.columns]
    gmap = torch.tensor([random.randrange(n) for _ in range(n * 2)],
                        dtype=torch.int64, device="cuda")
    G = gather(tbl, gmap)
    gm = gmap.cpu().tolist()
    for c, e in zip(G.columns, exp):
        assert c.to_pylist() == [e[i] for i in gm]


@pytest.mark.gpu
def test_nested_device_split_host_merge():
    from spark_rapids_jni_amd import shuffle_gpu
    n, nparts = 500, 4
    tbl = _mk_nested(n, "cuda")
    exp = [c.to_pylist() for c in tbl.columns]
    pids_h, offsets, perm = _partition(n, nparts, "cuda")
    buf, sizes = shuffle_gpu.split_and_serialize_to_device(tbl, offsets, perm)
    raw = buf.cpu().numpy().tobytes()
    pieces, pos = [], 0
    for s in sizes:
        pieces.append(raw[pos:pos + s])
        pos += s
    schema = [c.to("cpu") for c in tbl.columns]
    merged = kudo.merge_on_host(pieces, schema)
    permh = perm.cpu().tolist()
    for mc, e in zip(merged, exp):
        assert mc.to_pylist() == [e[i] for i in permh]


@pytest.mark.gpu
def test_nested_host_write_device_assemble():
    from spark_rapids_jni_amd import shuffle_gpu
    n = 300
    tbl_cpu = _mk_nested(n, "cpu")
    exp = [c.to_pylist() for c in tbl_cpu.columns]
    slices = [(0, 100), (100, 50), (150, 0), (150, 150)]
    bufs = []
    for off, cnt in slices:
        out = io.BytesIO()
        kudo.write_partition(tbl_cpu.columns, off, cnt, out)
        bufs.append(torch.frombuffer(bytearray(out.getvalue()),
                                     dtype=torch.uint8).to("cuda"))
    got = shuffle_gpu.assemble_from_device(bufs, tbl_cpu.columns)
    exp_rows = [i for off, cnt in slices for i in range(off, off + cnt)]
    for gc, e in zip(got.columns, exp):
        assert gc.to_pylist() == [e[i] for i in exp_rows]


@pytest.mark.gpu
def test_nested_device_roundtrip():
    from spark_rapids_jni_amd import shuffle_gpu
    n, nparts = 400, 3
    tbl = _mk_nested(n, "cuda")
    exp = [c.to_pylist() for c in tbl.columns]
    pids_h, offsets, perm = _partition(n, nparts, "cuda")
    buf, sizes = shuffle_gpu.split_and_serialize_to_device(tbl, offsets, perm)
    got = shuffle_gpu.assemble_from_device_raw(buf, sizes,
                                               [c.to("cpu") for c in
                                                tbl.columns])
    permh = perm.cpu().tolist()
    for gc, e in zip(got.columns, exp):
        assert gc.to_pylist() == [e[i] for i in permh]


@pytest.mark.gpu
def test_deep_nested_device_roundtrip():
    """LIST<LIST<INT64>> exercises the level-chained offset resolution."""
    from spark_rapids_jni_amd import shuffle_gpu
    n, nparts = 150, 3
    inner_vals, ioffs = [], [0]
    for _ in range(400):
        ln = random.randint(0, 3)
        inner_vals.extend(random.randint(0, 999) for _ in range(ln))
        ioffs.append(ioffs[-1] + ln)
    inner = Column(DType.LIST, 400, None, None,
                   torch.tensor(ioffs, dtype=torch.int32, device="cuda"),
                   [Column.from_pylist(inner_vals, DType.INT64, "cuda")])
    ooffs = [0]
    for i in range(n):
        ooffs.append(min(ooffs[-1] + random.randint(0, 5), 400))
    outer = Column(DType.LIST, n, None, None,
                   torch.tensor(ooffs, dtype=torch.int32, device="cuda"),
                   [inner])
    tbl = Table([outer])
    exp = outer.to_pylist()
    pids_h, offsets, perm = _partition(n, nparts, "cuda")
    buf, sizes = shuffle_gpu.split_and_serialize_to_device(tbl, offsets, perm)
    got = shuffle_gpu.assemble_from_device_raw(
        buf, sizes, [outer.to("cpu")])
    permh = perm.cpu().tolist()
    assert got.columns[0].to_pylist() == [exp[i] for i in permh]


@pytest.mark.gpu
def test_device_kudo_fuzz_roundtrip():
    """Random nested schemas + random partition maps through the DEVICE
    split/assemble pair."""
    import random as rnd
    r = rnd.Random(311)
    from spark_rapids_jni_amd import shuffle_gpu
    from spark_rapids_jni_amd.columnar import validity_from_bools

    def rand_col(n, depth=0):
        t = r.choice(["i64", "str", "f64"] +
                     (["struct", "list"] if depth < 2 else []))
        if t == "i64":
            vals = [None if r.random() < 0.15 else r.randint(-10**9, 10**9)
                    for _ in range(n)]
            return Column.from_pylist(vals, DType.INT64, "cuda")
        if t == "f64":
            vals = [None if r.random() < 0.1 else r.random() * 100
                    for _ in range(n)]
            return Column.from_pylist(vals, DType.FLOAT64, "cuda")
        if t == "str":
            vals = [None if r.random() < 0.2 else "x" * r.randrange(5)
                    for _ in range(n)]
            return Column.from_pylist(vals, DType.STRING, "cuda")
        if t == "struct":
            a = rand_col(n, depth + 1)
            b = rand_col(n, depth + 1)
            vmask = validity_from_bools([r.random() > 0.1 for _ in range(n)],
                                        "cuda")
            return Column(DType.STRUCT, n, None, vmask, None, [a, b],
                          null_count=None)
        offs = [0]
        total = 0
        for _ in range(n):
            total += r.randrange(3)
            offs.append(total)
        child = rand_col(total, depth + 1)
        return Column(DType.LIST, n, None, None,
                      torch.tensor(offs, dtype=torch.int32, device="cuda"),
                      [child])

    for trial in range(8):
        n = r.randrange(1, 400)
        tbl = Table([rand_col(n) for _ in range(r.randrange(1, 4))])
        exp = [c.to_pylist() for c in tbl.columns]
        nparts = r.randrange(1, 5)
        pids_h, offsets, perm = _partition(n, nparts, "cuda")
        buf, sizes = shuffle_gpu.split_and_serialize_to_device(tbl, offsets,
                                                               perm)
        got = shuffle_gpu.assemble_from_device_raw(
            buf, sizes, [c.to("cpu") for c in tbl.columns])
        permh = perm.cpu().tolist()
        for gc, e in zip(got.columns, exp):
            assert gc.to_pylist() == [e[i] for i in permh], f"trial {trial}"
