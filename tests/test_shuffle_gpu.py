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
