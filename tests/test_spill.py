"""SpillManager LRU bookkeeping (CPU) + device round-trip (GPU)."""
import pytest
import torch

from spark_rapids_jni_amd.columnar import Column, DType, Table
from spark_rapids_jni_amd.spill import SpillManager, _table_bytes


def _tbl(n, device="cpu"):
    return Table([Column.from_torch(
        torch.arange(n, dtype=torch.int64, device=device))])


def test_lru_order_and_accounting():
    m = SpillManager()
    a = m.register(_tbl(1000))
    b = m.register(_tbl(2000))
    c = m.register(_tbl(3000))
    assert m.spillable_bytes == (1000 + 2000 + 3000) * 8
    a.get(device="cpu")  # a becomes MRU
    freed = m.spill_until(2000 * 8)
    # b (oldest untouched) goes first, then c
    assert freed == 2000 * 8
    assert b.spilled and not a.spilled and not c.spilled
    freed = m.spill_until(10**9)
    assert freed == (1000 + 3000) * 8
    assert a.spilled and c.spilled
    # unspill restores data
    got = b.get(device="cpu")
    assert got.columns[0].to_pylist() == list(range(2000))
    assert not b.spilled


@pytest.mark.gpu
def test_device_spill_roundtrip():
    m = SpillManager()
    n = 200_000
    vals = [None if i % 17 == 3 else f"s{i}" for i in range(n)]
    t = Table([Column.from_pylist(vals, DType.STRING, "cuda"),
               Column.from_torch(torch.arange(n, dtype=torch.int64,
                                              device="cuda"))])
    st = m.register(t)
    nbytes = _table_bytes(t)
    freed = m.spill_until(1)
    assert freed == nbytes and st.spilled
    back = st.get()
    assert back.columns[0].to_pylist() == vals
    assert back.columns[1].data.device.type == "cuda"


@pytest.mark.gpu
def test_retry_loop_spills_device_tables():
    """End-to-end OOM protocol on device data: a forced GpuRetryOOM inside
    with_retry triggers SpillManager spilling real device tables to pinned
    host memory; the retried allocation then proceeds."""
    from spark_rapids_jni_amd.memory import RmmSpark, with_retry
    from spark_rapids_jni_amd.spill import SpillManager
    RmmSpark.clear_event_handler()
    RmmSpark.set_event_handler(pool_limit=1 << 30)
    tid = RmmSpark.current_thread_id()
    RmmSpark.start_dedicated_task_thread(tid, 500)
    m = SpillManager()
    tables = [m.register(Table([Column.from_torch(
        torch.arange(100_000, dtype=torch.int64, device="cuda"))]))
        for _ in range(3)]
    RmmSpark.force_retry_oom(tid, 1)

    def alloc():
        RmmSpark.alloc(1024, thread_id=tid)
        return torch.empty(1024, device="cuda")

    got = with_retry(
        alloc,
        spill=lambda: m.spill_until(100_000 * 8),
        split=lambda: None, thread_id=tid)
    assert got.numel() == 1024
    assert any(t.spilled for t in tables)
    # unspill still round-trips
    back = tables[0].get() if tables[0].spilled else tables[1].get()
    assert back.columns[0].data.device.type == "cuda"
    RmmSpark.task_done(500)
    RmmSpark.clear_event_handler()


_REAL_ALLOC_SCRIPT = r"""
import os, sys
sys.path.insert(0, os.environ["SRJ_REPO"])
from spark_rapids_jni_amd.memory import RmmSpark, with_retry, GpuRetryOOM
from spark_rapids_jni_amd.spill import SpillManager
from spark_rapids_jni_amd.columnar import Column, Table

# install the pluggable allocator BEFORE the first CUDA allocation
RmmSpark.use_real_allocator()
import torch

POOL = 512 << 20  # 512 MiB real-HBM cap
RmmSpark.set_event_handler(pool_limit=POOL, real=True)
tid = RmmSpark.current_thread_id()
RmmSpark.start_dedicated_task_thread(tid, 900)

m = SpillManager()
# ~384 MiB resident spillable table
big = Table([Column.from_torch(torch.zeros(48 << 20, dtype=torch.int64,
                                           device="cuda"))])
st = m.register(big)
del big
from spark_rapids_jni_amd import _native
used0 = _native.gpu().device_pool_used()
assert used0 >= (384 << 20), used0

attempts = []

def alloc():
    attempts.append(1)
    # needs ~256 MiB: only fits after the 384 MiB table spills to host
    return torch.zeros(32 << 20, dtype=torch.int64, device="cuda")

got = with_retry(alloc, spill=lambda: m.spill_until(1 << 40),
                 thread_id=tid)
assert got.numel() == 32 << 20
assert len(attempts) >= 2, "expected a real OOM retry"
assert st.spilled, "the spillable table must have spilled to host"
metrics = RmmSpark.get_and_reset_metrics(900)
assert metrics["num_retry"] >= 1, metrics
# device accounting went down when the table spilled
used1 = _native.gpu().device_pool_used()
assert used1 < used0 + (256 << 20), (used0, used1)
RmmSpark.task_done(900)
RmmSpark.clear_event_handler()
print("REAL_ALLOC_OK", len(attempts), metrics["num_retry"])
"""


@pytest.mark.gpu
def test_real_allocator_oom_spill_retry(tmp_path):
    """real HBM exhaustion (capped pluggable allocator
    over hipMallocAsync) raises GpuRetryOOM through the state machine, the
    retry loop spills an actual device table, and the allocation then
    succeeds."""
    import os
    import subprocess
    import sys
    script = tmp_path / "real_alloc.py"
    script.write_text(_REAL_ALLOC_SCRIPT)
    env = dict(os.environ)
    env["SRJ_REPO"] = os.path.dirname(os.path.dirname(
        os.path.abspath(__file__)))
    out = subprocess.run([sys.executable, str(script)], env=env,
                         capture_output=True, timeout=300)
    assert out.returncode == 0, \
        f"stdout={out.stdout.decode()[-2000:]}\n" \
        f"stderr={out.stderr.decode()[-3000:]}"
    assert b"REAL_ALLOC_OK" in out.stdout
