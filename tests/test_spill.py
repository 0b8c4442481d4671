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
