"""OOM-retry state machine tests (CPU-only; reference: RmmSparkTest.java).

These drive the full docs/memory_management.md machine: blocking on OOM,
wake-on-free, deadlock rollback (GpuRetryOOM), BUFN escalation to
split-and-retry (GpuSplitAndRetryOOM), forced-OOM injection, and metrics.
"""
import random
import threading
import time

import pytest

from spark_rapids_jni_amd.memory import (CpuRetryOOM, GpuRetryOOM,
                                         GpuSplitAndRetryOOM, RmmSpark,
                                         TaskPriority)


@pytest.fixture()
def rmm():
    RmmSpark.clear_event_handler()
    yield RmmSpark
    RmmSpark.clear_event_handler()


def _adaptor(rmm, pool=1000):
    return rmm.set_event_handler(pool_limit=pool)


def test_basic_alloc_dealloc(rmm):
    a = _adaptor(rmm)
    tid = 101
    rmm.start_dedicated_task_thread(tid, 1)
    rmm.alloc(100, thread_id=tid)
    assert a.pool_used() == 100
    rmm.dealloc(100, thread_id=tid)
    assert a.pool_used() == 0
    assert rmm.get_state_of(tid) == "THREAD_RUNNING"
    rmm.task_done(1)
    assert rmm.get_state_of(tid) == "UNKNOWN"


def test_untracked_thread_plain_failure(rmm):
    _adaptor(rmm, pool=50)
    from spark_rapids_jni_amd.memory import OffHeapOOM
    with pytest.raises(OffHeapOOM):
        rmm.alloc(100, thread_id=999)


def test_forced_retry_oom(rmm):
    _adaptor(rmm)
    tid = 7
    rmm.start_dedicated_task_thread(tid, 1)
    rmm.force_retry_oom(tid, 1)
    with pytest.raises(GpuRetryOOM):
        rmm.alloc(10, thread_id=tid)
    assert rmm.get_state_of(tid) == "THREAD_BUFN_WAIT"
    m = rmm.get_and_reset_metrics(1)
    assert m["num_retry"] == 1
    # after the rollback the thread blocks, then task_done-driven progress
    # (single thread: deadlock-escalates straight to split)
    rmm.force_split_and_retry_oom(tid, 1)
    with pytest.raises(GpuSplitAndRetryOOM):
        rmm.alloc(10, thread_id=tid)
    rmm.task_done(1)


def test_blocked_thread_woken_by_free(rmm):
    _adaptor(rmm, pool=100)
    t1, t2 = 11, 12
    rmm.start_dedicated_task_thread(t1, 1)
    rmm.start_dedicated_task_thread(t2, 2)
    rmm.alloc(80, thread_id=t1)
    results = {}

    def blocked_alloc():
        rmm.alloc(60, thread_id=t2)
        results["ok"] = True

    th = threading.Thread(target=blocked_alloc)
    th.start()
    deadline = time.time() + 5
    while rmm.get_state_of(t2) != "THREAD_BLOCKED" and time.time() < deadline:
        time.sleep(0.01)
    assert rmm.get_state_of(t2) == "THREAD_BLOCKED"
    rmm.dealloc(80, thread_id=t1)
    th.join(timeout=5)
    assert results.get("ok")
    m2 = rmm.get_and_reset_metrics(2)
    assert m2["block_time_ns"] > 0


def test_deadlock_rolls_back_lowest_priority(rmm):
    _adaptor(rmm, pool=100)
    t1, t2 = 21, 22
    rmm.start_dedicated_task_thread(t1, 1)   # registered first = higher prio
    rmm.start_dedicated_task_thread(t2, 2)
    rmm.alloc(50, thread_id=t1)
    rmm.alloc(40, thread_id=t2)
    errors = {}

    def try_alloc(tid, amount):
        try:
            rmm.alloc(amount, thread_id=tid)
            errors[tid] = None
        except Exception as e:
            errors[tid] = e

    th1 = threading.Thread(target=try_alloc, args=(t1, 40))
    th1.start()
    deadline = time.time() + 5
    while rmm.get_state_of(t1) != "THREAD_BLOCKED" and time.time() < deadline:
        time.sleep(0.01)
    # t1 blocked; t2 now also blocks -> deadlock -> t2 (lowest priority)
    # must be rolled back with GpuRetryOOM
    th2 = threading.Thread(target=try_alloc, args=(t2, 40))
    th2.start()
    th2.join(timeout=5)
    assert isinstance(errors.get(t2), GpuRetryOOM)
    # t2 rolls back: frees its memory; t1 should now proceed
    rmm.dealloc(40, thread_id=t2)
    th1.join(timeout=5)
    assert errors.get(t1, "unset") is None
    rmm.task_done(1)
    rmm.task_done(2)


def test_all_bufn_escalates_to_split(rmm):
    _adaptor(rmm, pool=100)
    tid = 31
    rmm.start_dedicated_task_thread(tid, 1)
    rmm.alloc(90, thread_id=tid)
    result = {}

    def worker():
        try:
            rmm.alloc(50, thread_id=tid)  # cannot fit -> only thread -> retry
        except GpuRetryOOM:
            try:
                # roll back (nothing actually freed) then block until ready:
                # still the only thread -> all BUFN -> split
                rmm.block_thread_until_ready(thread_id=tid)
                rmm.alloc(50, thread_id=tid)
            except GpuSplitAndRetryOOM:
                result["split"] = True
                return
        result["split"] = False

    th = threading.Thread(target=worker)
    th.start()
    th.join(timeout=5)
    assert result.get("split") is True


def test_task_metrics_footprint(rmm):
    _adaptor(rmm)
    tid = 41
    rmm.start_dedicated_task_thread(tid, 5)
    rmm.alloc(300, thread_id=tid)
    rmm.alloc(200, thread_id=tid)
    rmm.dealloc(100, thread_id=tid)
    m = rmm.get_and_reset_metrics(5)
    assert m["max_memory"] == 500
    assert m["current_memory"] == 400


def test_spill_range_excluded_from_footprint(rmm):
    _adaptor(rmm, pool=-1)
    tid = 51
    rmm.start_dedicated_task_thread(tid, 6)
    rmm.alloc(100, thread_id=tid)
    rmm.adaptor().spill_range_start(tid)
    rmm.alloc(1000, thread_id=tid)   # spill buffer: not counted
    rmm.dealloc(1000, thread_id=tid)
    rmm.adaptor().spill_range_done(tid)
    m = rmm.get_and_reset_metrics(6)
    assert m["max_memory"] == 100


def test_task_priority_monotonic():
    a = TaskPriority.get_task_priority(1001)
    b = TaskPriority.get_task_priority(1002)
    assert b > a
    assert TaskPriority.get_task_priority(1001) == a
    TaskPriority.task_done(1001)


def test_monte_carlo_no_livelock(rmm):
    """Mini port of RmmSparkMonteCarlo.java: concurrent tasks doing random
    alloc/free against a small pool must all complete."""
    _adaptor(rmm, pool=3000)
    nthreads = 8
    iters = 40
    held_cap = 600  # per-task working set cap (taskMaxMiB analog)
    done = []
    lock = threading.Lock()

    def task(task_id):
        tid = 1000 + task_id
        rmm.start_dedicated_task_thread(tid, task_id)
        rng = random.Random(task_id)
        held = []
        try:
            for _ in range(iters):
                attempts = 0
                size = rng.randint(10, 500)
                while held and sum(held) + size > held_cap:
                    rmm.dealloc(held.pop(), thread_id=tid)
                while True:
                    try:
                        rmm.alloc(size, thread_id=tid)
                        held.append(size)
                        break
                    except GpuRetryOOM:
                        # spill: free everything, block until ready, retry
                        for s in held:
                            rmm.dealloc(s, thread_id=tid)
                        held.clear()
                        try:
                            rmm.block_thread_until_ready(thread_id=tid)
                        except GpuSplitAndRetryOOM:
                            size = max(10, size // 2)
                    except GpuSplitAndRetryOOM:
                        # split point: input is spillable here too — release
                        # held buffers before retrying at half size
                        for s in held:
                            rmm.dealloc(s, thread_id=tid)
                        held.clear()
                        size = max(10, size // 2)
                    attempts += 1
                    assert attempts < 1000, "livelock"
                if rng.random() < 0.5 and held:
                    rmm.dealloc(held.pop(), thread_id=tid)
        finally:
            for s in held:
                rmm.dealloc(s, thread_id=tid)
            rmm.task_done(task_id)
        with lock:
            done.append(task_id)

    threads = [threading.Thread(target=task, args=(i,)) for i in range(nthreads)]
    for th in threads:
        th.start()
    for th in threads:
        th.join(timeout=30)
    assert len(done) == nthreads, f"only {len(done)}/{nthreads} finished"


def test_monte_carlo_cli_harness():
    from spark_rapids_jni_amd.memory import RmmSpark
    from spark_rapids_jni_amd.tools.rmm_monte_carlo import run
    RmmSpark.clear_event_handler()
    stats = run(6, 30, 2 * 2**20, 2**20 // 2, skewed=True)
    assert stats["done"] == 6 and stats["failed"] == 0


def test_with_retry_spills_then_succeeds(rmm):
    from spark_rapids_jni_amd.memory import with_retry
    _adaptor(rmm)
    tid = 41
    rmm.start_dedicated_task_thread(tid, 9)
    rmm.force_retry_oom(tid, 1)
    spills = []
    halves = []

    def alloc():
        rmm.alloc(10, thread_id=tid)
        return "ok"

    # single thread: the retry blocks, the deadlock-breaker escalates the
    # next attempt to split-and-retry — with_retry rides both transitions
    got = with_retry(alloc, spill=lambda: spills.append(1),
                     split=lambda: halves.append(1), thread_id=tid)
    assert got == "ok"
    assert len(spills) >= 1
    rmm.task_done(9)


def test_with_retry_split_path(rmm):
    from spark_rapids_jni_amd.memory import with_retry
    _adaptor(rmm)
    tid = 43
    rmm.start_dedicated_task_thread(tid, 11)
    rmm.force_split_and_retry_oom(tid, 1)
    halves = []

    def alloc():
        rmm.alloc(10, thread_id=tid)
        return len(halves)

    got = with_retry(alloc, split=lambda: halves.append(1), thread_id=tid)
    assert got == 1 and len(halves) == 1
    rmm.task_done(11)


def test_shuffle_thread_survives_deadlock(rmm):
    """Reference testShuffleBlocking semantics: shuffle pool threads get the
    HIGHEST priority (priority -1), so in a deadlock the dedicated task
    thread rolls back, never the shuffle thread."""
    _adaptor(rmm, pool=100)
    t_task, t_shuf = 31, 32
    rmm.start_dedicated_task_thread(t_task, 7)
    rmm.adaptor().pool_thread_working_on_tasks(t_shuf, [7], True)
    rmm.alloc(50, thread_id=t_shuf)
    rmm.alloc(40, thread_id=t_task)
    errors = {}

    def try_alloc(tid, amount):
        try:
            rmm.alloc(amount, thread_id=tid)
            errors[tid] = None
        except Exception as e:
            errors[tid] = e

    th_s = threading.Thread(target=try_alloc, args=(t_shuf, 40))
    th_s.start()
    deadline = time.time() + 5
    while rmm.get_state_of(t_shuf) != "THREAD_BLOCKED" and \
            time.time() < deadline:
        time.sleep(0.01)
    th_t = threading.Thread(target=try_alloc, args=(t_task, 40))
    th_t.start()
    th_t.join(timeout=5)
    # the DEDICATED thread (priority >= 0) is rolled back; shuffle survives
    assert isinstance(errors.get(t_task), GpuRetryOOM)
    rmm.dealloc(40, thread_id=t_task)
    th_s.join(timeout=5)
    assert errors.get(t_shuf, "unset") is None
    rmm.dealloc(40, thread_id=t_shuf)
    rmm.dealloc(50, thread_id=t_shuf)
    rmm.adaptor().pool_thread_finished_for_tasks(t_shuf, [7])
    rmm.task_done(7)
