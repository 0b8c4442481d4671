"""Memory/resource layer: RmmSpark facade + typed OOM exceptions.

Java API parity (reference): RmmSpark.java (static facade), GpuRetryOOM /
GpuSplitAndRetryOOM / CpuRetryOOM / CpuSplitAndRetryOOM / GpuOOM exception
hierarchy, SparkResourceAdaptor.java + SparkResourceAdaptorJni.cpp state
machine (see src/host/resource_adaptor.cpp and docs/memory_management.md).

The native core exposes two integration modes:
  * simulated pool (`alloc`/`dealloc`): full state machine against a byte
    budget — used by tests, the Monte Carlo stress harness, and host-memory
    tracking;
  * external hooks (`pre_alloc_external`/...): bracket real device
    allocations made through torch's caching allocator.
"""
import threading
from typing import Optional

from . import _native


class GpuOOM(RuntimeError):
    """Base of GPU OOM errors (reference GpuOOM.java)."""


class GpuRetryOOM(GpuOOM):
    """Roll back to a spillable state and retry (reference GpuRetryOOM.java:17)."""


class GpuSplitAndRetryOOM(GpuOOM):
    """Split the input and retry (reference GpuSplitAndRetryOOM.java)."""


class CpuRetryOOM(RuntimeError):
    pass


class CpuSplitAndRetryOOM(RuntimeError):
    pass


class OffHeapOOM(RuntimeError):
    pass


class ThreadRemovedError(RuntimeError):
    """Thread was unregistered while blocked (REMOVE_THROW)."""


_CODE_TO_EXC = {
    1: GpuRetryOOM,
    2: GpuSplitAndRetryOOM,
    3: CpuRetryOOM,
    4: CpuSplitAndRetryOOM,
    5: ThreadRemovedError,
    6: OffHeapOOM,
}


def _raise_for(code: int, what: str = "allocation"):
    if code == 0:
        return
    exc = _CODE_TO_EXC.get(code, RuntimeError)
    raise exc(f"{what} -> {exc.__name__}")


class TaskPriority:
    """Monotonic per-task-attempt priority (reference task_priority.hpp:26,
    TaskPriority.java:28). Lower number = higher priority."""
    _lock = threading.Lock()
    _next = 0
    _assigned = {}

    @classmethod
    def get_task_priority(cls, task_attempt_id: int) -> int:
        with cls._lock:
            if task_attempt_id not in cls._assigned:
                cls._assigned[task_attempt_id] = cls._next
                cls._next += 1
            return cls._assigned[task_attempt_id]

    @classmethod
    def task_done(cls, task_attempt_id: int) -> None:
        with cls._lock:
            cls._assigned.pop(task_attempt_id, None)


class RmmSpark:
    """Static facade over the resource adaptor (reference RmmSpark.java)."""

    _adaptor = None
    _lock = threading.Lock()

    # -- lifecycle (reference RmmSpark.setEventHandler / clearEventHandler) --
    @classmethod
    def set_event_handler(cls, pool_limit: int = -1, host_limit: int = -1,
                          real: bool = False):
        """Install the adaptor. With real=True the adaptor also becomes the
        target of the torch pluggable-allocator bridge: every REAL device
        allocation (hipMallocAsync under torch) runs the pre/post state
        machine and `pool_limit` caps actual HBM bytes — the reference's
        RMM-resource wrapping (SparkResourceAdaptorJni.cpp:2113). Requires
        use_real_allocator() to have been called before any CUDA work."""
        with cls._lock:
            if cls._adaptor is not None:
                raise RuntimeError("event handler already set")
            cls._adaptor = _native.host().SparkResourceAdaptor(
                -1 if real else pool_limit, host_limit)
            if real:
                host = _native.host()
                gpu = _native.gpu()
                host.install_as_current(cls._adaptor)
                gpu.install_ra_hooks(*host.hook_addrs())
                gpu.set_device_pool_limit(pool_limit)
            return cls._adaptor

    @classmethod
    def clear_event_handler(cls):
        with cls._lock:
            if cls._adaptor is not None:
                try:
                    _native.host().clear_current()
                    g = _native.gpu_or_none()
                    if g is not None:
                        g.clear_ra_hooks()
                        g.set_device_pool_limit(-1)
                except Exception:
                    pass
            cls._adaptor = None

    _real_allocator = None

    @classmethod
    def use_real_allocator(cls):
        """Route ALL torch device allocations through srj_torch_malloc/free
        (src/gpu/torch_alloc.hip). Must run before the first CUDA
        allocation in the process; typically paired with
        set_event_handler(real=True)."""
        if cls._real_allocator is not None:
            return
        import os
        import torch
        gpu = _native.gpu()  # ensure the .so is importable
        path = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                            "_gpu.so")
        alloc = torch.cuda.memory.CUDAPluggableAllocator(
            path, "srj_torch_malloc", "srj_torch_free")
        torch.cuda.memory.change_current_allocator(alloc)
        cls._real_allocator = alloc

    @classmethod
    def adaptor(cls):
        if cls._adaptor is None:
            raise RuntimeError("RmmSpark.set_event_handler was not called")
        return cls._adaptor

    @staticmethod
    def current_thread_id() -> int:
        return threading.get_ident()

    # -- thread/task association --------------------------------------------
    @classmethod
    def start_dedicated_task_thread(cls, thread_id: int, task_id: int):
        cls.adaptor().start_dedicated_task_thread(thread_id, task_id)

    @classmethod
    def current_thread_is_dedicated_to_task(cls, task_id: int):
        cls.start_dedicated_task_thread(cls.current_thread_id(), task_id)

    @classmethod
    def shuffle_thread_working_on_tasks(cls, task_ids):
        cls.adaptor().pool_thread_working_on_tasks(cls.current_thread_id(),
                                                   list(task_ids), True)

    @classmethod
    def pool_thread_working_on_tasks(cls, is_for_shuffle, task_ids,
                                     thread_id: Optional[int] = None):
        cls.adaptor().pool_thread_working_on_tasks(
            thread_id if thread_id is not None else cls.current_thread_id(),
            list(task_ids), bool(is_for_shuffle))

    @classmethod
    def pool_thread_finished_for_tasks(cls, task_ids,
                                       thread_id: Optional[int] = None):
        cls.adaptor().pool_thread_finished_for_tasks(
            thread_id if thread_id is not None else cls.current_thread_id(),
            list(task_ids))

    @classmethod
    def remove_current_thread_association(cls):
        cls.adaptor().remove_thread_association(cls.current_thread_id())

    @classmethod
    def task_done(cls, task_id: int):
        cls.adaptor().task_done(task_id)
        TaskPriority.task_done(task_id)

    # -- allocation (simulated pool; raises the typed OOMs) ------------------
    @classmethod
    def alloc(cls, nbytes: int, thread_id: Optional[int] = None):
        tid = thread_id if thread_id is not None else cls.current_thread_id()
        _raise_for(cls.adaptor().alloc_sim(tid, nbytes), f"alloc({nbytes})")

    @classmethod
    def dealloc(cls, nbytes: int, thread_id: Optional[int] = None):
        tid = thread_id if thread_id is not None else cls.current_thread_id()
        cls.adaptor().dealloc_sim(tid, nbytes)

    @classmethod
    def block_thread_until_ready(cls, thread_id: Optional[int] = None):
        tid = thread_id if thread_id is not None else cls.current_thread_id()
        _raise_for(cls.adaptor().block_thread_until_ready(tid),
                   "block_thread_until_ready")

    # -- test/fault injection (reference RmmSpark.forceRetryOOM:534) ---------
    @classmethod
    def force_retry_oom(cls, thread_id: int, num_ooms: int = 1):
        cls.adaptor().force_retry_oom(thread_id, num_ooms)

    @classmethod
    def force_split_and_retry_oom(cls, thread_id: int, num_ooms: int = 1):
        cls.adaptor().force_split_and_retry_oom(thread_id, num_ooms)

    # -- spill ranges (reference RmmSpark.spillRangeStart/Done:867-880) ------
    @classmethod
    def spill_range_start(cls):
        cls.adaptor().spill_range_start(cls.current_thread_id())

    @classmethod
    def spill_range_done(cls):
        cls.adaptor().spill_range_done(cls.current_thread_id())

    # -- metrics (reference RmmSpark.getAndReset*:663-767) -------------------
    @classmethod
    def get_and_reset_metrics(cls, task_id: int) -> dict:
        return cls.adaptor().get_and_reset_metrics(task_id)

    @classmethod
    def get_state_of(cls, thread_id: int) -> str:
        return cls.adaptor().get_state_of(thread_id)


def with_retry(fn, spill=None, split=None, max_retries: int = 16,
               thread_id: Optional[int] = None):
    """Run `fn` under the OOM-retry protocol (the plugin's
    RmmRapidsRetryIterator.withRetry over this library's state machine):

      * GpuRetryOOM: call `spill()` (e.g. SpillManager.spill_until), wait
        via blockThreadUntilReady, re-run `fn`.
      * GpuSplitAndRetryOOM: call `split()` to halve the working set first
        (raises if no split function was provided).
    """
    attempts = 0
    while True:
        try:
            return fn()
        except (GpuSplitAndRetryOOM, GpuRetryOOM, RuntimeError) as e:
            kind = _oom_kind(e)
            if kind == "split":
                if split is None:
                    raise
                attempts += 1
                if attempts > max_retries:
                    raise
                split()
                continue  # split made room; retry immediately
            if kind != "retry":
                raise
            attempts += 1
            if attempts > max_retries:
                raise
            if spill is not None:
                spill()
                # slow path hygiene (the JVM plugin does the same on OOM):
                # break python reference cycles that can briefly pin the
                # spilled device tensors, then flush any allocator cache
                import gc
                gc.collect()
                try:
                    import torch
                    if torch.cuda.is_initialized():
                        torch.cuda.empty_cache()
                except Exception:
                    pass
            try:
                RmmSpark.block_thread_until_ready(thread_id)
            except Exception:
                pass


def _oom_kind(e) -> Optional[str]:
    """Classify both the typed exceptions (simulated pool) and the marker-
    prefixed RuntimeErrors the real-allocator bridge raises through torch
    (torch_alloc.hip throw_code)."""
    if isinstance(e, GpuSplitAndRetryOOM):
        return "split"
    if isinstance(e, GpuRetryOOM):
        return "retry"
    msg = str(e)
    if "GpuSplitAndRetryOOM" in msg:
        return "split"
    if "GpuRetryOOM" in msg:
        return "retry"
    return None
