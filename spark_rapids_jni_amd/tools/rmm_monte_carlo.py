"""Monte-Carlo OOM-retry stress harness (reference RmmSparkMonteCarlo.java:
CLI-configurable concurrent fake tasks doing random alloc/free against a
small pool, asserting no deadlock/livelock and reporting retry overhead).

Run: python -m spark_rapids_jni_amd.tools.rmm_monte_carlo \\
       --tasks 16 --iters 200 --pool-mib 64 --task-max-mib 16 [--skewed]
"""
import argparse
import random
import threading
import time


def run(num_tasks: int, iters: int, pool_bytes: int, task_max_bytes: int,
        skewed: bool = False, seed: int = 1234) -> dict:
    from ..memory import (GpuRetryOOM, GpuSplitAndRetryOOM, RmmSpark)
    RmmSpark.clear_event_handler()
    RmmSpark.set_event_handler(pool_limit=pool_bytes)
    stats = {"retries": 0, "splits": 0, "done": 0, "failed": 0}
    lock = threading.Lock()

    def task(task_id):
        tid = 10_000 + task_id
        RmmSpark.start_dedicated_task_thread(tid, task_id)
        rng = random.Random(seed + task_id)
        held = []
        retries = splits = 0
        try:
            for _ in range(iters):
                hi = task_max_bytes // (8 if skewed and task_id % 4 else 1)
                size = rng.randint(256, max(hi, 512))
                while sum(held) + size > task_max_bytes and held:
                    RmmSpark.dealloc(held.pop(), thread_id=tid)
                attempts = 0
                while True:
                    attempts += 1
                    if attempts > 10_000:
                        raise RuntimeError("livelock")
                    try:
                        RmmSpark.alloc(size, thread_id=tid)
                        held.append(size)
                        break
                    except GpuRetryOOM:
                        retries += 1
                        for s in held:
                            RmmSpark.dealloc(s, thread_id=tid)
                        held.clear()
                        try:
                            RmmSpark.block_thread_until_ready(thread_id=tid)
                        except GpuSplitAndRetryOOM:
                            splits += 1
                            size = max(256, size // 2)
                    except GpuSplitAndRetryOOM:
                        splits += 1
                        for s in held:
                            RmmSpark.dealloc(s, thread_id=tid)
                        held.clear()
                        size = max(256, size // 2)
                if rng.random() < 0.4 and held:
                    RmmSpark.dealloc(held.pop(), thread_id=tid)
            with lock:
                stats["done"] += 1
        except Exception:
            with lock:
                stats["failed"] += 1
            raise
        finally:
            for s in held:
                RmmSpark.dealloc(s, thread_id=tid)
            RmmSpark.task_done(task_id)
            with lock:
                stats["retries"] += retries
                stats["splits"] += splits

    threads = [threading.Thread(target=task, args=(i,))
               for i in range(num_tasks)]
    t0 = time.time()
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    stats["wall_s"] = time.time() - t0
    RmmSpark.clear_event_handler()
    return stats


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--tasks", type=int, default=16)
    ap.add_argument("--iters", type=int, default=200)
    ap.add_argument("--pool-mib", type=float, default=4.0)
    ap.add_argument("--task-max-mib", type=float, default=1.0)
    ap.add_argument("--skewed", action="store_true")
    ap.add_argument("--seed", type=int, default=1234)
    args = ap.parse_args()
    stats = run(args.tasks, args.iters, int(args.pool_mib * 2**20),
                int(args.task_max_mib * 2**20), args.skewed, args.seed)
    print(stats)
    assert stats["failed"] == 0 and stats["done"] == args.tasks, stats


if __name__ == "__main__":
    main()
