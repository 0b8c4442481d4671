"""Multi-process (gloo, world_size 2) shuffle-exchange tests — the CPU tier
for the RCCL all-to-all path (same code, different backend)."""
import os

import pytest
import torch
import torch.multiprocessing as mp


def _worker(rank, world, port, q):
    import torch.distributed as dist
    try:
        dist.init_process_group(
            "gloo", init_method=f"tcp://127.0.0.1:{port}", rank=rank,
            world_size=world)
        from spark_rapids_jni_amd.columnar import Column, DType
        from spark_rapids_jni_amd.parallel.exchange import shuffle_table_host
        from spark_rapids_jni_amd.utils import sparkref

        n = 200
        vals = [rank * 10000 + i for i in range(n)]
        strs = [None if i % 5 == 3 else f"r{rank}v{i}" for i in range(n)]
        cols = [Column.from_pylist(vals, DType.INT64),
                Column.from_pylist(strs, DType.STRING)]
        # Spark partitioning: pmod(murmur3(key), world)
        pids = [sparkref.murmur3_row([v], [DType.INT64]) % world
                for v in vals]
        merged = shuffle_table_host(cols, pids)
        got_keys = merged[0].to_pylist()
        got_strs = merged[1].to_pylist()
        # every received key must belong to this rank
        for k in got_keys:
            assert sparkref.murmur3_row([k], [DType.INT64]) % world == rank
        q.put((rank, sorted(got_keys),
               sorted(s for s in got_strs if s is not None)))
        dist.destroy_process_group()
    except Exception as e:  # pragma: no cover
        q.put((rank, "ERROR", repr(e)))


def test_gloo_shuffle_exchange_world2():
    world = 2
    port = 29713
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_worker, args=(r, world, port, q))
             for r in range(world)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(world):
        rank, keys, strs = q.get(timeout=120)
        assert keys != "ERROR", strs
        results[rank] = (keys, strs)
    for p in procs:
        p.join(timeout=30)
    # conservation: all keys appear exactly once across ranks
    all_keys = sorted(results[0][0] + results[1][0])
    expect = sorted([r * 10000 + i for r in range(world) for i in range(200)])
    assert all_keys == expect


def test_bench_world2_dry_run(tmp_path):
    """`torchrun --nproc-per-node 2 bench.py --gpus 2`
    must reach the timed loop on a CPU/gloo dry run (rendezvous, build
    exchange, per-step all-to-all, max-over-ranks timing, JSON output)."""
    import json
    import os
    import subprocess
    import sys
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    env = dict(os.environ)
    env.update({"MASTER_ADDR": "127.0.0.1", "MASTER_PORT": "29533",
                "WORLD_SIZE": "2"})
    procs = []
    for r in range(2):
        e = dict(env)
        e["RANK"] = str(r)
        e["LOCAL_RANK"] = str(r)
        procs.append(subprocess.Popen(
            [sys.executable, os.path.join(repo, "bench.py"), "--gpus", "2",
             "--dry-run", "--steps", "1", "--warmup", "1",
             "--build-rows", "20000", "--probe-rows", "40000",
             "--chunk-rows", "20000"],
            env=e, stdout=subprocess.PIPE, stderr=subprocess.PIPE,
            cwd=repo))
    outs = []
    for p in procs:
        so, se = p.communicate(timeout=300)
        assert p.returncode == 0, se.decode()[-2000:]
        outs.append(so.decode())
    js = [l for l in outs[0].splitlines() if l.startswith("{")]
    assert js, outs[0]
    d = json.loads(js[-1])
    assert d["n_gpus"] == 2 and d["dry_run"] is True
    # probe keys cover the whole build key space: every row matches
    assert d["config"]["matches"] > 0
