#!/usr/bin/env python3
"""Flagship benchmark: hash-join build+probe on MI355X (BASELINE config[2]:
"hash-join build+probe 10B x 1B int64 keys, 1xMI355X"), scaled weakly over N
GPUs with the Spark-style shuffle (murmur3 hash partition -> RCCL all-to-all
over xGMI -> local probe), one rank per GPU.

One step (per rank):
  * build a hash table from the rank's ~1B-key build shard
  * for each of `chunks` probe chunks (1B rows each):
      - murmur3-hash the keys, pmod-partition into WORLD_SIZE buckets,
        gather + all_to_all_single over RCCL (skipped at N=1)
      - probe the local table, materializing the inner-join gather maps
        (int32 build idx + int64 probe idx)
Metric: probe rows/s over the whole job (all ranks), max step time over ranks.

Run: python bench.py [--gpus N] [--steps K] [--warmup W]
Multi-GPU: torchrun --nproc-per-node N bench.py --gpus N (driver contract).
"""
import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

# large-SF NDS runs allocate/free many odd-sized tensors; expandable
# segments avoid the fragmentation OOMs the default allocator hits
os.environ.setdefault("PYTORCH_ALLOC_CONF", "expandable_segments:True")

import torch

BUILD_ROWS_DEFAULT = 1_000_000_000
PROBE_ROWS_DEFAULT = 10_000_000_000
CHUNK_ROWS = 1_000_000_000


def log(rank, *a):
    if rank == 0:
        print(*a, file=sys.stderr, flush=True)


def setup_dist(args):
    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    if getattr(args, "dry_run", False):
        # CPU/gloo dry run: exercises the full world>1 control flow
        # (rendezvous, hash partition, all-to-all, max-over-ranks timing)
        # without a GPU — VERDICT r01 item 2 readiness check
        if world > 1:
            import torch.distributed as dist
            dist.init_process_group("gloo")
        return world, rank, local_rank
    if world > 1:
        import torch.distributed as dist
        torch.cuda.set_device(local_rank)
        dist.init_process_group("nccl")  # = RCCL on ROCm
    else:
        torch.cuda.set_device(local_rank)
    return world, rank, local_rank


def barrier_sync(world, dry_run=False):
    if world > 1:
        import torch.distributed as dist
        dist.barrier()
    if not dry_run:
        torch.cuda.synchronize()


def shuffle_exchange_dry(keys: torch.Tensor, world: int):
    """gloo/CPU stand-in for shuffle_exchange: same partition->a2a control
    flow, torch-side hash (the HIP murmur3 kernel needs a GPU)."""
    import torch.distributed as dist
    h = (keys * 0x9E3779B97F4A7C15) & (2**63 - 1)
    pids = (h % world).to(torch.int64)
    order = torch.argsort(pids, stable=True)
    send = keys[order]
    counts = torch.bincount(pids, minlength=world)
    recv_counts = torch.empty_like(counts)
    gathered = [torch.zeros_like(counts) for _ in range(world)]
    dist.all_gather(gathered, counts)
    me = dist.get_rank()
    recv_counts = torch.tensor([int(g[me]) for g in gathered])
    out = torch.empty(int(recv_counts.sum()), dtype=keys.dtype)
    in_splits = counts.tolist()
    out_splits = recv_counts.tolist()
    from spark_rapids_jni_amd.parallel.exchange import _exchange_bytes
    b, _os = _exchange_bytes(send.view(torch.uint8).view(-1),
                             [s * 8 for s in in_splits])
    return b.view(torch.int64)


def shuffle_exchange(keys: torch.Tensor, world: int):
    """Spark shuffle write+exchange: murmur3 -> pmod partition -> RCCL a2a."""
    from spark_rapids_jni_amd.columnar import Column
    from spark_rapids_jni_amd.ops import hashing
    from spark_rapids_jni_amd.ops.copying import partition_map, spark_partition_ids
    import torch.distributed as dist

    from spark_rapids_jni_amd.ops.copying import gather_column
    col = Column.from_torch(keys)
    h = hashing.murmur3([col])
    pids = spark_partition_ids(h, world)
    offsets, perm = partition_map(pids, world)
    send = gather_column(col, perm, has_nulls=False).data  # partition-contiguous
    counts = offsets.diff()
    recv_counts = torch.empty_like(counts)
    dist.all_to_all_single(recv_counts, counts)
    in_splits = counts.cpu().tolist()
    out_splits = recv_counts.cpu().tolist()
    out = torch.empty(sum(out_splits), dtype=keys.dtype, device=keys.device)
    dist.all_to_all_single(out, send, out_splits, in_splits)
    return out


def make_build_shard(n_build: int, world: int, rank: int, device,
                     dry_run=False):
    """Rank r's build shard: global key space [0, world*n_build) routed to the
    owning rank by pmod(murmur3), like Spark's build-side shuffle (one-time)."""
    local = torch.arange(rank * n_build, (rank + 1) * n_build, dtype=torch.int64,
                         device=device)
    if world == 1:
        return local
    if dry_run:
        return shuffle_exchange_dry(local, world)
    return shuffle_exchange(local, world)


def one_step_dry(build, chunks, world):
    """CPU stand-in for one_step: exchange + torch-side join probe count
    (control flow identical; HIP kernels need a GPU)."""
    bset = torch.zeros(1, dtype=torch.int64)
    for chunk in chunks:
        probe = shuffle_exchange_dry(chunk, world) if world > 1 else chunk
        bset += torch.isin(probe, build).sum()
    return int(bset.item())


def one_step(build_col, chunks, world, out_hint, comm_stream=None):
    from spark_rapids_jni_amd.columnar import Column
    from spark_rapids_jni_amd.ops.join import HashJoinTable

    # rebuild the table each step (config is build+probe)
    tbl = HashJoinTable.build(build_col)
    if world == 1:
        for chunk in chunks:
            pcol = Column.from_torch(chunk)
            bi, pi = tbl.inner_join(pcol, out_hint=chunk.numel() + out_hint)
            del pcol, bi, pi
        del tbl
        return

    # world > 1: run the shuffle of chunk i+1 on a side stream while chunk i
    # probes on the default stream (xGMI all-to-all overlaps with compute)
    main = torch.cuda.current_stream()

    def start_exchange(chunk):
        with torch.cuda.stream(comm_stream):
            probe = shuffle_exchange(chunk, world)
            ev = torch.cuda.Event()
            ev.record(comm_stream)
        return probe, ev

    nxt = start_exchange(chunks[0])
    for i in range(len(chunks)):
        probe, ev = nxt
        if i + 1 < len(chunks):
            nxt = start_exchange(chunks[i + 1])
        main.wait_event(ev)
        probe.record_stream(main)
        pcol = Column.from_torch(probe)
        bi, pi = tbl.inner_join(pcol, out_hint=probe.numel() + out_hint)
        del probe, pcol, bi, pi
    del tbl


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=3)
    ap.add_argument("--warmup", type=int, default=1)
    ap.add_argument("--build-rows", type=int, default=BUILD_ROWS_DEFAULT,
                    help="build rows per GPU (default = named config)")
    ap.add_argument("--probe-rows", type=int, default=PROBE_ROWS_DEFAULT,
                    help="probe rows per GPU per step (default = named config)")
    ap.add_argument("--chunk-rows", type=int, default=CHUNK_ROWS)
    ap.add_argument("--op", choices=["join", "groupby", "q3", "parquet",
                                     "nds"],
                    default="join",
                    help="flagship join (default), BASELINE config[1] "
                         "group-by, the NDS q3-shaped pipeline, the "
                         "config[3] parquet scan->filter->project, or the "
                         "config[4] NDS q1-q99 power run")
    ap.add_argument("--sf", type=float, default=10.0,
                    help="scale factor for --op nds")
    ap.add_argument("--queries", type=str, default="",
                    help="comma-separated query numbers for --op nds "
                         "(default all 99)")
    ap.add_argument("--dry-run", action="store_true",
                    help="CPU/gloo dry run of the world>1 control flow "
                         "(no GPU needed; readiness check only)")
    ap.add_argument("--groups", type=int, default=1_000_000,
                    help="distinct groups for --op groupby")
    ap.add_argument("--codec", choices=["none", "snappy", "zstd", "gzip"],
                    default="none", help="parquet codec for --op parquet")
    args = ap.parse_args()

    world, rank, local_rank = setup_dist(args)
    if args.op == "nds":
        return run_nds(args, world, rank, local_rank)
    if args.op != "join":
        return run_secondary(args, world, rank, local_rank)
    if args.gpus > 1 and world != args.gpus:
        raise RuntimeError(
            f"--gpus {args.gpus} needs torchrun with --nproc-per-node {args.gpus} "
            f"(WORLD_SIZE is {world})")
    if args.dry_run:
        return run_join_dry(args, world, rank)
    device = torch.device("cuda", local_rank)

    free, total = torch.cuda.mem_get_info(device)
    need = (args.build_rows * 84              # build shard + 25%-load slot table
            + args.probe_rows * 8             # resident probe chunks
            + args.chunk_rows * (48 if args.gpus > 1 else 0)  # 2 in-flight shuffles
            + args.chunk_rows * 13            # join output maps
            ) * 1.15
    if need > free:
        raise MemoryError(
            f"config needs ~{need/2**30:.0f} GiB, only {free/2**30:.0f} GiB free "
            "on this GPU — refusing to run a reduced config silently")

    from spark_rapids_jni_amd.columnar import Column

    log(rank, f"[bench] generating build shard ({args.build_rows} rows/GPU)")
    build = make_build_shard(args.build_rows, world, rank, device)
    build_col = Column.from_torch(build)
    key_space = world * args.build_rows
    # probe rows landing on this rank can exceed the chunk a bit at world>1
    out_hint = 0 if world == 1 else args.chunk_rows // 16

    # pre-generate probe chunks (synthetic data stays fixed across steps;
    # RNG cost is outside the timed region)
    chunks = []
    done = 0
    while done < args.probe_rows:
        m = min(args.chunk_rows, args.probe_rows - done)
        chunks.append(torch.randint(0, key_space, (m,), dtype=torch.int64,
                                    device=device))
        done += m

    comm_stream = torch.cuda.Stream(device=device) if world > 1 else None
    log(rank, f"[bench] warmup {args.warmup} steps")
    for _ in range(args.warmup):
        one_step(build_col, chunks, world, out_hint, comm_stream)
    barrier_sync(world)

    log(rank, f"[bench] timing {args.steps} steps")
    t0 = time.perf_counter()
    for _ in range(args.steps):
        one_step(build_col, chunks, world, out_hint, comm_stream)
    barrier_sync(world)
    elapsed = time.perf_counter() - t0

    # max over ranks
    if world > 1:
        import torch.distributed as dist
        t = torch.tensor([elapsed], dtype=torch.float64, device=device)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    ms_per_step = elapsed / args.steps * 1000.0
    total_rows = world * args.probe_rows * args.steps
    value = total_rows / elapsed
    if rank == 0:
        print(json.dumps({
            "metric": "hash_join_probe_rows_per_sec",
            "value": value,
            "unit": "rows/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_per_step,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "int64",
            "data": "synthetic",
            "config": {
                "model": "hash-join build+probe 10B x 1B int64 keys",
                "build_rows_per_gpu": args.build_rows,
                "probe_rows_per_gpu": args.probe_rows,
                "global_batch": world * args.probe_rows,
                "seq_len": 0,
                "parallelism": f"1 executor/GPU x{world}, RCCL all-to-all shuffle",
            },
        }))
    if world > 1:
        import torch.distributed as dist
        dist.destroy_process_group()


def run_join_dry(args, world, rank):
    """World>1 readiness dry run (gloo/CPU): same phases as the real
    bench — build-shard exchange, warmup, timed loop with barriers, max
    elapsed over ranks, one JSON line — with torch stand-ins for the HIP
    kernels."""
    build = make_build_shard(args.build_rows, world, rank, "cpu",
                             dry_run=True)
    chunks = []
    done = 0
    key_space = world * args.build_rows
    g = torch.Generator().manual_seed(11 + rank)
    while done < args.probe_rows:
        m = min(args.chunk_rows, args.probe_rows - done)
        chunks.append(torch.randint(0, key_space, (m,), dtype=torch.int64,
                                    generator=g))
        done += m
    for _ in range(args.warmup):
        one_step_dry(build, chunks, world)
    barrier_sync(world, dry_run=True)
    t0 = time.perf_counter()
    matches = 0
    for _ in range(args.steps):
        matches = one_step_dry(build, chunks, world)
    barrier_sync(world, dry_run=True)
    elapsed = time.perf_counter() - t0
    if world > 1:
        import torch.distributed as dist
        t = torch.tensor([elapsed], dtype=torch.float64)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())
    if rank == 0:
        print(json.dumps({
            "metric": "hash_join_probe_rows_per_sec", "dry_run": True,
            "value": world * args.probe_rows * args.steps / elapsed,
            "unit": "rows/s", "n_gpus": world, "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": elapsed / args.steps * 1000.0,
            "higher_is_better": True, "scaling": "weak",
            "vs_baseline": None, "dtype": "int64",
            "data": "synthetic (CPU/gloo DRY RUN — not a measurement)",
            "config": {"model": "hash-join dry run", "matches": matches,
                       "global_batch": world * args.probe_rows,
                       "seq_len": 0,
                       "parallelism": f"x{world} gloo dry-run"},
        }))
    if world > 1:
        import torch.distributed as dist
        dist.destroy_process_group()


def run_nds(args, world, rank, local_rank):
    """BASELINE config[4]: NDS q1-q99 power run. One step = one full power
    run of all (or --queries selected) queries; data is generated once per
    rank (facts sharded, dims replicated) and stays GPU-resident. Fixed SF
    split over N ranks = strong scaling."""
    device = torch.device("cuda", local_rank)
    from spark_rapids_jni_amd.nds import runner as nds_runner
    qs = ([int(x) for x in args.queries.split(",") if x] or None)
    log(rank, f"[bench] generating NDS catalog sf={args.sf} "
              f"(world={world}, rank={rank})")
    eng = nds_runner.make_engine(args.sf, device=f"cuda:{local_rank}",
                                 world=world, rank=rank)
    log(rank, "[bench] warmup")
    for _ in range(args.warmup):
        nds_runner.power_run(eng, qs, quiet=True)
    barrier_sync(world)
    t0 = time.perf_counter()
    per = None
    for _ in range(args.steps):
        per = nds_runner.power_run(eng, qs, quiet=True)
    barrier_sync(world)
    elapsed = time.perf_counter() - t0
    if world > 1:
        import torch.distributed as dist
        t = torch.tensor([elapsed], dtype=torch.float64, device=device)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())
    nq = per["n_queries"] if per else 0
    if rank == 0:
        print(json.dumps({
            "metric": "nds_power_run_seconds",
            "value": elapsed / args.steps,
            "unit": "s",
            "n_gpus": world, "steps": args.steps, "warmup": args.warmup,
            "ms_per_step": elapsed / args.steps * 1000.0,
            "higher_is_better": False,
            "scaling": "strong",
            "vs_baseline": None,
            "dtype": "int64+fp64",
            "data": "synthetic",
            "config": {"model": f"NDS q1-q99 power run (synthetic "
                                f"TPC-DS-shaped, SF{args.sf:g})",
                       "n_queries": nq,
                       "global_batch": 0, "seq_len": 0,
                       "parallelism": f"1 executor/GPU x{world}, "
                                      "RCCL exchange",
                       "per_query": per["queries"] if per else {}},
        }))
    if world > 1:
        import torch.distributed as dist
        dist.destroy_process_group()


def _write_store_sales_file(job):
    """Worker: write one store_sales-shaped parquet file (seeded by index)."""
    path, idx, rows, codec = job
    import numpy as np
    import pyarrow as pa
    import pyarrow.parquet as papq
    rng = np.random.default_rng(1000 + idx)
    tbl = pa.table({
        "ss_sold_date_sk": rng.integers(2450000, 2452555, rows,
                                        dtype=np.int64),
        "ss_item_sk": rng.integers(1, 300_000, rows, dtype=np.int64),
        "ss_quantity": rng.integers(1, 100, rows, dtype=np.int64),
        "ss_sales_price": rng.random(rows) * 100.0,
        "ss_ext_sales_price": rng.random(rows) * 1000.0,
    })
    kw = {"compression": codec, "row_group_size": 8_000_000}
    if codec == "ZSTD":
        kw["compression_level"] = 1
    papq.write_table(tbl, path, **kw)
    return path


def run_secondary(args, world, rank, local_rank):
    """Secondary benches: BASELINE config[1] hash-aggregate and the q3-shaped
    pipeline. Single-GPU metrics (weak-scaled aggregate when launched under
    torchrun)."""
    device = torch.device("cuda", local_rank)
    from spark_rapids_jni_amd.columnar import Column
    if args.op == "groupby":
        n = args.build_rows if args.build_rows != BUILD_ROWS_DEFAULT \
            else 1_000_000_000
        keys = torch.randint(0, args.groups, (n,), dtype=torch.int64,
                             device=device)
        vals = torch.randint(0, 1000, (n,), dtype=torch.int64, device=device)
        kc, vc = Column.from_torch(keys), Column.from_torch(vals)
        from spark_rapids_jni_amd.ops.aggregate import Agg, groupby

        def step():
            kt, res = groupby(kc, [(Agg.SUM, vc), (Agg.COUNT_ALL, None)],
                              num_groups_hint=args.groups * 2)
            del kt, res
        metric, model = "hash_aggregate_rows_per_sec", \
            "hash-aggregate sum/count group-by int64, 1B rows"
        rows_per_step = n
    elif args.op == "parquet":
        # BASELINE config[3]: parquet scan -> filter -> project over a
        # store_sales-shaped dataset. Files are written once (pyarrow,
        # parallel worker processes, outside the timed region) as a
        # multi-file dataset — SF1K (2.88B rows: --probe-rows 2880000000)
        # writes 50M-row files; each step decodes every file on the GPU,
        # filters a date window and projects a price sum, accumulating
        # across files (bounded device memory, Spark-partition style).
        from spark_rapids_jni_amd import parquet as srj_pq
        n = args.probe_rows if args.probe_rows != PROBE_ROWS_DEFAULT \
            else 100_000_000
        codec = {"snappy": "SNAPPY", "zstd": "ZSTD",
                 "gzip": "GZIP"}.get(args.codec, "NONE")
        per_file = min(n, 50_000_000)
        nfiles = (n + per_file - 1) // per_file
        pdir = f"/tmp/bench_store_sales_{rank}_{args.codec}"
        os.makedirs(pdir, exist_ok=True)
        paths = [os.path.join(pdir, f"part-{i:04d}.parquet")
                 for i in range(nfiles)]
        todo = [(p, i, per_file if i < nfiles - 1 or n % per_file == 0
                 else n % per_file, codec)
                for i, p in enumerate(paths) if not os.path.exists(p)]
        if todo:
            log(rank, f"[bench] writing {len(todo)} x {per_file}-row "
                      f"store_sales files ({codec}, parallel)")
            import concurrent.futures
            with concurrent.futures.ProcessPoolExecutor(
                    max_workers=min(16, len(todo))) as ex:
                list(ex.map(_write_store_sales_file, todo))

        def step():
            total = torch.zeros(1, dtype=torch.float64, device=device)
            rows = 0
            for p in paths:
                t = srj_pq.read_table(p, columns=["ss_sold_date_sk",
                                                  "ss_quantity",
                                                  "ss_sales_price"],
                                      device=str(device))
                d = t.columns[0].data
                mask = (d >= 2450500) & (d < 2451500)
                price = t.columns[2].data[mask]
                total += (price * t.columns[1].data[mask]).sum()
                rows += t.num_rows
                del t, mask, price
            assert rows == n, (rows, n)
        metric, model = "parquet_scan_rows_per_sec", \
            "parquet scan->filter->project, store_sales-shaped " \
            f"({n} rows, 5 cols, {args.codec}, {nfiles} files)"
        rows_per_step = n
    else:  # q3
        from spark_rapids_jni_amd import exec as ex
        n = args.probe_rows if args.probe_rows != PROBE_ROWS_DEFAULT \
            else 1_000_000_000
        ss = ex.gen_store_sales(n, device=device)
        dd = ex.gen_date_dim(device=device)
        it = ex.gen_item(device=device)

        def step():
            keys, res = ex.q3_like(ss, dd, it, year=2000, manufact_id=50)
            del keys, res
        metric, model = "nds_q3_rows_per_sec", \
            "q3-shape scan->filter->join x2->groupby over store_sales"
        rows_per_step = n

    for _ in range(args.warmup):
        step()
    barrier_sync(world)
    t0 = time.perf_counter()
    for _ in range(args.steps):
        step()
    barrier_sync(world)
    elapsed = time.perf_counter() - t0
    if world > 1:
        import torch.distributed as dist
        t = torch.tensor([elapsed], dtype=torch.float64, device=device)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())
    if rank == 0:
        print(json.dumps({
            "metric": metric,
            "value": world * rows_per_step * args.steps / elapsed,
            "unit": "rows/s", "n_gpus": world, "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": elapsed / args.steps * 1000.0,
            "higher_is_better": True, "scaling": "weak", "vs_baseline": None,
            "dtype": "int64", "data": "synthetic",
            "config": {"model": model, "global_batch": world * rows_per_step,
                       "seq_len": 0, "parallelism": f"x{world}"},
        }))
    if world > 1:
        import torch.distributed as dist
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
