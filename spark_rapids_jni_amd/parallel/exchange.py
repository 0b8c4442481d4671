"""Multi-GPU shuffle exchange: RCCL all-to-all over xGMI.

This is the NEW first-class component called out in SURVEY.md §5.8: the
reference ships only the serialization (kudo/shuffle_split); transport lives
in the plugin (netty/UCX over NVLink). Here, with one executor process per
MI355X GPU, the exchange is torch.distributed all_to_all_single over RCCL —
grouped point-to-point send/recv underneath, which is the right primitive for
the 7x ~153 GB/s xGMI point-to-point links (ring collectives would be bound
by a single link).

Flow per shuffle: murmur3(keys) -> pmod partition -> device kudo split ->
all_to_all_single (bytes) -> device kudo assemble. No host bounce.

On CPU (tests), the gloo backend has no all_to_all, so the same byte buffers
move via batched isend/irecv — semantics identical, letting the distributed
path run under multi-process CPU tests (world_size > 1, gloo).
"""
from typing import List, Sequence

import torch
import torch.distributed as dist

from ..columnar import Column, Table
from ..kudo import merge_on_host, write_partition
from .. import shuffle_gpu


def _exchange_bytes(send: torch.Tensor, in_splits: List[int]):
    """all-to-all of a byte buffer with per-rank splits; returns
    (recv buffer, out_splits)."""
    world = dist.get_world_size()
    dev = send.device
    counts = torch.tensor(in_splits, dtype=torch.int64,
                          device=dev if dist.get_backend() == "nccl" else "cpu")
    recv_counts = torch.empty_like(counts)
    if dist.get_backend() == "nccl":
        dist.all_to_all_single(recv_counts, counts)
    else:
        # gloo: exchange counts via all_gather
        gathered = [torch.zeros_like(counts) for _ in range(world)]
        dist.all_gather(gathered, counts)
        me = dist.get_rank()
        recv_counts = torch.tensor([int(g[me]) for g in gathered],
                                   dtype=torch.int64)
    out_splits = [int(x) for x in recv_counts.cpu()]
    out = torch.empty(sum(out_splits), dtype=torch.uint8, device=send.device)
    if dist.get_backend() == "nccl":
        dist.all_to_all_single(out, send, out_splits, in_splits)
    else:
        me = dist.get_rank()
        ops = []
        pos_in = 0
        for r, n in enumerate(in_splits):
            if r != me and n > 0:
                ops.append(dist.P2POp(dist.isend,
                                      send[pos_in:pos_in + n], r))
            pos_in += n
        pos_out = 0
        self_src = None
        for r, n in enumerate(out_splits):
            if r != me and n > 0:
                ops.append(dist.P2POp(dist.irecv, out[pos_out:pos_out + n], r))
            pos_out += n
        # local copy
        in_base = sum(in_splits[:me])
        out_base = sum(out_splits[:me])
        out[out_base:out_base + in_splits[me]] = \
            send[in_base:in_base + in_splits[me]]
        if ops:
            for req in dist.batch_isend_irecv(ops):
                req.wait()
    return out, out_splits


def shuffle_table_gpu(table: Table) -> Table:
    """Full device shuffle of a GPU table across all ranks by murmur3 of the
    FIRST column (Spark HashPartitioning on the key column)."""
    from ..ops import hashing
    from ..ops.copying import partition_map, spark_partition_ids
    world = dist.get_world_size()
    h = hashing.murmur3([table.columns[0]])
    pids = spark_partition_ids(h, world)
    offsets, perm = partition_map(pids, world)
    buf, sizes = shuffle_gpu.split_and_serialize_to_device(table, offsets, perm)
    recv, out_splits = _exchange_bytes(buf, sizes)
    views = []
    pos = 0
    for s in out_splits:
        views.append(recv[pos:pos + s])
        pos += s
    return shuffle_gpu.assemble_from_device(views, table.columns)


def shuffle_table_host(columns: Sequence[Column], pids, nparts_is_world=True
                       ) -> List[Column]:
    """Host-path shuffle (kudo serializer) for CPU tables / gloo tests:
    partition rows by precomputed partition ids, kudo-serialize each
    partition, exchange bytes, merge on host."""
    import io
    world = dist.get_world_size()
    n = columns[0].size
    # order rows by partition (stable) on host
    order = sorted(range(n), key=lambda i: pids[i])
    counts = [0] * world
    for p in pids:
        counts[p] += 1
    # build a reordered host table (python-level; test path only)
    reordered = []
    for c in columns:
        vals = c.to_pylist()
        reordered.append(Column.from_pylist([vals[i] for i in order], c.dtype))
    bufs = []
    start = 0
    for p in range(world):
        out = io.BytesIO()
        write_partition(reordered, start, counts[p], out)
        bufs.append(out.getvalue())
        start += counts[p]
    in_splits = [len(b) for b in bufs]
    send = torch.frombuffer(bytearray(b"".join(bufs)) or bytearray(1),
                            dtype=torch.uint8)[:sum(in_splits)]
    recv, out_splits = _exchange_bytes(send, in_splits)
    raw = recv.numpy().tobytes()
    pieces = []
    pos = 0
    for s in out_splits:
        pieces.append(raw[pos:pos + s])
        pos += s
    return merge_on_host(pieces, columns)
