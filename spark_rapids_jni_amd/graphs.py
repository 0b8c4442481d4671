"""hipGraph capture for launch-bound operator pipelines.

Spark tasks frequently process many small batches through the same operator
sequence (filter -> join -> aggregate); at a few hundred microseconds of
kernel time per batch, launch + Python dispatch overhead dominates. This
module captures one batch's kernel sequence into a hipGraph
(torch.cuda.CUDAGraph IS hipGraph on ROCm) and replays it per batch with a
single launch.

Contract (same as hipGraph): the captured callable must be shape-static and
read/write only tensors that stay at fixed addresses — stage each new batch
into the static input tensors with copy_(), then replay().
"""
from typing import Callable, Sequence

import torch


class CapturedPipeline:
    """Capture `fn(*static_inputs)` once; replay it per batch.

    fn must perform GPU work only (no host syncs: no .item()/.cpu() inside).
    """

    def __init__(self, fn: Callable, static_inputs: Sequence[torch.Tensor],
                 warmup: int = 3):
        from .columnar import ensure_pinned_arena
        ensure_pinned_arena()  # pinned allocation is illegal mid-capture
        self.inputs = list(static_inputs)
        self.fn = fn
        # warm up on a side stream so allocations settle before capture
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(warmup):
                out = fn(*self.inputs)
        torch.cuda.current_stream().wait_stream(s)
        self.graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.graph):
            self.outputs = fn(*self.inputs)

    def replay(self, *batch: torch.Tensor):
        """Stage a new batch into the static inputs and replay the graph."""
        for dst, src in zip(self.inputs, batch):
            dst.copy_(src, non_blocking=True)
        self.graph.replay()
        return self.outputs
