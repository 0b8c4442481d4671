"""hipGraph capture of a small-batch operator pipeline."""
import pytest
import torch

from spark_rapids_jni_amd.columnar import Column, DType

pytestmark = pytest.mark.gpu


def test_captured_pipeline_matches_eager():
    from spark_rapids_jni_amd.graphs import CapturedPipeline
    from spark_rapids_jni_amd.ops import hashing
    from spark_rapids_jni_amd.ops.misc import BloomFilter

    n = 8192
    bf = BloomFilter(3, 3, 1 << 16, seed=42)
    seed_keys = torch.arange(0, 4096, dtype=torch.int64, device="cuda")
    bf.put(Column.from_torch(seed_keys))

    def pipeline(keys):
        col = Column(DType.INT64, n, keys)
        h = hashing.murmur3([col])
        hits = bf.might_contain(col)
        return h.data, hits.data

    static = torch.zeros(n, dtype=torch.int64, device="cuda")
    cap = CapturedPipeline(pipeline, [static])

    for trial in range(3):
        batch = torch.randint(0, 8192, (n,), dtype=torch.int64,
                              device="cuda")
        gh, ghits = cap.replay(batch)
        col = Column.from_torch(batch)
        eh = hashing.murmur3([col]).data
        ehits = bf.might_contain(col).data
        torch.cuda.synchronize()
        assert torch.equal(gh, eh)
        assert torch.equal(ghits, ehits)


def test_captured_pipeline_faster_for_small_batches():
    """Replay must beat eager dispatch on launch-bound small batches."""
    import time
    from spark_rapids_jni_amd.graphs import CapturedPipeline
    from spark_rapids_jni_amd.ops import hashing
    from spark_rapids_jni_amd.ops.misc import BloomFilter

    n = 4096
    bf = BloomFilter(3, 3, 1 << 16, seed=7)
    bf.put(Column.from_torch(torch.arange(0, 2048, dtype=torch.int64,
                                          device="cuda")))

    def pipeline(keys):
        col = Column(DType.INT64, n, keys)
        h = hashing.murmur3([col])
        for _ in range(8):  # launch-bound chain
            h = hashing.murmur3([h.astype_int64() if hasattr(h, "astype_int64")
                                 else Column(DType.INT32, n, h.data)])
        hits = bf.might_contain(col)
        return h.data, hits.data

    static = torch.zeros(n, dtype=torch.int64, device="cuda")
    cap = CapturedPipeline(pipeline, [static])
    batch = torch.randint(0, 8192, (n,), dtype=torch.int64, device="cuda")

    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(200):
        cap.replay(batch)
    torch.cuda.synchronize()
    graph_t = time.perf_counter() - t0

    t0 = time.perf_counter()
    for _ in range(200):
        pipeline(batch)
    torch.cuda.synchronize()
    eager_t = time.perf_counter() - t0
    print(f"graph {graph_t*5:.3f} ms/iter eager {eager_t*5:.3f} ms/iter")
    assert graph_t < eager_t
