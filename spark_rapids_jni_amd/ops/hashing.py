"""Spark-exact hash ops (Java API parity: com.nvidia.spark.rapids.jni.Hash).

Reference behavior: Hash.java + src/main/cpp/src/hash/*.cu in
NVIDIA/spark-rapids-jni. Seeds chain column-to-column;
nulls pass the running seed/hash through (murmur3/xxhash64) or contribute 0
(hive). Defaults match Spark: murmur3 seed 42, xxhash64 seed 42.
"""
from typing import Sequence, Union

import torch

from .. import _native
from ..columnar import Column, Table, pack_descriptors

DEFAULT_MURMUR3_SEED = 42
DEFAULT_XXHASH64_SEED = 42

# Maximum nested depth accepted (mirrors Hash.java MAX_STACK_DEPTH behavior)
MAX_STACK_DEPTH = 8


def _cols(table_or_cols: Union[Table, Sequence[Column]]):
    cols = table_or_cols.columns if isinstance(table_or_cols, Table) else list(table_or_cols)
    assert cols, "need at least one column"
    dev = cols[0].device
    assert dev.type == "cuda", "hash ops run on GPU columns"
    return cols, dev


def murmur3(table_or_cols, seed: int = DEFAULT_MURMUR3_SEED) -> Column:
    cols, dev = _cols(table_or_cols)
    g = _native.gpu()
    n = cols[0].size
    out = torch.empty(n, dtype=torch.int32, device=dev)
    desc, top, keep = pack_descriptors(cols)
    g.murmur3(desc.data_ptr(), top.data_ptr(), len(cols), n, seed,
              out.data_ptr(), _native.current_stream())
    return Column.from_torch(out)


def xxhash64(table_or_cols, seed: int = DEFAULT_XXHASH64_SEED) -> Column:
    cols, dev = _cols(table_or_cols)
    g = _native.gpu()
    n = cols[0].size
    out = torch.empty(n, dtype=torch.int64, device=dev)
    desc, top, keep = pack_descriptors(cols)
    g.xxhash64(desc.data_ptr(), top.data_ptr(), len(cols), n, seed,
               out.data_ptr(), _native.current_stream())
    return Column.from_torch(out)


def hive_hash(table_or_cols) -> Column:
    cols, dev = _cols(table_or_cols)
    g = _native.gpu()
    n = cols[0].size
    out = torch.empty(n, dtype=torch.int32, device=dev)
    desc, top, keep = pack_descriptors(cols)
    g.hive_hash(desc.data_ptr(), top.data_ptr(), len(cols), n,
                out.data_ptr(), _native.current_stream())
    return Column.from_torch(out)
