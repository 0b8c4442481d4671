import json, time
import pyarrow as pa, pyarrow.parquet as pq
import numpy as np, torch
from spark_rapids_jni_amd import parquet as srj_pq

n = 100_000_000
rng = np.random.default_rng(7)
base = np.cumsum(rng.integers(-3, 8, n, dtype=np.int64))
t = pa.table({"a": pa.array(base)})
res = {}
for enc, kw in [("PLAIN", dict(use_dictionary=False)),
                ("DELTA_BINARY_PACKED",
                 dict(use_dictionary=False,
                      column_encoding={"a": "DELTA_BINARY_PACKED"}))]:
    p = f"/tmp/dbench_{enc}.parquet"
    pq.write_table(t, p, compression="NONE", row_group_size=10_000_000, **kw)
    srj_pq.read_table(p, device="cuda")  # warmup
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(3):
        tab = srj_pq.read_table(p, device="cuda")
        torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / 3
    assert int(tab.columns[0].data[-1].item()) == int(base[-1])
    import os
    res[enc] = {"rows_per_sec": n / dt, "seconds": dt,
                "file_bytes": os.path.getsize(p)}
print(json.dumps({"bench": "parquet_delta_vs_plain_scan", "rows": n,
                  "dtype": "int64", "results": res}))
