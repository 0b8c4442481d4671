"""NDS power-run driver.

Runs q1..q99 back-to-back on one engine (BASELINE configs[4] shape: the
reference's headline metric is the NDS SF3K power-run wall-clock). Data is
generated once per (sf, world, rank) and kept resident on the GPU across
queries — the engine caches scans — so the measured time is query
execution, not ingest. `verify(sf)` cross-checks every query's GPU result
against the CPU oracle backend (VERDICT r01 item 1 acceptance).
"""
import json
import time
from typing import Dict, List, Optional

import torch

from .plan import CpuBackend, Engine, Frame, GpuBackend
from .queries import QUERIES
from .schema import gen_catalog


def make_engine(sf: float, device: str = "cuda", world: int = 1,
                rank: int = 0) -> Engine:
    cat = gen_catalog(sf=sf, world=world, rank=rank)
    return Engine(cat, device=device, world=world, rank=rank)


def power_run(engine: Engine, queries: Optional[List[int]] = None,
              quiet: bool = False) -> Dict:
    """Run the listed queries (default all 99); returns timings."""
    qs = queries or sorted(QUERIES)
    per_query = {}
    total0 = time.time()
    for n in qs:
        # fresh temps per query; scans stay cached on device
        engine.temps.clear()
        if engine.device.type == "cuda":
            torch.cuda.synchronize()
        t0 = time.time()
        f = QUERIES[n](engine)
        if engine.device.type == "cuda":
            torch.cuda.synchronize()
        dt = time.time() - t0
        per_query[n] = {"seconds": round(dt, 4), "rows": f.nrows}
        if not quiet:
            print(f"  q{n}: {dt*1000:.1f} ms, {f.nrows} rows", flush=True)
    total = time.time() - total0
    return {"total_seconds": round(total, 3),
            "queries": per_query,
            "n_queries": len(qs)}


def frame_sorted_rows(f: Frame, round_floats: int = 4):
    rows = f.to_rows()
    def norm(v):
        if isinstance(v, float):
            if v != v:
                return "nan"
            return round(v, round_floats)
        return v
    return sorted((tuple(norm(v) for v in r) for r in rows),
                  key=lambda t: tuple((x is None, str(x)) for x in t))


def compare_frames(got: Frame, exp: Frame, q: int, rel_tol: float = 1e-6):
    assert got.nrows == exp.nrows, \
        f"q{q}: row count {got.nrows} != {exp.nrows}"
    assert list(got.cols.keys()) == list(exp.cols.keys()), \
        f"q{q}: column mismatch {got.names()} vs {exp.names()}"
    g = frame_sorted_rows(got)
    x = frame_sorted_rows(exp)
    for i, (rg, rx) in enumerate(zip(g, x)):
        for a, b in zip(rg, rx):
            if isinstance(a, float) and isinstance(b, float):
                # rows carry values rounded to 4 decimals; a true 1e-9
                # accumulation difference can still land across a rounding
                # boundary, so allow one ulp of the rounding grid
                tol = max(2e-4, rel_tol * max(abs(a), abs(b)))
                assert abs(a - b) <= tol, \
                    f"q{q} row {i}: {a} != {b}\n got={rg}\n exp={rx}"
            else:
                assert a == b, f"q{q} row {i}: {a!r} != {b!r}\n" \
                    f" got={rg}\n exp={rx}"


def verify(sf: float = 0.05, queries: Optional[List[int]] = None,
           device: str = "cuda") -> List[int]:
    """Cross-check GPU results against the CPU oracle; returns failures."""
    qs = queries or sorted(QUERIES)
    cat = gen_catalog(sf=sf)
    failures = []
    for n in qs:
        eg = Engine(cat, device=device)
        ec = Engine(cat, device="cpu", backend=CpuBackend())
        try:
            got = QUERIES[n](eg)
            exp = QUERIES[n](ec)
            compare_frames(got, exp, n)
        except AssertionError as ex:
            failures.append(n)
            print(f"q{n}: MISMATCH {ex}")
        except Exception as ex:
            failures.append(n)
            print(f"q{n}: ERROR {type(ex).__name__}: {ex}")
    return failures
