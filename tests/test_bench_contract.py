"""bench.py driver-contract guard: tiny-config runs must emit the JSON line
with every required field (the round-end driver depends on this shape)."""
import json
import subprocess
import sys

import pytest

pytestmark = pytest.mark.gpu

REQUIRED = {"metric", "value", "unit", "n_gpus", "steps", "warmup",
            "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
            "dtype", "data", "config"}


def _run(args):
    out = subprocess.run(
        [sys.executable, "bench.py"] + args,
        capture_output=True, text=True, timeout=420)
    assert out.returncode == 0, out.stderr[-2000:]
    line = out.stdout.strip().splitlines()[-1]
    d = json.loads(line)
    assert REQUIRED <= set(d), sorted(REQUIRED - set(d))
    assert d["n_gpus"] == 1 and d["value"] > 0
    return d


def test_join_contract_tiny():
    d = _run(["--build-rows", "1000000", "--probe-rows", "4000000",
              "--chunk-rows", "2000000", "--steps", "2", "--warmup", "1"])
    assert d["metric"] == "hash_join_probe_rows_per_sec"
    assert d["config"]["global_batch"] == 4000000


def test_groupby_contract_tiny():
    d = _run(["--op", "groupby", "--build-rows", "2000000", "--steps", "2",
              "--warmup", "1"])
    assert d["metric"] == "hash_aggregate_rows_per_sec"


def test_q3_contract_tiny():
    d = _run(["--op", "q3", "--probe-rows", "2000000", "--steps", "2",
              "--warmup", "1"])
    assert d["metric"] == "nds_q3_rows_per_sec"
