"""Fault-injection driver (reference faultinj/README.md workflow).

The native shim is spark_rapids_jni_amd/libsrjfaultinj.so (built by
build_native.py from src/tools/faultinj.cpp); load it into any HIP process
via LD_PRELOAD with SRJ_FAULT_INJECT_CONFIG pointing at a JSON config — the
LD_PRELOAD equivalent of the reference's CUDA_INJECTION64_PATH library.
"""
import json
import os
import subprocess
import sys
from typing import List, Optional


def shim_path() -> str:
    return os.path.join(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))), "libsrjfaultinj.so")


def write_config(path: str, faults: List[dict], seed: int = 12345,
                 dynamic: bool = False):
    """faults: [{"name": "hipMalloc", "code": 2, "percent": 100, "count": -1}]"""
    with open(path, "w") as f:
        json.dump({"seed": seed, "dynamic": dynamic, "faults": faults}, f)


def run_with_faults(cmd: List[str], config_path: str,
                    extra_env: Optional[dict] = None, **popen_kw):
    """Run a subprocess with the shim preloaded."""
    env = dict(os.environ)
    env["LD_PRELOAD"] = shim_path() + (
        ":" + env["LD_PRELOAD"] if env.get("LD_PRELOAD") else "")
    env["SRJ_FAULT_INJECT_CONFIG"] = config_path
    if extra_env:
        env.update(extra_env)
    return subprocess.run(cmd, env=env, **popen_kw)
