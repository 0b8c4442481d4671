"""Native extension loaders.

Policy (fail loudly on GPU boxes): if this process can see a GPU, a missing or
broken _gpu.so is a hard error -- GPU ops must never fall back silently to an
eager/CPU path. On CPU-only machines the GPU extension stays optional so host
subsystems (kudo, OOM state machine, parquet footer) remain importable.
"""
import importlib
import os

_gpu_mod = None
_gpu_err = None
_host_mod = None
_host_err = None


def _cuda_available():
    try:
        import torch
        return torch.cuda.is_available()
    except Exception:
        return False


def gpu():
    """Return the _gpu extension module, raising if unavailable."""
    global _gpu_mod, _gpu_err
    if _gpu_mod is None and _gpu_err is None:
        try:
            _gpu_mod = importlib.import_module("spark_rapids_jni_amd._gpu")
        except Exception as e:  # pragma: no cover
            _gpu_err = e
    if _gpu_mod is None:
        raise RuntimeError(
            "spark_rapids_jni_amd._gpu native extension is not available "
            "(build with `python build_native.py`); GPU ops refuse to run "
            f"without it. Import error: {_gpu_err}")
    return _gpu_mod


def gpu_or_none():
    try:
        return gpu()
    except RuntimeError:
        if _cuda_available():
            raise  # GPU present: never run without the native kernels
        return None


def host():
    """Return the _host extension (CPU-only subsystems). SRJ_HOST_SO
    overrides the module path — the ASAN sanitizer lane (ci/sanitize.sh)
    points it at the AddressSanitizer build."""
    global _host_mod, _host_err
    if _host_mod is None and _host_err is None:
        import importlib.util
        try:
            override = os.environ.get("SRJ_HOST_SO")
            if override:
                spec = importlib.util.spec_from_file_location(
                    "spark_rapids_jni_amd._host", override)
                _host_mod = importlib.util.module_from_spec(spec)
                spec.loader.exec_module(_host_mod)
            else:
                _host_mod = importlib.import_module(
                    "spark_rapids_jni_amd._host")
        except Exception as e:  # pragma: no cover
            _host_err = e
    if _host_mod is None:
        raise RuntimeError(
            "spark_rapids_jni_amd._host native extension is not available "
            f"(build with `python build_native.py`). Import error: {_host_err}")
    return _host_mod


def current_stream():
    """The HIP stream of torch's current CUDA stream (0 on CPU)."""
    import torch
    if torch.cuda.is_available():
        return torch.cuda.current_stream().cuda_stream
    return 0
