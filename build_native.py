#!/usr/bin/env python3
"""In-tree native build driver for spark_rapids_jni_amd.

Builds two extension modules, committed next to the Python package so the
built .so travels to GPU boxes with the repo snapshot:

  spark_rapids_jni_amd/_gpu.so   -- HIP/CDNA4 kernels (gfx950) + pybind11 bindings
  spark_rapids_jni_amd/_host.so  -- CPU-only C++ (OOM state machine, kudo host
                                    serializer, parquet footer, ...) + bindings

Design (MI355X-first): kernel translation units are pure HIP
with a C ABI (no torch headers), compiled by hipcc for gfx950 only; binding
TUs are pybind11 compiled by g++. Memory is owned by the Python layer (torch
caching allocator); kernels receive raw device pointers + hipStream_t.
"""
import hashlib
import json
import os
import subprocess
import sys
from concurrent.futures import ThreadPoolExecutor
from pathlib import Path

ROOT = Path(__file__).resolve().parent
SRC_GPU = ROOT / "src" / "gpu"
SRC_HOST = ROOT / "src" / "host"
PKG = ROOT / "spark_rapids_jni_amd"
OBJ = ROOT / ".build" / "obj"
CACHE = ROOT / ".build" / "cache.json"

HIPCC = os.environ.get("HIPCC", "hipcc")
CXX = os.environ.get("CXX", "g++")
ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")
ROCM = os.environ.get("ROCM_PATH", "/opt/rocm")


def _pyb_includes():
    import pybind11
    import sysconfig
    return [pybind11.get_include(), sysconfig.get_paths()["include"]]


# NOTE: no -ffast-math — Spark-exact semantics require IEEE NaN/-0.0 behavior
# (float normalization, NaN-greatest ordering). Individual kernels may opt in.
HIP_FLAGS = [f"--offload-arch={ARCH}", "-O3", "-std=c++17", "-fPIC",
             "-fno-gpu-rdc", "-Wall"]
CXX_FLAGS = ["-O2", "-std=c++17", "-fPIC", "-Wall",
             "-D__HIP_PLATFORM_AMD__", f"-I{ROCM}/include"]
LINK_FLAGS = ["-shared", f"-L{ROCM}/lib", "-lamdhip64",
              "-lrocprofiler-sdk-roctx", "-pthread"]


def _load_cache():
    try:
        return json.loads(CACHE.read_text())
    except Exception:
        return {}


def _key(src: Path, flags, headers):
    h = hashlib.sha256()
    h.update(src.read_bytes())
    h.update(" ".join(flags).encode())
    for hdr in headers:
        h.update(hdr.read_bytes())
    return h.hexdigest()


def _compile(src: Path, flags, headers, cache, log):
    obj = OBJ / (src.stem + ".o")
    key = _key(src, flags, headers)
    if obj.exists() and cache.get(str(obj)) == key:
        return obj, None
    cc = HIPCC if src.suffix == ".hip" else CXX
    cmd = [cc] + flags + ["-c", str(src), "-o", str(obj)]
    log.append(" ".join(cmd))
    r = subprocess.run(cmd, capture_output=True, text=True)
    if r.returncode != 0:
        raise RuntimeError(f"compile failed: {src}\n{r.stdout}\n{r.stderr}")
    cache[str(obj)] = key
    return obj, r.stderr or None


def build(verbose=False, asan_host=False):
    OBJ.mkdir(parents=True, exist_ok=True)
    cache = _load_cache()
    log = []
    pyb = [f"-I{p}" for p in _pyb_includes()]

    def batch(srcs, flags):
        outs = []
        with ThreadPoolExecutor(max_workers=os.cpu_count()) as ex:
            futs = [(s, ex.submit(_compile, s, flags, headers, cache, log))
                    for s, headers, flags in srcs]
            for s, f in futs:
                obj, warn = f.result()
                if warn and verbose:
                    print(f"-- {s.name}:\n{warn}", file=sys.stderr)
                outs.append(obj)
        return outs

    gpu_headers = sorted(SRC_GPU.glob("*.hpp")) + sorted(SRC_GPU.glob("*.inc"))
    host_headers = sorted(SRC_HOST.glob("*.hpp"))

    jobs = []
    for s in sorted(SRC_GPU.glob("*.hip")):
        jobs.append((s, gpu_headers, HIP_FLAGS))
    for s in sorted(SRC_GPU.glob("*.cpp")):
        jobs.append((s, gpu_headers, CXX_FLAGS + pyb))
    gpu_objs = batch(jobs, None)

    if asan_host:
        # sanitizer tier (reference pom.xml test-with-sanitizer profile;
        # the MI355X analog is an ASAN lane over the host-native code):
        # compile the host sources with AddressSanitizer into a side .so
        # the tests load via SRJ_HOST_SO (see ci/sanitize.sh)
        asan_dir = ROOT / ".build" / "asan"
        asan_dir.mkdir(parents=True, exist_ok=True)
        aflags = CXX_FLAGS + pyb + ["-fsanitize=address",
                                    "-fno-omit-frame-pointer", "-g"]
        objs = []
        for s in sorted(SRC_HOST.glob("*.cpp")):
            obj = asan_dir / (s.stem + ".o")
            r = subprocess.run([CXX] + aflags + ["-c", str(s), "-o",
                                str(obj)], capture_output=True, text=True)
            if r.returncode != 0:
                raise RuntimeError(f"asan compile failed:\n{r.stderr}")
            objs.append(obj)
        out = asan_dir / "_host_asan.so"
        r = subprocess.run([CXX, "-shared", "-fsanitize=address"] +
                           [str(o) for o in objs] + ["-o", str(out),
                            "-pthread"], capture_output=True, text=True)
        if r.returncode != 0:
            raise RuntimeError(f"asan link failed:\n{r.stderr}")
        print(f"asan host extension: {out}")
        return

    host_jobs = [(s, host_headers, CXX_FLAGS + pyb) for s in sorted(SRC_HOST.glob("*.cpp"))]
    host_objs = batch(host_jobs, None)

    def link(objs, out):
        cmd = [CXX] + [str(o) for o in objs] + LINK_FLAGS + ["-o", str(out)]
        r = subprocess.run(cmd, capture_output=True, text=True)
        if r.returncode != 0:
            raise RuntimeError(f"link failed: {out}\n{r.stderr}")

    if gpu_objs:
        link(gpu_objs, PKG / "_gpu.so")
    # fault-injection preload shim (reference faultinj/ — LD_PRELOAD analog)
    fi = ROOT / "src" / "tools" / "faultinj.cpp"
    if fi.exists():
        out = PKG / "libsrjfaultinj.so"
        key = _key(fi, CXX_FLAGS, [])
        ck = str(out)
        if not out.exists() or cache.get(ck) != key:
            r = subprocess.run([CXX] + CXX_FLAGS + ["-shared", str(fi), "-o",
                                str(out), "-ldl"], capture_output=True,
                               text=True)
            if r.returncode != 0:
                raise RuntimeError(f"faultinj build failed:\n{r.stderr}")
            cache[ck] = key
    # kernel-activity capture lib (roctracer; reference CUPTI activity)
    ka = ROOT / "src" / "tools" / "kernel_activity.cpp"
    if ka.exists():
        out = PKG / "libsrjktrace.so"
        key = _key(ka, CXX_FLAGS, [])
        ck = str(out)
        if not out.exists() or cache.get(ck) != key:
            r = subprocess.run(
                [CXX] + CXX_FLAGS + ["-I/opt/rocm/include",
                                     "-I/opt/rocm/include/roctracer",
                                     "-shared", str(ka), "-o", str(out),
                                     "-L/opt/rocm/lib", "-lroctracer64",
                                     "-Wl,-rpath,/opt/rocm/lib"],
                capture_output=True, text=True)
            if r.returncode != 0:
                raise RuntimeError(f"kernel_activity build failed:\n{r.stderr}")
            cache[ck] = key
    # JNI-mirroring C ABI surface (docs/JNI_ABI.md)
    ja = ROOT / "src" / "host" / "jni_abi.cpp"
    if ja.exists():
        out = PKG / "_jniabi.so"
        gen = ROOT / "src" / "host" / "jni_abi_generated.inc"
        key = _key(ja, CXX_FLAGS, []) + (
            _key(gen, [], []) if gen.exists() else "")
        ck = str(out)
        if not out.exists() or cache.get(ck) != key:
            r = subprocess.run([CXX] + CXX_FLAGS + ["-shared", str(ja),
                                "-o", str(out)], capture_output=True,
                               text=True)
            if r.returncode != 0:
                raise RuntimeError(f"jni_abi build failed:\n{r.stderr}")
            cache[ck] = key
    if host_objs:
        link(host_objs, PKG / "_host.so")
    CACHE.write_text(json.dumps(cache))
    if verbose:
        print(f"built: {len(gpu_objs)} gpu objs, {len(host_objs)} host objs")


if __name__ == "__main__":
    build(verbose=True, asan_host="--asan-host" in sys.argv)
