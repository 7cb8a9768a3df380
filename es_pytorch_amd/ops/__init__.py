"""Native ops loader.

Two in-tree shared libraries, built from ``csrc/`` (see ``build.py``):

* ``_cpu_ops.so``  — g++-compiled host ops (Philox noise fill, gather-GEMV);
* ``_hip_ops.so``  — hipcc-compiled gfx950 kernels (noise fill, batched pheno,
  fused population MLP forward, gather-GEMV gradient, fused Adam/SGD).

Loaded via ctypes against raw pointers (``tensor.data_ptr()``) so the HIP
library has no torch-ABI dependency and cross-compiles on a GPU-less box.

Policy: on a machine WITH a GPU the HIP library is REQUIRED — ops fail loudly
rather than falling back to eager torch, so a passing GPU test means the
native path ran. On CPU-only machines the CPU library is auto-built on demand
(g++, seconds).
"""
from __future__ import annotations

import ctypes
import os
import subprocess
from typing import Optional

_DIR = os.path.dirname(os.path.abspath(__file__))
_CPU_SO = os.path.join(_DIR, "_cpu_ops.so")
# ES_HIP_SO: load an alternative prebuilt kernel library (A/B experiments
# with different -DES_DEPTH_* builds on one box; see tools/build_variants.py)
_HIP_SO = os.environ.get("ES_HIP_SO", os.path.join(_DIR, "_hip_ops.so"))

_cpu_lib: Optional[ctypes.CDLL] = None
_hip_lib: Optional[ctypes.CDLL] = None


class _BuildLock:
    """Serialize .so builds across processes (N torchrun ranks share the
    tree; concurrent compiler invocations writing one output would corrupt
    it). Uses filelock when available, else a best-effort O_EXCL spinlock."""

    def __init__(self, path: str):
        self.path = path + ".lock"
        self._fl = None

    def __enter__(self):
        try:
            from filelock import FileLock
            self._fl = FileLock(self.path)
            self._fl.acquire(timeout=600)
        except ImportError:
            import time
            for _ in range(6000):
                try:
                    fd = os.open(self.path, os.O_CREAT | os.O_EXCL | os.O_WRONLY)
                    os.close(fd)
                    self._fl = "posix"
                    break
                except FileExistsError:
                    time.sleep(0.1)
        return self

    def __exit__(self, *a):
        if self._fl == "posix":
            try:
                os.unlink(self.path)
            except OSError:
                pass
        elif self._fl is not None:
            self._fl.release()


def build_cpu(force: bool = False) -> str:
    src = os.path.join(_DIR, "csrc", "cpu_ops.cpp")

    def stale():
        return (force or not os.path.exists(_CPU_SO)
                or os.path.getmtime(_CPU_SO) < os.path.getmtime(src))

    if stale():
        with _BuildLock(_CPU_SO):
            if stale():  # re-check under the lock: another rank may have built
                tmp = _CPU_SO + ".tmp"
                cmd = ["g++", "-O3", "-std=c++17", "-shared", "-fPIC", "-pthread", src,
                       "-o", tmp]
                subprocess.run(cmd, check=True, capture_output=True, text=True)
                os.replace(tmp, _CPU_SO)
    return _CPU_SO


def hip_sources():
    d = os.path.join(_DIR, "csrc", "hip")
    return sorted(os.path.join(d, f) for f in os.listdir(d) if f.endswith(".hip"))


def build_hip(force: bool = False, arch: str = "gfx950", extra_flags=None,
              out: Optional[str] = None) -> str:
    srcs = hip_sources()
    hdrs = [os.path.join(_DIR, "csrc", "philox.h")] + \
        [os.path.join(_DIR, "csrc", "hip", h)
         for h in os.listdir(os.path.join(_DIR, "csrc", "hip")) if h.endswith(".h")]
    newest = max(os.path.getmtime(f) for f in srcs + hdrs)
    target = out or _HIP_SO

    def stale():
        return force or not os.path.exists(target) or os.path.getmtime(target) < newest

    if stale():
        with _BuildLock(target):
            if stale():  # re-check under the lock: another rank may have built
                hipcc = os.environ.get("HIPCC", "/opt/rocm/bin/hipcc")
                tmp = target + ".tmp"
                cmd = [hipcc, f"--offload-arch={arch}", "-O3", "-std=c++17", "-shared",
                       "-fPIC", *(extra_flags or []), *srcs, "-o", tmp]
                r = subprocess.run(cmd, check=False, capture_output=True, text=True)
                if r.returncode != 0:
                    raise RuntimeError(f"hipcc build failed:\n{r.stdout}\n{r.stderr}")
                os.replace(tmp, target)
    return target


def cpu() -> ctypes.CDLL:
    """The CPU ops library, auto-building if stale/missing."""
    global _cpu_lib
    if _cpu_lib is None:
        build_cpu()
        _cpu_lib = ctypes.CDLL(_CPU_SO)
        _bind_cpu(_cpu_lib)
    return _cpu_lib


def hip() -> ctypes.CDLL:
    """The HIP ops library. On a GPU machine this must exist and load.

    Raises (loudly) if the library is missing on a machine with a visible
    GPU — a silent eager fallback would defeat the native-path contract.
    """
    global _hip_lib
    if _hip_lib is None:
        if not os.path.exists(_HIP_SO):
            # cross-compiling is cheap and possible without a GPU; try it
            build_hip()
        _hip_lib = ctypes.CDLL(_HIP_SO)
        _bind_hip(_hip_lib)
    return _hip_lib


def hip_available() -> bool:
    try:
        return hip() is not None
    except Exception:
        return False


c_f32p = ctypes.POINTER(ctypes.c_float)
c_i64p = ctypes.POINTER(ctypes.c_int64)


def _bind_cpu(lib):
    lib.es_noise_fill_cpu.argtypes = [ctypes.c_void_p, ctypes.c_int64, ctypes.c_uint64,
                                      ctypes.c_uint32]
    lib.es_noise_fill_cpu.restype = None
    lib.es_grad_gather_cpu.argtypes = [ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p,
                                       ctypes.c_void_p, ctypes.c_int64, ctypes.c_int64]
    lib.es_grad_gather_cpu.restype = None


def _bind_hip(lib):
    u64, i64, u32, i32, f32, p = (ctypes.c_uint64, ctypes.c_int64, ctypes.c_uint32,
                                  ctypes.c_int32, ctypes.c_float, ctypes.c_void_p)
    lib.es_noise_fill.argtypes = [p, i64, u64, u32, p]
    lib.es_pheno_bf16.argtypes = [p, p, p, p, p, i64, i64, i64, f32, p]
    lib.es_pheno_fp8.argtypes = [p, p, p, p, i32, i64, i64, i64, f32, p]
    lib.es_mlp_fwd.argtypes = [p, p, p, p, p, p, i32, p, u64, i32, f32, p, i64, i32,
                               i32, i32, i32, i32, p, p, p]
    lib.es_grad_gather.argtypes = [p, p, p, p, i64, i64, f32, p]
    lib.es_adam_step.argtypes = [p, p, p, p, i64, f32, f32, f32, f32, f32, f32, p]
    lib.es_sgd_step.argtypes = [p, p, p, i64, f32, f32, f32, f32, p]
    lib.es_loco_step.argtypes = [p, p, p, p, i32, p, u64, f32, p, i64,
                                 p, p, p, p, p, p, p, p, p, p,
                                 p, p, p, p, p, p,
                                 i32, i32, i32, i32, i32, i32, i32, i32, i32,
                                 f32, f32, f32, f32, f32, p]
    lib.es_loco_step_split.argtypes = [p, p, p, p, i32, p, u64, f32, p, i64,
                                       p, p, p, p, p, p, p, p, p, p,
                                       p, p, p, p, p, p,
                                       i32, i32, i32, i32, i32, i32, i32, i32, i32,
                                       f32, f32, f32, f32, f32, p, i32, p]
    lib.es_loco_pair_step.argtypes = [p, p, p, p, p, i32, p, u64, f32, p, i64,
                                      p, p, p, p, p, p, p, p, p, p,
                                      p, p, p, p, p, p,
                                      i32, i32, i32, i32, i32, i32, i32, i32, i32,
                                      f32, f32, f32, f32, f32, p]
    lib.es_loco_pair_step_fp8.argtypes = lib.es_loco_pair_step.argtypes
    lib.es_loco_pair_episode.argtypes = [p, p, p, p, p, i32, p, f32, p, i64,
                                         p, p, p, p, p, p, p, p, p, p,
                                         p, p, p, p, p, p,
                                         i32, i32, i32, i32, i32, i32, i32, i32, i32,
                                         f32, f32, f32, f32, f32, i32, i32, p]
    lib.es_loco_episode.argtypes = [p, p, p, p, i32, p, i32, f32, p, i64,
                                    p, p, p, p, p, p, p, p, p, p,
                                    p, p, p, p, p, p,
                                    i32, i32, i32, i32, i32, i32, i32, i32, i32, i32,
                                    i32, i32, f32, f32, f32, f32, f32, i32, p]
    for fn in ["es_noise_fill", "es_pheno_bf16", "es_pheno_fp8", "es_mlp_fwd",
               "es_grad_gather", "es_adam_step", "es_sgd_step", "es_loco_step",
               "es_loco_step_split", "es_loco_pair_step", "es_loco_pair_step_fp8",
               "es_loco_pair_episode", "es_loco_episode"]:
        getattr(lib, fn).restype = i32


def check(ret: int, name: str):
    if ret != 0:
        raise RuntimeError(f"HIP op {name} failed with hipError_t={ret}")
