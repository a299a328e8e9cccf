"""ctypes bindings to the in-tree CDNA4 GPU library (libcsp_gpu.so).

The library (covalent_ssh_plugin_amd/ops/hip/csp_gpu.hip, built for
gfx950 only) provides:

* ``csp_probe_json`` — device probe: properties + measured HBM bandwidth
  and bf16 MFMA throughput from the hand-written warm-up kernels,
* ``csp_warmup`` — clock/cache warm-up spin (MFMA + HBM sweep),
* ``csp_staging_get`` / ``csp_host_alloc`` / ``csp_host_free`` /
  ``csp_memcpy_d2h`` / ``csp_memcpy_h2d`` — hipHostMalloc-pinned staging.

On a GPU box these bindings FAIL LOUDLY if the extension is missing or
broken — there is no eager/CPU fallback for the device path.
"""

from __future__ import annotations

import ctypes
import json
import os
from pathlib import Path
from typing import Optional

_LIB_NAME = "libcsp_gpu.so"
_lib: Optional[ctypes.CDLL] = None


class GpuLibError(RuntimeError):
    """The CDNA4 GPU extension is missing or a call into it failed."""


def local_gpu_lib_path() -> str:
    """Path of the built in-tree library, or '' if not built."""
    candidate = Path(__file__).resolve().parent.parent / "ops" / _LIB_NAME
    return str(candidate) if candidate.exists() else ""


def load(path: str = "") -> ctypes.CDLL:
    """Load (once) and return the GPU library with typed signatures."""
    global _lib
    if _lib is not None:
        return _lib
    lib_path = path or local_gpu_lib_path()
    if not lib_path:
        raise GpuLibError(
            f"{_LIB_NAME} not built — run __graft_entry__.build() / "
            "python -m covalent_ssh_plugin_amd.ops.build"
        )
    # One HSA runtime per process: bind torch's bundled ROCm runtime
    # first when torch is installed, so this library shares it instead of
    # initializing the system runtime (which would break a later
    # torch.cuda init with "No HIP GPUs are available").
    import importlib.util

    if importlib.util.find_spec("torch") is not None:
        import torch

        if torch.cuda.is_available():
            torch.cuda.init()
    lib = ctypes.CDLL(lib_path)

    lib.csp_device_count.restype = ctypes.c_int
    lib.csp_probe_json.restype = ctypes.c_int
    lib.csp_probe_json.argtypes = [ctypes.c_int, ctypes.c_char_p, ctypes.c_size_t]
    lib.csp_probe_props_json.restype = ctypes.c_int
    lib.csp_probe_props_json.argtypes = [ctypes.c_int, ctypes.c_char_p, ctypes.c_size_t]
    lib.csp_warmup.restype = ctypes.c_int
    lib.csp_warmup.argtypes = [ctypes.c_int, ctypes.c_int]
    lib.csp_host_alloc.restype = ctypes.c_void_p
    lib.csp_host_alloc.argtypes = [ctypes.c_size_t]
    lib.csp_host_free.restype = ctypes.c_int
    lib.csp_host_free.argtypes = [ctypes.c_void_p]
    lib.csp_staging_get.restype = ctypes.c_void_p
    lib.csp_staging_get.argtypes = [ctypes.c_size_t]
    lib.csp_staging_alloc.restype = ctypes.c_void_p
    lib.csp_staging_alloc.argtypes = [ctypes.c_size_t]
    lib.csp_staging_release_all.restype = ctypes.c_int
    lib.csp_staging_reset.restype = ctypes.c_int
    lib.csp_memcpy_d2h.restype = ctypes.c_int
    lib.csp_memcpy_d2h.argtypes = [ctypes.c_void_p, ctypes.c_void_p, ctypes.c_size_t]
    lib.csp_memcpy_h2d.restype = ctypes.c_int
    lib.csp_memcpy_h2d.argtypes = [ctypes.c_void_p, ctypes.c_void_p, ctypes.c_size_t]
    lib.csp_mem_info.restype = ctypes.c_int
    lib.csp_mem_info.argtypes = [
        ctypes.c_int,
        ctypes.POINTER(ctypes.c_double),
        ctypes.POINTER(ctypes.c_double),
    ]
    lib.csp_last_error.restype = ctypes.c_char_p

    _lib = lib
    return lib


def _check(lib: ctypes.CDLL, rc: int, what: str) -> None:
    if rc != 0:
        err = lib.csp_last_error().decode(errors="replace")
        raise GpuLibError(f"{what} failed (rc={rc}): {err}")


def device_count() -> int:
    lib = load()
    n = lib.csp_device_count()
    if n < 0:
        _check(lib, n, "csp_device_count")
    return n


def probe(device: int = 0) -> dict:
    """Probe + measure one GPU.  Returns the JSON dict produced by the
    HIP library (name, gfx arch, CUs, HBM size/bandwidth, bf16 MFMA
    TFLOP/s from the warm-up kernels)."""
    lib = load()
    buf = ctypes.create_string_buffer(8192)
    _check(lib, lib.csp_probe_json(device, buf, len(buf)), "csp_probe_json")
    return json.loads(buf.value.decode())


def probe_props(device: int = 0) -> dict:
    """Device properties only (no measurement kernels, no big allocs)."""
    lib = load()
    buf = ctypes.create_string_buffer(8192)
    _check(lib, lib.csp_probe_props_json(device, buf, len(buf)), "csp_probe_props_json")
    return json.loads(buf.value.decode())


def mem_info(device: int = 0) -> dict:
    """HBM occupancy via csp_mem_info (the per-task telemetry call)."""
    lib = load()
    free_gb = ctypes.c_double()
    total_gb = ctypes.c_double()
    _check(lib, lib.csp_mem_info(device, ctypes.byref(free_gb), ctypes.byref(total_gb)),
           "csp_mem_info")
    return {"hbm_free_gb": free_gb.value, "hbm_total_gb": total_gb.value}


def warmup(device: int = 0, budget_ms: int = 50) -> None:
    lib = load()
    _check(lib, lib.csp_warmup(device, budget_ms), "csp_warmup")


def staged_d2h_bytes(data_ptr: int, nbytes: int) -> bytes:
    """Copy ``nbytes`` from device pointer ``data_ptr`` to host through
    the pooled pinned staging buffer and return them as bytes."""
    lib = load()
    dst = lib.csp_staging_alloc(nbytes)
    if not dst:
        _check(lib, -1, "csp_staging_alloc")
    try:
        _check(
            lib,
            lib.csp_memcpy_d2h(ctypes.c_void_p(dst), ctypes.c_void_p(data_ptr), nbytes),
            "csp_memcpy_d2h",
        )
        return ctypes.string_at(dst, nbytes)
    finally:
        lib.csp_staging_release_all()
