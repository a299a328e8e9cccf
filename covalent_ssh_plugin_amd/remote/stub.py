"""Renderer for the remote execution stub.

The reference renders its exec.py template with ``str.format`` and three
placeholders (reference ssh.py:160-171, exec.py:12-14).  This build uses
token substitution (``__CSP_*__``) instead so the template stays a valid,
brace-safe Python file, and adds tokens for the GPU library path, the
meta/timings file and staging policy.
"""

from __future__ import annotations

import functools
from pathlib import Path
from typing import Optional

_TEMPLATE_PATH = Path(__file__).parent / "stub_template.py"
_WORKER_TEMPLATE_PATH = Path(__file__).parent / "worker_template.py"


@functools.lru_cache(maxsize=2)
def _template_text(path: str) -> str:
    return Path(path).read_text()

# 64 MiB: below this a plain torch .cpu() copy is cheap; above it the
# hipHostMalloc-pinned D2H path wins (PCIe Gen5 x16 ~63 GB/s vs pageable).
DEFAULT_STAGING_THRESHOLD = 64 * 1024 * 1024


def render_stub(
    *,
    remote_result_file: str,
    remote_function_file: str,
    current_remote_workdir: str,
    remote_meta_file: str = "",
    gpu_lib_path: str = "",
    warmup: bool = True,
    staging_threshold: Optional[int] = None,
) -> str:
    """Return the per-task stub script text."""
    text = _template_text(str(_TEMPLATE_PATH))
    thr = DEFAULT_STAGING_THRESHOLD if staging_threshold is None else int(staging_threshold)
    replacements = {
        "__CSP_RESULT_FILE__": remote_result_file,
        "__CSP_FUNCTION_FILE__": remote_function_file,
        "__CSP_WORKDIR__": current_remote_workdir,
        "__CSP_META_FILE__": remote_meta_file,
        "__CSP_GPU_LIB__": gpu_lib_path,
        "__CSP_WARMUP__": repr(bool(warmup)),
        "__CSP_STAGING_THRESHOLD__": str(thr),
    }
    for token, value in replacements.items():
        text = text.replace(token, value)
    return text


@functools.lru_cache(maxsize=64)
def render_worker(
    *,
    gpu_lib_path: str = "",
    warmup: bool = True,
    staging_threshold: Optional[int] = None,
    idle_timeout: float = 0.0,
    isolate: bool = False,
    isolate_preload: str = "torch",
    telemetry_every: int = 0,
) -> str:
    """Return the persistent worker script text (one per endpoint; the
    GPU slot arrives via the CSP_GPU_SLOT env var at launch).  With
    ``isolate`` the worker is a fork-server zygote: every electron runs
    in a freshly forked child process; ``isolate_preload`` picks what
    the zygote imports pre-fork ("torch" or "none")."""
    text = _template_text(str(_WORKER_TEMPLATE_PATH))
    thr = DEFAULT_STAGING_THRESHOLD if staging_threshold is None else int(staging_threshold)
    replacements = {
        "__CSP_GPU_LIB__": gpu_lib_path,
        "__CSP_WARMUP__": repr(bool(warmup)),
        "__CSP_STAGING_THRESHOLD__": str(thr),
        "__CSP_IDLE_TIMEOUT__": repr(float(idle_timeout)),
        "__CSP_ISOLATE__": repr(bool(isolate)),
        "__CSP_ISOLATE_PRELOAD__": str(isolate_preload),
        "__CSP_TELEMETRY_EVERY__": str(int(telemetry_every)),
    }
    for token, value in replacements.items():
        text = text.replace(token, value)
    return text
