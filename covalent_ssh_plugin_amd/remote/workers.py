"""Dispatcher-side pool of persistent remote workers.

One warm worker process per GPU slot per endpoint (plus a small
round-robin set of CPU workers when no GPU policy is active), launched
over a long-lived transport channel and reused across electrons.  This
removes the per-electron python + HIP-runtime start the classic stub
pays (the reference architecture pays it per task by design —
/root/reference/covalent_ssh_plugin/ssh.py:377-383).

Module-level, keyed per endpoint: all executor instances targeting the
same host share workers, mirroring the transport pool and slot tables.
"""

from __future__ import annotations

import asyncio
from dataclasses import dataclass, field
from typing import Dict, Optional, Tuple

import cloudpickle

from ..compat import app_log
from ..transport.channel import Channel, ChannelClosed

WorkerKey = Tuple[object, ...]


@dataclass
class WorkerHandle:
    channel: Channel
    startup_meta: dict
    key: WorkerKey

    @property
    def alive(self) -> bool:
        return self.channel.alive


class WorkerStartupError(RuntimeError):
    pass


_workers: Dict[WorkerKey, WorkerHandle] = {}
_locks: Dict[WorkerKey, Tuple[asyncio.Lock, object]] = {}
_cpu_rr = 0  # round-robin cursor for CPU worker sets


def _lock_for(key: WorkerKey) -> asyncio.Lock:
    loop = asyncio.get_running_loop()
    entry = _locks.get(key)
    if entry is None or entry[1] is not loop:
        entry = (asyncio.Lock(), loop)
        _locks[key] = entry
    return entry[0]


def cpu_worker_index(pool_size: int) -> int:
    global _cpu_rr
    idx = _cpu_rr % max(1, pool_size)
    _cpu_rr += 1
    return idx


def pick_cpu_tag(pool_key, pool_size: int) -> str:
    """Pick a CPU worker tag: prefer an already-running IDLE worker
    (avoids head-of-line blocking behind a long task), else round-robin
    (which also spreads initial spawns across the set)."""
    candidates = [f"cpu{i}" for i in range(max(1, pool_size))]
    for tag in candidates:
        for key, handle in _workers.items():
            if len(key) >= 2 and key[0] == pool_key and key[1] == tag:
                if handle.alive and handle.channel.inflight == 0:
                    return tag
    return f"cpu{cpu_worker_index(pool_size)}"


async def get_worker(key: WorkerKey, launcher, startup_timeout: float = 180.0) -> WorkerHandle:
    """Return the live worker for ``key``, starting it with ``launcher``
    (an async callable returning a Channel) if needed."""
    loop = asyncio.get_running_loop()
    async with _lock_for(key):
        handle = _workers.get(key)
        if handle is not None and handle.alive and handle.channel.loop is loop:
            return handle
        if handle is not None and handle.channel.loop is not loop:
            # worker belongs to a closed event loop: reap it synchronously
            handle.channel.kill()
            _workers.pop(key, None)
        channel = await launcher()
        try:
            ready = await channel.recv_frame(timeout=startup_timeout)
        except (ChannelClosed, asyncio.TimeoutError) as e:
            await channel.close()
            raise WorkerStartupError(f"worker {key} failed to start: {e}") from e
        tag, meta = cloudpickle.loads(ready)
        if tag != "READY" or meta.get("error"):
            await channel.close()
            raise WorkerStartupError(
                f"worker {key} startup failed: {meta.get('error')}"
            )
        handle = WorkerHandle(channel=channel, startup_meta=meta, key=key)
        _workers[key] = handle
        return handle


def extract_arg_buffers(args, kwargs, threshold: int):
    """Pull large contiguous CPU torch tensors out of (args, kwargs) and
    replace them with out-of-band buffer markers (the request-direction
    mirror of the worker's result staging).  Returns
    (new_args, new_kwargs, buffer_meta, buffers); no-op (and no torch
    import) unless torch is already loaded in the dispatcher."""
    import sys

    if "torch" not in sys.modules:
        return args, kwargs, [], []
    import ctypes

    import torch

    buffer_meta = []
    buffers = []

    def extract(t):
        src_t = t.contiguous()
        nbytes = src_t.numel() * src_t.element_size()
        view = (ctypes.c_char * nbytes).from_address(src_t.data_ptr())
        buffers.append((view, src_t))  # keep tensor alive until sent
        buffer_meta.append(
            {"dtype": str(src_t.dtype).replace("torch.", ""), "shape": list(src_t.shape)}
        )
        return ("__csp_tensor_buffer_v1__", len(buffers) - 1)

    def walk(obj):
        if isinstance(obj, torch.Tensor):
            if (
                not obj.is_cuda
                and obj.numel() * obj.element_size() >= threshold
            ):
                return extract(obj)
            return obj
        if isinstance(obj, dict):
            return {k: walk(v) for k, v in obj.items()}
        if isinstance(obj, tuple):
            vals = [walk(v) for v in obj]
            return type(obj)(*vals) if hasattr(obj, "_fields") else tuple(vals)
        if isinstance(obj, list):
            return [walk(v) for v in obj]
        return obj

    new_args = walk(list(args))
    new_kwargs = walk(dict(kwargs))
    return new_args, new_kwargs, buffer_meta, buffers


def _reconstruct(result, buffer_meta, buffers):
    """Replace out-of-band tensor-buffer markers with rebuilt torch
    tensors (workers ship large tensors as raw frames; see
    remote/worker_template.py protocol note)."""
    import torch

    def build(i: int):
        info = buffer_meta[i]
        dtype = getattr(torch, info["dtype"])
        raw = buffers[i]
        # large frames arrive as a preallocated bytearray (zero-copy
        # here); small ones as bytes (copy once into writable memory)
        data = raw if isinstance(raw, bytearray) else bytearray(raw)
        t = torch.frombuffer(data, dtype=dtype)
        return t.reshape(info["shape"])

    def walk(obj):
        if (
            isinstance(obj, tuple)
            and len(obj) == 2
            and obj[0] == "__csp_tensor_buffer_v1__"
        ):
            return build(obj[1])
        if isinstance(obj, dict):
            return {k: walk(v) for k, v in obj.items()}
        if isinstance(obj, tuple):
            vals = [walk(v) for v in obj]
            return type(obj)(*vals) if hasattr(obj, "_fields") else tuple(vals)
        if isinstance(obj, list):
            return [walk(v) for v in obj]
        return obj

    return walk(result)


async def run_task(
    handle: WorkerHandle,
    op_id: str,
    workdir: str,
    function_blob: bytes,
    timeout: Optional[float] = None,
    arg_buffer_meta=None,
    arg_buffers=None,
    ack_state: Optional[dict] = None,
):
    """One electron through a worker.  Returns (result, exception, meta).

    ``ack_state`` (a mutable dict) gets ``{"started": True}`` the moment
    the worker's A1 ack arrives — i.e. once user code may have begun.
    On ChannelClosed the caller reads it to decide whether a retry could
    re-execute a partially-run task.
    """
    request = cloudpickle.dumps(
        {
            "op_id": op_id,
            "workdir": workdir,
            "function_blob": function_blob,
            "arg_buffers": arg_buffer_meta or [],
        }
    )
    frames = [request] + [view for view, _keep in (arg_buffers or [])]
    parsed = {}

    def reply_frames(main: bytes):
        decoded = cloudpickle.loads(main)
        if decoded[0] == "A1":
            if ack_state is not None:
                ack_state["started"] = True
            return None  # control frame: the R1 reply follows
        tag, result_blob, meta, nbuf = decoded
        assert tag == "R1", f"unexpected worker response tag {tag!r}"
        parsed["value"] = (result_blob, meta, nbuf)
        return nbuf

    # pipelined: the channel's write lock is held only while sending, so
    # queued electrons on one worker overlap their wire round trips
    _main, buffers = await handle.channel.exchange(
        frames, reply_frames, timeout=timeout
    )
    assert "value" in parsed, "worker reply ended at the ack frame"
    result_blob, meta, nbuf = parsed["value"]
    result, exception = cloudpickle.loads(result_blob)
    if nbuf:
        result = _reconstruct(result, meta.get("buffers", []), buffers)
    return result, exception, meta


def drop(key: WorkerKey) -> None:
    _workers.pop(key, None)


async def kill(key: WorkerKey) -> None:
    """Hard-kill the worker for ``key`` (task cancellation): its
    in-flight request fails with ChannelClosed; the next electron for
    the key spawns a fresh worker."""
    handle = _workers.pop(key, None)
    if handle is not None:
        handle.channel.kill()


async def close_all() -> None:
    handles = list(_workers.values())
    _workers.clear()
    _locks.clear()
    for h in handles:
        try:
            await h.channel.close()
        except Exception:  # noqa: BLE001
            app_log.debug("worker close failed", exc_info=True)


def reset() -> None:
    """Synchronous test hook (leaks processes if any are live — tests
    that start workers must close them)."""
    _workers.clear()
    _locks.clear()
