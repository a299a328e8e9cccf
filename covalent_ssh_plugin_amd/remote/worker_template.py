"""Persistent remote worker for the MI355X SSH executor.

Like stub_template.py this file is NEVER imported by the plugin: it is
rendered (CSP token substitution) and shipped to the remote host once
per endpoint (content-addressed), then launched once per GPU slot over a
long-lived transport channel.  Where the classic stub pays a fresh
python + HIP-runtime start per electron (the reference's architecture,
/root/reference/covalent_ssh_plugin/exec.py), a worker pays it once:

  * starts, optionally runs the CDNA4 warm-up/device-probe prologue on
    its pinned GPU (the launcher injects CSP_GPU_SLOT; _resolve_gpu_slot
    maps it into the ambient HIP_VISIBLE_DEVICES),
  * then serves electrons over a length-framed binary protocol on
    stdin/stdout: each request carries the cloudpickled
    ``(fn, args, kwargs)`` and a workdir; each reply carries the pickled
    ``(result, exception)`` 2-tuple (same payload contract as the
    result file) plus a meta JSON dict.

Protocol (all frames: 8-byte big-endian length + payload):
  request  = pickle dict {"op_id", "workdir", "function_blob",
             "arg_buffers": [{dtype, shape}, ...]} followed by one RAW
             frame per entry (large CPU-tensor arguments, mirrored from
             the dispatcher without pickle copies)
  ack      = pickle tuple ("A1", op_id) — written as soon as the request
             (and its arg frames) has been fully received, BEFORE any
             user code runs.  The dispatcher uses it to distinguish
             "worker died before starting the task" (safe to retry on a
             fresh worker) from "worker died mid-execution" (retrying
             would re-run a possibly non-idempotent task — surfaced as
             an error instead, unless retry_on_worker_death opts in)
  response = pickle tuple ("R1", result_blob: bytes, meta: dict, nbuf: int)
             followed by nbuf RAW tensor-buffer frames.  Large tensors in
             the result are replaced by ("__csp_tensor_buffer_v1__", i)
             markers and shipped out-of-band as raw frames — CUDA tensors
             straight from the hipHostMalloc-pinned staging block, large
             CPU tensors zero-copy from their own storage — skipping both
             pickle copies on the payload.  meta["buffers"][i] carries
             dtype/shape.  A zero-length request frame means: shut down.

The worker's REAL stdout is reserved for the protocol; fd 1 is
re-pointed at stderr before any user code runs, so user prints cannot
corrupt frames.
"""


import json
import os
import struct
import sys
import time

GPU_LIB = "__CSP_GPU_LIB__"
DO_WARMUP = bool(__CSP_WARMUP__)
STAGING_THRESHOLD = int(__CSP_STAGING_THRESHOLD__)
IDLE_TIMEOUT = float(__CSP_IDLE_TIMEOUT__)  # seconds; 0 = never exit
# Fork-isolation mode: the worker is a warm ZYGOTE (python + cloudpickle
# imported, HIP **not** initialized — initializing HIP pre-fork would
# break the children) and every electron executes in a freshly forked
# child that does its own GPU prologue.  Fresh-process semantics at fork
# cost instead of a full interpreter + import start per task.
ISOLATE = bool(__CSP_ISOLATE__)
# What the zygote binds before forking.  "torch": children inherit a
# loaded torch — saves the multi-second import for torch-using electrons
# but makes each fork copy the torch-ROCm address space (measured
# ~250 ms/fork on MI355X vs ~ms without).  "none": cheap forks; torch
# electrons pay their own import.  Choose per workload.
ISOLATE_PRELOAD = "__CSP_ISOLATE_PRELOAD__"
# Per-task GPU telemetry: sample HBM occupancy (hipMemGetInfo) every Nth
# electron into the task meta.  0 = off (the hot no-op path stays
# syscall-minimal); enable for GPU-heavy workloads where ~0.1 ms of
# sampling is noise.
TELEMETRY_EVERY = int(__CSP_TELEMETRY_EVERY__)
_last_mem = None

if GPU_LIB:
    GPU_LIB = os.path.abspath(os.path.expanduser(GPU_LIB))


def _resolve_gpu_slot():
    """Map CSP_GPU_SLOT to HIP_VISIBLE_DEVICES within the ambient
    visibility list (composes with pod GPU isolation).  Must run before
    any HIP/torch initialization."""
    slot = os.environ.get("CSP_GPU_SLOT")
    if slot is None:
        return None
    ambient = os.environ.get("HIP_VISIBLE_DEVICES") or ""
    ids = [x for x in ambient.split(",") if x.strip() != ""]
    if ids:
        os.environ["HIP_VISIBLE_DEVICES"] = ids[int(slot) % len(ids)]
    else:
        os.environ["HIP_VISIBLE_DEVICES"] = slot
    return slot


GPU_SLOT = _resolve_gpu_slot()

# --- claim the protocol channel, push user stdout to stderr ---------------
_proto_fd = os.dup(1)
os.dup2(2, 1)
sys.stdout = sys.stderr

# widen the protocol pipes (default 64 KiB caps large-tensor frames)
try:
    import fcntl

    F_SETPIPE_SZ = 1031  # linux
    for _fd in (0, _proto_fd):
        try:
            fcntl.fcntl(_fd, F_SETPIPE_SZ, 1 << 20)
        except OSError:
            pass
except ImportError:
    pass

import cloudpickle as pickle  # noqa: E402  (after fd surgery on purpose)

_gpu_lib = None
_gpu_info = None
_served = 0  # electrons completed by this worker (observability)


def _ensure_torch_runtime_first():
    """One process can host only ONE HSA runtime.  torch bundles its own
    ROCm runtime; if our ctypes library initializes the system runtime
    first, a later torch.cuda init finds no agents ("No HIP GPUs are
    available").  So when torch is installed, import it (binding its
    runtime) BEFORE loading the CDNA4 library — the library then resolves
    libamdhip64.so.7 to torch's already-loaded copy (soname dedup) and
    both share one runtime.  Measured on-box: probe-then-torch breaks,
    torch-then-probe runs the probe at full rate."""
    import importlib.util

    if importlib.util.find_spec("torch") is None:
        return
    import torch

    if torch.cuda.is_available():
        torch.cuda.init()


def _load_gpu_lib():
    import ctypes

    _ensure_torch_runtime_first()

    lib = ctypes.CDLL(GPU_LIB)
    lib.csp_probe_json.restype = ctypes.c_int
    lib.csp_probe_json.argtypes = [ctypes.c_int, ctypes.c_char_p, ctypes.c_size_t]
    lib.csp_warmup.restype = ctypes.c_int
    lib.csp_warmup.argtypes = [ctypes.c_int, ctypes.c_int]
    lib.csp_staging_alloc.restype = ctypes.c_void_p
    lib.csp_staging_alloc.argtypes = [ctypes.c_size_t]
    lib.csp_staging_release_all.restype = ctypes.c_int
    lib.csp_memcpy_d2h.restype = ctypes.c_int
    lib.csp_memcpy_d2h.argtypes = [ctypes.c_void_p, ctypes.c_void_p, ctypes.c_size_t]
    lib.csp_mem_info.restype = ctypes.c_int
    lib.csp_mem_info.argtypes = [
        ctypes.c_int,
        ctypes.POINTER(ctypes.c_double),
        ctypes.POINTER(ctypes.c_double),
    ]
    lib.csp_last_error.restype = ctypes.c_char_p
    return lib


def _sample_hbm():
    """Cheap HBM occupancy sample for the task meta (telemetry row in
    SURVEY.md §5; sampled every TELEMETRY_EVERY electrons)."""
    global _last_mem
    if _gpu_lib is None:
        return
    import ctypes

    free_gb = ctypes.c_double()
    total_gb = ctypes.c_double()
    if _gpu_lib.csp_mem_info(0, ctypes.byref(free_gb), ctypes.byref(total_gb)) == 0:
        _last_mem = {
            "hbm_free_gb": round(free_gb.value, 2),
            "hbm_total_gb": round(total_gb.value, 2),
            "sampled_at_serial": _served,
        }


def _prologue():
    """One-time GPU warm-up + probe for this worker's pinned device."""
    global _gpu_lib, _gpu_info
    import ctypes

    _gpu_lib = _load_gpu_lib()
    buf = ctypes.create_string_buffer(8192)
    rc = _gpu_lib.csp_probe_json(0, buf, len(buf))
    if rc != 0:
        raise RuntimeError(
            "csp_probe_json failed: %s"
            % _gpu_lib.csp_last_error().decode(errors="replace")
        )
    _gpu_info = json.loads(buf.value.decode())
    if DO_WARMUP:
        rc = _gpu_lib.csp_warmup(0, 50)
        if rc != 0:
            raise RuntimeError(
                "csp_warmup failed: %s"
                % _gpu_lib.csp_last_error().decode(errors="replace")
            )


def _stage_result(result, stats, buffers, buffer_meta):
    """Replace large tensors in ``result`` with out-of-band buffer
    markers; CUDA tensors are staged through hipHostMalloc-pinned memory
    (fast D2H), large contiguous CPU tensors are shipped zero-copy from
    their own storage.  Small tensors stay inline in the pickle."""
    if "torch" not in sys.modules:
        return result
    import ctypes

    import torch

    lib = _gpu_lib
    if lib is None and GPU_LIB and os.path.exists(GPU_LIB):
        try:
            lib = _load_gpu_lib()
        except Exception:  # noqa: BLE001
            lib = None

    def emit_buffer(view, keepalive, dtype, shape):
        buffers.append((view, keepalive))
        buffer_meta.append(
            {"dtype": str(dtype).replace("torch.", ""), "shape": list(shape)}
        )
        return ("__csp_tensor_buffer_v1__", len(buffers) - 1)

    def to_host(t):
        stats["tensors"] += 1
        nbytes = t.numel() * t.element_size()
        stats["bytes"] += nbytes
        if nbytes < STAGING_THRESHOLD:
            return t.cpu() if t.is_cuda else t
        if t.is_cuda:
            if lib is None:
                raise RuntimeError(
                    "CUDA tensor result but the CDNA4 staging library is "
                    "not available on this GPU host"
                )
            src_t = t.contiguous()
            torch.cuda.synchronize()
            dst = lib.csp_staging_alloc(nbytes)
            if not dst:
                raise RuntimeError("csp_staging_alloc failed")
            rc = lib.csp_memcpy_d2h(
                ctypes.c_void_p(dst), ctypes.c_void_p(src_t.data_ptr()), nbytes
            )
            if rc != 0:
                raise RuntimeError("csp_memcpy_d2h failed")
            view = (ctypes.c_char * nbytes).from_address(dst)
            stats["pinned_tensors"] += 1
            stats["mode"] = "pinned"
            return emit_buffer(view, None, src_t.dtype, src_t.shape)
        # large CPU tensor: zero-copy out-of-band (keep tensor alive)
        src_t = t.contiguous()
        view = (ctypes.c_char * nbytes).from_address(src_t.data_ptr())
        if stats["mode"] != "pinned":
            stats["mode"] = "cpu-oob"
        return emit_buffer(view, src_t, src_t.dtype, src_t.shape)

    def walk(obj):
        if isinstance(obj, torch.Tensor):
            return to_host(obj)
        if isinstance(obj, dict):
            return {k: walk(v) for k, v in obj.items()}
        if isinstance(obj, tuple):
            vals = [walk(v) for v in obj]
            return type(obj)(*vals) if hasattr(obj, "_fields") else tuple(vals)
        if isinstance(obj, list):
            return [walk(v) for v in obj]
        return obj

    return walk(result)


def _read_frame(fd, idle_timeout=0.0):
    if idle_timeout > 0:
        import select

        ready, _, _ = select.select([fd], [], [], idle_timeout)
        if not ready:
            return None  # idle too long: orderly exit
    header = b""
    while len(header) < 8:
        chunk = os.read(fd, 8 - len(header))
        if not chunk:
            return None
        header += chunk
    (length,) = struct.unpack(">Q", header)
    if length == 0:
        return b""
    parts = []
    remaining = length
    while remaining:
        chunk = os.read(fd, min(remaining, 1 << 20))
        if not chunk:
            return None
        parts.append(chunk)
        remaining -= len(chunk)
    return b"".join(parts)


def _write_frame(fd, payload):
    view = memoryview(payload)
    if view.nbytes <= 64 * 1024:
        # small frames: single write (header + payload) — halves the
        # per-frame syscalls on the no-op hot path
        os.write(fd, struct.pack(">Q", view.nbytes) + bytes(view))
        return
    os.write(fd, struct.pack(">Q", view.nbytes))
    while view:
        written = os.write(fd, view[: 1 << 20])
        view = view[written:]


def _rebuild_arg_tensors(obj, buffers, meta_list):
    """Mirror of the dispatcher-side marker walk for request buffers."""
    import torch

    def build(i):
        info = meta_list[i]
        dtype = getattr(torch, info["dtype"])
        return torch.frombuffer(buffers[i], dtype=dtype).reshape(info["shape"])

    def walk(o):
        if isinstance(o, tuple) and len(o) == 2 and o[0] == "__csp_tensor_buffer_v1__":
            return build(o[1])
        if isinstance(o, dict):
            return {k: walk(v) for k, v in o.items()}
        if isinstance(o, tuple):
            vals = [walk(v) for v in o]
            return type(o)(*vals) if hasattr(o, "_fields") else tuple(vals)
        if isinstance(o, list):
            return [walk(v) for v in o]
        return o

    return walk(obj)


def _serve_one(request):
    global _served
    _served += 1
    if TELEMETRY_EVERY > 0 and _served % TELEMETRY_EVERY == 1 % TELEMETRY_EVERY:
        _sample_hbm()
    t0 = time.monotonic()
    meta = {"phases_ms": {}, "gpu": _gpu_info, "staging": None,
            "hbm": _last_mem,
            "gpu_slot": GPU_SLOT, "buffers": [],
            "hip_visible_devices": os.environ.get("HIP_VISIBLE_DEVICES"),
            "pid": os.getpid(), "worker": True, "served": _served}
    result = None
    exception = None
    buffers = []
    try:
        fn, args, kwargs = pickle.loads(request["function_blob"])
        arg_bufs = request.get("_arg_buffer_frames") or []
        if arg_bufs:
            args = _rebuild_arg_tensors(args, arg_bufs, request["arg_buffers"])
            kwargs = _rebuild_arg_tensors(kwargs, arg_bufs, request["arg_buffers"])
    except Exception as e:  # noqa: BLE001
        exception = e
        fn = None

    home = os.getcwd()
    if exception is None:
        workdir = request.get("workdir") or "."
        try:
            os.makedirs(workdir, exist_ok=True)
            os.chdir(workdir)
        except OSError as e:
            exception = e

    if exception is None:
        t_fn = time.monotonic()
        # capture the task's python-level stdout/stderr into the meta
        # (fd-level output still flows to the worker's stderr stream)
        import contextlib
        import io as _io

        out_buf, err_buf = _io.StringIO(), _io.StringIO()
        try:
            with contextlib.redirect_stdout(out_buf), contextlib.redirect_stderr(err_buf):
                result = fn(*args, **kwargs)
        except Exception as e:  # noqa: BLE001
            exception = e
        meta["phases_ms"]["user_fn"] = round((time.monotonic() - t_fn) * 1000, 3)
        meta["stdout"] = out_buf.getvalue()[-65536:]
        meta["stderr"] = err_buf.getvalue()[-65536:]
        if meta["stdout"]:
            sys.stderr.write(meta["stdout"])
        if meta["stderr"]:
            sys.stderr.write(meta["stderr"])
    os.chdir(home)

    if exception is None and result is not None:
        t_stage = time.monotonic()
        stats = {"tensors": 0, "pinned_tensors": 0, "bytes": 0, "mode": "none"}
        try:
            result = _stage_result(result, stats, buffers, meta["buffers"])
            meta["staging"] = stats
        except Exception as e:  # noqa: BLE001
            result, exception, buffers = None, e, []
            meta["buffers"] = []
        meta["phases_ms"]["staging"] = round((time.monotonic() - t_stage) * 1000, 3)

    try:
        result_blob = pickle.dumps((result, exception))
    except Exception as e:  # noqa: BLE001
        result_blob = pickle.dumps((None, e))
        buffers = []
        meta["buffers"] = []
    meta["phases_ms"]["total"] = round((time.monotonic() - t0) * 1000, 3)
    return result_blob, meta, buffers


def _serve_isolated(request):
    """Run one electron in a freshly forked child (fresh-process
    semantics: no state, no HIP context, no module cache survives into
    the next electron).  The child writes its response frames to a pipe;
    the parent relays them frame-by-frame so a child crash can never
    leave a half-written frame on the protocol stream."""
    global _served
    _served += 1
    serial = _served
    r, w = os.pipe()
    pid = os.fork()
    if pid == 0:
        # ---- child: the per-task process ----
        os.close(r)
        try:
            import ctypes

            libc = ctypes.CDLL(None, use_errno=True)
            libc.prctl(1, 9, 0, 0, 0)  # PR_SET_PDEATHSIG=SIGKILL: die with parent
        except Exception:  # noqa: BLE001
            pass
        wrote_reply = False
        try:
            if GPU_LIB and GPU_SLOT is not None and DO_WARMUP:
                _prologue()  # fresh HIP init belongs to THIS child
            result_blob, meta, buffers = _serve_one(request)
            meta["isolated"] = True
            meta["served"] = serial
            _write_frame(w, pickle.dumps(("R1", result_blob, meta, len(buffers))))
            wrote_reply = True
            for view, _keep in buffers:
                _write_frame(w, view)
        except BaseException as e:  # noqa: BLE001 - report anything reportable
            # Fallback error reply ONLY if the real R1 never went out: a
            # failure mid-buffer-stream must NOT append a second R1 that
            # the parent would miscount as a raw buffer frame (the parent
            # detects the short stream and drops the channel instead).
            if not wrote_reply:
                try:
                    blob = pickle.dumps(
                        (None, RuntimeError(f"isolated task failed: {e!r}"))
                    )
                    _write_frame(
                        w,
                        pickle.dumps(
                            ("R1", blob, {"phases_ms": {}, "isolated": True}, 0)
                        ),
                    )
                except BaseException:  # noqa: BLE001
                    pass
        os._exit(0)

    # ---- parent: frame-by-frame relay ----
    os.close(w)
    relayed = 0
    expected = None
    while True:
        frame = _read_frame(r)
        if frame is None or frame == b"":
            break
        if relayed == 0:
            _tag, _blob, _meta, nbuf = pickle.loads(frame)
            expected = 1 + nbuf
        _write_frame(_proto_fd, frame)
        relayed += 1
        if expected is not None and relayed >= expected:
            break
    os.close(r)
    _, status = os.waitpid(pid, 0)
    if relayed == 0:
        # child died before completing even the main response frame:
        # synthesize the error so the electron fails cleanly and the
        # zygote keeps serving (no worker respawn needed)
        blob = pickle.dumps(
            (None, RuntimeError(
                f"isolated task process died (wait status {status}) "
                "before reporting a result"
            ))
        )
        _write_frame(
            _proto_fd,
            pickle.dumps(("R1", blob, {"phases_ms": {}, "isolated": True,
                                       "served": serial}, 0)),
        )
    elif expected is not None and relayed < expected:
        # partial multi-frame response already relayed: the stream is
        # unrecoverable — exit so the dispatcher sees ChannelClosed
        os._exit(4)


def main():
    startup_meta = {"pid": os.getpid(), "gpu": None, "error": None,
                    "gpu_slot": GPU_SLOT, "isolate": ISOLATE,
                    "hip_visible_devices": os.environ.get("HIP_VISIBLE_DEVICES")}
    if ISOLATE:
        # zygote warm-up: bind the configured imports pre-fork, but
        # touch NO GPU state (HIP contexts do not survive fork)
        if ISOLATE_PRELOAD == "torch":
            try:
                import importlib.util

                if importlib.util.find_spec("torch") is not None:
                    import torch  # noqa: F401
            except Exception as e:  # noqa: BLE001
                startup_meta["error"] = repr(e)
    elif GPU_LIB and GPU_SLOT is not None:
        try:
            _prologue()
            startup_meta["gpu"] = _gpu_info
        except Exception as e:  # noqa: BLE001
            startup_meta["error"] = repr(e)
    _write_frame(_proto_fd, pickle.dumps(("READY", startup_meta)))
    if startup_meta["error"]:
        sys.exit(3)

    while True:
        frame = _read_frame(0, idle_timeout=IDLE_TIMEOUT)
        if frame is None or frame == b"":
            break
        request = pickle.loads(frame)
        n_arg_bufs = len(request.get("arg_buffers") or [])
        if n_arg_bufs:
            frames = []
            for _ in range(n_arg_bufs):
                f = _read_frame(0)
                if f is None:
                    return
                frames.append(bytearray(f))
            request["_arg_buffer_frames"] = frames
        # ack AFTER the full request is in hand, BEFORE user code runs:
        # the dispatcher's retry-on-death policy keys off this frame
        _write_frame(_proto_fd, pickle.dumps(("A1", request.get("op_id"))))
        if ISOLATE:
            _serve_isolated(request)
            continue
        result_blob, meta, buffers = _serve_one(request)
        _write_frame(_proto_fd, pickle.dumps(("R1", result_blob, meta, len(buffers))))
        for view, _keep in buffers:
            _write_frame(_proto_fd, view)
        del buffers
        if _gpu_lib is not None:
            _gpu_lib.csp_staging_release_all()


if __name__ == "__main__":
    main()
