"""Persistent worker mode: warm worker processes fed over framed
channels — the executor's high-throughput dispatch path."""

import asyncio

import pytest


def test_worker_roundtrip_and_reuse(local_executor):
    ex = local_executor(persistent_workers=True, cpu_workers=1)

    def fn(x):
        import os

        return (x * 2, os.getpid())

    async def main():
        r1 = await ex.execute(fn, [21], {}, dispatch_id="w", node_id=0)
        r2 = await ex.execute(fn, [100], {}, dispatch_id="w", node_id=1)
        return r1, r2

    (r1, pid1), (r2, pid2) = asyncio.run(main())
    assert (r1, r2) == (42, 200)
    # same worker process served both electrons (warm reuse)
    assert pid1 == pid2
    assert ex.last_task_record.remote_meta["worker"] is True


def test_worker_exception_roundtrip(local_executor):
    ex = local_executor(persistent_workers=True)

    def boom():
        raise ValueError("worker task failed")

    with pytest.raises(ValueError, match="worker task failed"):
        asyncio.run(ex.execute(boom, [], {}))

    # worker survives a task exception and serves the next electron
    def ok():
        return "fine"

    assert asyncio.run(ex.execute(ok, [], {})) == "fine"


def test_worker_stdout_pollution_safe(local_executor):
    ex = local_executor(persistent_workers=True)

    def chatty():
        print("x" * 10000)
        print("\x00\xff frames? \x00\x00\x00\x05")
        return 7

    assert asyncio.run(ex.execute(chatty, [], {})) == 7


def test_worker_concurrency_cpu_pool(local_executor):
    ex = local_executor(persistent_workers=True, cpu_workers=4)

    def slow(i):
        import os
        import time

        time.sleep(0.1)
        return i, os.getpid()

    async def main():
        return await asyncio.gather(
            *[ex.execute(slow, [i], {}, dispatch_id="c", node_id=i) for i in range(4)]
        )

    out = asyncio.run(main())
    assert [i for i, _ in out] == [0, 1, 2, 3]
    # four DISTINCT worker processes served the batch: true 4-way
    # parallelism (each worker channel serializes its own requests, so
    # distinct pids == concurrent capability; wall-clock assertions are
    # flaky on loaded CI machines)
    pids = {pid for _, pid in out}
    assert len(pids) == 4, out


def test_worker_death_respawn(local_executor):
    ex = local_executor(persistent_workers=True, cpu_workers=1)

    def suicide():
        import os

        os._exit(9)  # hard-kill the worker mid-task

    def fine():
        return 5

    with pytest.raises(RuntimeError):
        asyncio.run(ex.execute(suicide, [], {}, dispatch_id="k", node_id=0))
    # a fresh worker replaces the dead one
    assert asyncio.run(ex.execute(fine, [], {}, dispatch_id="k", node_id=1)) == 5


def test_worker_workdir(local_executor):
    ex = local_executor(persistent_workers=True, create_unique_workdir=True)

    def cwd():
        import os

        return os.getcwd()

    out = asyncio.run(ex.execute(cwd, [], {}, dispatch_id="wd", node_id=3))
    assert out.endswith("covalent-workdir/wd/node_3")


def test_worker_large_payload(local_executor):
    ex = local_executor(persistent_workers=True)

    def big(n):
        return b"z" * n

    out = asyncio.run(ex.execute(big, [8 * 1024 * 1024], {}))
    assert len(out) == 8 * 1024 * 1024


def test_worker_out_of_band_tensor_buffers(local_executor):
    """Large CPU tensors return via raw out-of-band frames (the same path
    CUDA tensors take through pinned staging on a GPU box)."""
    torch = __import__("pytest").importorskip("torch")
    ex = local_executor(
        persistent_workers=True, pinned_staging_threshold_bytes=1024
    )

    def fn(n):
        import torch

        big = torch.arange(n, dtype=torch.float32).reshape(4, -1)
        small = torch.ones(3, dtype=torch.bfloat16)
        return {"big": big, "nested": [small, (big * 2, "tag")], "n": n}

    out = asyncio.run(ex.execute(fn, [4096], {}, dispatch_id="oob", node_id=0))
    assert out["n"] == 4096
    assert torch.equal(
        out["big"], torch.arange(4096, dtype=torch.float32).reshape(4, -1)
    )
    assert torch.equal(out["nested"][1][0], out["big"] * 2)
    assert out["nested"][0].dtype == torch.bfloat16
    meta = ex.last_task_record.remote_meta
    assert meta["staging"]["mode"] == "cpu-oob"
    assert len(meta["buffers"]) == 2  # the two large tensors
    assert meta["staging"]["tensors"] == 3


def test_worker_large_tensor_roundtrip_exact(local_executor):
    torch = __import__("pytest").importorskip("torch")
    ex = local_executor(
        persistent_workers=True, pinned_staging_threshold_bytes=1 << 20
    )

    def fn():
        import torch

        g = torch.Generator().manual_seed(7)
        return torch.randn(1024, 1024, generator=g)

    out = asyncio.run(ex.execute(fn, [], {}))
    g = torch.Generator().manual_seed(7)
    assert torch.equal(out, torch.randn(1024, 1024, generator=g))


def test_cancel_kills_worker_task(local_executor):
    """cancel() on a worker-dispatched task kills the serving worker; the
    task fails and the next electron gets a fresh worker (reference
    parity: stub tasks stay uncancellable)."""
    ex = local_executor(persistent_workers=True, cpu_workers=1)

    def hang():
        import time

        time.sleep(60)
        return "never"

    async def main():
        task = asyncio.ensure_future(
            ex.execute(hang, [], {}, dispatch_id="cx", node_id=0)
        )
        # wait until the task is actually in flight on a worker
        for _ in range(200):
            if "cx_0" in ex._inflight:
                await asyncio.sleep(0.2)  # let the request frame land
                break
            await asyncio.sleep(0.01)
        await ex.cancel({"dispatch_id": "cx", "node_id": 0})
        with pytest.raises(RuntimeError):
            await task
        # executor recovers: next electron runs on a fresh worker
        return await ex.execute(lambda: 123, [], {}, dispatch_id="cx", node_id=1)

    assert asyncio.run(main()) == 123


def test_cancel_unknown_task_not_implemented(local_executor):
    ex = local_executor()
    with pytest.raises(NotImplementedError):
        asyncio.run(ex.cancel({"dispatch_id": "nope", "node_id": 9}))


def test_prewarm_starts_workers(local_executor):
    ex = local_executor(persistent_workers=True, cpu_workers=3)

    async def main():
        n = await ex.prewarm()
        assert n == 3
        # electrons reuse prewarmed workers (no spawn in the timed path)
        import time

        t0 = time.perf_counter()
        out = await ex.execute(lambda: 1, [], {})
        dt = time.perf_counter() - t0
        return out, dt

    out, dt = asyncio.run(main())
    assert out == 1


def test_pick_cpu_tag_prefers_idle(local_executor):
    """A long task must not block a following short task behind the same
    worker: the short task picks a different (idle or fresh) worker."""
    ex = local_executor(persistent_workers=True, cpu_workers=2)

    def slow():
        import time

        time.sleep(1.0)
        return "slow"

    def quick():
        return "quick"

    async def main():
        slow_task = asyncio.ensure_future(
            ex.execute(slow, [], {}, dispatch_id="p", node_id=0)
        )
        await asyncio.sleep(0.2)  # slow task is in flight on one worker
        import time

        t0 = time.perf_counter()
        q = await ex.execute(quick, [], {}, dispatch_id="p", node_id=1)
        dt = time.perf_counter() - t0
        s = await slow_task
        return q, s, dt

    q, s, dt = asyncio.run(main())
    assert (q, s) == ("quick", "slow")
    assert dt < 0.7, f"short task was blocked behind the long one ({dt:.2f}s)"


def test_worker_captures_task_output(local_executor):
    """Per-task stdout/stderr land in the task record's remote meta."""
    ex = local_executor(persistent_workers=True)

    def chatty():
        import sys

        print("captured stdout line")
        sys.stderr.write("captured stderr line\n")
        return 1

    assert asyncio.run(ex.execute(chatty, [], {})) == 1
    meta = ex.last_task_record.remote_meta
    assert "captured stdout line" in meta["stdout"]
    assert "captured stderr line" in meta["stderr"]


def test_large_tensor_arguments_out_of_band(local_executor):
    """Large CPU-tensor ARGUMENTS travel as raw request frames and
    reconstruct exactly in the worker (mirror of result staging)."""
    torch = __import__("pytest").importorskip("torch")
    ex = local_executor(
        persistent_workers=True, pinned_staging_threshold_bytes=1024
    )

    def consume(x, small, named=None):
        import torch

        assert isinstance(x, torch.Tensor)
        return {
            "sum": float(x.double().sum()),
            "shape": tuple(x.shape),
            "small_ok": bool((small == 1).all()),
            "named_sum": float(named.double().sum()),
        }

    big = torch.arange(6000, dtype=torch.float32).reshape(3, -1)
    small = torch.ones(4, dtype=torch.int64)  # below threshold: inline
    named = torch.full((512,), 2.0)
    out = asyncio.run(
        ex.execute(consume, [big, small], {"named": named}, dispatch_id="ab", node_id=0)
    )
    assert out["sum"] == float(big.double().sum())
    assert out["shape"] == (3, 2000)
    assert out["small_ok"] is True
    assert out["named_sum"] == 1024.0
    # the original dispatcher-side tensors are untouched
    assert torch.equal(big, torch.arange(6000, dtype=torch.float32).reshape(3, -1))


def test_worker_requests_pipeline_on_one_channel(local_executor):
    """Multiple queued electrons on ONE worker are all written to the
    channel without waiting for earlier replies (pipelining): the
    channel's in-flight count exceeds 1 while the first task runs."""
    from covalent_ssh_plugin_amd.remote import workers as worker_pool

    ex = local_executor(persistent_workers=True, cpu_workers=1)

    def slow(i):
        import time

        time.sleep(0.3)
        return i

    async def main():
        tasks = [
            asyncio.ensure_future(
                ex.execute(slow, [i], {}, dispatch_id="pl", node_id=i)
            )
            for i in range(3)
        ]
        peak = 0
        for _ in range(300):
            for handle in worker_pool._workers.values():
                peak = max(peak, handle.channel.inflight)
            if all(t.done() for t in tasks):
                break
            await asyncio.sleep(0.01)
        results = await asyncio.gather(*tasks)
        return results, peak

    results, peak = asyncio.run(main())
    assert results == [0, 1, 2]
    assert peak >= 2, f"requests were not pipelined (peak in-flight {peak})"


def test_task_timeout_worker_mode(local_executor):
    """A task exceeding task_timeout fails promptly; the wedged worker is
    killed and replaced for the next electron."""
    ex = local_executor(persistent_workers=True, cpu_workers=1, task_timeout=1.0)

    def hang():
        import time

        time.sleep(60)

    def quick():
        return "ok"

    import time

    t0 = time.perf_counter()
    with pytest.raises(RuntimeError, match="task_timeout"):
        asyncio.run(ex.execute(hang, [], {}, dispatch_id="to", node_id=0))
    assert time.perf_counter() - t0 < 10
    assert asyncio.run(ex.execute(quick, [], {}, dispatch_id="to", node_id=1)) == "ok"


def test_task_timeout_fused_mode(local_executor):
    ex = local_executor(task_timeout=1.0)

    def hang():
        import time

        time.sleep(60)

    import time

    t0 = time.perf_counter()
    with pytest.raises(RuntimeError, match="task_timeout"):
        asyncio.run(ex.execute(hang, [], {}, dispatch_id="tf", node_id=0))
    assert time.perf_counter() - t0 < 10


def test_unpicklable_result_surfaces_as_error(local_executor):
    """A result cloudpickle cannot serialize (e.g. a generator) comes
    back as a pickling exception, not a dead worker."""
    ex = local_executor(persistent_workers=True)

    def gen():
        return (i for i in range(3))

    with pytest.raises(Exception) as exc_info:
        asyncio.run(ex.execute(gen, [], {}, dispatch_id="up", node_id=0))
    assert "pickle" in str(exc_info.value).lower() or "generator" in str(
        exc_info.value
    ).lower(), exc_info.value

    # worker survives and serves the next electron
    assert asyncio.run(ex.execute(lambda: 3, [], {}, dispatch_id="up", node_id=1)) == 3


def test_worker_idle_timeout_exit_and_respawn(local_executor):
    """With worker_idle_timeout set, an idle worker exits on its own; the
    next electron transparently respawns one."""
    import time

    ex = local_executor(
        persistent_workers=True, cpu_workers=1, worker_idle_timeout=0.5
    )

    def pidof():
        import os

        return os.getpid()

    async def main():
        pid1 = await ex.execute(pidof, [], {}, dispatch_id="it", node_id=0)
        from covalent_ssh_plugin_amd.remote import workers as worker_pool

        (handle,) = worker_pool._workers.values()
        for _ in range(100):  # wait for the idle exit
            if not handle.alive:
                break
            await asyncio.sleep(0.1)
        assert not handle.alive, "worker did not exit on idle timeout"
        pid2 = await ex.execute(pidof, [], {}, dispatch_id="it", node_id=1)
        return pid1, pid2

    pid1, pid2 = asyncio.run(main())
    assert pid1 != pid2


def test_no_orphan_workers_when_dispatcher_dies(tmp_path):
    """A dispatcher that exits WITHOUT close_pool must not leave worker
    processes behind: the worker sees EOF on its stdin and exits."""
    import subprocess
    import sys
    import time
    from pathlib import Path

    repo = Path(__file__).resolve().parent.parent
    code = f"""
import asyncio, os, sys, tempfile
sys.path.insert(0, {str(repo)!r})
from covalent_ssh_plugin_amd import SSHExecutor

async def main():
    home = tempfile.mkdtemp()
    cache = tempfile.mkdtemp()
    ex = SSHExecutor(transport="local", local_home=home, cache_dir=cache,
                     python_path=sys.executable, persistent_workers=True,
                     cpu_workers=1)
    def pidof():
        import os
        return os.getpid()
    pid = await ex.execute(pidof, [], {{}})
    print(pid, flush=True)
    os._exit(0)  # die abruptly: no close_pool, no cleanup

asyncio.run(main())
"""
    proc = subprocess.run(
        [sys.executable, "-c", code], capture_output=True, text=True, timeout=120
    )
    assert proc.returncode == 0, proc.stderr[-1500:]
    worker_pid = int(proc.stdout.strip().splitlines()[-1])
    # the worker must exit on its own (EOF on stdin) within a few seconds
    for _ in range(50):
        try:
            import os

            os.kill(worker_pid, 0)  # still alive?
        except ProcessLookupError:
            break
        time.sleep(0.1)
    else:
        import os

        os.kill(worker_pid, 9)  # clean up before failing
        raise AssertionError(f"worker {worker_pid} orphaned after dispatcher death")


def test_worker_standalone_protocol_and_slot_resolution(tmp_path):
    """Drive a rendered worker directly over its framed protocol (no
    executor): READY handshake, slot resolution within ambient
    HIP_VISIBLE_DEVICES, task round trip, orderly shutdown."""
    import os
    import struct
    import subprocess
    import sys

    import cloudpickle

    from covalent_ssh_plugin_amd.remote.stub import render_worker

    script = tmp_path / "worker.py"
    script.write_text(render_worker())
    env = dict(os.environ)
    env["CSP_GPU_SLOT"] = "1"
    env["HIP_VISIBLE_DEVICES"] = "4,6"
    proc = subprocess.Popen(
        [sys.executable, str(script)],
        stdin=subprocess.PIPE,
        stdout=subprocess.PIPE,
        stderr=subprocess.DEVNULL,
        env=env,
        cwd=tmp_path,
    )

    def send(payload: bytes):
        proc.stdin.write(struct.pack(">Q", len(payload)) + payload)
        proc.stdin.flush()

    def recv() -> bytes:
        header = proc.stdout.read(8)
        (n,) = struct.unpack(">Q", header)
        return proc.stdout.read(n)

    try:
        tag, startup = cloudpickle.loads(recv())
        assert tag == "READY"
        assert startup["gpu_slot"] == "1"
        assert startup["hip_visible_devices"] == "6"  # ambient[1]

        def fn():
            import os

            return os.environ["HIP_VISIBLE_DEVICES"]

        send(
            cloudpickle.dumps(
                {
                    "op_id": "x_0",
                    "workdir": str(tmp_path / "wd"),
                    "function_blob": cloudpickle.dumps((fn, [], {})),
                    "arg_buffers": [],
                }
            )
        )
        # ack frame first (execution-start marker), then the response
        ack_tag, ack_op = cloudpickle.loads(recv())
        assert (ack_tag, ack_op) == ("A1", "x_0")
        tag, blob, meta, nbuf = cloudpickle.loads(recv())
        assert tag == "R1" and nbuf == 0
        result, exception = cloudpickle.loads(blob)
        assert exception is None
        assert result == "6"

        send(b"")  # zero-length frame: orderly shutdown
        assert proc.wait(timeout=10) == 0
    finally:
        if proc.poll() is None:
            proc.kill()
