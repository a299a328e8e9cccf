"""Round-2 reliability policies (VERDICT r1 items 3/9 + ADVICE r1).

* Worker A1-ack protocol: a task whose worker died AFTER execution began
  is surfaced, never silently re-executed; a task the worker never
  started is retried transparently; re-execution is opt-in.
* GPU slot pinning independent of warm-up-library provisioning.
* Negative environment-check TTL + invalidation on transport reconnect.
* Slot-table capacity guard for mismatched executors.
"""

import asyncio
import os
import sys

import pytest

from covalent_ssh_plugin_amd import SSHExecutor
from covalent_ssh_plugin_amd.ssh import SSHTaskError
from covalent_ssh_plugin_amd.transport import pool as transport_pool


def _suicide_fn_factory(marker_path: str, die_always: bool):
    """Returns a task that records each execution in ``marker_path`` and
    SIGKILLs its own worker process (always, or only on the first run)."""

    def task():
        import os
        import signal
        import time

        with open(marker_path, "a") as f:
            f.write(f"{os.getpid()}\n")
        runs = sum(1 for _ in open(marker_path))
        if die_always or runs == 1:
            time.sleep(0.5)  # let pipelined co-resident requests queue up
            os.kill(os.getpid(), signal.SIGKILL)
        return "survived"

    return task


def test_worker_death_after_start_is_surfaced_not_rerun(local_executor, tmp_path):
    """non-idempotent tasks must not execute twice.
    The worker acks before running user code; death after the ack raises
    instead of retrying."""
    marker = tmp_path / "exec_count"
    ex = local_executor(persistent_workers=True, cpu_workers=1)
    fn = _suicide_fn_factory(str(marker), die_always=True)

    async def go():
        try:
            with pytest.raises(SSHTaskError, match="execution had started"):
                await ex.execute(fn, [], {}, dispatch_id="kill", node_id=0)
        finally:
            await SSHExecutor.close_pool()

    asyncio.run(go())
    # executed exactly once — no silent re-execution
    assert marker.read_text().count("\n") == 1
    assert ex.counters["worker_respawns"] == 0


def test_worker_death_retry_opt_in(local_executor, tmp_path):
    """retry_on_worker_death=True restores the old retry-once behavior
    for explicitly idempotent workloads."""
    marker = tmp_path / "exec_count"
    ex = local_executor(
        persistent_workers=True, cpu_workers=1, retry_on_worker_death=True
    )
    fn = _suicide_fn_factory(str(marker), die_always=False)

    async def go():
        try:
            return await ex.execute(fn, [], {}, dispatch_id="kill", node_id=0)
        finally:
            await SSHExecutor.close_pool()

    assert asyncio.run(go()) == "survived"
    assert marker.read_text().count("\n") == 2  # first run died, retry ran
    assert ex.counters["worker_respawns"] == 1


def test_unstarted_coresident_task_retries_safely(local_executor, tmp_path):
    """The advisor's exact scenario: two electrons pipelined on ONE
    worker; the first kills the worker mid-execution.  The first must
    surface an error (it had started); the second — queued but never
    started — must transparently re-dispatch and succeed."""
    marker = tmp_path / "exec_count"
    ex = local_executor(persistent_workers=True, cpu_workers=1)
    killer = _suicide_fn_factory(str(marker), die_always=True)

    def innocent(x):
        return x * 10

    async def go():
        try:
            t_killer = asyncio.create_task(
                ex.execute(killer, [], {}, dispatch_id="pair", node_id=0)
            )
            await asyncio.sleep(0.25)  # killer is executing (sleeps 0.5)
            t_victim = asyncio.create_task(
                ex.execute(innocent, [7], {}, dispatch_id="pair", node_id=1)
            )
            res = await asyncio.gather(t_killer, t_victim, return_exceptions=True)
            return res
        finally:
            await SSHExecutor.close_pool()

    killer_res, victim_res = asyncio.run(go())
    assert isinstance(killer_res, SSHTaskError)
    assert "execution had started" in str(killer_res)
    assert victim_res == 70
    assert marker.read_text().count("\n") == 1


def test_slot_pinning_without_gpu_lib(local_executor):
    """an endpoint with a GPU stack gets slot
    scheduling + HIP_VISIBLE_DEVICES pinning even when the dispatcher
    never built libcsp_gpu.so."""
    ex = local_executor(hip_visible_devices_policy="roundrobin", gpu_slots=2)
    # simulate the env probe's verdict: has_gpu=True, no library shipped
    transport_pool.store_check(ex._pool_key(), "env", (True, "", "", True))

    def report():
        import os

        return (os.environ.get("CSP_GPU_SLOT"), os.environ.get("HIP_VISIBLE_DEVICES"))

    async def go():
        out = []
        for i in range(2):
            out.append(await ex.execute(report, [], {}, dispatch_id="pin", node_id=i))
        return out

    results = asyncio.run(go())
    # both electrons were pinned, and round-robin spread them over GPUs
    assert {slot for slot, _ in results} == {"0", "1"}
    for slot, hip in results:
        assert hip == slot  # stub maps CSP_GPU_SLOT -> HIP_VISIBLE_DEVICES
    assert ex.last_task_record.gpu_id in (0, 1)


def test_fixed_policy_pins_to_fixed_gpu(local_executor):
    """`fixed` policy (per-rank bench drivers): every electron pins to
    fixed_gpu with no slot accounting."""
    ex = local_executor(hip_visible_devices_policy="fixed", fixed_gpu=3, gpu_slots=8)
    transport_pool.store_check(ex._pool_key(), "env", (True, "", "", True))

    def report():
        import os

        return (os.environ.get("CSP_GPU_SLOT"), os.environ.get("HIP_VISIBLE_DEVICES"))

    slot, hip = asyncio.run(ex.execute(report, [], {}, dispatch_id="fx", node_id=0))
    assert (slot, hip) == ("3", "3")
    assert ex.last_task_record.gpu_id == 3


def test_oversubscription_runs_distinct_workers(local_executor):
    """slots_per_gpu=N runs N warm worker PROCESSES per GPU (one per
    sub-slot), not N electrons pipelined through one process (VERDICT r1
    item 2a, CPU-side readiness half)."""
    ex = local_executor(
        persistent_workers=True,
        hip_visible_devices_policy="roundrobin",
        gpu_slots=1,
        slots_per_gpu=4,
    )
    transport_pool.store_check(ex._pool_key(), "env", (True, "", "", True))

    def report():
        import os
        import time

        time.sleep(0.2)  # hold the slot so the fan spreads over sub-slots
        return (os.getpid(), os.environ.get("CSP_GPU_SLOT"))

    async def go():
        try:
            return await asyncio.gather(
                *[
                    ex.execute(report, [], {}, dispatch_id="over", node_id=i)
                    for i in range(16)
                ]
            )
        finally:
            await SSHExecutor.close_pool()

    results = asyncio.run(go())
    pids = {pid for pid, _ in results}
    assert len(pids) == 4, pids  # one worker process per sub-slot
    assert {slot for _, slot in results} == {"0"}  # all pinned to GPU 0


def test_negative_env_check_expires(local_executor, tmp_path, monkeypatch):
    """a failing env check must heal after its TTL —
    a python-missing endpoint recovers once the remote is fixed."""
    monkeypatch.setattr(transport_pool, "NEGATIVE_CHECK_TTL", 0.0)
    pybin = tmp_path / "pybin"  # does not exist yet

    def add(x, y):
        return x + y

    ex = local_executor(python_path=str(pybin))
    with pytest.raises(RuntimeError):
        asyncio.run(ex.execute(add, [1, 2], {}))

    pybin.symlink_to(sys.executable)  # "remote" is fixed
    assert asyncio.run(ex.execute(add, [1, 2], {})) == 3


def test_negative_env_check_invalidated_on_reconnect(local_executor, tmp_path):
    """A reconnecting transport drops cached FAILURES (but a still-down
    endpoint keeps failing fast from cache within the TTL)."""
    pybin = tmp_path / "pybin"

    def add(x, y):
        return x + y

    ex = local_executor(python_path=str(pybin))
    with pytest.raises(RuntimeError):
        asyncio.run(ex.execute(add, [1, 2], {}))

    pybin.symlink_to(sys.executable)
    # within the TTL the cached failure still short-circuits
    with pytest.raises(RuntimeError):
        asyncio.run(ex.execute(add, [1, 2], {}))

    # simulate a dropped connection: the pool reconnects and must
    # invalidate the cached failure
    transport = transport_pool._pool[ex._pool_key()]
    transport._connected = False
    assert asyncio.run(ex.execute(add, [1, 2], {})) == 3


def test_slot_table_capacity_guard():
    """conflicting capacities no longer silently
    share the first-created table."""
    from covalent_ssh_plugin_amd.gpu import slots

    key = ("ssh", "guard-host", "u", "", "22", "")
    t1 = slots.get_slot_table(key, num_gpus=4, slots_per_gpu=1)
    assert t1.capacity == 4

    # idle mismatch: rebuilt to the new shape (warned)
    t2 = slots.get_slot_table(key, num_gpus=8, slots_per_gpu=2)
    assert t2 is not t1
    assert (t2.num_gpus, t2.slots_per_gpu) == (8, 2)

    # matching request returns the same table
    assert slots.get_slot_table(key, num_gpus=8, slots_per_gpu=2) is t2

    # in-use mismatch: hard error
    async def hold_and_conflict():
        slot = await t2.acquire()
        try:
            with pytest.raises(ValueError, match="in use"):
                slots.get_slot_table(key, num_gpus=2, slots_per_gpu=1)
        finally:
            await slot.release()

    asyncio.run(hold_and_conflict())
    # released: mismatch resizes again
    t3 = slots.get_slot_table(key, num_gpus=2, slots_per_gpu=1)
    assert (t3.num_gpus, t3.slots_per_gpu) == (2, 1)
