"""End-to-end executor pipeline tests over the loopback transport.

Covers the behaviors the reference unit tests assert (reference
tests/ssh_test.py — orchestration, failure policy, unique workdir,
staging filename contract) but against the real pipeline: files are
actually staged, a real stub subprocess runs the task, the result pickle
round-trips.
"""

import asyncio
import json
import sys

import pytest

from covalent_ssh_plugin_amd import SSHExecutor


def _make_add():
    # defined inside a function so cloudpickle serializes it BY VALUE —
    # the stub process cannot import the test module
    def add(x, y):
        return x + y

    return add


_add = _make_add()


def test_fused_roundtrip(local_executor):
    ex = local_executor()
    result = asyncio.run(ex.execute(_add, [3, 4], {}, dispatch_id="d", node_id=0))
    assert result == 7
    rec = ex.last_task_record
    assert rec.operation_id == "d_0"
    assert rec.phases["dispatch"] > 0
    assert rec.remote_meta is not None and "total" in rec.remote_meta["phases_ms"]
    # every fused return is integrity-checked against the stub's hash
    assert len(rec.remote_meta["result_sha256"]) == 64


def test_template_path_roundtrip(local_executor):
    """The discrete upload/submit/poll/fetch path (reference §3.1 flow)."""
    ex = local_executor(batch_roundtrips=False, poll_freq=1)
    result = asyncio.run(ex.execute(_add, [1], {"y": 2}, dispatch_id="d", node_id=1))
    assert result == 3
    assert "upload" in ex.last_task_record.phases
    assert "poll" in ex.last_task_record.phases


def test_task_exception_reraised(local_executor):
    def boom():
        raise ValueError("inner failure")

    ex = local_executor()
    with pytest.raises(ValueError, match="inner failure"):
        asyncio.run(ex.execute(boom, [], {}))


def test_kwargs_and_rich_types(local_executor):
    def fn(a, b=None, scale=1):
        return {"sum": (a + sum(b)) * scale, "list": [a, b]}

    ex = local_executor()
    out = asyncio.run(ex.execute(fn, [1], {"b": [2, 3], "scale": 2}))
    assert out == {"sum": 12, "list": [1, [2, 3]]}


def test_closure_serialization(local_executor):
    offset = 100

    def fn(x):
        return x + offset

    ex = local_executor()
    assert asyncio.run(ex.execute(fn, [1], {})) == 101


def test_workdir_default_and_unique(local_executor):
    def cwd_name():
        import os

        return os.getcwd()

    ex = local_executor()
    out = asyncio.run(ex.execute(cwd_name, [], {}, dispatch_id="dx", node_id=5))
    assert out.endswith("covalent-workdir")

    ex2 = local_executor(create_unique_workdir=True)
    out2 = asyncio.run(ex2.execute(cwd_name, [], {}, dispatch_id="dx", node_id=5))
    # reference layout: {workdir}/{dispatch_id}/node_{node_id}
    # (reference ssh.py:486-491, asserted at ssh_test.py:309)
    assert out2.endswith("covalent-workdir/dx/node_5")


def test_remote_python_missing_raises(local_executor):
    ex = local_executor(python_path="/nonexistent/python3")
    with pytest.raises(RuntimeError):
        asyncio.run(ex.execute(_add, [1, 2], {}))


def test_remote_python_missing_falls_back_local(local_executor):
    ex = local_executor(python_path="/nonexistent/python3", run_local_on_ssh_fail=True)
    # reference behavior: compute locally on the dispatcher
    # (reference ssh.py:202-204, asserted at ssh_test.py:72-110)
    assert asyncio.run(ex.execute(_add, [5, 20], {})) == 25


def test_task_stdout_does_not_corrupt_result(local_executor):
    def chatty():
        print("binary-ish output \x00\xff and markers --CSP-RESULT-- etc")
        sys.stderr.write("stderr noise\n")
        return 42

    ex = local_executor()
    assert asyncio.run(ex.execute(chatty, [], {})) == 42


def test_cleanup_removes_files(local_executor):
    ex = local_executor()
    asyncio.run(ex.execute(_add, [1, 2], {}, dispatch_id="dc", node_id=0))
    leftovers = [p.name for p in local_executor.cache.glob("*dc_0*")]
    assert leftovers == []
    remote_leftovers = [
        p.name for p in (local_executor.home / ".cache/covalent").glob("*dc_0*")
    ]
    assert remote_leftovers == []


def test_no_cleanup_keeps_files(local_executor):
    ex = local_executor(do_cleanup=False)
    asyncio.run(ex.execute(_add, [1, 2], {}, dispatch_id="dk", node_id=0))
    remote = local_executor.home / ".cache/covalent"
    names = {p.name for p in remote.glob("*dk_0*")}
    assert "function_dk_0.pkl" in names
    assert "exec_dk_0.py" in names
    assert "result_dk_0.pkl" in names
    # local result copy kept as well
    assert (local_executor.cache / "result_dk_0.pkl").exists()


def test_staging_filename_contract(local_executor):
    """Filenames must match the reference contract exactly (SURVEY.md
    §2.3; reference ssh_test.py:319-360)."""
    ex = local_executor()
    paths = ex._task_paths("dispatch_3")
    assert paths["function_remote"] == ".cache/covalent/function_dispatch_3.pkl"
    assert paths["script_remote"] == ".cache/covalent/exec_dispatch_3.py"
    assert paths["result_remote"] == ".cache/covalent/result_dispatch_3.pkl"
    assert paths["function_local"].endswith("function_dispatch_3.pkl")
    assert paths["script_local"].endswith("exec_dispatch_3.py")


def test_result_pickle_is_two_tuple(local_executor):
    """On-disk result format parity: pickle of (result, exception)
    (reference exec.py:44-46, ssh.py:455-458)."""
    import pickle

    ex = local_executor(do_cleanup=False)
    asyncio.run(ex.execute(_add, [2, 3], {}, dispatch_id="fmt", node_id=0))
    blob = (local_executor.home / ".cache/covalent/result_fmt_0.pkl").read_bytes()
    result, exception = pickle.loads(blob)
    assert result == 5 and exception is None


def test_concurrent_electrons(local_executor):
    ex = local_executor()

    async def main():
        tasks = [
            ex.execute(_add, [i, i], {}, dispatch_id="cc", node_id=i) for i in range(12)
        ]
        return await asyncio.gather(*tasks)

    results = asyncio.run(main())
    assert results == [2 * i for i in range(12)]
    assert len(ex.task_records) == 12


def test_stats_summary(local_executor):
    ex = local_executor()

    async def main():
        for i in range(4):
            await ex.execute(_add, [i, 1], {}, dispatch_id="st", node_id=i)

    asyncio.run(main())
    s = ex.stats()
    assert s["count"] == 4
    assert s["p50_ms"] > 0
    assert "dispatch" in s["phase_mean_ms"]


def test_task_log_jsonl(local_executor, tmp_path, monkeypatch):
    log = tmp_path / "tasks.jsonl"
    monkeypatch.setenv("CSP_AMD_TASK_LOG", str(log))
    ex = local_executor()

    async def main():
        for i in range(3):
            await ex.execute(_add, [i, 1], {}, dispatch_id="log", node_id=i)

    asyncio.run(main())
    lines = [json.loads(l) for l in log.read_text().splitlines()]
    assert len(lines) == 3
    assert lines[0]["operation_id"] == "log_0"
    assert "dispatch" in lines[0]["phases"]


def test_cold_start_env_checks_run_once(local_executor):
    """A cold-start fan of electrons must run the per-endpoint env checks
    once, not once per electron."""
    ex = local_executor()
    calls = []
    orig = ex._ensure_environment_locked

    async def counting(transport, key):
        calls.append(1)
        return await orig(transport, key)

    ex._ensure_environment_locked = counting

    async def main():
        return await asyncio.gather(
            *[ex.execute(_add, [i, 1], {}, dispatch_id="cold", node_id=i) for i in range(8)]
        )

    out = asyncio.run(main())
    assert out == [i + 1 for i in range(8)]
    assert len(calls) == 1, calls


def test_counters(local_executor):
    ex = local_executor()

    def boom():
        raise ValueError("x")

    async def main():
        await ex.execute(_add, [1, 2], {}, dispatch_id="cn", node_id=0)
        try:
            await ex.execute(boom, [], {}, dispatch_id="cn", node_id=1)
        except ValueError:
            pass

    asyncio.run(main())
    s = ex.stats()
    assert s["counters"]["tasks"] == 2
    assert s["counters"]["task_exceptions"] == 1
    assert s["counters"]["ssh_failures"] == 0
