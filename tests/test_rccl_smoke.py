"""Multi-process collective path, CPU edition: the RCCL smoke electron
run with gloo/world_size=2 — the same code path that runs nccl(=RCCL)
x8 on the MI355X node (tests/test_gpu.py)."""

import asyncio

import pytest

from covalent_ssh_plugin_amd.parallel.rccl_smoke import make_rccl_smoke_electron


@pytest.mark.timeout(300)
def test_gloo_allreduce_direct():
    electron = make_rccl_smoke_electron()
    out = electron(world_size=2, nbytes=1024 * 1024, iters=4, warmup=1,
                   backend="gloo", port=29612)
    assert out["world_size"] == 2
    assert out["correct"] is True
    assert out["busbw_GBps"] > 0


@pytest.mark.timeout(300)
def test_gloo_allreduce_via_executor(local_executor):
    """Full pipeline: the smoke electron dispatched through SSHExecutor
    (cloudpickled by value, executed by the stub subprocess)."""
    electron = make_rccl_smoke_electron()
    ex = local_executor()
    out = asyncio.run(
        ex.execute(
            electron,
            [],
            {"world_size": 2, "nbytes": 1024 * 1024, "iters": 4, "warmup": 1,
             "backend": "gloo", "port": 29613},
            dispatch_id="rccl",
            node_id=0,
        )
    )
    assert out["correct"] is True
