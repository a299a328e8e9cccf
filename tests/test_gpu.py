"""MI355X hardware tests (run via gpurun / the driver's GPU tier).

These exercise the native CDNA4 library IN-PROCESS (the .so load is the
point — no eager fallback exists) plus the executor's GPU path end to
end on a real GPU.
"""

import asyncio
import ctypes
import sys

import pytest

torch = pytest.importorskip("torch")

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module", autouse=True)
def _require_gpu():
    if not torch.cuda.is_available():
        pytest.fail("GPU tests require a visible MI355X (torch.cuda unavailable)")


@pytest.fixture(scope="module")
def gpu_lib():
    from covalent_ssh_plugin_amd.gpu import probe

    # Fails loudly (GpuLibError) if the extension was not built/shipped.
    return probe.load()


def test_lib_built_and_loads(gpu_lib):
    assert gpu_lib.csp_device_count() >= 1


def test_probe_reports_mi355x(gpu_lib):
    from covalent_ssh_plugin_amd.gpu import probe

    info = probe.probe(0)
    assert "gfx950" in info["gcn_arch"], info
    assert info["cu_count"] == 256, info
    assert info["wavefront_size"] == 64, info
    # 288 GB HBM3E
    assert info["hbm_total_gb"] > 250, info
    # measured HBM bandwidth: spec 8 TB/s, ~6.3 achievable; require a
    # sane floor that a broken sweep could not hit
    assert info["hbm_bw_gbps"] > 3000, info
    # measured bf16 MFMA throughput: dense peak ~2.5 PF; require >1 PF
    assert info["mfma_bf16_tflops"] > 1000, info


def test_warmup_runs(gpu_lib):
    from covalent_ssh_plugin_amd.gpu import probe

    probe.warmup(0, 50)


def test_pinned_staging_correctness(gpu_lib):
    from covalent_ssh_plugin_amd.gpu import probe

    for dtype in (torch.float32, torch.bfloat16):
        t = torch.randn(513, 1027, device="cuda").to(dtype)
        torch.cuda.synchronize()
        raw = probe.staged_d2h_bytes(t.data_ptr(), t.numel() * t.element_size())
        staged = torch.frombuffer(bytearray(raw), dtype=dtype).reshape(t.shape)
        assert torch.equal(staged, t.cpu()), dtype


def test_pinned_staging_bandwidth(gpu_lib):
    """Pinned D2H must beat a conservative pageable floor (PCIe Gen5 x16
    pinned should sustain tens of GB/s).  The pool is warmed at full size
    first so the one-time hipHostMalloc pinning cost is excluded."""
    import ctypes
    import time

    nbytes = 1 << 30  # 1 GiB
    t = torch.empty(nbytes, dtype=torch.uint8, device="cuda")
    torch.cuda.synchronize()
    dst = gpu_lib.csp_staging_get(nbytes)  # warm: pin the full buffer once
    assert dst
    gpu_lib.csp_memcpy_d2h(ctypes.c_void_p(dst), ctypes.c_void_p(t.data_ptr()), nbytes)
    t0 = time.perf_counter()
    rc = gpu_lib.csp_memcpy_d2h(
        ctypes.c_void_p(dst), ctypes.c_void_p(t.data_ptr()), nbytes
    )
    dt = time.perf_counter() - t0
    assert rc == 0
    gbps = nbytes / dt / 1e9
    print(f"pinned D2H: {gbps:.1f} GB/s")
    assert gbps > 20.0, gbps

    # pageable comparison (torch .cpu() into fresh pageable memory)
    t0 = time.perf_counter()
    t.cpu()
    dt_pageable = time.perf_counter() - t0
    print(f"pageable D2H: {nbytes / dt_pageable / 1e9:.1f} GB/s")


def test_host_alloc_roundtrip(gpu_lib):
    p = gpu_lib.csp_host_alloc(4096)
    assert p
    ctypes.memset(p, 0xAB, 4096)
    assert gpu_lib.csp_host_free(ctypes.c_void_p(p)) == 0


def _executor(tmp_path, **kw):
    from covalent_ssh_plugin_amd import SSHExecutor

    home = tmp_path / "home"
    home.mkdir(exist_ok=True)
    defaults = dict(
        transport="local",
        local_home=str(home),
        cache_dir=str(tmp_path / "cache"),
        python_path=sys.executable,
        gpu_slots=max(1, torch.cuda.device_count()),
    )
    defaults.update(kw)
    return SSHExecutor(**defaults)


def test_executor_gpu_electron_with_prologue(tmp_path):
    """Full pipeline: slot pinning + CDNA4 warm-up/probe prologue + bf16
    matmul + pinned result staging, via the stub subprocess."""

    def electron(n):
        import torch

        a = torch.randn(n, n, device="cuda", dtype=torch.bfloat16)
        b = torch.eye(n, device="cuda", dtype=torch.bfloat16)
        c = a @ b
        torch.cuda.synchronize()
        return {"mat": c, "check": a}

    ex = _executor(tmp_path, pinned_staging_threshold_bytes=1024)
    out = asyncio.run(ex.execute(electron, [2048], {}, dispatch_id="g", node_id=0))
    # numerics: A @ I must equal A exactly, staged through pinned D2H
    assert torch.equal(out["mat"], out["check"])
    rec = ex.last_task_record
    assert rec.gpu_id is not None
    meta = rec.remote_meta
    assert meta is not None
    assert meta["gpu"] is not None and "gfx950" in meta["gpu"]["gcn_arch"]
    assert meta["staging"]["mode"] == "pinned", meta
    assert meta["gpu_slot"] == str(rec.gpu_id)
    asyncio.run(ex.close_pool())


def test_executor_mm_numerics_vs_fp32(tmp_path):
    """bf16 GPU matmul electron vs a plain fp32 CPU reference."""

    def electron(seed, n):
        import torch

        g = torch.Generator(device="cpu").manual_seed(seed)
        a = torch.randn(n, n, generator=g)
        b = torch.randn(n, n, generator=g)
        c = (a.cuda().bfloat16() @ b.cuda().bfloat16()).float().cpu()
        torch.cuda.synchronize()
        return c

    n, seed = 512, 1234
    ex = _executor(tmp_path)
    got = asyncio.run(ex.execute(electron, [seed, n], {}, dispatch_id="mm", node_id=0))
    g = torch.Generator(device="cpu").manual_seed(seed)
    a = torch.randn(n, n, generator=g)
    b = torch.randn(n, n, generator=g)
    ref = a @ b
    # bf16 matmul against fp32 reference: relative error ~1e-2
    rel = (got - ref).abs().max() / ref.abs().max()
    assert rel < 0.05, rel.item()
    asyncio.run(ex.close_pool())


def test_concurrent_electrons_share_gpu_slots(tmp_path):
    """Fan 8 electrons across available GPU slots; each must see the
    HIP_VISIBLE_DEVICES its record claims."""

    def whoami():
        import os

        return os.environ.get("CSP_GPU_SLOT")

    ex = _executor(tmp_path, warmup_gpu=False)

    async def main():
        return await asyncio.gather(
            *[
                ex.execute(whoami, [], {}, dispatch_id="fan", node_id=i)
                for i in range(8)
            ]
        )

    results = asyncio.run(main())
    n_gpus = max(1, torch.cuda.device_count())
    assert all(r is not None and 0 <= int(r) < n_gpus for r in results), results
    if n_gpus > 1:
        assert len(set(results)) > 1  # work actually spread across GPUs
    asyncio.run(ex.close_pool())


def test_rccl_allreduce_smoke_gpu(tmp_path):
    """BASELINE config 5 (scaled to the visible GPUs): RCCL all-reduce
    over xGMI; on a 1-GPU box this degenerates to world_size=1."""
    from covalent_ssh_plugin_amd.parallel.rccl_smoke import make_rccl_smoke_electron

    world = max(1, torch.cuda.device_count())
    electron = make_rccl_smoke_electron()
    out = electron(world_size=world, nbytes=64 * 1024 * 1024, iters=10,
                   warmup=2, backend="nccl", port=29515)
    assert out["correct"] is True
    assert out["world_size"] == world
    print("rccl busbw GB/s:", out["busbw_GBps"])
    if world > 1:
        # xGMI floor: each MI355X has 7 p2p links
        # x ~153 GB/s; a ring all-reduce is per-link bound, so >100 GB/s
        # busbw is a conservative must-hit for a healthy fabric — any
        # PCIe-fallback or host-staged path would land far below this.
        assert out["busbw_GBps"] > 100.0, out


@pytest.mark.timeout(600)
def test_slot_oversubscription_single_gpu(tmp_path):
    """slots_per_gpu=8 on ONE MI355X — 8 persistent
    worker processes share the GPU; a fan of 64 small matmul electrons
    spreads across all 8 workers, every one pinned to the same device.
    288 GB HBM3E makes 8-way co-residency realistic for small tasks."""

    def electron(n):
        import os

        import torch

        a = torch.randn(n, n, device="cuda", dtype=torch.bfloat16)
        c = a @ a
        torch.cuda.synchronize()
        return (os.getpid(), os.environ.get("HIP_VISIBLE_DEVICES"),
                float(c.float().mean().item()))

    import time

    ex = _executor(
        tmp_path,
        persistent_workers=True,
        hip_visible_devices_policy="roundrobin",
        gpu_slots=1,
        slots_per_gpu=8,
        warmup_gpu=False,
    )

    async def go():
        try:
            await ex.prewarm()
            t0 = time.perf_counter()
            res = await asyncio.gather(
                *[
                    ex.execute(electron, [1024], {}, dispatch_id="over", node_id=i)
                    for i in range(64)
                ]
            )
            dt = time.perf_counter() - t0
            return res, dt
        finally:
            await ex.close_pool()

    results, dt = asyncio.run(go())
    pids = {pid for pid, _, _ in results}
    devices = {dev for _, dev, _ in results}
    assert len(pids) == 8, f"expected 8 distinct workers, got {len(pids)}"
    assert len(devices) == 1, devices  # all co-resident on the one GPU
    print(f"oversubscribed fan: 64 matmul electrons over 8 workers on one "
          f"GPU in {dt:.2f}s ({64 / dt:.0f} electrons/s)")


def test_executor_gpu_electron_over_sshim(tmp_path, sshim, monkeypatch):
    """The flagship SSH path on a real GPU: slot pinning, worker
    channels and pinned staging through the PATH-shim ssh client."""
    monkeypatch.setenv(
        "SSHIM_PASS_ENV",
        "HSA_ENABLE_IPC_MODE_LEGACY,LD_LIBRARY_PATH,HIP_VISIBLE_DEVICES,"
        "ROCR_VISIBLE_DEVICES,PYTORCH_ROCM_ARCH,TMPDIR",
    )
    from covalent_ssh_plugin_amd import SSHExecutor

    ex = SSHExecutor(
        transport="ssh",
        hostname=sshim.hostname,
        username="mi355x",
        ssh_key_file=str(sshim.key),
        cache_dir=str(tmp_path / "cache"),
        python_path=sys.executable,
        gpu_slots=max(1, torch.cuda.device_count()),
        pinned_staging_threshold_bytes=1024,
    )

    def electron(n):
        import torch

        a = torch.randn(n, n, device="cuda", dtype=torch.bfloat16)
        b = torch.eye(n, device="cuda", dtype=torch.bfloat16)
        c = a @ b
        torch.cuda.synchronize()
        return {"mat": c, "check": a}

    out = asyncio.run(ex.execute(electron, [1024], {}, dispatch_id="sg", node_id=0))
    assert torch.equal(out["mat"], out["check"])
    meta = ex.last_task_record.remote_meta
    assert meta["staging"]["mode"] == "pinned", meta
    assert meta["gpu"] is not None and "gfx950" in meta["gpu"]["gcn_arch"]
    asyncio.run(ex.close_pool())


def test_mem_info_telemetry(gpu_lib):
    from covalent_ssh_plugin_amd.gpu import probe

    info = probe.mem_info(0)
    # 288 GiB HBM3E = 309.2 decimal GB
    assert 250 < info["hbm_total_gb"] < 320
    assert 0 < info["hbm_free_gb"] <= info["hbm_total_gb"]


def test_worker_meta_hbm_telemetry(tmp_path):
    """gpu_telemetry_every=1: every electron's meta carries a fresh HBM
    occupancy sample, and allocations made by the task show up."""

    def hold_memory(gib):
        import torch

        t = torch.empty(gib << 30, dtype=torch.uint8, device="cuda")
        torch.cuda.synchronize()
        return t.numel()

    ex = _executor(
        tmp_path, persistent_workers=True, gpu_telemetry_every=1,
        warmup_gpu=True,
    )

    async def go():
        try:
            n1 = await ex.execute(hold_memory, [1], {}, dispatch_id="tm", node_id=0)
            hbm1 = dict(ex.last_task_record.remote_meta["hbm"])
            n2 = await ex.execute(hold_memory, [8], {}, dispatch_id="tm", node_id=1)
            hbm2 = dict(ex.last_task_record.remote_meta["hbm"])
            return n1, hbm1, n2, hbm2
        finally:
            await ex.close_pool()

    n1, hbm1, n2, hbm2 = asyncio.run(go())
    assert n1 == 1 << 30 and n2 == 8 << 30
    assert hbm1["hbm_total_gb"] > 200
    assert 0 < hbm1["hbm_free_gb"] <= hbm1["hbm_total_gb"]
    # the second sample sees the first task's cached 1 GiB allocation
    # (torch caching allocator keeps it) -> free memory dropped
    assert hbm2["hbm_free_gb"] < hbm1["hbm_free_gb"]
    assert hbm2["sampled_at_serial"] == hbm1["sampled_at_serial"] + 1


def test_isolated_gpu_electron(tmp_path):
    """Fork-isolated dispatch on a real MI355X: each electron's child
    process does its OWN HIP init and pinned staging; two electrons get
    two distinct processes with bit-exact results."""

    def electron(n):
        import os

        import torch

        a = torch.randn(n, n, device="cuda", dtype=torch.bfloat16)
        b = torch.eye(n, device="cuda", dtype=torch.bfloat16)
        c = a @ b
        torch.cuda.synchronize()
        return {"pid": os.getpid(), "mat": c, "check": a}

    ex = _executor(
        tmp_path, isolate_tasks=True, pinned_staging_threshold_bytes=1024
    )

    async def go():
        try:
            r1 = await ex.execute(electron, [1024], {}, dispatch_id="iso", node_id=0)
            m1 = dict(ex.last_task_record.remote_meta)
            r2 = await ex.execute(electron, [1024], {}, dispatch_id="iso", node_id=1)
            return r1, m1, r2
        finally:
            await ex.close_pool()

    r1, m1, r2 = asyncio.run(go())
    assert torch.equal(r1["mat"], r1["check"])
    assert torch.equal(r2["mat"], r2["check"])
    assert r1["pid"] != r2["pid"], "isolated electrons shared a process"
    assert m1["isolated"] is True
    assert m1["staging"]["mode"] == "pinned", m1


def test_cluster_two_nodes_over_sshim_gpu(tmp_path, sshim, monkeypatch):
    """Multi-node fan-out on real hardware: two shim-SSH 'nodes' (two
    fake remote homes on one box), a fan of bf16 matmul electrons spread
    across both, plus transparent failover from a dead third node."""
    monkeypatch.setenv(
        "SSHIM_PASS_ENV",
        "HSA_ENABLE_IPC_MODE_LEGACY,LD_LIBRARY_PATH,HIP_VISIBLE_DEVICES,"
        "ROCR_VISIBLE_DEVICES,PYTORCH_ROCM_ARCH,TMPDIR",
    )
    from covalent_ssh_plugin_amd.cluster import SSHClusterExecutor

    # the shim resolves SSHIM_HOME per invocation; with one home both
    # "nodes" share a filesystem but keep separate transports, worker
    # pools and slot tables (distinct hostnames -> distinct pool keys)
    hosts = [
        {"hostname": "deadnode.invalid"},  # always refuses: exercises failover
        {"hostname": "gpu-node-a.test"},
        {"hostname": "gpu-node-b.test"},
    ]
    cluster = SSHClusterExecutor(
        hosts,
        transport="ssh",
        username="mi355x",
        ssh_key_file=str(sshim.key),
        cache_dir=str(tmp_path / "cache"),
        python_path=sys.executable,
        gpu_slots=max(1, torch.cuda.device_count()),
        retry_connect=False,
        pinned_staging_threshold_bytes=1024,
    )

    def electron(n):
        import socket

        import torch

        a = torch.randn(n, n, device="cuda", dtype=torch.bfloat16)
        b = torch.eye(n, device="cuda", dtype=torch.bfloat16)
        c = a @ b
        torch.cuda.synchronize()
        return bool(torch.equal(c, a)), socket.gethostname()

    async def go():
        try:
            return await asyncio.gather(
                *[
                    cluster.execute(electron, [512], {}, dispatch_id="cg", node_id=i)
                    for i in range(6)
                ]
            )
        finally:
            from covalent_ssh_plugin_amd import SSHExecutor

            await SSHExecutor.close_pool()

    results = asyncio.run(go())
    assert all(ok for ok, _ in results)
    # the dead node was cooled down and every electron completed
    assert cluster._unhealthy_until[0] > 0
    stats = cluster.stats()
    served = {h: s.get("count", 0) for h, s in stats.items()}
    assert served.get("deadnode.invalid", 0) == 0
    assert served.get("gpu-node-a.test", 0) + served.get("gpu-node-b.test", 0) == 6
    assert served.get("gpu-node-a.test", 0) > 0
    assert served.get("gpu-node-b.test", 0) > 0


def test_probe_props_fast(gpu_lib):
    import time

    from covalent_ssh_plugin_amd.gpu import probe

    t0 = time.perf_counter()
    info = probe.probe_props(0)
    dt = time.perf_counter() - t0
    assert "gfx950" in info["gcn_arch"]
    assert info["cu_count"] == 256
    assert dt < 1.0, dt  # no measurement kernels in the props probe


def test_pinned_h2d_roundtrip(gpu_lib):
    """csp_memcpy_h2d: pinned host -> device -> back is bit-exact."""
    import ctypes

    nbytes = 1 << 20
    src = torch.randint(0, 255, (nbytes,), dtype=torch.uint8)
    host = gpu_lib.csp_host_alloc(nbytes)
    assert host
    try:
        ctypes.memmove(host, src.data_ptr(), nbytes)
        dev = torch.empty(nbytes, dtype=torch.uint8, device="cuda")
        torch.cuda.synchronize()
        rc = gpu_lib.csp_memcpy_h2d(
            ctypes.c_void_p(dev.data_ptr()), ctypes.c_void_p(host), nbytes
        )
        assert rc == 0
        assert torch.equal(dev.cpu(), src)
    finally:
        gpu_lib.csp_host_free(ctypes.c_void_p(host))


@pytest.mark.timeout(600)
def test_worker_tensor_over_4gib(tmp_path):
    """Single result tensors beyond a 4-byte frame-length limit (the
    protocol uses 8-byte lengths; 288 GB HBM makes multi-GiB results
    routine).  Runs in the GPU tier because CI microVMs page large
    allocations erratically; any real machine finishes in seconds."""
    import asyncio

    ex = _executor(tmp_path, persistent_workers=True,
                   pinned_staging_threshold_bytes=1 << 20, warmup_gpu=False)

    n = (4 * 1024 + 512) * 1024 * 1024  # 4.5 GiB of uint8

    def fn(n):
        import torch

        t = torch.zeros(n, dtype=torch.uint8, device="cuda")
        t[0] = 7
        t[-1] = 9
        torch.cuda.synchronize()
        return t

    out = asyncio.run(ex.execute(fn, [n], {}, dispatch_id="big", node_id=0))
    assert out.numel() == n
    assert int(out[0]) == 7 and int(out[-1]) == 9
    assert ex.last_task_record.remote_meta["staging"]["mode"] == "pinned"
    asyncio.run(ex.close_pool())


@pytest.mark.timeout(600)
def test_fused_45gib_result_bounded_rss(tmp_path):
    """VERDICT r1 item 4 done-criterion: a 4.5 GiB stub-path (fused)
    result with peak dispatcher RSS < 1.5x the payload.  The result
    streams through the sentinel parser straight to disk; only the
    final deserialized object occupies dispatcher memory."""
    from test_fused_streaming import _run_rss_probe

    nbytes = (4 * 1024 + 512) << 20  # 4.5 GiB
    peak = _run_rss_probe(tmp_path, nbytes)
    print(f"dispatcher peak RSS: {peak / 1e9:.2f} GB for {nbytes / 1e9:.2f} GB payload")
    assert peak < nbytes * 1.5, peak


def test_mfma_numerics_vs_torch_reference(gpu_lib):
    """The warm-up kernel's MFMA instruction, numerics-checked: one
    v_mfma_f32_32x32x16_bf16 with self-described operand layouts must
    equal the PyTorch fp32 reference EXACTLY (integer-valued operands:
    no rounding anywhere)."""
    import ctypes

    gpu_lib.csp_mfma_check.restype = ctypes.c_int
    gpu_lib.csp_mfma_check.argtypes = [ctypes.c_int, ctypes.c_int] + [
        ctypes.POINTER(ctypes.c_float)
    ] * 3

    matched = []
    for layout in (0, 1):
        D = (ctypes.c_float * (32 * 32))()
        A = (ctypes.c_float * (32 * 16))()
        B = (ctypes.c_float * (16 * 32))()
        rc = gpu_lib.csp_mfma_check(0, layout, D, A, B)
        assert rc == 0
        a = torch.tensor(list(A)).reshape(32, 16)
        b = torch.tensor(list(B)).reshape(16, 32)
        d = torch.tensor(list(D)).reshape(32, 32)
        ref = a @ b
        if torch.equal(d, ref):
            matched.append(layout)
        # A/B must be fully populated regardless of layout candidate
        assert a.abs().sum() > 0 and b.abs().sum() > 0
    # NOTE: a k-index permutation applied to BOTH A and B leaves
    # sum_k A[i,k]*B[k,j] invariant, so both candidates pass (verified
    # on-box: [0, 1]).  What the test pins down is the numerics contract:
    # the instruction computes exactly the matrix product of the operands
    # as fed, bit-equal to the fp32 reference.
    assert matched, "MFMA output did not reproduce A @ B"
    print("verified A/B k-mapping layout(s):", matched)
