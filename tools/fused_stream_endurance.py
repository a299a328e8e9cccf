"""Endurance of the streamed fused (stub) path: repeated 1 GiB GPU
tensor returns through sentinel streaming to disk; dispatcher peak RSS
must stay flat and far below the cumulative payload."""
import asyncio
import os
import sys
import tempfile
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from covalent_ssh_plugin_amd import SSHExecutor


def vmhwm_mb() -> float:
    for line in open("/proc/self/status"):
        if line.startswith("VmHWM:"):
            return int(line.split()[1]) / 1024
    return -1.0


def electron(nbytes):
    import torch

    t = torch.ones(nbytes // 2, device="cuda", dtype=torch.bfloat16)
    torch.cuda.synchronize()
    return t


async def main(iters=15, nbytes=1 << 30):
    home = tempfile.mkdtemp()
    cache = tempfile.mkdtemp()
    ex = SSHExecutor(
        transport="local", local_home=home, cache_dir=cache,
        python_path=sys.executable, pinned_staging_threshold_bytes=1 << 20,
        gpu_slots=1,
    )
    for i in range(iters):
        t0 = time.perf_counter()
        out = await ex.execute(electron, [nbytes], {}, dispatch_id="fe", node_id=i)
        dt = time.perf_counter() - t0
        assert out.numel() == nbytes // 2
        assert float(out[0]) == 1.0 and float(out[-1]) == 1.0
        del out
        print(f"[{i:2d}] {dt:6.2f}s  {nbytes / dt / 1e9:5.2f} GB/s  "
              f"dispatcher VmHWM {vmhwm_mb():7.1f} MB", flush=True)
    await SSHExecutor.close_pool()

asyncio.run(main())
