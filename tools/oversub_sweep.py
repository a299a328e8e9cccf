"""Oversubscription contention sweep on ONE MI355X.

For slots_per_gpu in {1, 2, 4, 8}: run a fan of bf16 matmul electrons
through that many co-resident persistent workers on one GPU and report
aggregate throughput.  288 GB HBM3E makes co-residency cheap memory-wise;
this measures the compute/scheduling contention instead.
"""

import argparse
import asyncio
import os
import sys
import tempfile
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from covalent_ssh_plugin_amd import SSHExecutor


def electron(n):
    import torch

    a = torch.randn(n, n, device="cuda", dtype=torch.bfloat16)
    c = a @ a
    torch.cuda.synchronize()
    return float(c.float().mean().item())


async def sweep_one(slots_per_gpu: int, fan: int, n: int) -> dict:
    home = tempfile.mkdtemp()
    cache = tempfile.mkdtemp()
    ex = SSHExecutor(
        transport="local",
        local_home=home,
        cache_dir=cache,
        python_path=sys.executable,
        persistent_workers=True,
        hip_visible_devices_policy="roundrobin",
        gpu_slots=1,
        slots_per_gpu=slots_per_gpu,
        warmup_gpu=False,
    )
    await ex.prewarm()
    # warm the torch/HIP path in every worker
    await asyncio.gather(
        *[
            ex.execute(electron, [256], {}, dispatch_id="warm", node_id=i)
            for i in range(slots_per_gpu * 2)
        ]
    )
    t0 = time.perf_counter()
    await asyncio.gather(
        *[
            ex.execute(electron, [n], {}, dispatch_id="sweep", node_id=i)
            for i in range(fan)
        ]
    )
    dt = time.perf_counter() - t0
    await SSHExecutor.close_pool()
    return {"slots": slots_per_gpu, "fan": fan, "n": n, "secs": dt,
            "electrons_per_sec": fan / dt}


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--fan", type=int, default=128)
    ap.add_argument("--n", type=int, default=2048)
    ap.add_argument("--slots", default="1,2,4,8")
    args = ap.parse_args()
    for s in [int(x) for x in args.slots.split(",")]:
        out = asyncio.run(sweep_one(s, args.fan, args.n))
        print(
            f"slots_per_gpu={out['slots']}: {out['fan']} x mm{args.n}^2 bf16 "
            f"in {out['secs']:.2f}s = {out['electrons_per_sec']:.1f} electrons/s",
            flush=True,
        )


if __name__ == "__main__":
    main()
