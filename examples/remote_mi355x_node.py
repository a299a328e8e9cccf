#!/usr/bin/env python3
"""Dispatching onto a remote 8xMI355X node over SSH (pooled ControlMaster).

Requires: key-based SSH access to the node; python + cloudpickle (and
torch-ROCm for GPU electrons) on the node.  The CDNA4 library
(libcsp_gpu.so) ships automatically, content-addressed, on first use.

    python examples/remote_mi355x_node.py user@mi355x-node
"""

import asyncio
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

from covalent_ssh_plugin_amd import SSHExecutor


def main():
    target = sys.argv[1] if len(sys.argv) > 1 else "user@mi355x-node"
    username, hostname = target.split("@", 1)

    ex = SSHExecutor(
        username=username,
        hostname=hostname,
        gpu_slots=8,                 # one slot per MI355X GPU
        persistent_workers=True,     # warm worker + HIP context per GPU
        warmup_gpu=True,             # CDNA4 MFMA/HBM warm-up before tasks
    )

    def gpu_electron(n):
        import torch

        a = torch.randn(n, n, device="cuda", dtype=torch.bfloat16)
        b = torch.randn(n, n, device="cuda", dtype=torch.bfloat16)
        c = a @ b
        torch.cuda.synchronize()
        return c  # returned via hipHostMalloc-pinned staging

    async def run():
        # 16 electrons fan out round-robin across the node's 8 GPUs
        outs = await asyncio.gather(
            *[
                ex.execute(gpu_electron, [4096], {}, dispatch_id="fan", node_id=i)
                for i in range(16)
            ]
        )
        print("shapes:", [tuple(o.shape) for o in outs])
        print("gpu spread:", ex.stats()["gpu_spread"])
        await SSHExecutor.close_pool()

    asyncio.run(run())


if __name__ == "__main__":
    main()
