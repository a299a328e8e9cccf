#!/usr/bin/env python3
"""Standalone usage (no covalent server): dispatch electrons through the
executor on this machine via the loopback transport.

    python examples/standalone_dispatch.py
"""

import asyncio
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

from covalent_ssh_plugin_amd import SSHExecutor


def main():
    ex = SSHExecutor(
        transport="local",
        python_path=sys.executable,
        persistent_workers=True,  # warm worker per GPU slot / CPU pool
    )

    def electron(n):
        # runs in the worker process; uses the GPU slot if one is visible
        try:
            import torch

            if torch.cuda.is_available():
                a = torch.randn(n, n, device="cuda", dtype=torch.bfloat16)
                return float((a @ a).float().mean())
        except ImportError:
            pass
        return sum(i * i for i in range(n))

    async def run():
        results = await asyncio.gather(
            *[
                ex.execute(electron, [256], {}, dispatch_id="demo", node_id=i)
                for i in range(8)
            ]
        )
        print("results:", results)
        print("stats:", ex.stats())
        await SSHExecutor.close_pool()

    asyncio.run(run())


if __name__ == "__main__":
    main()
