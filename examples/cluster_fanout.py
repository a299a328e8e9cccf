#!/usr/bin/env python3
"""Fan electrons across SEVERAL 8xMI355X nodes with least-loaded
placement (SSHClusterExecutor).

    python examples/cluster_fanout.py alice@node0 alice@node1 ...
"""

import asyncio
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

from covalent_ssh_plugin_amd import SSHClusterExecutor


def main():
    hosts = sys.argv[1:] or ["user@node0", "user@node1"]
    cluster = SSHClusterExecutor(
        hosts, gpu_slots=8, persistent_workers=True, warmup_gpu=True
    )

    def electron(n):
        import torch

        a = torch.randn(n, n, device="cuda", dtype=torch.bfloat16)
        return float((a @ a).float().mean())

    async def run():
        await cluster.prewarm()  # all nodes x all GPU workers warm
        outs = await asyncio.gather(
            *[
                cluster.execute(electron, [4096], {}, dispatch_id="fan", node_id=i)
                for i in range(len(hosts) * 8 * 2)
            ]
        )
        print(f"{len(outs)} electrons done across {len(hosts)} nodes "
              f"({cluster.capacity} GPU slots)")
        for host, stats in cluster.stats().items():
            print(host, stats["counters"])

    asyncio.run(run())


if __name__ == "__main__":
    main()
