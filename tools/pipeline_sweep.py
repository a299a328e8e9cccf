"""Pipelining sweep: aggregate no-op throughput through ONE persistent
worker channel at increasing dispatcher-side concurrency.

The worker protocol is strictly ordered but PIPELINED (the channel's
write lock is held only while sending; a FIFO reader pump resolves
replies), so concurrent electrons overlap their wire round trips on one
channel.  This quantifies that overlap.
"""

import argparse
import asyncio
import os
import sys
import tempfile
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from covalent_ssh_plugin_amd import SSHExecutor


def noop():
    return 0


async def run_level(ex, concurrency: int, total: int) -> float:
    sem = asyncio.Semaphore(concurrency)

    async def one(i):
        async with sem:
            await ex.execute(noop, [], {}, dispatch_id="pipe", node_id=i)

    t0 = time.perf_counter()
    await asyncio.gather(*[one(i) for i in range(total)])
    return total / (time.perf_counter() - t0)


async def main(total: int, levels):
    home = tempfile.mkdtemp()
    cache = tempfile.mkdtemp()
    ex = SSHExecutor(
        transport="local",
        local_home=home,
        cache_dir=cache,
        python_path=sys.executable,
        persistent_workers=True,
        cpu_workers=1,  # ONE worker channel: overlap is pure pipelining
        hip_visible_devices_policy="none",
    )
    await run_level(ex, 4, 256)  # warm
    for c in levels:
        eps = await run_level(ex, c, total)
        print(f"concurrency {c:4d}: {eps:9.1f} electrons/s through one channel",
              flush=True)
    await SSHExecutor.close_pool()


if __name__ == "__main__":
    ap = argparse.ArgumentParser()
    ap.add_argument("--total", type=int, default=4096)
    ap.add_argument("--levels", default="1,4,16,64,256")
    args = ap.parse_args()
    asyncio.run(main(args.total, [int(x) for x in args.levels.split(",")]))
