#!/usr/bin/env python3
"""Soak test: sustained electron dispatch, leak detection.

Drives N electrons through the persistent-worker pipeline while
sampling dispatcher and worker RSS plus open-fd counts.  Flat curves
after warm-up = no leaks in the channel framing, staging freelist, task
records, or transport pool.

    python tools/soak.py --electrons 10000 [--tensor-bytes 0]
"""

from __future__ import annotations

import argparse
import asyncio
import os
import resource
import sys
import tempfile
import time
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(REPO))

from covalent_ssh_plugin_amd import SSHExecutor  # noqa: E402


def rss_mb(pid: int = 0) -> float:
    try:
        path = f"/proc/{pid or os.getpid()}/status"
        for line in open(path):
            if line.startswith("VmRSS"):
                return int(line.split()[1]) / 1024.0
    except OSError:
        pass
    return -1.0


def fd_count() -> int:
    try:
        return len(os.listdir("/proc/self/fd"))
    except OSError:
        return -1


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--electrons", type=int, default=10000)
    ap.add_argument("--report-every", type=int, default=1000)
    ap.add_argument("--tensor-bytes", type=int, default=0,
                    help="per-electron tensor result size (0 = no-op)")
    ap.add_argument("--concurrency", type=int, default=1)
    ap.add_argument("--mix", action="store_true",
                    help="mixed workload: no-ops, bf16 matmuls, tensor returns, "
                         "exceptions (GPU electrons when cuda is visible)")
    ap.add_argument("--isolate", action="store_true",
                    help="fork-isolated dispatch: fresh child per electron")
    ap.add_argument("--isolate-preload", default="none",
                    help="zygote preload for --isolate: torch|none")
    args = ap.parse_args()

    if args.mix:

        def electron(i):
            kind = i % 5
            if kind == 0:
                return i
            if kind == 1:
                raise ValueError(f"intentional {i}")
            import torch

            dev = "cuda" if torch.cuda.is_available() else "cpu"
            if kind == 2:  # matmul
                a = torch.randn(512, 512, device=dev, dtype=torch.bfloat16)
                c = a @ a
                if dev == "cuda":
                    torch.cuda.synchronize()
                return float(c.float().mean())
            if kind == 3:  # tensor return through staging
                return torch.ones(4 << 20, device=dev, dtype=torch.bfloat16)
            return {"nested": [torch.arange(1000), (i, "tag")]}

        fargs = None  # per-task arg
    elif args.tensor_bytes:

        def electron(nbytes):
            import torch

            if torch.cuda.is_available():
                return torch.ones(nbytes // 2, device="cuda", dtype=torch.bfloat16)
            return torch.ones(nbytes // 2, dtype=torch.bfloat16)

        fargs = [args.tensor_bytes]
    else:

        def electron():
            return 0

        fargs = []

    async def run():
        with tempfile.TemporaryDirectory() as home, tempfile.TemporaryDirectory() as cache:
            ex = SSHExecutor(
                transport="local",
                local_home=home,
                cache_dir=cache,
                python_path=sys.executable,
                persistent_workers=True,
                isolate_tasks=args.isolate,
                isolate_preload=args.isolate_preload,
                cpu_workers=args.concurrency,
                pinned_staging_threshold_bytes=1 << 20,
                warmup_gpu=False,
            )
            t0 = time.perf_counter()
            sem = asyncio.Semaphore(args.concurrency)

            errors = [0]

            async def one(i):
                async with sem:
                    try:
                        await ex.execute(
                            electron,
                            [i] if fargs is None else list(fargs),
                            {},
                            dispatch_id="soak",
                            node_id=i,
                        )
                    except ValueError:
                        errors[0] += 1  # the mix's intentional failures

            done = 0
            worker_pid = None
            for batch_start in range(0, args.electrons, args.report_every):
                batch = min(args.report_every, args.electrons - batch_start)
                await asyncio.gather(*[one(batch_start + i) for i in range(batch)])
                done += batch
                meta = ex.last_task_record.remote_meta or {}
                worker_pid = meta.get("pid", worker_pid)
                rate = done / (time.perf_counter() - t0)
                print(
                    f"[{done:>6}] {rate:8.1f} e/s | dispatcher rss {rss_mb():.1f} MB "
                    f"fds {fd_count()} | worker rss {rss_mb(worker_pid):.1f} MB "
                    f"| records {len(ex.task_records)}",
                    flush=True,
                )
            elapsed = time.perf_counter() - t0
            print(f"TOTAL {args.electrons} electrons in {elapsed:.1f}s = "
                  f"{args.electrons/elapsed:.1f} e/s; "
                  f"intentional-failure roundtrips: {errors[0]}")
            await SSHExecutor.close_pool()

    asyncio.run(run())


if __name__ == "__main__":
    main()
