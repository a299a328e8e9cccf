"""RCCL-over-xGMI all-reduce smoke-test electron.

The only collective in this system: the plugin itself
has no inherent cross-GPU communication, so RCCL/xGMI is exercised by a
purpose-built electron that fans an N-rank ``torch.distributed``
all-reduce across the remote node's MI355X GPUs (one process per GPU,
backend "nccl" — which IS RCCL on ROCm) and reports measured bus
bandwidth.  xGMI is point-to-point (7 links x ~153 GB/s per GPU), so the
ring all-reduce bus bandwidth ceiling is per-link bound; the smoke test
uses a large bucket (256 MiB default) to sit in the bandwidth-bound
regime.

``make_rccl_smoke_electron`` returns a fully self-contained closure:
cloudpickle serializes it by value, so the remote host needs torch but
NOT this package.  Rank processes are launched as plain subprocesses of
the electron with a generated worker script (no dependence on
``torch.multiprocessing`` picklability).
"""

from __future__ import annotations


def make_rccl_smoke_electron():
    """Build the smoke-test electron.

    Dispatch with ``hip_visible_devices_policy="none"`` (or let the
    electron clear the slot pinning itself, which it does): it needs the
    whole node's GPUs, not one slot.
    """

    def rccl_allreduce_smoke(
        world_size: int = 8,
        nbytes: int = 256 * 1024 * 1024,
        iters: int = 20,
        warmup: int = 5,
        backend: str = "nccl",
        port: int = 29513,
    ):
        import json
        import os
        import subprocess
        import sys
        import tempfile

        worker_src = r'''
import json, os, sys, time

rank = int(sys.argv[1]); world = int(sys.argv[2]); port = sys.argv[3]
backend = sys.argv[4]; nbytes = int(sys.argv[5])
iters = int(sys.argv[6]); warmup = int(sys.argv[7])

import torch
import torch.distributed as dist

os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
os.environ["MASTER_PORT"] = port
dist.init_process_group(backend=backend, rank=rank, world_size=world)

if backend == "nccl":
    torch.cuda.set_device(rank)
    device = torch.device("cuda", rank)
else:
    device = torch.device("cpu")

n = nbytes // 4
x = torch.ones(n, dtype=torch.float32, device=device)

def sync():
    if backend == "nccl":
        torch.cuda.synchronize()

for _ in range(warmup):
    dist.all_reduce(x)
sync()
dist.barrier()
t0 = time.perf_counter()
for _ in range(iters):
    dist.all_reduce(x)
sync()
dist.barrier()
dt = time.perf_counter() - t0

# ring all-reduce moves 2*(W-1)/W * nbytes per rank per iteration
busbw = (2.0 * (world - 1) / world) * nbytes * iters / dt / 1e9
algbw = nbytes * iters / dt / 1e9
ok = bool(torch.allclose(x[:4].float().cpu(),
                          torch.full((4,), float(world) ** (warmup + iters))))
if rank == 0:
    print("CSP_RCCL_JSON " + json.dumps({
        "world_size": world, "nbytes": nbytes, "iters": iters,
        "seconds": dt, "busbw_GBps": busbw, "algbw_GBps": algbw,
        "correct": ok, "backend": backend,
    }), flush=True)
dist.destroy_process_group()
'''
        with tempfile.NamedTemporaryFile(
            "w", suffix="_rccl_worker.py", delete=False
        ) as f:
            f.write(worker_src)
            worker_path = f.name

        env = dict(os.environ)
        # The electron may have been pinned to one GPU slot; the
        # collective needs every GPU on the node.
        env.pop("HIP_VISIBLE_DEVICES", None)
        env.pop("ROCR_VISIBLE_DEVICES", None)
        env.pop("CUDA_VISIBLE_DEVICES", None)
        env.setdefault("MASTER_ADDR", "127.0.0.1")
        env.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")

        procs = [
            subprocess.Popen(
                [
                    sys.executable,
                    worker_path,
                    str(rank),
                    str(world_size),
                    str(port),
                    backend,
                    str(nbytes),
                    str(iters),
                    str(warmup),
                ],
                env=env,
                stdout=subprocess.PIPE,
                stderr=subprocess.PIPE,
            )
            for rank in range(world_size)
        ]
        outs = []
        try:
            for p in procs:
                out, err = p.communicate(timeout=600)
                outs.append((p.returncode, out.decode(), err.decode()))
        finally:
            for p in procs:
                if p.poll() is None:
                    p.kill()
            os.unlink(worker_path)

        for rc, out, err in outs:
            if rc != 0:
                raise RuntimeError(f"rccl smoke rank failed (rc={rc}): {err[-2000:]}")
        for rc, out, err in outs:
            for line in out.splitlines():
                if line.startswith("CSP_RCCL_JSON "):
                    return json.loads(line[len("CSP_RCCL_JSON "):])
        raise RuntimeError("rccl smoke produced no result line")

    return rccl_allreduce_smoke
