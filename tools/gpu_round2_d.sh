#!/bin/bash
set -x
mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0
python -m pytest tests -m gpu -q 2>&1 | tail -3 > gpurun_out/pytest_gpu_d.log
timeout 300 python bench.py --steps 4096 --warmup 256 --no-secondary > gpurun_out/bench_noop_d.json 2> gpurun_out/bench_noop_d.err
timeout 300 python bench.py --config noop-isolated --steps 256 --warmup 16 --no-secondary > gpurun_out/bench_isolated_d.json 2> gpurun_out/bench_isolated_d.err
timeout 200 python - > gpurun_out/fork_micro.log 2>&1 <<'PYEOF'
import os, time

def bench_fork(n=50):
    t0 = time.perf_counter()
    for _ in range(n):
        pid = os.fork()
        if pid == 0:
            os._exit(0)
        os.waitpid(pid, 0)
    return (time.perf_counter() - t0) / n * 1000

print(f"fork (plain python):      {bench_fork():8.2f} ms")
import cloudpickle
print(f"fork (+cloudpickle):      {bench_fork():8.2f} ms")
import torch
print(f"fork (+torch-rocm):       {bench_fork():8.2f} ms")
PYEOF
echo "=== noop 4096 ==="; cat gpurun_out/bench_noop_d.json
echo "=== isolated preload=none ==="; cat gpurun_out/bench_isolated_d.json
echo "=== fork micro ==="; cat gpurun_out/fork_micro.log
tail -1 gpurun_out/pytest_gpu_d.log
echo DONE
