#!/bin/bash
set -x
mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0
python -m pytest tests -m gpu -q 2>&1 | tail -3 > gpurun_out/pytest_gpu_f.log
timeout 600 python bench.py > gpurun_out/bench_default_rehearsal.json 2> gpurun_out/bench_default_rehearsal.err
timeout 400 python tools/oversub_sweep.py --fan 64 --n 6144 > gpurun_out/oversub_sweep.log 2>&1
tail -1 gpurun_out/pytest_gpu_f.log
echo "=== default bench (driver rehearsal) ==="; cat gpurun_out/bench_default_rehearsal.json
echo "=== oversub sweep ==="; grep slots_per_gpu gpurun_out/oversub_sweep.log
echo DONE
