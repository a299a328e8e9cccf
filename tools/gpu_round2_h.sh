#!/bin/bash
set -x
mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0
python -m pytest tests -m gpu -q 2>&1 | tail -3 > gpurun_out/pytest_gpu_h.log
timeout 300 python tools/pipeline_sweep.py --total 8192 > gpurun_out/pipeline_sweep.log 2>&1
tail -1 gpurun_out/pytest_gpu_h.log
echo "=== pipeline sweep ==="; grep concurrency gpurun_out/pipeline_sweep.log
echo DONE
