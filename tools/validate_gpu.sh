#!/bin/bash
# On-box validation bundle (run via gpurun): mirrors the driver's
# round-end checks plus the benches.  Writes logs under gpurun_out/.
set -x
mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0
timeout 900 python -m pytest tests -x -q -m gpu > gpurun_out/pytest_gpu.log 2>&1; echo "pytest rc=$?"
timeout 300 python __graft_entry__.py smoke > gpurun_out/smoke.log 2>&1; echo "smoke rc=$?"
timeout 300 python bench.py > gpurun_out/bench_default.log 2>&1; echo "default rc=$?"
tail -n 2 gpurun_out/pytest_gpu.log
grep -h metric gpurun_out/bench_default.log
