#!/bin/bash
set -x
mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0
timeout 300 python __graft_entry__.py smoke > gpurun_out/smoke_shim.log 2>&1; echo "smoke rc=$?" >> gpurun_out/smoke_shim.log
timeout 300 python tools/fused_stream_endurance.py > gpurun_out/fused_endurance.log 2>&1
timeout 540 python tools/soak.py --electrons 600000 --report-every 100000 --concurrency 8 --mix --tensor-bytes 268435456 > gpurun_out/soak_long.log 2>&1
tail -2 gpurun_out/smoke_shim.log
echo "=== fused endurance ==="; tail -6 gpurun_out/fused_endurance.log
echo "=== long soak ==="; tail -4 gpurun_out/soak_long.log
echo DONE
