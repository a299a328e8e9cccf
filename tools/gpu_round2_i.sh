#!/bin/bash
set -x
mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0
timeout 540 python tools/soak.py --electrons 200000 --report-every 50000 --concurrency 8 --isolate > gpurun_out/soak_isolated_long.log 2>&1
# zombie / leaked-process audit after 200k forks
ps -eo stat,comm | awk '$1 ~ /Z/' | wc -l >> gpurun_out/soak_isolated_long.log
tail -6 gpurun_out/soak_isolated_long.log
echo DONE
