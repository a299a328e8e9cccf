#!/bin/bash
# Grand final: full tier + high-resolution bench + sustained soak
set -x
mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0
python -m pytest tests -m gpu -q 2>&1 | tail -2 > gpurun_out/grand_pytest.log
timeout 420 python bench.py --steps 8192 --warmup 256 > gpurun_out/grand_bench.json 2> gpurun_out/grand_bench.err
timeout 780 python tools/soak.py --electrons 1200000 --report-every 300000 --concurrency 8 --mix --tensor-bytes 268435456 > gpurun_out/grand_soak.log 2>&1
timeout 120 python __graft_entry__.py smoke > gpurun_out/grand_smoke.log 2>&1; echo "smoke rc=$?" >> gpurun_out/grand_smoke.log
tail -1 gpurun_out/grand_pytest.log
python -c "import json; d=json.load(open('gpurun_out/grand_bench.json')); print('bench:', d['value'], 'e/s p50', d['p50_ms'], 'p99', d['p99_ms'], 'p999', d['p999_ms'])"
tail -3 gpurun_out/grand_soak.log
tail -1 gpurun_out/grand_smoke.log
echo DONE
