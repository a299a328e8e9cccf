#!/bin/bash
# Round-2 third GPU pass: isolate-mode validation + bench + HBM spread
set -x
mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0
python -m pytest tests -m gpu -q -rs 2>&1 | tail -8 > gpurun_out/pytest_gpu_c.log
timeout 400 python bench.py --config noop-isolated --steps 512 --warmup 16 --no-secondary > gpurun_out/bench_isolated.json 2> gpurun_out/bench_isolated.err
timeout 200 python tools/soak.py --electrons 2000 --report-every 500 --concurrency 4 --isolate > gpurun_out/soak_isolated.log 2>&1
timeout 300 python - > gpurun_out/hbm_spread.log 2>&1 <<'PYEOF'
import sys, time
sys.path.insert(0, "/root/repo")
import torch; torch.cuda.init()
from covalent_ssh_plugin_amd.gpu import probe
for i in range(6):
    info = probe.probe(0)
    print(f"run {i}: hbm {info['hbm_bw_gbps']:.0f} GB/s, mfma {info['mfma_bf16_tflops']:.0f} TF/s")
    time.sleep(2)
PYEOF
tail -3 gpurun_out/pytest_gpu_c.log
echo "=== isolated bench ==="; cat gpurun_out/bench_isolated.json
echo "=== isolated soak ==="; tail -3 gpurun_out/soak_isolated.log
echo "=== hbm spread ==="; cat gpurun_out/hbm_spread.log
echo DONE
