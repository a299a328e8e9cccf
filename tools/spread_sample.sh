#!/bin/bash
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out
timeout 200 python - <<'PYEOF'
import sys
sys.path.insert(0, "/root/repo")
import torch; torch.cuda.init()
from covalent_ssh_plugin_amd.gpu import probe
for i in range(3):
    info = probe.probe(0)
    print(f"probe: hbm {info['hbm_bw_gbps']:.0f} GB/s mfma {info['mfma_bf16_tflops']:.0f} TF/s")
PYEOF
timeout 240 python bench.py --steps 2048 --warmup 128 --no-secondary 2>/dev/null | tail -1 | python -c 'import json,sys; d=json.loads(sys.stdin.read()); print("bench:", d["value"], "e/s p50", d["p50_ms"], "ms")'
