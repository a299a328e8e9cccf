#!/bin/bash
set -x
mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0
timeout 1020 python tools/soak.py --electrons 1500000 --report-every 250000 --concurrency 8 --mix --tensor-bytes 268435456 > gpurun_out/soak_15min.log 2>&1
timeout 120 python -c "import sys; sys.path.insert(0,'/root/repo'); import torch; torch.cuda.init(); from covalent_ssh_plugin_amd.gpu import probe; info=probe.probe(0); print('post-soak probe:', info['gcn_arch'], f\"{info['hbm_bw_gbps']:.0f} GB/s\", f\"{info['mfma_bf16_tflops']:.0f} TF/s\")" >> gpurun_out/soak_15min.log 2>&1
tail -10 gpurun_out/soak_15min.log
echo DONE
