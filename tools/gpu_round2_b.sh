#!/bin/bash
# Round-2 second GPU pass: regression gate for the streaming/ack/cancel
# rewrite + statistical weight + fresh PMC evidence + soak
set -x
mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0
python -m pytest tests -m gpu -q -rs 2>&1 | tail -20 > gpurun_out/pytest_gpu_b.log
timeout 500 python bench.py --steps 8192 --warmup 256 --dump-latencies gpurun_out/lat_shim.txt > gpurun_out/bench_shim_8k.json 2> gpurun_out/bench_shim_8k.err
timeout 300 python bench.py --config staging --steps 16 --warmup 2 --no-secondary > gpurun_out/bench_staging16.json 2> gpurun_out/bench_staging16.err
timeout 360 python tools/soak.py --electrons 60000 --report-every 20000 --concurrency 8 --mix --tensor-bytes 268435456 > gpurun_out/soak_b.log 2>&1
cd /tmp && export TMPDIR=/tmp
timeout 240 rocprofv3 --pmc SQ_VALU_MFMA_BUSY_CYCLES,SQ_WAVE_CYCLES,GRBM_GUI_ACTIVE -d /tmp/pmc -o r2pmc -- python -c "import sys; sys.path.insert(0,'/root/repo'); import torch; torch.cuda.init(); from covalent_ssh_plugin_amd.gpu import probe; print(probe.probe(0))" > /root/repo/gpurun_out/rocprof_pmc.log 2>&1
cp -r /tmp/pmc /root/repo/gpurun_out/rocprof_pmc 2>/dev/null || true
tail -4 /root/repo/gpurun_out/pytest_gpu_b.log
echo "=== bench 8k ==="; cat /root/repo/gpurun_out/bench_shim_8k.json
echo "=== soak tail ==="; tail -6 /root/repo/gpurun_out/soak_b.log
echo DONE
