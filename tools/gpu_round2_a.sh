#!/bin/bash
# Round-2 first GPU pass: tests + benches + rocprof trace
set -x
mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0
python -m pytest tests -m gpu -q -rs 2>&1 | tail -25 > gpurun_out/pytest_gpu.log
timeout 400 python bench.py --steps 2048 --warmup 128 > gpurun_out/bench_shim.json 2> gpurun_out/bench_shim.err
timeout 300 python bench.py --steps 2048 --warmup 128 --transport local --no-secondary > gpurun_out/bench_local.json 2> gpurun_out/bench_local.err
timeout 300 python bench.py --config noop-stub --steps 256 --warmup 16 --no-secondary > gpurun_out/bench_stub.json 2> gpurun_out/bench_stub.err
timeout 300 python bench.py --config fan --steps 48 --warmup 4 --no-secondary > gpurun_out/bench_fan.json 2> gpurun_out/bench_fan.err
cd /tmp && export TMPDIR=/tmp
timeout 240 rocprofv3 --kernel-trace --stats -d /tmp/prof -o r2probe -- python -c "import sys; sys.path.insert(0,'/root/repo'); import torch; torch.cuda.init(); from covalent_ssh_plugin_amd.gpu import probe; probe.warmup(0,50); print(probe.probe(0))" > /root/repo/gpurun_out/rocprof_probe.log 2>&1
cp -r /tmp/prof /root/repo/gpurun_out/rocprof_prof 2>/dev/null || true
tail -5 /root/repo/gpurun_out/pytest_gpu.log
echo "=== bench_shim ==="; cat /root/repo/gpurun_out/bench_shim.json
echo DONE
