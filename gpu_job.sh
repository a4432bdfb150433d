#!/bin/bash
# GPU run 2: SDMA fix validation + ceilings + rocprof
set -x
cd /root/repo
mkdir -p gpurun_out
timeout 600 python -m pytest tests -m gpu -q > gpurun_out/pytest_gpu.log 2>&1
echo "pytest: $?" >> gpurun_out/pytest_gpu.log
timeout 300 python __graft_entry__.py --smoke > gpurun_out/smoke.log 2>&1
echo "smoke: $?" >> gpurun_out/smoke.log
timeout 300 python tools/pcie_probe.py > gpurun_out/pcie_probe.log 2>&1
timeout 300 python bench.py --steps 10 --warmup 2 > gpurun_out/bench_staged.json 2> gpurun_out/bench_staged.log
timeout 300 python bench.py --steps 10 --warmup 2 --copy-path zero_copy > gpurun_out/bench_zc.json 2>/dev/null
timeout 300 python bench.py --steps 10 --warmup 2 --io-threads 32 > gpurun_out/bench_t32.json 2>/dev/null
timeout 300 python bench.py --steps 10 --warmup 2 --io-threads 8 > gpurun_out/bench_t8.json 2>/dev/null
cd /tmp && export TMPDIR=/tmp
timeout 300 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/prof2 -- python /root/repo/bench.py --steps 3 --warmup 1 > /root/repo/gpurun_out/rocprof.log 2>&1
echo "rocprof: $?" >> /root/repo/gpurun_out/rocprof.log
