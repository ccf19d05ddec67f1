#!/bin/bash
set -x
mkdir -p gpurun_out
cd /root/repo
timeout 300 python -m pytest tests/test_gpu_kernels.py -q -m gpu > gpurun_out/s7_pytest.log 2>&1
echo "pytest rc=$?" >> gpurun_out/s7_pytest.log
timeout 300 python tools/gemm_probe.py > gpurun_out/s7_gemm_probe.txt 2>&1
timeout 300 python bench.py --gpus 1 --steps 60 --warmup 10 \
  > gpurun_out/s7_bench.json 2>/dev/null
timeout 300 python bench.py --gpus 1 --steps 60 --warmup 10 \
  > gpurun_out/s7_bench_b.json 2>/dev/null
cat gpurun_out/s7_gemm_probe.txt
cat gpurun_out/s7_bench.json gpurun_out/s7_bench_b.json
tail -n 4 gpurun_out/s7_pytest.log
