#!/bin/bash
# Round-2 GPU session 4: forward-pipeline A/B (fwd census + bench) +
# full gpu test suite on the new kernels.
set -x
mkdir -p gpurun_out
cd /root/repo
timeout 900 python -m pytest tests -m gpu -q > gpurun_out/s4_pytest.log 2>&1
echo "pytest rc=$?" >> gpurun_out/s4_pytest.log
timeout 240 ./tools/fwd_census 1500 35 20 50 > gpurun_out/s4_fwd_census.txt 2>&1
timeout 300 python bench.py --gpus 1 --steps 60 --warmup 10 \
  > gpurun_out/s4_bench.json 2>gpurun_out/s4_bench.err
echo "bench rc=$?" >> gpurun_out/s4_bench.err
timeout 300 python bench.py --gpus 1 --steps 60 --warmup 10 \
  > gpurun_out/s4_bench_b.json 2>/dev/null
tail -n 14 gpurun_out/s4_fwd_census.txt
cat gpurun_out/s4_bench.json gpurun_out/s4_bench_b.json
tail -n 4 gpurun_out/s4_pytest.log
