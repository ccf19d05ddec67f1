#!/bin/bash
# Round-2 GPU session 3: 8-wave fused-bwd A/B (census + bench), parity
# tests for the new variants.
set -x
mkdir -p gpurun_out
cd /root/repo
timeout 300 python -m pytest tests/test_gpu_kernels.py -q -m gpu -k fused_bwd \
  > gpurun_out/s3_pytest.log 2>&1
echo "pytest rc=$?" >> gpurun_out/s3_pytest.log
timeout 240 ./tools/bwd_census 1500 35 20 30 2 4 > gpurun_out/s3_census_w4.txt 2>&1
timeout 240 ./tools/bwd_census 1500 35 20 30 2 8 > gpurun_out/s3_census_w8.txt 2>&1
timeout 240 ./tools/bwd_census 1500 35 20 30 4 8 > gpurun_out/s3_census_ks4w8.txt 2>&1
timeout 300 python bench.py --gpus 1 --steps 60 --warmup 10 \
  > gpurun_out/s3_bench_w4.json 2>/dev/null
ZAREMBA_AMD_BWD_WAVES=8 timeout 300 python bench.py --gpus 1 --steps 60 --warmup 10 \
  > gpurun_out/s3_bench_w8.json 2>gpurun_out/s3_bench_w8.err
echo "w8 rc=$?" >> gpurun_out/s3_bench_w8.err
tail -n 12 gpurun_out/s3_census_w4.txt gpurun_out/s3_census_w8.txt gpurun_out/s3_census_ks4w8.txt
cat gpurun_out/s3_bench_w4.json gpurun_out/s3_bench_w8.json
tail -n 4 gpurun_out/s3_pytest.log
