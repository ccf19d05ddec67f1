#!/bin/bash
# 12-wave fused-bwd A/B + convergence re-certification on the final
# kernel stack (bf16 HIP vs fp32 eager trajectory).
set -x
mkdir -p gpurun_out
cd /root/repo
timeout 240 ./tools/bwd_census 1500 35 20 30 2 8  > gpurun_out/s10_census_w8.txt 2>&1
timeout 240 ./tools/bwd_census 1500 35 20 30 2 12 > gpurun_out/s10_census_w12.txt 2>&1
ZAREMBA_AMD_BWD_WAVES=12 timeout 300 python bench.py --gpus 1 --steps 60 --warmup 10 \
  > gpurun_out/s10_bench_w12.json 2>gpurun_out/s10_bench_w12.err
ZAREMBA_AMD_BWD_WAVES=12 timeout 300 python -m pytest tests/test_gpu_train.py -q -m gpu -k parity \
  > gpurun_out/s10_parity_w12.log 2>&1
echo "parity rc=$?" >> gpurun_out/s10_parity_w12.log
timeout 1200 python tools/convergence_check.py 4 hip > gpurun_out/s10_conv_hip.log 2>&1
echo "conv hip rc=$?" >> gpurun_out/s10_conv_hip.log
ZAREMBA_AMD_FORCE_EAGER=1 timeout 1200 python tools/convergence_check.py 4 eager > gpurun_out/s10_conv_eager.log 2>&1
echo "conv eager rc=$?" >> gpurun_out/s10_conv_eager.log
tail -n 12 gpurun_out/s10_census_w8.txt gpurun_out/s10_census_w12.txt
cat gpurun_out/s10_bench_w12.json
tail -2 gpurun_out/s10_parity_w12.log
grep -iE "epoch|ppl|rc=" gpurun_out/s10_conv_hip.log | tail -8
grep -iE "epoch|ppl|rc=" gpurun_out/s10_conv_eager.log | tail -8
