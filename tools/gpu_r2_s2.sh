#!/bin/bash
# Round-2 GPU session 2: full gpu pytest (fixed RCCL skip + ksplit
# parity + softmax_acc), bwd census 2-way vs 4-way K-split, bench A/B,
# then the CPX partition probe for real-RCCL DP evidence.
set -x
mkdir -p gpurun_out
cd /root/repo

timeout 900 python -m pytest tests -m gpu -q > gpurun_out/s2_pytest.log 2>&1
echo "pytest rc=$?" >> gpurun_out/s2_pytest.log

timeout 240 ./tools/bwd_census 1500 35 20 30 2 > gpurun_out/s2_census_ks2.txt 2>&1
timeout 240 ./tools/bwd_census 1500 35 20 30 4 > gpurun_out/s2_census_ks4.txt 2>&1

timeout 300 python bench.py --gpus 1 --steps 60 --warmup 10 \
  > gpurun_out/s2_bench_ks2.json 2>/dev/null
ZAREMBA_AMD_BWD_KSPLIT=4 timeout 300 python bench.py --gpus 1 --steps 60 --warmup 10 \
  > gpurun_out/s2_bench_ks4.json 2>gpurun_out/s2_bench_ks4.err
echo "ks4 bench rc=$?" >> gpurun_out/s2_bench_ks4.err

bash tools/gpu_cpx_dp.sh > gpurun_out/s2_cpx.out 2>&1

tail -n 16 gpurun_out/s2_census_ks2.txt gpurun_out/s2_census_ks4.txt
cat gpurun_out/s2_bench_ks2.json gpurun_out/s2_bench_ks4.json
tail -n 6 gpurun_out/s2_pytest.log
tail -n 25 gpurun_out/cpx_dp.log
