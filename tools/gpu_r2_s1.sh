#!/bin/bash
# Round-2 GPU session 1: full gpu test suite (incl. the new RCCL
# 2-rank-1-GPU test), comparison baselines (eager fp32, nn.LSTM/MIOpen),
# dp2-on-one-GPU bench, hip headline sanity, fused-bwd entry census.
# Everything lands in gpurun_out/.
set -x
mkdir -p gpurun_out
cd /root/repo

echo "=== pytest -m gpu ===" > gpurun_out/s1_pytest.log
timeout 900 python -m pytest tests -m gpu -x -q >> gpurun_out/s1_pytest.log 2>&1
echo "pytest rc=$?" >> gpurun_out/s1_pytest.log

# hip headline sanity (new .so)
timeout 300 python bench.py --gpus 1 --steps 60 --warmup 10 \
  > gpurun_out/s1_bench_hip.json 2> gpurun_out/s1_bench_hip.err
echo "hip bench rc=$?" >> gpurun_out/s1_bench_hip.err

# comparison baselines on the same box
timeout 600 python bench.py --gpus 1 --steps 30 --warmup 5 --engine eager \
  > gpurun_out/s1_bench_eager.json 2> gpurun_out/s1_bench_eager.err
echo "eager rc=$?" >> gpurun_out/s1_bench_eager.err
timeout 600 python tools/bench_nnlstm.py --steps 40 --warmup 8 --dtype bf16 \
  > gpurun_out/s1_bench_nnlstm_bf16.json 2> gpurun_out/s1_bench_nnlstm.err
echo "nnlstm bf16 rc=$?" >> gpurun_out/s1_bench_nnlstm.err
timeout 600 python tools/bench_nnlstm.py --steps 40 --warmup 8 --dtype fp32 \
  > gpurun_out/s1_bench_nnlstm_fp32.json 2>> gpurun_out/s1_bench_nnlstm.err
echo "nnlstm fp32 rc=$?" >> gpurun_out/s1_bench_nnlstm.err

# dp2 on one GPU over real RCCL (ZAREMBA_AMD_ONE_GPU pins both ranks to 0)
ZAREMBA_AMD_ONE_GPU=1 timeout 420 python -m torch.distributed.run \
  --nnodes=1 --nproc-per-node 2 --master-addr 127.0.0.1 --master-port 29571 \
  bench.py --gpus 2 --steps 30 --warmup 5 \
  > gpurun_out/s1_bench_dp2.json 2> gpurun_out/s1_bench_dp2.err
echo "dp2 rc=$?" >> gpurun_out/s1_bench_dp2.err

# fused-backward phase census with the new entry/launch-gap probes
timeout 240 ./tools/bwd_census 1500 35 20 30 \
  > gpurun_out/s1_bwd_census.txt 2>&1
echo "census rc=$?" >> gpurun_out/s1_bwd_census.txt

tail -n 3 gpurun_out/s1_bench_hip.json gpurun_out/s1_bench_eager.json \
  gpurun_out/s1_bench_nnlstm_bf16.json gpurun_out/s1_bench_nnlstm_fp32.json \
  gpurun_out/s1_bench_dp2.json gpurun_out/s1_bwd_census.txt \
  gpurun_out/s1_pytest.log
