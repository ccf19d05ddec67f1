#!/bin/bash
# Fallback-path full suite (persistent + fused-bwd OFF -> hipGraph
# per-step trains) and config-breadth benches (Medium / Non-reg).
set -x
mkdir -p gpurun_out
cd /root/repo
ZAREMBA_AMD_PERSISTENT=0 ZAREMBA_AMD_FUSED_BWD=0 timeout 600 \
  python -m pytest tests -q -m gpu > gpurun_out/s26_pytest_fallback.log 2>&1
echo rc=$? >> gpurun_out/s26_pytest_fallback.log
timeout 300 python bench.py --gpus 1 --steps 120 --warmup 10 --hidden_size 650 --dropout 0.5 \
  2>/dev/null | tail -1 > gpurun_out/s26_bench_medium.json
timeout 300 python bench.py --gpus 1 --steps 200 --warmup 10 --hidden_size 200 --dropout 0.0 --seq_length 20 \
  2>/dev/null | tail -1 > gpurun_out/s26_bench_nonreg.json
ZAREMBA_AMD_PERSISTENT=0 ZAREMBA_AMD_FUSED_BWD=0 timeout 300 \
  python bench.py --gpus 1 --steps 60 --warmup 10 2>/dev/null | tail -1 > gpurun_out/s26_bench_fallback.json
tail -3 gpurun_out/s26_pytest_fallback.log
echo MEDIUM:; cat gpurun_out/s26_bench_medium.json
echo NONREG:; cat gpurun_out/s26_bench_nonreg.json
echo FALLBACK-LARGE:; cat gpurun_out/s26_bench_fallback.json
