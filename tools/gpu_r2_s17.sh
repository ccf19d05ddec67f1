#!/bin/bash
# Same-box A/B matrix: splitk 2 vs 4, norm chunk 64K vs 32K. 3 bench
# reps per config to average out step noise.
set -x
mkdir -p gpurun_out
cd /root/repo
for cfg in "2 65536" "4 65536" "2 32768" "4 32768"; do
  set -- $cfg
  echo "=== splitk_nz=$1 norm_chunk=$2 ===" >> gpurun_out/s17_ab.txt
  for r in 1 2 3; do
    ZAMD_SPLITK_NZ=$1 ZAREMBA_AMD_NORM_CHUNK=$2 timeout 300 \
      python bench.py --gpus 1 --steps 80 --warmup 10 2>/dev/null \
      | python -c "import json,sys; d=json.load(sys.stdin); print(f\"{d['value']:.0f} tok/s  {d['ms_per_step']:.4f} ms\")" \
      >> gpurun_out/s17_ab.txt
  done
done
cat gpurun_out/s17_ab.txt
