#!/bin/bash
# Final-state PMC evidence (north-star deliverable: rocprof counters on
# the shipped kernels). Counters-only run per the pool rules.
set -x
mkdir -p gpurun_out
cd /tmp && export TMPDIR=/tmp
timeout 420 rocprofv3 --pmc SQ_WAVE_CYCLES,SQ_WAIT_ANY,SQ_ACTIVE_INST_ANY,TCC_HIT,TCC_MISS \
  --stats -d /root/repo/gpurun_out/s23_pmc -o s23 -- \
  python /root/repo/bench.py --gpus 1 --steps 12 --warmup 6 \
  > /root/repo/gpurun_out/s23_pmc_bench.json 2>/root/repo/gpurun_out/s23_pmc.err
echo "pmc rc=$?" >> /root/repo/gpurun_out/s23_pmc.err
cd /root/repo
DB=$(ls gpurun_out/s23_pmc/*results.db 2>/dev/null | head -1)
python tools/prof_summary.py "$DB" > gpurun_out/s23_pmc_summary.txt 2>&1
head -60 gpurun_out/s23_pmc_summary.txt
tail -3 gpurun_out/s23_pmc.err
