#!/bin/bash
# Re-profile the current state (per-kernel stats) + endurance (20K steps
# through the spin-kernel protocols, abort-checked) on one box.
set -x
mkdir -p gpurun_out
cd /root/repo
cd /tmp && export TMPDIR=/tmp
timeout 420 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/s13_prof -o s13 -- \
  python /root/repo/bench.py --gpus 1 --steps 30 --warmup 8 > /root/repo/gpurun_out/s13_prof_bench.json 2>/root/repo/gpurun_out/s13_prof.err
cd /root/repo
DB=$(ls gpurun_out/s13_prof/*results.db 2>/dev/null | head -1)
python tools/prof_summary.py "$DB" > gpurun_out/s13_kernel_stats.txt 2>&1 || true
timeout 300 python bench.py --gpus 1 --steps 20000 --warmup 10 > gpurun_out/s13_endurance.json 2>gpurun_out/s13_endurance.err
echo "endurance rc=$?" >> gpurun_out/s13_endurance.err
head -22 gpurun_out/s13_kernel_stats.txt
cat gpurun_out/s13_endurance.json
