#!/bin/bash
set -x
mkdir -p gpurun_out
cd /root/repo
timeout 300 python -m pytest tests/test_gpu_kernels.py -q -m gpu > gpurun_out/s6_pytest.log 2>&1
echo "pytest rc=$?" >> gpurun_out/s6_pytest.log
timeout 240 ./tools/fwd_census 1500 35 20 50 > gpurun_out/s6_fwd_census.txt 2>&1
timeout 300 python bench.py --gpus 1 --steps 60 --warmup 10 \
  > gpurun_out/s6_bench.json 2>gpurun_out/s6_bench.err
echo "bench rc=$?" >> gpurun_out/s6_bench.err
cd /tmp && export TMPDIR=/tmp
timeout 420 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/s6_prof -o s6 -- \
  python /root/repo/bench.py --gpus 1 --steps 30 --warmup 8 > /root/repo/gpurun_out/s6_prof_bench.json 2>/root/repo/gpurun_out/s6_prof.err
echo "prof rc=$?" >> /root/repo/gpurun_out/s6_prof.err
cd /root/repo
DB=$(ls gpurun_out/s6_prof/*results.db gpurun_out/s6_prof/**/*results.db 2>/dev/null | head -1)
python tools/prof_summary.py "$DB" > gpurun_out/s6_kernel_stats.txt 2>&1 || true
tail -n 12 gpurun_out/s6_fwd_census.txt
cat gpurun_out/s6_bench.json
head -n 24 gpurun_out/s6_kernel_stats.txt
tail -n 4 gpurun_out/s6_pytest.log
