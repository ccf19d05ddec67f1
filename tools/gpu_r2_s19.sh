#!/bin/bash
set -x
mkdir -p gpurun_out
cd /root/repo
timeout 300 python -m pytest tests/test_gpu_kernels.py -q -m gpu > gpurun_out/s19_pytest.log 2>&1
echo rc=$? >> gpurun_out/s19_pytest.log
for r in 1 2 3; do
  timeout 300 python bench.py --gpus 1 --steps 80 --warmup 10 2>/dev/null | tail -1 >> gpurun_out/s19_bench.txt
done
cd /tmp && export TMPDIR=/tmp
timeout 420 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/s19_prof -o s19 -- \
  python /root/repo/bench.py --gpus 1 --steps 30 --warmup 8 >/dev/null 2>&1
cd /root/repo
DB=$(ls gpurun_out/s19_prof/*results.db | head -1)
python tools/prof_summary.py "$DB" > gpurun_out/s19_kernel_stats.txt 2>&1
cat gpurun_out/s19_bench.txt
grep -E "colsum|lsm" gpurun_out/s19_kernel_stats.txt
tail -3 gpurun_out/s19_pytest.log
