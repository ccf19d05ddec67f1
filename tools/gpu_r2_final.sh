#!/bin/bash
# Final validation: fresh box, full gpu suite, smoke(), bench reps,
# final kernel stats snapshot.
set -x
mkdir -p gpurun_out
cd /root/repo
timeout 600 python -m pytest tests -q -m gpu > gpurun_out/final_pytest.log 2>&1
echo rc=$? >> gpurun_out/final_pytest.log
timeout 300 python -c "import __graft_entry__ as g; g.smoke()" > gpurun_out/final_smoke.log 2>&1
echo "smoke rc=$?" >> gpurun_out/final_smoke.log
for r in 1 2 3; do
  timeout 300 python bench.py --gpus 1 --steps 100 --warmup 10 2>/dev/null | tail -1 >> gpurun_out/final_bench.txt
done
cd /tmp && export TMPDIR=/tmp
timeout 300 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/final_prof -o fin -- \
  python /root/repo/bench.py --gpus 1 --steps 25 --warmup 8 >/dev/null 2>&1
cd /root/repo
DB=$(ls gpurun_out/final_prof/*results.db | head -1)
python tools/prof_summary.py "$DB" > gpurun_out/final_kernel_stats.txt 2>&1
tail -3 gpurun_out/final_pytest.log
tail -2 gpurun_out/final_smoke.log
cat gpurun_out/final_bench.txt
head -24 gpurun_out/final_kernel_stats.txt
