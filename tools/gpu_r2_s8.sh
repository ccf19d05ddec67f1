#!/bin/bash
# GEMM tile sweep: force each tile on the probe shapes, then bench the
# best-looking default. Correctness guarded by the gemm pytest subset.
set -x
mkdir -p gpurun_out
cd /root/repo
timeout 240 python -m pytest tests/test_gpu_kernels.py -q -m gpu -k gemm > gpurun_out/s8_pytest.log 2>&1
echo "pytest rc=$?" >> gpurun_out/s8_pytest.log
for t in 0 64 12864 128 256; do
  echo "=== ZAMD_GEMM_TILE=$t ===" >> gpurun_out/s8_sweep.txt
  ZAMD_GEMM_TILE=$t timeout 300 python tools/gemm_probe.py >> gpurun_out/s8_sweep.txt 2>&1
done
timeout 300 python bench.py --gpus 1 --steps 60 --warmup 10 > gpurun_out/s8_bench.json 2>/dev/null
cat gpurun_out/s8_sweep.txt
cat gpurun_out/s8_bench.json
tail -n 3 gpurun_out/s8_pytest.log
