#!/bin/bash
# Convergence re-certification on the FINAL kernel stack (several
# reduction associations changed since s10: colsum 4-chain, lsm 2-chain,
# fwd K-parity split) + a 5-minute sustained endurance + full suite.
set -x
mkdir -p gpurun_out
cd /root/repo
timeout 600 python -m pytest tests -q -m gpu > gpurun_out/s25_pytest.log 2>&1
echo rc=$? >> gpurun_out/s25_pytest.log
timeout 1200 python tools/convergence_check.py 4 hip > gpurun_out/s25_conv_hip.log 2>&1
echo "conv hip rc=$?" >> gpurun_out/s25_conv_hip.log
ZAREMBA_AMD_FORCE_EAGER=1 timeout 1200 python tools/convergence_check.py 4 eager > gpurun_out/s25_conv_eager.log 2>&1
echo "conv eager rc=$?" >> gpurun_out/s25_conv_eager.log
timeout 400 python bench.py --gpus 1 --steps 120000 --warmup 10 > gpurun_out/s25_endurance.json 2>gpurun_out/s25_endurance.err
echo "endurance rc=$?" >> gpurun_out/s25_endurance.err
tail -3 gpurun_out/s25_pytest.log
grep -E "epoch|rc=" gpurun_out/s25_conv_hip.log | tail -6
grep -E "epoch|rc=" gpurun_out/s25_conv_eager.log | tail -6
cat gpurun_out/s25_endurance.json
