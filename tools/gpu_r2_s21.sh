#!/bin/bash
# End-to-end CLI runs on the HIP engine: the reference's Large command
# (markov data stands in for the missing PTB train blob) and a small
# ensemble; full gpu test suite; bench reps.
set -x
mkdir -p gpurun_out
cd /root/repo
timeout 600 python -m pytest tests -q -m gpu > gpurun_out/s21_pytest.log 2>&1
echo rc=$? >> gpurun_out/s21_pytest.log
timeout 600 python main.py --layer_num 2 --hidden_size 1500 --dropout 0.65 \
  --winit 0.04 --batch_size 20 --seq_length 35 --learning_rate 1 \
  --total_epochs 6 --factor_epoch 3 --factor 1.15 --max_grad_norm 10 \
  --lstm_type custom --seed 3 --data synthetic_markov:vocab=2000,branch=20 \
  --save gpurun_out/s21_large.pt --jsonl gpurun_out/s21_large.jsonl \
  > gpurun_out/s21_large_cli.log 2>&1
echo "large cli rc=$?" >> gpurun_out/s21_large_cli.log
timeout 600 python ensemble.py --ensemble_num 2 --hidden_size 650 \
  --dropout 0.5 --total_epochs 1 --seed 5 \
  --data synthetic_markov:vocab=500,branch=20 --lstm_type custom \
  --save_dir gpurun_out/s21_ens > gpurun_out/s21_ens_cli.log 2>&1
echo "ensemble cli rc=$?" >> gpurun_out/s21_ens_cli.log
for r in 1 2; do
  timeout 300 python bench.py --gpus 1 --steps 80 --warmup 10 2>/dev/null | tail -1 >> gpurun_out/s21_bench.txt
done
tail -4 gpurun_out/s21_pytest.log
grep -E "Epoch|Test set|rc=" gpurun_out/s21_large_cli.log | tail -10
grep -E "averaged|rc=" gpurun_out/s21_ens_cli.log | tail -6
cat gpurun_out/s21_bench.txt
