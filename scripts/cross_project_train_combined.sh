#!/bin/bash
cd "$(dirname "$0")/.."
# Cross-project DeepDFA+LineVul training (reference
# scripts/cross_project_train_combined.sh).
seed=${1:-1}
python -m deepdfa_amd.train.linevul_main \
  --do_train --do_test \
  --split cross_project \
  --output_dir saved_models/cross_project_combined \
  --epochs 10 --block_size 512 --train_batch_size 16 --eval_batch_size 16 \
  --learning_rate 2e-5 --max_grad_norm 1.0 --seed "$seed" "${@:2}" \
  2>&1 | tee "train_cross_project_combined_${seed}.log"
