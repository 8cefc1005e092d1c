#!/bin/bash
cd "$(dirname "$0")/.."
# UniXcoder RQ1 training (reference LineVul/unixcoder/rq1_train_uxc.sh and
# the _noflowgnn / _size / _crossproject variants — variant selection via
# extra flags, e.g. --num_layers 6 for the size ablation).
seed=${1:-1}
python -m deepdfa_amd.train.unixcoder_main \
  --do_train --do_test \
  --output_dir saved_models/unixcoder \
  --epochs 10 --block_size 512 --train_batch_size 16 --eval_batch_size 16 \
  --learning_rate 2e-5 --max_grad_norm 1.0 --seed "$seed" "${@:2}" \
  2>&1 | tee "uxc_rq1_train_${seed}.log"
