#!/bin/bash
# DeepDFA+LineVul combined training (msr_train_combined.sh parity: identical
# to msr_train_linevul.sh minus --no_flowgnn)
seed="${1:-1}"
python -m deepdfa_amd.train.linevul_main --do_train --do_test \
  --epochs 10 --block_size 512 --train_batch_size 16 --eval_batch_size 16 \
  --learning_rate 2e-5 --max_grad_norm 1.0 --seed "$seed" "${@:2}"
