#!/bin/bash
# LineVul baseline training (reference LineVul/linevul/scripts/msr_train_linevul.sh parity)
seed="${1:-1}"
python -m deepdfa_amd.train.linevul_main --do_train --do_test --no_flowgnn \
  --epochs 10 --block_size 512 --train_batch_size 16 --eval_batch_size 16 \
  --learning_rate 2e-5 --max_grad_norm 1.0 --seed "$seed" "${@:2}"
