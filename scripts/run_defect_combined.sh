#!/bin/bash
# DeepDFA+CodeT5 defect training (CodeT5/sh/run_exp.py --task defect
# --flowgnn_model parity: bs 8, accum 4, lr 2e-5, epochs 10, patience 2)
python -m deepdfa_amd.train.run_defect --do_train --do_eval --do_test \
  --flowgnn_data --flowgnn_model --num_train_epochs 10 --max_source_length 512 \
  --train_batch_size 8 --gradient_accumulation_steps 4 --learning_rate 2e-5 "$@"
