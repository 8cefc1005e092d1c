#!/bin/bash
# Headline experiments (reference scripts/performance_evaluation.sh parity):
# DeepDFA, LineVul, DeepDFA+LineVul
set -e
bash scripts/train.sh
bash scripts/msr_train_linevul.sh 1
bash scripts/msr_train_combined.sh 1
