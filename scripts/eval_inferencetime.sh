#!/bin/bash
cd "$(dirname "$0")/.."
# Inference-time evaluation (reference eval_inferencetime_{linevul,combined}[_cpu].sh):
#   bash scripts/eval_inferencetime.sh [combined|linevul] [gpu|cpu]
variant=${1:-combined}
dev=${2:-gpu}
flags=""
[ "$variant" = "linevul" ] && flags="--no_flowgnn"
if [ "$dev" = "cpu" ]; then export CUDA_VISIBLE_DEVICES=""; fi
python -m deepdfa_amd.train.linevul_main \
  --do_test --time $flags \
  --block_size 512 --eval_batch_size 16 "${@:3}" \
  2>&1 | tee "eval_time_${variant}_${dev}.log"
