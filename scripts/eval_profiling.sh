#!/bin/bash
cd "$(dirname "$0")/.."
# FLOPs-profiling evaluation (reference eval_profiling_{linevul,combined}[_cpu].sh):
#   bash scripts/eval_profiling.sh [combined|linevul] [gpu|cpu]
variant=${1:-combined}
dev=${2:-gpu}
flags=""
[ "$variant" = "linevul" ] && flags="--no_flowgnn"
if [ "$dev" = "cpu" ]; then export CUDA_VISIBLE_DEVICES=""; fi
python -m deepdfa_amd.train.linevul_main \
  --do_test --profile $flags \
  --block_size 512 --eval_batch_size 16 "${@:3}" \
  2>&1 | tee "eval_profiling_${variant}_${dev}.log"
python scripts/report_profiling.py 2>/dev/null || true
