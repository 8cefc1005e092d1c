#!/bin/bash
cd "$(dirname "$0")/.."
# DbgBench prediction export (reference run_all_eval_export_dbgbench[_combined].sh):
# evaluates on the DbgBench-shaped held-out set and dumps per-example
# predictions + the CodeT5-format dataset.
python -m deepdfa_amd.train.unixcoder_main \
  --do_test --dbgbench --eval_export --export_codet5 \
  --output_dir saved_models/unixcoder_dbgbench "$@" \
  2>&1 | tee uxc_eval_export_dbgbench.log
