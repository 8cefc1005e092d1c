#!/bin/bash
# Profiling runs (reference DDFA/scripts/run_profiling.sh parity)
for metric in profile time; do
  python -m deepdfa_amd.train.linevul_main --do_test --seed 1 --"$metric" "$@"
done
python scripts/report_profiling.py saved_models
