#!/bin/bash
cd "$(dirname "$0")/.."
# Line-level localization metrics sweep (reference rq ... localization runs):
# Effort@TopK, Recall@TopK%LOC, Top-k accuracy, IFA per reasoning method.
for method in attention saliency; do
  python -m deepdfa_amd.train.unixcoder_main \
    --do_local_explanation --reasoning_method "$method" \
    --output_dir saved_models/unixcoder_loc "$@" \
    2>&1 | tee "uxc_localization_${method}.log"
done
