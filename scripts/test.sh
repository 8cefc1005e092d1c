#!/bin/bash
# DeepDFA evaluation (reference DDFA/scripts/test.sh parity): test.sh <ckpt>
ckpt="$1"; shift
python -m deepdfa_amd.train.main_cli test --config configs/config_bigvul.yaml --config configs/config_ggnn.yaml --ckpt_path "$ckpt" "$@"
