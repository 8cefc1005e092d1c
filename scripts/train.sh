#!/bin/bash
# DeepDFA training (reference DDFA/scripts/train.sh parity)
python -m deepdfa_amd.train.main_cli fit --config configs/config_bigvul.yaml --config configs/config_ggnn.yaml "$@"
