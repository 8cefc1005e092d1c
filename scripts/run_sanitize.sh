#!/bin/bash
# Sanitizer-style debug pass (the reference has none; this
# framework's compute-sanitizer analog on ROCm):
#  * AMD_SERIALIZE_KERNEL=3  — synchronize before+after every kernel so a
#    fault is attributed to the kernel that raised it
#  * AMD_LOG_LEVEL on failures, HSA queue error reporting
#  * autograd anomaly mode via DEEPDFA_DETECT_ANOMALY for the numerics suite
# Run on a GPU box: bash scripts/run_sanitize.sh
set -e
export AMD_SERIALIZE_KERNEL=3
export AMD_SERIALIZE_COPY=3
export HSA_ENABLE_DEBUG=1
python -m pytest tests -m gpu -x -q "$@"
