#!/bin/bash
# Next-round promotion gate for experimental kernels: run ON A GPU BOX.
#   1. numerics:   pytest -m gpu_experimental
#   2. perf:       scripts/gpu_attn_bench.py (EXPERIMENTAL lines)
# Promote by switching ops/dispatch.py (and csrc/forward.h attend) to the
# new kernel, then re-run pytest -m gpu and bench.py.
set -e
python -m pytest tests -m gpu_experimental -q
python scripts/gpu_attn_bench.py
