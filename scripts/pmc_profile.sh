#!/bin/bash
# PMC counter collection for the attention kernels (run on a GPU box).
#
# IMPORTANT (pool policy): a rocprofv3 invocation may combine --pmc with
# --kernel-trace/--stats ONLY — never with -s/--sys-trace, -r/--runtime-
# trace, or the hip/hsa/memory-copy/scratch-memory/marker trace domains.
# Collect counters in their own runs, write under gpurun_out/, then copy
# the summaries you keep into profiles/.
set -e
cd /tmp && export TMPDIR=/tmp
OUT=${1:-/root/repo/gpurun_out/pmc}
mkdir -p "$OUT"

run_pmc () {
  local name="$1"; shift
  rocprofv3 --pmc "$@" -d "$OUT/$name" --output-format csv -- \
    python /root/repo/scripts/gpu_attn_bench.py > "$OUT/$name.log" 2>&1 || true
}

# one counter set per run (hardware counter slots are limited)
run_pmc mfma   SQ_VALU_MFMA_BUSY_CYCLES SQ_BUSY_CYCLES
run_pmc lds    SQ_LDS_BANK_CONFLICT SQ_LDS_IDX_ACTIVE
run_pmc mem    TCC_REQ_sum TCC_HIT_sum TCC_MISS_sum
run_pmc waves  SQ_WAVES SQ_WAIT_ANY_CYCLES
echo "PMC outputs in $OUT — keep only the small CSV summaries for profiles/"
