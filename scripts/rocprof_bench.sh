#!/bin/bash
# Capture rocprofv3 kernel-trace + PMC evidence for the flagship bench and
# extract compact summaries (SURVEY §5 tracing parity; profiles/ workflow).
#
#   ./scripts/rocprof_bench.sh out_dir [batch]
#
# Two separate captures (never combine --pmc with trace domains):
#   1. --kernel-trace --stats  -> per-kernel time table
#   2. --pmc <counters> --kernel-trace -> MFMA/VALU/LDS counters per kernel
set -e
OUT=${1:-profiles}
B=${2:-512}
export TMPDIR=/tmp
mkdir -p "$OUT" /tmp/rocp_k /tmp/rocp_c
ROOT="$(cd "$(dirname "$0")/.." && pwd)"
cd /tmp
rocprofv3 --kernel-trace --stats -d /tmp/rocp_k -- \
    python "$ROOT/bench.py" --steps 4 --warmup 2 --batch-per-gpu "$B"
rocprofv3 --pmc SQ_INSTS_MFMA SQ_INSTS_VALU SQ_WAVES SQ_LDS_BANK_CONFLICT \
    --kernel-trace -d /tmp/rocp_c -- \
    python "$ROOT/bench.py" --steps 2 --warmup 1 --batch-per-gpu 256
cd "$ROOT"
python3 tools/extract_rocprof.py kernels /tmp/rocp_k "$OUT/rocprof_kernels_b$B.txt"
python3 tools/extract_rocprof.py pmc /tmp/rocp_c "$OUT/pmc_counters_b256.txt"
echo "wrote $OUT/rocprof_kernels_b$B.txt and $OUT/pmc_counters_b256.txt"
