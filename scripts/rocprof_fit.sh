#!/usr/bin/env bash
# rocprofv3 harness for the scheduling-simulation hot path:
# captures per-kernel time stats for the flagship decision benchmark so
# the LDS-staged fit kernel and the MFMA segmented sums are visible in
# a committed profile.
#
# Usage (on an MI355X box):
#   scripts/rocprof_fit.sh [outdir] [bench args...]
# PMC counters (MFMA util, LDS conflicts) need a SEPARATE run:
#   scripts/rocprof_fit.sh --pmc
# If rocprofv3 segfaults on the box image (seen late round 1), use the
# event-timed fallback: python scripts/probe_phase_times.py
set -euo pipefail
cd "$(dirname "$0")/.."
REPO="$PWD"
OUT="${1:-gpurun_out/rocprof_fit}"
shift || true
mkdir -p "$OUT"
cd /tmp && export TMPDIR=/tmp

if [ "${OUT}" = "--pmc" ]; then
  OUT="$REPO/gpurun_out/rocprof_fit_pmc"
  mkdir -p "$OUT"
  # counters-only run (never combined with trace domains)
  exec rocprofv3 --pmc SQ_INSTS_MFMA SQ_INSTS_VALU SQ_LDS_BANK_CONFLICT SQ_WAVES \
    --output-format csv -d "$OUT" -o fit_pmc -- \
    python "$REPO/bench.py" --steps 5 --warmup 2 "$@"
fi

exec rocprofv3 --kernel-trace --stats --output-format csv -d "$OUT" -o fit \
  -- python "$REPO/bench.py" --steps 10 --warmup 3 "$@"
