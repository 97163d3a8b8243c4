#!/usr/bin/env bash
# rocprofv3 profiling recipes for the HIP kernels (MI355X / gfx950).
# Parity: the reference's cudaProfilerStop + -lineinfo profiling hooks
# (cudapolisher.cpp:71, CMakeLists.txt:25) — here as a first-class target.
#
# Usage:
#   tools/profile.sh trace [bench args...]   # kernel trace + per-kernel stats
#   tools/profile.sh pmc   [bench args...]   # SQ wait/issue counters
#   tools/profile.sh hbm   [bench args...]   # L2 fetch/write bytes
#   tools/profile.sh lds   [bench args...]   # LDS issue/stall counters
# Output lands in profiles/<mode>/ (CSV + summary).
set -euo pipefail
cd "$(dirname "$0")/.."
MODE=${1:-trace}
shift || true
OUT=profiles/$MODE
mkdir -p "$OUT"
export TMPDIR=${TMPDIR:-/tmp}

BENCH=(python bench.py --genome-mbp 2 --steps 1 --warmup 0 "$@")

case "$MODE" in
  trace)
    (cd /tmp && rocprofv3 --kernel-trace --stats -d "$OLDPWD/$OUT" -- "${BENCH[@]}")
    ;;
  pmc)
    (cd /tmp && rocprofv3 --pmc SQ_WAVES SQ_WAVE_CYCLES SQ_WAIT_ANY SQ_WAIT_INST_ANY \
        SQ_ACTIVE_INST_ANY SQ_INSTS_VALU SQ_INSTS_LDS --stats -d "$OLDPWD/$OUT" \
        -- "${BENCH[@]}")
    ;;
  hbm)
    (cd /tmp && rocprofv3 --pmc TCC_EA0_RDREQ_sum TCC_EA0_WRREQ_sum --stats \
        -d "$OLDPWD/$OUT" -- "${BENCH[@]}")
    ;;
  lds)
    (cd /tmp && rocprofv3 --pmc SQ_INSTS_LDS SQ_WAIT_INST_LDS SQ_WAVE_CYCLES SQ_WAVES \
        --stats -d "$OLDPWD/$OUT" -- "${BENCH[@]}")
    ;;
  *)
    echo "unknown mode: $MODE (trace|pmc|hbm|lds)" >&2
    exit 1
    ;;
esac
echo "profile written to $OUT"
