#!/usr/bin/env bash
# GPU CI (MI355X): GPU test suite + a short flagship bench + determinism
# double-run byte-diff (parity: reference ci/gpu/cuda_test.sh golden diff).
set -euo pipefail
cd "$(dirname "$0")/.."
python -m pytest tests/ -x -q -m gpu
python bench.py --steps 2 --warmup 1

# determinism: two CLI runs over the same synthetic sample must be byte-equal
WORK=$(mktemp -d)
trap 'rm -rf "$WORK"' EXIT
python - "$WORK" <<'PY'
import sys
from racon_amd import synth
synth.make_sample(sys.argv[1], genome_bp=1000000, coverage=20, seed=5)
PY
./build/racon -t 4 -c 2 --cudaaligner-batches 2 "$WORK/reads.fasta" \
    "$WORK/overlaps.paf" "$WORK/layout.fasta" > "$WORK/run1.fasta"
./build/racon -t 8 -c 1 --cudaaligner-batches 1 "$WORK/reads.fasta" \
    "$WORK/overlaps.paf" "$WORK/layout.fasta" > "$WORK/run2.fasta"
cmp "$WORK/run1.fasta" "$WORK/run2.fasta"
echo "determinism OK"
