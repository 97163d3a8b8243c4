#!/usr/bin/env bash
# TSan job: threaded CPU pipeline race check (batch queue / pool / collect).
set -euo pipefail
cd "$(dirname "$0")/.."
cmake -B build-tsan -G Ninja -DRACON_BUILD_HIP=OFF -DRACON_BUILD_PYTHON=OFF \
      -DCMAKE_BUILD_TYPE=Debug -DRACON_SANITIZE=thread
ninja -C build-tsan
python3 - <<'PY'
from racon_amd import synth
synth.make_sample("/tmp/tsan_sample", genome_bp=60000, coverage=20, seed=9)
PY
./build-tsan/racon -t 8 /tmp/tsan_sample/reads.fasta /tmp/tsan_sample/overlaps.paf \
    /tmp/tsan_sample/layout.fasta > /dev/null
echo "tsan OK"
