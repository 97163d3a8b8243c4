#!/usr/bin/env bash
# ASan CI job (CPU path; parity: reference `make debug` -Db_sanitize=address).
set -euo pipefail
cd "$(dirname "$0")/.."
cmake -B build-asan -G Ninja -DRACON_BUILD_HIP=OFF -DRACON_BUILD_PYTHON=OFF \
      -DCMAKE_BUILD_TYPE=Debug -DRACON_SANITIZE=address
ninja -C build-asan
python - <<'PY'
from racon_amd import synth
synth.make_sample("/tmp/asan_sample", genome_bp=50000, coverage=15, seed=3)
PY
./build-asan/racon -t 4 /tmp/asan_sample/reads.fasta /tmp/asan_sample/overlaps.paf \
    /tmp/asan_sample/layout.fasta > /dev/null
echo "asan OK"

# deeper pass: the reference golden workload end-to-end under ASan
if [ -d /root/reference/test/data ]; then
  ./build-asan/racon -t 8 -m 5 -x -4 -g -8 \
      /root/reference/test/data/sample_reads.fastq.gz \
      /root/reference/test/data/sample_overlaps.paf.gz \
      /root/reference/test/data/sample_layout.fasta.gz > /dev/null
  echo "asan golden OK"
fi
