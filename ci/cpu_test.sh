#!/usr/bin/env bash
# CPU CI: build everything (HIP cross-compiles without a GPU) and run the
# full non-GPU test suite, including the reference-sample golden parity set.
# Parity: reference ci/cpu/build.sh.
set -euo pipefail
cd "$(dirname "$0")/.."
python __graft_entry__.py build
python -m pytest tests/ -x -q -m "not gpu"
