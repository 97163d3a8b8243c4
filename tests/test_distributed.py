"""Multi-process (gloo, world_size=2) tests of the distributed gather path
used by bench.py — runs on CPU here; the same code runs over RCCL on the
GPU node (backend name "nccl" is RCCL on ROCm)."""

import json
import os
import subprocess
import sys
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent


def test_bench_two_ranks_gloo(tmp_path):
    """bench.py end-to-end at world_size=2 on CPU (tiny shards)."""
    env = dict(os.environ)
    env["MASTER_ADDR"] = "127.0.0.1"
    env["MASTER_PORT"] = "29511"
    env["TMPDIR"] = str(tmp_path)
    procs = []
    for rank in range(2):
        e = dict(env, RANK=str(rank), LOCAL_RANK=str(rank), WORLD_SIZE="2")
        procs.append(subprocess.Popen(
            [sys.executable, str(REPO / "bench.py"), "--gpus", "2", "--steps", "1",
             "--warmup", "0", "--genome-mbp", "0.05", "--coverage", "10", "--cpu"],
            stdout=subprocess.PIPE, stderr=subprocess.PIPE, env=e, text=True))
    outs = []
    for p in procs:
        out, err = p.communicate(timeout=600)
        assert p.returncode == 0, err[-2000:]
        outs.append(out)
    # rank 0 prints exactly one JSON line with whole-job aggregates
    line = [l for l in outs[0].splitlines() if l.startswith("{")]
    assert len(line) == 1
    rec = json.loads(line[0])
    assert rec["n_gpus"] == 2
    assert rec["scaling"] == "weak"
    assert rec["value"] > 0
    # only rank 0 prints the JSON record (gloo may chat about peers)
    assert not [l for l in outs[1].splitlines() if l.startswith("{")]
