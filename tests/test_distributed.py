"""Multi-process tests of the engine's own distributed component
(src/hip/comm.cpp: TCP control plane + RCCL data plane) and of bench.py's
multi-rank flow — run on CPU here over the TCP plane; the same code runs
the RCCL plane over xGMI on the GPU node."""

import json
import os
import subprocess
import sys
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent

COMM_WORKER = r'''
import os, sys
sys.path.insert(0, sys.argv[4])
import _racon
rank = int(sys.argv[1]); world = int(sys.argv[2]); port = int(sys.argv[3])
_racon.comm_init(rank, world, "127.0.0.1", port, False)
# variable-length gather: rank r contributes (r+1) copies of its tag
payload = (f"<{rank}>" * (rank + 1)).encode()
parts = _racon.comm_gather(payload, 0)
if rank == 0:
    assert [p.decode().count("<") for p in parts] == list(range(1, world + 1)), parts
    assert parts[2].decode() == "<2>" * 3
else:
    assert parts == []
# reductions visible on every rank
assert _racon.comm_allreduce_sum(float(rank + 1)) == world * (world + 1) / 2
assert _racon.comm_allreduce_max(float(rank * 7)) == (world - 1) * 7.0
_racon.comm_barrier()
# a second gather on the same communicator (state is reusable across steps)
parts = _racon.comm_gather(b"x" * (1000 * (rank + 1)), 0)
if rank == 0:
    assert [len(p) for p in parts] == [1000 * (r + 1) for r in range(world)]
_racon.comm_finalize()
print("COMM_WORKER_OK")
'''


def test_comm_gather_allreduce_three_ranks():
    """Gather/allreduce/barrier correctness on the TCP plane, world=3."""
    procs = [
        subprocess.Popen(
            [sys.executable, "-c", COMM_WORKER, str(r), "3", "29531", str(REPO / "build")],
            stdout=subprocess.PIPE, stderr=subprocess.PIPE, text=True)
        for r in range(3)
    ]
    for p in procs:
        out, err = p.communicate(timeout=120)
        assert p.returncode == 0, err[-2000:]
        assert "COMM_WORKER_OK" in out


def test_bench_two_ranks_cpu(tmp_path):
    """bench.py end-to-end at world_size=2 on CPU (tiny shards): bootstrap,
    per-rank polish, length-prefixed gather, max/sum stat reduction."""
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
    # only rank 0 prints the JSON record
    assert not [l for l in outs[1].splitlines() if l.startswith("{")]


COMM_BIG_WORKER = r'''
import os, sys
sys.path.insert(0, sys.argv[4])
import _racon
rank = int(sys.argv[1]); world = int(sys.argv[2]); port = int(sys.argv[3])
_racon.comm_init(rank, world, "127.0.0.1", port, False)
# multi-megabyte uneven payloads across several rounds (bench does one
# gather per step; sizes vary as contigs change)
for step in range(3):
    payload = bytes([rank * 7 + step]) * (1_000_000 * (rank + 1) + step)
    parts = _racon.comm_gather(payload, 0)
    if rank == 0:
        assert [len(p) for p in parts] == [1_000_000 * (r + 1) + step for r in range(world)]
        assert all(p[0] == r * 7 + step for r, p in enumerate(parts))
    _racon.comm_barrier()
_racon.comm_finalize()
print("COMM_BIG_OK")
'''


def test_comm_multi_round_large_payloads():
    """World=4 TCP-plane gather with MB-scale uneven payloads over several
    rounds (the shape of bench.py's per-step consensus gather)."""
    procs = [
        subprocess.Popen(
            [sys.executable, "-c", COMM_BIG_WORKER, str(r), "4", "29537", str(REPO / "build")],
            stdout=subprocess.PIPE, stderr=subprocess.PIPE, text=True)
        for r in range(4)
    ]
    for p in procs:
        out, err = p.communicate(timeout=180)
        assert p.returncode == 0, err[-2000:]
        assert "COMM_BIG_OK" in out


def test_comm_init_errors():
    """Misuse must raise, not crash: world < 2, and double init."""
    code = r'''
import sys
sys.path.insert(0, sys.argv[1])
import _racon
try:
    _racon.comm_init(0, 1, "127.0.0.1", 29541, False)
    raise SystemExit("expected world<2 to raise")
except RuntimeError:
    pass
print("COMM_ERRORS_OK")
'''
    out = subprocess.run([sys.executable, "-c", code, str(REPO / "build")],
                         capture_output=True, text=True, timeout=60)
    assert out.returncode == 0, out.stderr[-1000:]
    assert "COMM_ERRORS_OK" in out.stdout
