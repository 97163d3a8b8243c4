"""Tests for the M5 tooling: rampler (subsample/split), racon_wrapper,
racon_preprocess. CPU-only."""

import gzip
import os
import subprocess
import sys
from pathlib import Path

import pytest

REPO = Path(__file__).resolve().parent.parent
RAMPLER = REPO / "build" / "rampler"
WRAPPER = REPO / "scripts" / "racon_wrapper.py"
PREPROCESS = REPO / "scripts" / "racon_preprocess.py"


@pytest.fixture(scope="module")
def rampler(racon):  # racon fixture builds the tree
    assert RAMPLER.exists()
    return str(RAMPLER)


def write_fasta(path, seqs):
    with open(path, "w") as f:
        for name, data in seqs:
            f.write(f">{name}\n{data}\n")


def read_fasta_names(path):
    names = []
    with open(path) as f:
        for line in f:
            if line.startswith(">"):
                names.append(line[1:].strip())
    return names


def test_rampler_split(tmp_path, rampler):
    seqs = [(f"s{i}", "ACGT" * 250) for i in range(10)]  # 1000 bp each
    src = tmp_path / "targets.fasta"
    write_fasta(src, seqs)
    subprocess.run([rampler, "-o", str(tmp_path), "split", str(src), "2500"], check=True)
    chunks = sorted(tmp_path.glob("targets_*.fasta"))
    assert [c.name for c in chunks] == [f"targets_{i}.fasta" for i in range(5)]
    got = [n for c in chunks for n in read_fasta_names(c)]
    assert got == [f"s{i}" for i in range(10)]  # order preserved, none lost


def test_rampler_subsample(tmp_path, rampler):
    seqs = [(f"r{i}", "ACGT" * 250) for i in range(100)]  # 100 kbp total
    src = tmp_path / "reads.fasta"
    write_fasta(src, seqs)
    # ask for 10x of a 2500 bp reference = 25 kbp ~ 25% of reads
    subprocess.run([rampler, "-o", str(tmp_path), "subsample", str(src), "2500", "10"],
                   check=True)
    out = tmp_path / "reads_10x.fasta"
    assert out.exists()
    n = len(read_fasta_names(out))
    assert 5 <= n <= 60, n  # binomial around 25

    # full coverage keeps everything
    subprocess.run([rampler, "-o", str(tmp_path), "subsample", str(src), "2500", "100"],
                   check=True)
    assert len(read_fasta_names(tmp_path / "reads_100x.fasta")) == 100


def test_rampler_gz_fastq(tmp_path, rampler):
    src = tmp_path / "reads.fastq.gz"
    with gzip.open(src, "wt") as f:
        for i in range(4):
            f.write(f"@q{i}\nACGTACGT\n+\nIIIIIIII\n")
    subprocess.run([rampler, "-o", str(tmp_path), "split", str(src), "16"], check=True)
    chunks = sorted(tmp_path.glob("reads_*.fastq"))
    assert len(chunks) == 2
    text = chunks[0].read_text().splitlines()
    assert text[0] == "@q0" and text[2] == "+" and text[3] == "IIIIIIII"


def test_wrapper_split_end_to_end(tmp_path, racon, sample):
    env = dict(os.environ, RACON_BIN=str(REPO / "build" / "racon"), RAMPLER_BIN=str(RAMPLER))
    out = subprocess.run(
        [sys.executable, str(WRAPPER), "--split", "8000", "-t", "2",
         sample["reads"], sample["overlaps"], sample["layout"]],
        capture_output=True, text=True, env=env, cwd=tmp_path, check=True)
    polished = [l for l in out.stdout.splitlines() if l.startswith(">")]
    assert len(polished) >= 1
    # single-contig layout: split granularity is whole records -> 1 chunk
    assert "total number of splits: 1" in out.stderr


def test_wrapper_forwards_gpu_flags(tmp_path):
    """The wrapper must forward GPU flags (the reference drops them)."""
    fake = tmp_path / "fake_racon.sh"
    fake.write_text("#!/bin/sh\necho \"$@\"\n")
    fake.chmod(0o755)
    for f in ("reads.fa", "ovl.paf", "tgt.fa"):
        (tmp_path / f).write_text("")
    env = dict(os.environ, RACON_BIN=str(fake), RAMPLER_BIN=str(fake))
    out = subprocess.run(
        [sys.executable, str(WRAPPER), "-c", "4", "--cudaaligner-batches", "2", "-b",
         str(tmp_path / "reads.fa"), str(tmp_path / "ovl.paf"), str(tmp_path / "tgt.fa")],
        capture_output=True, text=True, env=env, cwd=tmp_path, check=True)
    assert "-c 4" in out.stdout
    assert "--cudaaligner-batches 2" in out.stdout
    assert "-b" in out.stdout


def test_preprocess_uniquifies_pairs(tmp_path):
    fq = tmp_path / "pe.fastq"
    fq.write_text("@p1 extra\nACGT\n+\nIIII\n@p1 extra\nTTTT\n+\nIIII\n@p2\nGGGG\n+\nIIII\n")
    out = subprocess.run([sys.executable, str(PREPROCESS), str(fq)],
                         capture_output=True, text=True, check=True)
    lines = out.stdout.splitlines()
    assert lines[0] == "@p11"
    assert lines[4] == "@p12"
    assert lines[8] == "@p21"


def test_rampler_edge_cases(tmp_path, rampler):
    # empty input: subsample errors cleanly; split writes nothing
    empty = tmp_path / "empty.fasta"
    empty.write_text("")
    r = subprocess.run([rampler, "-o", str(tmp_path), "subsample", str(empty), "1000", "5"],
                       capture_output=True, text=True)
    assert r.returncode != 0
    r = subprocess.run([rampler, "-o", str(tmp_path), "split", str(empty), "1000"],
                       capture_output=True, text=True)
    assert r.returncode == 0
    assert "wrote 0 chunks" in r.stderr

    # single record larger than the chunk size still lands in one chunk
    single = tmp_path / "single.fasta"
    write_fasta(single, [("big", "ACGT" * 1000)])
    r = subprocess.run([rampler, "-o", str(tmp_path), "split", str(single), "100"],
                       capture_output=True, text=True)
    assert r.returncode == 0
    assert (tmp_path / "single_0.fasta").exists()
    assert not (tmp_path / "single_1.fasta").exists()


def test_wrapper_in_process_split_matches_subprocess(racon, sample, tmp_path):
    """--in-process chunk polishing (single process, pooled GPU arenas) must
    be byte-identical to the reference-parity one-subprocess-per-chunk mode."""
    import subprocess
    import sys as _sys
    from pathlib import Path
    repo = Path(__file__).resolve().parent.parent
    base = [_sys.executable, str(repo / "scripts" / "racon_wrapper.py"),
            "--split", "8000", "-t", "4",
            sample["reads"], sample["overlaps"], sample["layout"]]
    a = subprocess.run(base, capture_output=True, text=True, timeout=600)
    assert a.returncode == 0, a.stderr[-800:]
    b = subprocess.run(base + ["--in-process"], capture_output=True, text=True, timeout=600)
    assert b.returncode == 0, b.stderr[-800:]
    assert a.stdout == b.stdout
