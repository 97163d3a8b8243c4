#!/usr/bin/env python3
"""Quick golden comparison harness against the reference sample data.

Usage: python3 tests/golden_check.py [builddir]
Prints one line per scenario: <golden> <ours> <delta>.
"""
import sys
import gzip

BUILD = sys.argv[1] if len(sys.argv) > 1 else "build"
sys.path.insert(0, BUILD)
import _racon  # noqa: E402

D = "/root/reference/test/data/"


def read_fasta(path):
    op = gzip.open if path.endswith(".gz") else open
    seqs = {}
    name, chunks = None, []
    with op(path, "rt") as f:
        for line in f:
            line = line.rstrip()
            if line.startswith(">"):
                if name:
                    seqs[name] = "".join(chunks)
                name, chunks = line[1:].split()[0], []
            else:
                chunks.append(line)
    if name:
        seqs[name] = "".join(chunks)
    return seqs


REF = list(read_fasta(D + "sample_reference.fasta.gz").values())[0].upper()


def polish_ed(reads, ovl, **kw):
    out = _racon.polish(D + reads, D + ovl, D + "sample_layout.fasta.gz", threads=4, **kw)
    assert len(out) == 1, len(out)
    rc = _racon.reverse_complement(out[0][1])
    return _racon.edit_distance(rc, REF)


def frag(reads, ovl, **kw):
    out = _racon.polish(D + reads, D + ovl, D + reads, threads=4, match=1, mismatch=-1, gap=-1, **kw)
    return len(out), sum(len(s) for _, s in out)


def main():
    rows = []
    rows.append(("polish fastq+paf", 1312,
                 polish_ed("sample_reads.fastq.gz", "sample_overlaps.paf.gz", match=5, mismatch=-4, gap=-8)))
    rows.append(("polish fasta+paf", 1566,
                 polish_ed("sample_reads.fasta.gz", "sample_overlaps.paf.gz", match=5, mismatch=-4, gap=-8)))
    rows.append(("polish fastq+sam", 1317,
                 polish_ed("sample_reads.fastq.gz", "sample_overlaps.sam.gz", match=5, mismatch=-4, gap=-8)))
    rows.append(("polish fasta+sam", 1770,
                 polish_ed("sample_reads.fasta.gz", "sample_overlaps.sam.gz", match=5, mismatch=-4, gap=-8)))
    rows.append(("polish w=1000", 1289,
                 polish_ed("sample_reads.fastq.gz", "sample_overlaps.paf.gz", match=5, mismatch=-4, gap=-8,
                           window_length=1000)))
    rows.append(("polish unit scores", 1321,
                 polish_ed("sample_reads.fastq.gz", "sample_overlaps.paf.gz", match=1, mismatch=-1, gap=-1)))

    n, t = frag("sample_reads.fastq.gz", "sample_ava_overlaps.paf.gz")
    rows.append(("frag kC count", 39, n))
    rows.append(("frag kC total", 389394, t))
    n, t = frag("sample_reads.fastq.gz", "sample_ava_overlaps.paf.gz", fragment_correction=True,
                include_unpolished=True)
    rows.append(("frag kF fastq count", 236, n))
    rows.append(("frag kF fastq total", 1658216, t))
    n, t = frag("sample_reads.fasta.gz", "sample_ava_overlaps.paf.gz", fragment_correction=True,
                include_unpolished=True)
    rows.append(("frag kF fasta count", 236, n))
    rows.append(("frag kF fasta total", 1663982, t))
    n, t = frag("sample_reads.fastq.gz", "sample_ava_overlaps.mhap.gz", fragment_correction=True,
                include_unpolished=True)
    rows.append(("frag kF mhap count", 236, n))
    rows.append(("frag kF mhap total", 1658216, t))

    exact = 0
    for name, golden, ours in rows:
        mark = "OK " if golden == ours else "DIFF"
        exact += golden == ours
        print(f"{mark} {name:24s} golden={golden:9d} ours={ours:9d} delta={ours - golden:+d}")
    print(f"{exact}/{len(rows)} exact")


if __name__ == "__main__":
    main()
