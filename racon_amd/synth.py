"""Synthetic ONT-style polishing workloads.

Generates, from a fixed seed: a random genome, a draft assembly (the polishing
target: genome + substitution errors, no indels so true read coordinates stay
exact), long reads with an ONT-like error profile (substitutions + insertions
+ deletions), and a PAF overlap file mapping each read to the draft.

Used by tests/ (small configs) and bench.py (C. elegans-scale configs);
there is no dataset download in this environment (synthetic
long reads of the named genome size / coverage / error profile).
"""

from pathlib import Path

import numpy as np

BASES = np.frombuffer(b"ACGT", dtype=np.uint8)
COMP = np.zeros(256, dtype=np.uint8)
for a, b in zip(b"ACGT", b"TGCA"):
    COMP[a] = b


def random_genome(rng, n):
    return BASES[rng.integers(0, 4, size=n)]


def mutate_reads(rng, genome, starts, lengths, sub=0.02, ins=0.02, dele=0.02):
    """Vectorized read mutation. Returns list of uint8 arrays."""
    reads = []
    for s, L in zip(starts, lengths):
        seg = genome[s : s + L].copy()
        n = seg.size
        r = rng.random(n)
        # substitutions: xor-shift into a different base
        sub_mask = r < sub
        seg[sub_mask] = BASES[(np.searchsorted(BASES, seg[sub_mask]) + rng.integers(1, 4, size=int(sub_mask.sum()))) % 4]
        # indels via repeat counts (0 = deletion, 2 = insertion after)
        counts = np.ones(n, dtype=np.int64)
        counts[(r >= sub) & (r < sub + dele)] = 0
        ins_mask = (r >= sub + dele) & (r < sub + dele + ins)
        counts[ins_mask] = 2
        out = np.repeat(seg, counts)
        # replace the duplicated (inserted) copy with a random base
        ins_pos = np.cumsum(counts)[ins_mask] - 1
        out[ins_pos] = BASES[rng.integers(0, 4, size=ins_pos.size)]
        reads.append(out)
    return reads


def revcomp(arr):
    return COMP[arr[::-1]]


def write_fasta(path, records):
    with open(path, "w") as f:
        for name, seq in records:
            f.write(f">{name}\n{seq}\n")


def make_sample(outdir, genome_bp=20000, coverage=20, seed=0, read_len_mean=12000,
                read_len_sd=4000, sub=0.02, ins=0.02, dele=0.02, draft_sub=0.02,
                paired_files=True):
    """Builds a polishing workload; returns a dict of file paths + truths."""
    outdir = Path(outdir)
    outdir.mkdir(parents=True, exist_ok=True)
    rng = np.random.default_rng(seed)

    genome = random_genome(rng, genome_bp)

    # draft: substitution-only errors keep read->draft coordinates exact
    draft = genome.copy()
    dmask = rng.random(genome_bp) < draft_sub
    draft[dmask] = BASES[(np.searchsorted(BASES, draft[dmask]) + rng.integers(1, 4, size=int(dmask.sum()))) % 4]

    n_reads = max(3, int(genome_bp * coverage / read_len_mean))
    lengths = np.clip(rng.normal(read_len_mean, read_len_sd, n_reads), 500, None).astype(np.int64)
    lengths = np.minimum(lengths, genome_bp)
    starts = rng.integers(0, np.maximum(1, genome_bp - lengths + 1))

    reads = mutate_reads(rng, genome, starts, lengths, sub, ins, dele)
    strands = rng.random(n_reads) < 0.5

    read_records = []
    paf_lines = []
    for i, (arr, s, L, rc) in enumerate(zip(reads, starts, lengths, strands)):
        name = f"read{i:06d}"
        out = revcomp(arr) if rc else arr
        read_records.append((name, out.tobytes().decode()))
        qlen = arr.size
        paf_lines.append(
            f"{name}\t{qlen}\t0\t{qlen}\t{'-' if rc else '+'}\tdraft0\t{genome_bp}\t{s}\t{s + L}\t{qlen}\t{max(qlen, L)}\t255"
        )

    paths = {
        "reads": str(outdir / "reads.fasta"),
        "overlaps": str(outdir / "overlaps.paf"),
        "layout": str(outdir / "layout.fasta"),
        "reference": str(outdir / "reference.fasta"),
        "genome_bp": genome_bp,
        "n_reads": n_reads,
    }
    write_fasta(paths["reads"], read_records)
    with open(paths["overlaps"], "w") as f:
        f.write("\n".join(paf_lines) + "\n")
    write_fasta(paths["layout"], [("draft0", draft.tobytes().decode())])
    write_fasta(paths["reference"], [("truth0", genome.tobytes().decode())])

    # all-vs-all read overlaps (for -f fragment correction); approximate
    # genome-projected coordinates — racon realigns the spans itself
    ava_lines = []
    order = np.argsort(starts)
    min_olap = 500
    rl = [int(r.size) for r in reads]  # actual (indel-shifted) read lengths
    for oi in range(n_reads):
        i = order[oi]
        for oj in range(oi + 1, n_reads):
            j = order[oj]
            a = max(starts[i], starts[j])
            b = min(starts[i] + lengths[i], starts[j] + lengths[j])
            if b - a < min_olap:
                if starts[j] >= starts[i] + lengths[i]:
                    break
                continue
            qlen, tlen = rl[i], rl[j]

            def span(s0, L, gl, rc, lo, hi):
                # genome offsets scaled into the (indel-shifted) read
                lo_r = min(L, (lo - s0) * L // gl)
                hi_r = min(L, (hi - s0) * L // gl)
                if rc:
                    return L - hi_r, L - lo_r
                return lo_r, hi_r

            qs, qe = span(starts[i], qlen, int(lengths[i]), strands[i], a, b)
            ts, te = span(starts[j], tlen, int(lengths[j]), strands[j], a, b)
            rel = "-" if strands[i] != strands[j] else "+"
            ava_lines.append(
                f"read{i:06d}\t{qlen}\t{qs}\t{qe}\t{rel}\tread{j:06d}\t{tlen}\t{ts}\t{te}"
                f"\t{b - a}\t{b - a}\t255")
    paths["ava_overlaps"] = str(outdir / "ava_overlaps.paf")
    with open(paths["ava_overlaps"], "w") as f:
        f.write("\n".join(ava_lines) + "\n")
    return paths
