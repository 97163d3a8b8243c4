"""Builds the config-5 workload (BASELINE.json: HG002-class via
racon_wrapper --split): an N-contig synthetic genome with ONT-style reads
and PAF overlaps, concatenated from per-contig shards with unique names.

Usage: python tools/make_config5.py OUTDIR N_CONTIGS MBP_PER_CONTIG COVERAGE
"""
import multiprocessing
import pathlib
import sys

sys.path.insert(0, str(pathlib.Path(__file__).resolve().parent.parent))
from racon_amd import synth  # noqa: E402


def gen_shard(args):
    k, outroot, mbp, cov = args
    d = pathlib.Path(outroot) / f"shard{k}"
    synth.make_sample(d, genome_bp=int(mbp * 1e6), coverage=cov, seed=9000 + k,
                      read_len_mean=15000, read_len_sd=5000)
    # uniquify names across shards (sequence bytes are ACGT-only, so plain
    # text replacement cannot touch the data lines)
    for fname, subs in (
        ("reads.fasta", (("read", f"s{k}read"),)),
        ("layout.fasta", (("draft0", f"draft{k}"),)),
        ("reference.fasta", (("truth0", f"truth{k}"),)),
        ("overlaps.paf", (("read", f"s{k}read"), ("draft0", f"draft{k}"))),
    ):
        p = d / fname
        t = p.read_text()
        for a, b in subs:
            t = t.replace(a, b)
        p.write_text(t)
    return str(d)


def main():
    outroot = pathlib.Path(sys.argv[1])
    n = int(sys.argv[2])
    mbp = float(sys.argv[3])
    cov = int(sys.argv[4])
    outroot.mkdir(parents=True, exist_ok=True)

    with multiprocessing.Pool(min(n, 8)) as pool:
        shards = pool.map(gen_shard, [(k, str(outroot), mbp, cov) for k in range(n)])

    for fname in ("reads.fasta", "overlaps.paf", "layout.fasta", "reference.fasta"):
        with open(outroot / fname, "w") as out:
            for s in shards:
                with open(pathlib.Path(s) / fname) as f:
                    for chunk in iter(lambda: f.read(1 << 24), ""):
                        out.write(chunk)
    print(f"config5 ready: {n} x {mbp} Mbp x {cov}x under {outroot}")


if __name__ == "__main__":
    main()
