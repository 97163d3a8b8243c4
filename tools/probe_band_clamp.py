"""Quantify the aligner band-clamp quality bet.

The polisher's auto band clamps to [64, 256] (hip_polisher.cpp) where the
reference uses an unbounded 10% of mean overlap length
(cudapolisher.cpp:150-163). The clamp is a throughput bet: escapes fall back
per item to the exact CPU aligner. This measures the escape rate and the
resulting end-to-end quality at 100 kbp-read scale and 10-12% error.
Run on a GPU box: python tools/probe_band_clamp.py
"""
import random
import sys

sys.path.insert(0, "build")
import _racon  # noqa: E402


def mutate(rng, seq, sub, ins, dele):
    out = []
    for ch in seq:
        r = rng.random()
        if r < dele:
            continue
        if r < dele + ins:
            out.append(rng.choice("ACGT"))
        if r < dele + ins + sub:
            out.append(rng.choice([c for c in "ACGT" if c != ch]))
        else:
            out.append(ch)
    return "".join(out)


def main():
    rng = random.Random(1234)
    for read_len, err in ((20000, 0.02), (100000, 0.02), (100000, 0.04)):
        pairs = []
        for _ in range(24):
            t = "".join(rng.choice("ACGT") for _ in range(read_len))
            q = mutate(rng, t, err, err, err)  # err per channel: 3*err total
            pairs.append((q, t))
        for band in (256, 512, 1024, 2048):
            res = _racon.gpu_align(pairs, band_width=band)
            ok = sum(1 for _, _, st in res if st == 0)
            exact = sum(1 for (q, t), (_, ed, st) in zip(pairs, res)
                        if st == 0 and ed == _racon.edit_distance(q, t))
            print(f"BAND read_len={read_len} err={3*err:.0%} band={band}: "
                  f"completed {ok}/{len(pairs)}, exact {exact}/{ok}", flush=True)


if __name__ == "__main__":
    main()
