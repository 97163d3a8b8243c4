#!/usr/bin/env python3
"""Parser robustness fuzzer: mutates valid FASTA/PAF inputs (truncation,
byte flips, span deletion/duplication, junk injection) and runs the ASan
racon binary — any exit other than clean success/clean error, or any
sanitizer report, is a finding. Run ci/asan_test.sh first to build.
Usage: python tools/fuzz_parsers.py [trials]"""
import random, subprocess, sys, tempfile, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from racon_amd import synth

def mutate(rng, data):
    data = bytearray(data)
    mode = rng.randrange(5)
    if mode == 0 and data:
        del data[rng.randrange(len(data)):]
    elif mode == 1 and data:
        for _ in range(rng.randrange(1, 30)):
            data[rng.randrange(len(data))] = rng.randrange(256)
    elif mode == 2 and data:
        a = rng.randrange(len(data)); b = min(len(data), a + rng.randrange(1, 500))
        del data[a:b]
    elif mode == 3:
        a = rng.randrange(max(1, len(data))); b = min(len(data), a + rng.randrange(1, 500))
        data[a:a] = data[a:b]
    else:
        pos = rng.randrange(max(1, len(data)))
        data[pos:pos] = b"\n" + bytes(rng.randrange(32, 127)
                                      for _ in range(rng.randrange(200))) + b"\t\t\t-\n"
    return bytes(data)

def main():
    trials = int(sys.argv[1]) if len(sys.argv) > 1 else 120
    rng = random.Random(1234)
    binary = "./build-asan/racon"
    crashes = 0
    with tempfile.TemporaryDirectory() as d:
        s = synth.make_sample(d, genome_bp=15000, coverage=10, seed=4)
        base = {k: open(s[k], "rb").read() for k in ("reads", "overlaps", "layout")}
        for trial in range(trials):
            victim = rng.choice(list(base))
            files = dict(base)
            files[victim] = mutate(rng, files[victim])
            paths = {}
            for k, v in files.items():
                p = os.path.join(d, f"f_{k}." + ("paf" if k == "overlaps" else "fasta"))
                open(p, "wb").write(v)
                paths[k] = p
            r = subprocess.run([binary, "-t", "2", paths["reads"], paths["overlaps"],
                                paths["layout"]], capture_output=True, text=True, timeout=120)
            if r.returncode not in (0, 1) or "Sanitizer" in r.stderr:
                crashes += 1
                print("FINDING at trial", trial, victim, "rc", r.returncode)
                print(r.stderr[-1200:])
    print(f"fuzz done: {trials} trials, {crashes} findings")
    return 1 if crashes else 0

if __name__ == "__main__":
    sys.exit(main())
