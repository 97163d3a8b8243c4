"""Window-level CPU-vs-GPU differ sweep over backbone lengths (bug hunt for
the w=1000 divergence)."""
import sys
sys.path.insert(0, "build")
import _racon, random

rng = random.Random(5)
def mutate(seq, sub, ins, dele):
    out=[]
    for ch in seq:
        r=rng.random()
        if r<dele: continue
        if r<dele+ins: out.append(rng.choice("ACGT"))
        if r<dele+ins+sub: out.append(rng.choice([c for c in "ACGT" if c!=ch]))
        else: out.append(ch)
    return "".join(out)

for blen in (500, 640, 768, 832, 900, 960, 1000, 1023):
    bb = "".join(rng.choice("ACGT") for _ in range(blen))
    layers = [(bb, "!"*blen, 0, 0)]
    for _ in range(20):
        layers.append((mutate(bb, 0.02, 0.02, 0.02), "", 0, blen))
    for trim in (True, False):
        cpu = _racon.poa_windows_cpu([layers], trim=trim)
        gpu = _racon.poa_windows_gpu([layers], trim=trim)
        dv = _racon.edit_distance(cpu[0][0], gpu[0][0])
        print(f"PROBE blen={blen} trim={int(trim)}: cpu_len={len(cpu[0][0])} "
              f"gpu_len={len(gpu[0][0])} ed={dv} gpu_ok={gpu[0][1]}", flush=True)
