"""Find layout-dependent windows: same windows, shuffled batch order."""
import random, sys
sys.path.insert(0, "build")
import _racon

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

windows = []
for i in range(48):
    blen = rng.choice([200, 350, 500, 500, 500, 520, 575, 700, 1000])
    bb = "".join(rng.choice("ACGT") for _ in range(blen))
    layers = [(bb, "!"*blen, 0, 0)]
    for _ in range(rng.choice([5, 20, 30])):
        layers.append((mutate(bb, 0.02, 0.02, 0.02)[:1023], "", 0, blen))
    windows.append(layers)

a = _racon.poa_windows_gpu(windows)
order = list(range(len(windows)))
rng.shuffle(order)
b = _racon.poa_windows_gpu([windows[i] for i in order])
bad = []
for k, i in enumerate(order):
    if a[i] != b[k]:
        bad.append((i, len(windows[i][0][0]), len(windows[i])-1,
                    len(a[i][0]), len(b[k][0]),
                    _racon.edit_distance(a[i][0], b[k][0])))
print("ORDER bad:", len(bad))
for t in bad[:12]:
    print("ORDER  idx=%d blen=%d depth=%d lenA=%d lenB=%d ed=%d" % t)
# repeat same order twice (pure run-to-run determinism)
c = _racon.poa_windows_gpu(windows)
print("ORDER rerun identical:", a == c)
