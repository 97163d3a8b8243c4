import sys, pathlib, tempfile
sys.path.insert(0, "build"); sys.path.insert(0, ".")
from racon_amd import synth
import _racon, gzip

def read_fa(p):
    seqs = {}; name=None; buf=[]
    with open(p) as f:
        for line in f:
            if line.startswith(">"):
                if name: seqs[name]="".join(buf)
                name=line[1:].split()[0]; buf=[]
            else: buf.append(line.strip())
    if name: seqs[name]="".join(buf)
    return seqs

for tag, kw in [("ins-only", dict(sub=0.0, ins=0.02, dele=0.0)),
                ("default", dict())]:
    d = pathlib.Path(tempfile.mkdtemp())
    s = synth.make_sample(d, genome_bp=30000, coverage=30, seed=33, **kw)
    truth = list(read_fa(s["reference"]).values())[0]
    for w in (500, 1000):
        cpu = _racon.polish(s["reads"], s["overlaps"], s["layout"], threads=4, window_length=w)
        gpu = _racon.polish(s["reads"], s["overlaps"], s["layout"], threads=4, window_length=w, poa_batches=1)
        ec = _racon.edit_distance(cpu[0][1], truth)
        eg = _racon.edit_distance(gpu[0][1], truth)
        dv = _racon.edit_distance(cpu[0][1], gpu[0][1])
        print(f"PROBE {tag} w={w}: cpu={ec} gpu={eg} cpu-vs-gpu={dv} len_cpu={len(cpu[0][1])} len_gpu={len(gpu[0][1])}", flush=True)
