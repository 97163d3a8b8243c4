"""Map where the GPU e2e w=1000 output loses bases vs CPU: align the two
consensus strings, report big indel runs and their positions mod window."""
import sys, re, pathlib, tempfile
sys.path.insert(0, "build"); sys.path.insert(0, ".")
from racon_amd import synth
import _racon

d = pathlib.Path(tempfile.mkdtemp())
s = synth.make_sample(d, genome_bp=30000, coverage=30, seed=33)
cpu = _racon.polish(s["reads"], s["overlaps"], s["layout"], threads=4, window_length=1000)
gpu = _racon.polish(s["reads"], s["overlaps"], s["layout"], threads=4, window_length=1000, poa_batches=1)
c, g = cpu[0][1], gpu[0][1]
print(f"LOSS len_cpu={len(c)} len_gpu={len(g)}", flush=True)
cig = _racon.align_cigar(g, c)  # query=gpu, target=cpu: D = missing-in-gpu
ci = 0
runs = []
for num, op in re.findall(r"(\d+)([MID])", cig):
    num = int(num)
    if op == "D":
        if num >= 4:
            runs.append((ci, num))
        ci += num
    elif op == "M":
        ci += num
print("LOSS big-deletion runs (cpu_pos, len, pos_mod_1000):", flush=True)
for pos, num in runs[:40]:
    print(f"LOSS   {pos} {num} mod={pos % 1000}", flush=True)
# count of big runs and total lost in them
print(f"LOSS n_runs={len(runs)} total_in_runs={sum(n for _, n in runs)}", flush=True)
