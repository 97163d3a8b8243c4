#!/usr/bin/env python3
"""Dev-time simulator of the banded blocked-Myers kernel in
src/hip/aligner_kernel.hip — validates the recurrence, band slide, score
reconstruction and traceback logic against brute-force DP on random pairs.
Run: python tools/sim_myers.py [K] [trials]
"""

import random
import sys

M64 = (1 << 64) - 1


def base_code(ch):
    return {"A": 0, "C": 1, "G": 2, "T": 3}.get(ch, 4)


def brute_edit(q, t):
    n, m = len(q), len(t)
    prev = list(range(m + 1))
    for i in range(1, n + 1):
        cur = [i] + [0] * m
        for j in range(1, m + 1):
            cur[j] = min(prev[j - 1] + (q[i - 1] != t[j - 1]), prev[j] + 1, cur[j - 1] + 1)
        prev = cur
    return prev[m]


def btop_of(j, n, m, nbt, K):
    step = (n << 32) // m  # 32.32 fixed point, as in the kernel
    center_blk = (((j * step) >> 32) & 0xFFFFFFFF) >> 6
    top = center_blk - K // 2
    hi = nbt - K if nbt > K else 0
    return max(0, min(top, hi))


def score_at(S, Pv, Mv, k):
    mask = 0 if k == 63 else (M64 << (k + 1)) & M64
    return S - (bin(Pv & mask).count("1") - bin(Mv & mask).count("1"))


def align(q, t, K):
    n, m = len(q), len(t)
    nbt = (n + 63) >> 6
    # peq
    nb = max(K, nbt)
    peq = [[0] * 4 for _ in range(nb)]
    for i, ch in enumerate(q):
        c = base_code(ch)
        if c < 4:
            peq[i >> 6][c] |= 1 << (i & 63)

    Pv = [M64] * K
    Mv = [0] * K
    S = [(b + 1) * 64 for b in range(K)]
    btop = 0
    tb = {}  # (j, rb) -> (Pv, Mv, S)
    for b in range(K):
        tb[(0, b)] = (Pv[b], Mv[b], S[b])

    for j in range(1, m + 1):
        c = base_code(t[j - 1])
        btn = btop_of(j, n, m, nbt, K)
        while btop < btn:
            for b in range(K - 1):
                Pv[b], Mv[b], S[b] = Pv[b + 1], Mv[b + 1], S[b + 1]
            Pv[K - 1], Mv[K - 1], S[K - 1] = M64, 0, S[K - 2] + 64
            btop += 1
        hin = 1
        for b in range(K):
            Eq = peq[btop + b][c] if (c < 4 and btop + b < nb) else 0
            hin_neg = 1 if hin < 0 else 0
            hin_pos = 1 if hin > 0 else 0
            Xv = Eq | Mv[b]
            Eq |= hin_neg
            Xh = ((((Eq & Pv[b]) + Pv[b]) & M64) ^ Pv[b]) | Eq
            Ph = Mv[b] | (~(Xh | Pv[b]) & M64)
            Mh = Pv[b] & Xh
            hout = ((Ph >> 63) & 1) - ((Mh >> 63) & 1)
            Ph = ((Ph << 1) | hin_pos) & M64
            Mh = ((Mh << 1) | hin_neg) & M64
            Pv[b] = Mh | (~(Xv | Ph) & M64)
            Mv[b] = Ph & Xv
            S[b] += hout
            hin = hout
            tb[(j, b)] = (Pv[b], Mv[b], S[b])

    rb = ((n - 1) >> 6) - btop
    if rb < 0 or rb >= K:
        return None, None, "bandedge-final"
    D = score_at(S[rb], Pv[rb], Mv[rb], (n - 1) & 63)
    ed = D

    # traceback
    path = []
    i, j = n, m
    while i > 0 and j > 0:
        babs = (i - 1) >> 6
        k = (i - 1) & 63
        btj = btop_of(j, n, m, nbt, K)
        btj1 = btop_of(j - 1, n, m, nbt, K)
        rbj = babs - btj
        rbj1 = babs - btj1
        if not (0 <= rbj < K and 0 <= rbj1 < K):
            return ed, None, "bandedge-tb"
        Pvj, Mvj, _ = tb[(j, rbj)]
        Pvj1, Mvj1, S1 = tb[(j - 1, rbj1)]
        vd = 1 if (Pvj >> k) & 1 else (-1 if (Mvj >> k) & 1 else 0)
        D_left = score_at(S1, Pvj1, Mvj1, k)
        vd1 = 1 if (Pvj1 >> k) & 1 else (-1 if (Mvj1 >> k) & 1 else 0)
        D_diag = D_left - vd1
        sub = 0 if (base_code(q[i - 1]) == base_code(t[j - 1]) and base_code(q[i - 1]) < 4) else 1
        if D_diag + sub == D:
            path.append("M")
            i -= 1
            j -= 1
            D = D_diag
        elif D_left + 1 == D:
            path.append("D")
            j -= 1
            D = D_left
        elif vd == 1:
            path.append("I")
            i -= 1
            D = D - vd
        else:
            return ed, None, "stuck"
    path.extend("I" * i)
    path.extend("D" * j)
    return ed, "".join(reversed(path)), "ok"


def check_path(q, t, path, ed):
    qi = ti = cost = 0
    for op in path:
        if op == "M":
            cost += q[qi] != t[ti]
            qi += 1
            ti += 1
        elif op == "I":
            cost += 1
            qi += 1
        else:
            cost += 1
            ti += 1
    assert qi == len(q) and ti == len(t), (qi, len(q), ti, len(t))
    assert cost == ed, (cost, ed)


def mutate(t, rng, rate):
    out = []
    for ch in t:
        r = rng.random()
        if r < rate / 3:
            continue
        if r < 2 * rate / 3:
            out.append(rng.choice("ACGT"))
        if r < rate:
            out.append(rng.choice([c for c in "ACGT" if c != ch]))
        else:
            out.append(ch)
    return "".join(out)


def main():
    K = int(sys.argv[1]) if len(sys.argv) > 1 else 4
    trials = int(sys.argv[2]) if len(sys.argv) > 2 else 60
    rng = random.Random(1)
    fails = edges = 0
    for trial in range(trials):
        m = rng.randint(3, 700)
        t = "".join(rng.choice("ACGT") for _ in range(m))
        if trial % 4 == 0:
            q = "".join(rng.choice("ACGTN") for _ in range(rng.randint(3, 700)))
        else:
            q = mutate(t, rng, rng.choice([0.02, 0.06, 0.15]))
        if not q:
            continue
        ref = brute_edit(q, t)
        ed, path, st = align(q, t, K)
        if st != "ok":
            edges += 1
            continue
        if ed != ref:
            # band may truncate the optimum for wildly divergent pairs:
            # only equal-or-worse is acceptable, and only when divergent
            if ed < ref or ref <= 64 * K // 2 - 64:
                print(f"FAIL trial {trial}: ed={ed} ref={ref} n={len(q)} m={m}")
                fails += 1
                continue
        check_path(q, t, path, ed)
    print(f"K={K}: {trials} trials, {fails} fails, {edges} band-edge")
    return 1 if fails else 0


if __name__ == "__main__":
    sys.exit(main())
