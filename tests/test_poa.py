"""POA consensus engine (spoa-equivalent) unit tests."""

import random


def rand_seq(rng, n):
    return "".join(rng.choice("ACGT") for _ in range(n))


def mutate(rng, s, rate):
    out = []
    for c in s:
        r = rng.random()
        if r < rate / 3:
            continue
        if r < 2 * rate / 3:
            out.append(rng.choice("ACGT"))
            out.append(c)
        elif r < rate:
            out.append(rng.choice("ACGT"))
        else:
            out.append(c)
    return "".join(out) or "A"


def test_consensus_of_identical_sequences(racon):
    seqs = ["ACGTTGCAAGGTC"] * 5
    assert racon.poa_consensus(seqs) == seqs[0]


def test_consensus_recovers_truth_from_noisy_copies(racon):
    rng = random.Random(3)
    truth = rand_seq(rng, 400)
    seqs = [mutate(rng, truth, 0.08) for _ in range(12)]
    cons = racon.poa_consensus([seqs[0]] + seqs)
    ed = sum(1 for a, b in zip(cons, truth) if a != b) + abs(len(cons) - len(truth))
    # POA consensus over 12 copies at 8% error should be near-perfect
    assert racon.edit_distance(cons, truth) < 0.01 * len(truth)


def test_consensus_majority_substitution(racon):
    # 1 backbone with an error + 4 reads agreeing on the truth
    backbone = "AAAATAAAA"
    read = "AAAACAAAA"
    cons = racon.poa_consensus([backbone, read, read, read, read])
    assert cons == read


def test_consensus_quality_weighting(racon):
    # low-quality disagreeing reads should lose to high-quality agreement
    backbone = "AAAATAAAA"
    good = "AAAACAAAA"
    seqs = [backbone, good, good, backbone, backbone]
    quals = ["!!!!!!!!!", "IIIIIIIII", "IIIIIIIII", '"""""""""', '"""""""""']
    cons = racon.poa_consensus(seqs, quals)
    assert cons == good
