"""Pairwise aligner (edlib-equivalent) unit tests: exact NW edit distance and
CIGAR validity against a plain python DP reference."""

import random

import pytest


def py_edit_distance(a, b):
    prev = list(range(len(b) + 1))
    for i, ca in enumerate(a, 1):
        cur = [i] + [0] * len(b)
        for j, cb in enumerate(b, 1):
            cur[j] = min(prev[j] + 1, cur[j - 1] + 1, prev[j - 1] + (ca != cb))
        prev = cur
    return prev[-1]


def parse_cigar(cigar):
    ops, n = [], 0
    for c in cigar:
        if c.isdigit():
            n = n * 10 + int(c)
        else:
            ops.append((n, c))
            n = 0
    return ops


def rand_seq(rng, n):
    return "".join(rng.choice("ACGT") for _ in range(n))


def mutate(rng, s, rate):
    out = []
    for c in s:
        r = rng.random()
        if r < rate / 3:
            continue  # deletion
        if r < 2 * rate / 3:
            out.append(rng.choice("ACGT"))  # insertion
        if r < rate and r >= 2 * rate / 3:
            out.append(rng.choice("ACGT"))  # substitution
        else:
            out.append(c)
    return "".join(out)


@pytest.mark.parametrize("seed", range(5))
def test_edit_distance_matches_python_dp(racon, seed):
    rng = random.Random(seed)
    a = rand_seq(rng, rng.randint(1, 300))
    b = mutate(rng, a, 0.15) or "A"
    assert racon.edit_distance(a, b) == py_edit_distance(a, b)


def test_edit_distance_long_block_boundaries(racon):
    rng = random.Random(42)
    for n in (63, 64, 65, 127, 128, 129, 1000):
        a = rand_seq(rng, n)
        b = mutate(rng, a, 0.1) or "A"
        assert racon.edit_distance(a, b) == py_edit_distance(a, b)


@pytest.mark.parametrize("seed", range(5))
def test_cigar_is_consistent_optimal_path(racon, seed):
    rng = random.Random(100 + seed)
    q = rand_seq(rng, rng.randint(10, 400))
    t = mutate(rng, q, 0.2) or "A"
    cigar = racon.align_cigar(q, t)
    ops = parse_cigar(cigar)
    q_consumed = sum(n for n, op in ops if op in "MI")
    t_consumed = sum(n for n, op in ops if op in "MD")
    assert q_consumed == len(q)
    assert t_consumed == len(t)

    # replay the path and verify its cost equals the true edit distance
    cost, qi, ti = 0, 0, 0
    for n, op in ops:
        if op == "M":
            for _ in range(n):
                cost += q[qi] != t[ti]
                qi += 1
                ti += 1
        elif op == "I":
            cost += n
            qi += n
        else:
            cost += n
            ti += n
    assert cost == py_edit_distance(q, t)


def test_cigar_identical_sequences(racon):
    assert racon.align_cigar("ACGTACGT", "ACGTACGT") == "8M"


def test_cigar_empty_cases(racon):
    assert racon.align_cigar("", "ACG") == "3D"
    assert racon.align_cigar("ACG", "") == "3I"
