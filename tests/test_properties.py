"""Property-based tests (hypothesis) over the CPU engines: invariants that
must hold for arbitrary inputs, complementing the exact golden tests."""

import re

from hypothesis import given, settings, strategies as st

DNA = st.text(alphabet="ACGT", min_size=1, max_size=300)


def cigar_spans(cigar):
    q = t = 0
    for num, op in re.findall(r"(\d+)([MID])", cigar):
        n = int(num)
        if op == "M":
            q += n
            t += n
        elif op == "I":
            q += n
        else:
            t += n
    return q, t


@settings(max_examples=200, deadline=None)
@given(DNA, DNA)
def test_cigar_consumes_exact_lengths(racon, q, t):
    cigar = racon.align_cigar(q, t)
    qs, ts = cigar_spans(cigar)
    assert qs == len(q) and ts == len(t)


@settings(max_examples=200, deadline=None)
@given(DNA, DNA)
def test_cigar_cost_equals_edit_distance(racon, q, t):
    """The CIGAR's implied cost must reproduce the optimal edit distance."""
    cigar = racon.align_cigar(q, t)
    ed = racon.edit_distance(q, t)
    qi = ti = cost = 0
    for num, op in re.findall(r"(\d+)([MID])", cigar):
        n = int(num)
        if op == "M":
            cost += sum(1 for k in range(n) if q[qi + k] != t[ti + k])
            qi += n
            ti += n
        elif op == "I":
            cost += n
            qi += n
        else:
            cost += n
            ti += n
    assert cost == ed


@settings(max_examples=100, deadline=None)
@given(DNA, DNA)
def test_edit_distance_symmetry_and_bounds(racon, a, b):
    ed = racon.edit_distance(a, b)
    assert ed == racon.edit_distance(b, a)
    assert abs(len(a) - len(b)) <= ed <= max(len(a), len(b))
    assert racon.edit_distance(a, a) == 0


@settings(max_examples=100, deadline=None)
@given(DNA)
def test_reverse_complement_involution(racon, s):
    assert racon.reverse_complement(racon.reverse_complement(s)) == s


@settings(max_examples=50, deadline=None)
@given(DNA, st.integers(min_value=2, max_value=8))
def test_poa_consensus_of_identical_sequences(racon, s, n):
    """POA of n copies of one sequence must return that sequence."""
    assert racon.poa_consensus([s] * n) == s


@settings(max_examples=50, deadline=None)
@given(st.lists(DNA, min_size=1, max_size=6))
def test_poa_consensus_is_deterministic(racon, seqs):
    a = racon.poa_consensus(seqs)
    b = racon.poa_consensus(seqs)
    assert a == b


@given(st.text(alphabet="ACGT", min_size=0, max_size=400),
       st.text(alphabet="ACGT", min_size=0, max_size=400))
@settings(max_examples=80, deadline=None)
def test_banded_distance_matches_definition(racon, a, b):
    """edit_distance (banded, band-doubling) must equal the textbook DP."""
    import functools
    @functools.lru_cache(maxsize=None)
    def dp(i, j):
        if i == 0:
            return j
        if j == 0:
            return i
        return min(dp(i - 1, j) + 1, dp(i, j - 1) + 1,
                   dp(i - 1, j - 1) + (a[i - 1] != b[j - 1]))
    import sys
    sys.setrecursionlimit(10000)
    assert racon.edit_distance(a, b) == dp(len(a), len(b))


@given(st.integers(min_value=0, max_value=2**31))
@settings(max_examples=20, deadline=None)
def test_banded_distance_skewed_lengths(racon, seed):
    """Very unequal lengths (band dominated by the length gap)."""
    import random
    rng = random.Random(seed)
    a = "".join(rng.choice("ACGT") for _ in range(rng.randint(0, 50)))
    b = a + "".join(rng.choice("ACGT") for _ in range(rng.randint(500, 2000)))
    assert racon.edit_distance(a, b) >= len(b) - len(a) - 2 * len(a)
    # appending to a superstring: distance is exactly the length gap when a
    # is a prefix of b
    assert racon.edit_distance(a, b) <= len(b) - len(a) + 2 * len(a)
    assert racon.edit_distance(b, b) == 0
