"""Combinatorics vs itertools."""

import itertools
import math

from sboxgates_amd.ops import (combination_rank, decode_pair, n_choose_k,
                               nth_combination)


def test_n_choose_k():
    for n in (1, 2, 7, 20, 100, 500):
        for k in range(8):
            assert n_choose_k(n, k) == math.comb(n, k)


def test_nth_combination_roundtrip():
    n, k = 9, 4
    combos = list(itertools.combinations(range(n), k))
    for rank, want in enumerate(combos):
        got = tuple(nth_combination(rank, n, k))
        assert got == want
        assert combination_rank(list(want), n) == rank


def test_nth_combination_large():
    n, k = 500, 5
    total = math.comb(n, k)
    for rank in (0, 1, total // 3, total - 2, total - 1):
        combo = nth_combination(rank, n, k)
        assert combination_rank(combo, n) == rank
        assert all(combo[i] < combo[i + 1] for i in range(k - 1))


def test_decode_pair():
    for m in (2, 3, 10, 257, 499):
        pairs = list(itertools.combinations(range(m), 2))
        step = max(1, len(pairs) // 200)
        for q in range(0, len(pairs), step):
            assert tuple(decode_pair(q, m)) == pairs[q]
