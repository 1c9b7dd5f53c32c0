"""Cell-algebra LUT solver vs reference-style brute force.

The 2-coloring decomposition solver (sbg/lutcover.hpp) replaces the
reference's 10x256 (5-LUT) / 70x256x256 (7-LUT) function brute force; this
file proves solution-space equivalence on random instances and validity of
every solution produced.
"""

import random

from conftest import rand_sparse_tt, rand_tt, tt_bit

from sboxgates_amd import _core
from sboxgates_amd.ops import (gen_lut_ttable, lut5_solve, lut7_solve,
                               naive_check_n_lut_possible,
                               naive_get_lut_function, splits5, tt_eq_mask)

FULL = b"\xff" * 32


def brute_5lut(tables, target, mask):
    """Reference-semantics brute force: 10 splits x 256 outer functions,
    inner derived by constraint propagation (lut.c:174-246)."""
    for sp in splits5():
        t0, t1, t2 = tables[sp[0]], tables[sp[1]], tables[sp[2]]
        t3, t4 = tables[sp[3]], tables[sp[4]]
        for fo in range(256):
            t_outer = gen_lut_ttable(fo, t0, t1, t2)
            ok, fi = naive_get_lut_function(t_outer, t3, t4, target, mask)
            if ok:
                t_inner = gen_lut_ttable(fi, t_outer, t3, t4)
                if tt_eq_mask(target, t_inner, mask):
                    return True
    return False


def check_5_solution(tables, target, mask, fo, fi, split):
    sp = splits5()[split]
    t_outer = gen_lut_ttable(fo, tables[sp[0]], tables[sp[1]], tables[sp[2]])
    t_inner = gen_lut_ttable(fi, t_outer, tables[sp[3]], tables[sp[4]])
    return tt_eq_mask(target, t_inner, mask)


def test_lut5_constructed_decomposable():
    """Targets built as LUT(LUT(a,b,c),d,e) must always be solved, and the
    returned functions must verify."""
    rng = random.Random(11)
    for trial in range(40):
        tables = [rand_tt(rng) for _ in range(5)]
        fo = rng.randrange(256)
        fi = rng.randrange(256)
        t_outer = gen_lut_ttable(fo, tables[0], tables[1], tables[2])
        target = gen_lut_ttable(fi, t_outer, tables[3], tables[4])
        found, gfo, gfi, split = lut5_solve(tables, target, FULL, rng.getrandbits(64))
        assert found, trial
        assert check_5_solution(tables, target, FULL, gfo, gfi, split)


def test_lut5_equivalence_random_sparse_mask():
    """On random instances with sparse masks, solver existence must equal
    the reference brute force, and solutions must verify."""
    rng = random.Random(12)
    found_cnt = 0
    for trial in range(60):
        tables = [rand_tt(rng) for _ in range(5)]
        target = rand_tt(rng)
        mask = rand_sparse_tt(rng, rng.choice([4, 8, 16, 32]))
        found, fo, fi, split = lut5_solve(tables, target, mask, rng.getrandbits(64))
        want = brute_5lut(tables, target, mask)
        assert found == want, trial
        if found:
            found_cnt += 1
            assert check_5_solution(tables, target, mask, fo, fi, split)
    assert found_cnt > 5  # the case mix must actually exercise both sides


def test_lut5_infeasible_dense():
    """A random dense target over a full mask is essentially never a 5-LUT
    composition of random tables."""
    rng = random.Random(13)
    for _ in range(10):
        tables = [rand_tt(rng) for _ in range(5)]
        target = rand_tt(rng)
        found, *_ = lut5_solve(tables, target, FULL, 1)
        assert not found


def test_lut7_constructed_decomposable():
    rng = random.Random(14)
    for trial in range(15):
        tables = [rand_tt(rng) for _ in range(7)]
        fo, fm, fi = (rng.randrange(256) for _ in range(3))
        t_outer = gen_lut_ttable(fo, tables[0], tables[1], tables[2])
        t_middle = gen_lut_ttable(fm, tables[3], tables[4], tables[5])
        target = gen_lut_ttable(fi, t_outer, t_middle, tables[6])
        found, sol = lut7_solve(tables, target, FULL, rng.getrandbits(64))
        assert found, trial
        gfo, gfm, gfi = sol[0], sol[1], sol[2]
        ordv = sol[3:]
        t_o = gen_lut_ttable(gfo, tables[ordv[0]], tables[ordv[1]], tables[ordv[2]])
        t_m = gen_lut_ttable(gfm, tables[ordv[3]], tables[ordv[4]], tables[ordv[5]])
        t_i = gen_lut_ttable(gfi, t_o, t_m, tables[ordv[6]])
        assert tt_eq_mask(target, t_i, FULL)


def test_lut7_sparse_solutions_verify():
    rng = random.Random(15)
    found_cnt = 0
    for trial in range(25):
        tables = [rand_tt(rng) for _ in range(7)]
        target = rand_tt(rng)
        mask = rand_sparse_tt(rng, rng.choice([4, 8, 12]))
        found, sol = lut7_solve(tables, target, mask, rng.getrandbits(64))
        if found:
            found_cnt += 1
            gfo, gfm, gfi = sol[0], sol[1], sol[2]
            ordv = sol[3:]
            t_o = gen_lut_ttable(gfo, tables[ordv[0]], tables[ordv[1]], tables[ordv[2]])
            t_m = gen_lut_ttable(gfm, tables[ordv[3]], tables[ordv[4]], tables[ordv[5]])
            t_i = gen_lut_ttable(gfi, t_o, t_m, tables[ordv[6]])
            assert tt_eq_mask(target, t_i, mask)
    assert found_cnt > 3


def test_lut7_orderings_table():
    seen = set()
    for i in range(70):
        ordv = tuple(_core.lut7_ordering(i))
        outer, middle, g = ordv[:3], ordv[3:6], ordv[6]
        assert sorted(set(ordv)) == list(range(7))
        assert list(outer) == sorted(outer)
        assert list(middle) == sorted(middle)
        assert outer[0] < middle[0]
        key = (frozenset([outer, middle]), g)
        assert key not in seen
        seen.add(key)
    assert len(seen) == 70


def test_naive_check_matches_derivation():
    """naive_check_n_lut_possible(3) must agree with the existence of a
    derived function."""
    rng = random.Random(16)
    for _ in range(100):
        tables = [rand_tt(rng) for _ in range(3)]
        target = rand_tt(rng)
        mask = rand_sparse_tt(rng, rng.choice([2, 4, 8, 64]))
        possible = naive_check_n_lut_possible(3, target, mask, tables)
        ok, fun = naive_get_lut_function(tables[0], tables[1], tables[2], target, mask)
        assert possible == ok
        if ok:
            t = gen_lut_ttable(fun, *tables)
            assert tt_eq_mask(target, t, mask)


def test_no_zero_function_bytes_emitted():
    """Function byte 00 does not round-trip through the gates.xsd loader
    (the reference's validator rejects it); solvers must never emit it.
    Constructed degenerate targets force the all-zero corner."""
    rng = random.Random(99)
    zero = b"\x00" * 32
    ones = b"\xff" * 32
    for trial in range(200):
        tables = [rand_tt(rng) for _ in range(5)]
        # Degenerate targets: constant 0 / constant 1 / sparse masks.
        target = rng.choice([zero, ones, rand_tt(rng)])
        mask = rng.choice([ones, rand_sparse_tt(rng, 4)])
        found, fo, fi, split = lut5_solve(tables, target, mask, rng.getrandbits(64))
        if found:
            assert fo != 0 and fi != 0, trial
        t7 = [rand_tt(rng) for _ in range(7)]
        found7, sol = lut7_solve(t7, target, mask, rng.getrandbits(64))
        if found7:
            assert sol[0] != 0 and sol[1] != 0 and sol[2] != 0, trial
