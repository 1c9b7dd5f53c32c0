"""Gate-vocabulary construction vs a pure-Python closure oracle."""

import itertools

from sboxgates_amd.ops import function_lists, make_2_input_fun


def val2(fun, a, b):
    """Value of a 2-input function (reversed 4-bit encoding)."""
    return (fun >> (3 - ((a << 1) | b))) & 1


def compose3(f1, f2, na, nb, nc):
    """8-bit table of f2(f1(A^na, B^nb), C^nc), bit p = pattern A<<2|B<<1|C."""
    out = 0
    for p in range(8):
        a, b, c = (p >> 2) & 1 ^ na, (p >> 1) & 1 ^ nb, (p & 1) ^ nc
        out |= val2(f2, val2(f1, a, b), c) << p
    return out


def py_closure_3(gates, try_nots):
    """Oracle: set of reachable 3-input functions."""
    funs = set()
    not_patterns = range(8) if try_nots else [0]
    for nv in not_patterns:
        for f1 in gates:
            for f2 in gates:
                funs.add(compose3(f1, f2, (nv >> 2) & 1, (nv >> 1) & 1, nv & 1))
    if try_nots:
        funs |= {(~f) & 0xFF for f in funs}
    return funs


BITFIELDS = [
    2 + 64 + 128,       # default AND|OR|XOR
    10694,              # the CI-restricted set (.travis.yml:43)
    0xFFFF,             # everything
    2,                  # AND only
]


def test_2_input_commutativity():
    for fun in range(16):
        f = make_2_input_fun(fun)
        want = all(val2(fun, a, b) == val2(fun, b, a)
                   for a, b in itertools.product((0, 1), repeat=2))
        assert f["ab_commutative"] == want


def test_3_input_closure_matches_oracle():
    for bf in BITFIELDS:
        gates = [i for i in range(16) if bf & (1 << i)]
        for try_nots in (False, True):
            _, _, threes = function_lists(bf, try_nots)
            got = {f["fun"] for f in threes}
            assert got == py_closure_3(gates, try_nots), (bf, try_nots)


def test_3_input_entries_realize_their_table():
    _, _, threes = function_lists(0xFFFF, True)
    for f in threes:
        na, nb, nc = int(f["not_a"]), int(f["not_b"]), int(f["not_c"])
        t = compose3(f["fun1"], f["fun2"], na, nb, nc)
        if f["not_out"]:
            t = (~t) & 0xFF
        assert t == f["fun"], f


def test_not_functions():
    gates, nots, _ = function_lists(2 + 64 + 128, True)
    gate_funs = {g["fun"] for g in gates}
    not_funs = {g["fun"] for g in nots}
    # Complements of available gates not already available.
    want = {(~f) & 0xF for f in gate_funs} - gate_funs
    assert not_funs == want
    for g in nots:
        assert g["not_out"]


def test_commutativity_flags_3():
    _, _, threes = function_lists(0xFFFF, True)

    def lut_val(fun, a, b, c):
        return (fun >> ((a << 2) | (b << 1) | c)) & 1

    for f in threes[:64]:
        fun = f["fun"]
        ab = all(lut_val(fun, a, b, c) == lut_val(fun, b, a, c)
                 for a in (0, 1) for b in (0, 1) for c in (0, 1))
        ac = all(lut_val(fun, a, b, c) == lut_val(fun, c, b, a)
                 for a in (0, 1) for b in (0, 1) for c in (0, 1))
        bc = all(lut_val(fun, a, b, c) == lut_val(fun, a, c, b)
                 for a in (0, 1) for b in (0, 1) for c in (0, 1))
        assert f["ab_commutative"] == ab
        assert f["ac_commutative"] == ac
        assert f["bc_commutative"] == bc
