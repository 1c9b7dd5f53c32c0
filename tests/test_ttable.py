"""Truth-table substrate vs pure-Python oracles."""

import random
import struct

from conftest import rand_tt, tt_bit

from sboxgates_amd import _core
from sboxgates_amd import models


def test_generate_target_input_bits():
    # Input-bit tables: bit i of table for input bit b = (i >> b) & 1.
    for b in range(8):
        t = _core.generate_target(b, None)
        for i in range(256):
            assert tt_bit(t, i) == (i >> b) & 1


def test_generate_target_sbox():
    sbox, n = models.load("rijndael")
    assert n == 8
    for b in range(8):
        t = _core.generate_target(b, sbox)
        for i in range(256):
            assert tt_bit(t, i) == (sbox[i] >> b) & 1


def test_mask_for_inputs():
    for n in range(1, 9):
        m = _core.mask_for_inputs(n)
        for i in range(256):
            assert tt_bit(m, i) == (1 if i < (1 << n) else 0), (n, i)


def test_eq_mask():
    rng = random.Random(1)
    for _ in range(50):
        a = rand_tt(rng)
        b = rand_tt(rng)
        mask = rand_tt(rng)
        want = all(
            (tt_bit(a, i) == tt_bit(b, i)) or not tt_bit(mask, i) for i in range(256))
        assert _core.tt_eq_mask(a, b, mask) == want


def test_gen_ttable_2_all_functions():
    rng = random.Random(2)
    a, b = rand_tt(rng), rand_tt(rng)
    for fun in range(16):
        t = _core.gen_ttable_2(fun, a, b)
        for i in range(0, 256, 7):
            # reversed 4-bit encoding: value at pattern p is fun bit (3-p)
            p = (tt_bit(a, i) << 1) | tt_bit(b, i)
            assert tt_bit(t, i) == (fun >> (3 - p)) & 1, (fun, i)


def test_gen_lut_ttable():
    rng = random.Random(3)
    a, b, c = rand_tt(rng), rand_tt(rng), rand_tt(rng)
    for fun in (0x00, 0xFF, 0xAC, 0x96, 0xE8, 0x17):
        t = _core.gen_lut_ttable(fun, a, b, c)
        for i in range(256):
            p = (tt_bit(a, i) << 2) | (tt_bit(b, i) << 1) | tt_bit(c, i)
            assert tt_bit(t, i) == (fun >> p) & 1


def test_permute_load():
    sbox, n = models.load("rijndael", permute=0x63)
    plain, _ = models.load("rijndael")
    for i in range(256):
        assert sbox[i] == plain[i ^ 0x63]
