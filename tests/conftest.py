import os
import random
import struct
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an AMD GPU (run on an MI355X box)")


def rand_tt(rng: random.Random) -> bytes:
    return struct.pack("<4Q", *(rng.getrandbits(64) for _ in range(4)))


def rand_sparse_tt(rng: random.Random, density_bits: int) -> bytes:
    """Truth table with ~density_bits random set bits (for sparse masks)."""
    words = [0, 0, 0, 0]
    for _ in range(density_bits):
        i = rng.randrange(256)
        words[i // 64] |= 1 << (i % 64)
    return struct.pack("<4Q", *words)


def tt_bit(tt: bytes, i: int) -> int:
    w = struct.unpack("<4Q", tt)
    return (w[i // 64] >> (i % 64)) & 1


@pytest.fixture
def rng():
    return random.Random(0xC0FFEE)
