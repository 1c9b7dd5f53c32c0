"""S-box "model" registry: the bundled tables the reference ships
(/root/reference/sboxes — plain data) plus loaders for user tables.

Bundled: rijndael (AES, 8->8), sodark (8->8), des_s1 (6->4),
crypto1_fa/fb (4->1), crypto1_fc (5->1), identity and linear (8->8 test
vectors). Reference parity: README.md:70-74.
"""

import os
from typing import Tuple

from .. import _core

_SBOX_DIR = os.path.join(os.path.dirname(os.path.dirname(__file__)), "sboxes")

BUNDLED = (
    "crypto1_fa",
    "crypto1_fb",
    "crypto1_fc",
    "des_s1",
    "identity",
    "linear",
    "rijndael",
    "sodark",
)


def sbox_path(name: str) -> str:
    """Path of a bundled S-box table file."""
    if name not in BUNDLED:
        raise KeyError(f"unknown bundled S-box {name!r}; have {BUNDLED}")
    return os.path.join(_SBOX_DIR, name + ".txt")


def load(name_or_path: str, permute: int = 0) -> Tuple[bytes, int]:
    """Loads an S-box (bundled name or file path).

    Returns (sbox, num_inputs): a 256-byte table (tail zeroed for smaller
    S-boxes) and the number of input bits. `permute` XORs the input index
    (the reference's --permute).
    """
    path = sbox_path(name_or_path) if name_or_path in BUNDLED else name_or_path
    return _core.load_sbox_file(path, permute)


def load_table(table, permute: int = 0) -> Tuple[bytes, int]:
    """Loads an S-box from a Python sequence of ints."""
    return _core.load_sbox_table(list(table), permute)
