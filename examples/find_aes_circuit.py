#!/usr/bin/env python3
"""Example: find a 3-LUT circuit for one output bit of the AES S-box and
emit C, CUDA, HIP and DOT renderings.

    python examples/find_aes_circuit.py [bit]

On an MI355X this takes ~15 s per bit through the gfx950 kernels; on CPU
it is a long run — use des_s1 instead for a quick CPU demo.
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from sboxgates_amd import _core  # noqa: E402
from sboxgates_amd.search import find_circuit  # noqa: E402

bit = int(sys.argv[1]) if len(sys.argv) > 1 else 0
name = "rijndael" if _core.gpu_available() else "des_s1"

st = find_circuit(name, bit=bit, lut=True, seed=1)
print(f"{name} bit {bit}: {st.num_gates - st.num_inputs} LUT-graph gates")
print("XML checkpoint name:", st.file_name())
print("\n--- CUDA ---")
print(_core.graph_to_source(st, "cuda"))
print("--- HIP ---")
print(_core.graph_to_source(st, "hip"))
