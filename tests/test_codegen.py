"""Codegen: emitted C compiles and computes the S-box (bitsliced); DOT is
well-formed; CUDA/HIP text contains the expected constructs.

The compile-and-run check is the reference CI's correctness oracle
(.travis.yml:44-51) made stronger: the compiled bitsliced function is
actually executed against the S-box table via ctypes.
"""

import ctypes
import os
import re
import shutil
import subprocess

import pytest

from sboxgates_amd import _core, models
from sboxgates_amd.ops import make_engine, mask_for_inputs


def build_circuit(name, bit, lut=False, seed=31):
    sbox, n = models.load(name)
    eng = make_engine(lut_graph=lut, seed=seed, gpu="off", save_states=False)
    eng.set_sbox(sbox, n)
    st = eng.initial_state()
    out = eng.create_circuit(st, eng.target(bit), mask_for_inputs(n))
    assert out >= 0
    st.set_output(bit, out)
    return st, sbox, n


def compile_and_eval_c(src, st, sbox, n, bit):
    """Compiles emitted C to a shared lib and evaluates it bitsliced."""
    gcc = shutil.which("gcc") or shutil.which("cc")
    if gcc is None:
        pytest.skip("no C compiler")
    import tempfile
    with tempfile.TemporaryDirectory() as d:
        cpath = os.path.join(d, "sbox.c")
        # Wrap: the emitted function is `bit_t sN(bits in)`.
        with open(cpath, "w") as f:
            f.write(src)
            f.write(f"""
unsigned long long drive(unsigned long long* ins) {{
  bits b;
""")
            for i in range(n):
                f.write(f"  b.b{i} = ins[{i}];\n")
            f.write(f"  return s{bit}(b);\n}}\n")
        so = os.path.join(d, "sbox.so")
        subprocess.run([gcc, "-O2", "-shared", "-fPIC", "-Wall", "-Werror",
                        cpath, "-o", so], check=True)
        lib = ctypes.CDLL(so)
        lib.drive.restype = ctypes.c_ulonglong
        lib.drive.argtypes = [ctypes.POINTER(ctypes.c_ulonglong)]
        # Bitslice the first 64 input patterns into one word per input bit.
        ins = (ctypes.c_ulonglong * n)()
        for i in range(n):
            w = 0
            for x in range(64):
                w |= ((x >> i) & 1) << x
            ins[i] = w
        out = lib.drive(ins)
        # Only the S-box's 2^n real input patterns are specified; beyond
        # them the circuit output is don't-care (mask semantics).
        for x in range(min(64, 1 << n)):
            assert (out >> x) & 1 == (sbox[x] >> bit) & 1, x


def test_c_output_compiles_and_computes():
    st, sbox, n = build_circuit("des_s1", 0)
    src = _core.graph_to_source(st, "auto")
    assert "typedef unsigned long long int bit_t;" in src
    assert "LUT(" not in src
    compile_and_eval_c(src, st, sbox, n, 0)


def test_c_output_multi_output():
    """Full multi-output graph converts; every output present."""
    sbox, n = models.load("des_s1")
    eng = make_engine(seed=77, gpu="off", save_states=False)
    eng.set_sbox(sbox, n)
    eng.generate_graph(eng.initial_state())
    # generate_graph saves nothing here (save_states False) — rebuild two
    # outputs by hand instead.
    st = eng.initial_state()
    o0 = eng.create_circuit(st, eng.target(0), mask_for_inputs(n))
    st.set_output(0, o0)
    o1 = eng.create_circuit(st, eng.target(1), mask_for_inputs(n))
    st.set_output(1, o1)
    src = _core.graph_to_source(st, "auto")
    assert "void s(bits in, bit_t *out0, bit_t *out1)" in src


def test_cuda_output_for_lut_graph():
    st, sbox, n = build_circuit("crypto1_fa", 0, lut=True)
    src = _core.graph_to_source(st, "auto")
    assert "lop3.b32" in src
    assert "__device__ __forceinline__" in src


def test_hip_output_compiles():
    st, sbox, n = build_circuit("crypto1_fa", 0, lut=True)
    src = _core.graph_to_source(st, "hip")
    assert "lop3" not in src
    # Evaluate the HIP emission on the host: bit_t is u64, LUT gates are
    # plain bitwise expressions, so it compiles as C too.
    compile_and_eval_c(src.replace("__device__ __forceinline__ ", ""), st,
                       sbox, n, 0)


def test_dot_output():
    st, _, _ = build_circuit("des_s1", 0)
    dot = _core.graph_to_dot(st)
    assert dot.startswith("digraph sbox {")
    assert dot.rstrip().endswith("}")
    assert "-> out0;" in dot
    # Every gate node declared.
    for i in range(st.num_gates):
        assert f"gt{i} " in dot


def test_ttable_to_string():
    t = _core.generate_target(0, None)
    s = _core.ttable_to_string(t)
    lines = s.strip().split("\n")
    assert len(lines) == 16
    assert lines[0] == "0101010101010101"


def test_no_outputs_rejected():
    st = _core.State(3)
    st.add_gate(6, 0, 1)
    with pytest.raises(RuntimeError):
        _core.graph_to_source(st, "auto")


def test_c_request_on_lut_graph_promotes_to_cuda():
    st, sbox, n = build_circuit("crypto1_fa", 0, lut=True)
    src = _core.graph_to_source(st, "c")
    assert "lop3.b32" in src  # the reference's rule: LUT graphs emit CUDA
