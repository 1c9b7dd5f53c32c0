"""Circuit state + XML persistence (gates.xsd format)."""

import os
import re
import xml.etree.ElementTree as ET

import pytest

from sboxgates_amd import _core
from sboxgates_amd.ops import make_engine, mask_for_inputs
from sboxgates_amd import models

GATE_NAMES = {"FALSE", "AND", "A_AND_NOT_B", "A", "NOT_A_AND_B", "B", "XOR",
              "OR", "NOR", "XNOR", "NOT_B", "A_OR_NOT_B", "NOT_A",
              "NOT_A_OR_B", "NAND", "TRUE", "NOT", "IN", "LUT"}


def small_state():
    st = _core.State(4)
    g = st.add_gate(6, 0, 1)          # XOR
    g2 = st.add_gate(1, g, 2)         # AND
    g3 = st.add_lut(0xAC, 3, g, g2)   # LUT
    st.set_output(0, g3)
    st.set_output(2, g2)
    return st


def test_xml_roundtrip_exact():
    st = small_state()
    xml = st.to_xml()
    st2 = _core.State.from_xml(xml)
    assert st2.to_xml() == xml
    assert st2.num_gates == st.num_gates
    assert st2.outputs == st.outputs
    for i in range(st.num_gates):
        assert st.gate(i) == st2.gate(i)


def test_xml_schema_conformance():
    """The emitted document must parse and respect gates.xsd constraints."""
    st = small_state()
    root = ET.fromstring(st.to_xml())
    assert root.tag == "gates"
    for el in root:
        assert el.tag in ("output", "gate")
        if el.tag == "output":
            assert 0 <= int(el.get("bit")) < 8
            assert 0 <= int(el.get("gate")) < 500
        else:
            assert el.get("type") in GATE_NAMES
            fn = el.get("function")
            if fn is not None:
                assert re.fullmatch(r"[0-9a-f]{2}", fn)
            assert len(el.findall("input")) <= 3


def test_reference_format_loads():
    """A hand-written file in the exact reference text format must load."""
    xml = """<?xml version="1.0" encoding="UTF-8" ?>
<gates>
  <output bit="0" gate="5" />
  <gate type="IN" />
  <gate type="IN" />
  <gate type="IN" />
  <gate type="XOR">
    <input gate="0" />
    <input gate="1" />
  </gate>
  <gate type="NOT">
    <input gate="3" />
  </gate>
  <gate type="LUT" function="ac">
    <input gate="2" />
    <input gate="3" />
    <input gate="4" />
  </gate>
</gates>
"""
    st = _core.State.from_xml(xml)
    assert st.num_gates == 6
    assert st.num_inputs == 3
    assert st.outputs[0] == 5
    # LUT 0xac = mux(sel=in2, a=g3, b=g4): check by evaluation.
    st2 = st
    for x in range(8):
        a, b, c = x & 1, (x >> 1) & 1, (x >> 2) & 1
        xor_v = a ^ b
        not_v = 1 - xor_v
        want = not_v if c else xor_v
        assert st2.eval(x) & 1 == want


@pytest.mark.parametrize("bad", [
    # input references a later gate
    '<gates><output bit="0" gate="0" /><gate type="NOT"><input gate="1" /></gate>'
    '<gate type="IN" /></gates>',
    # IN after non-IN
    '<gates><output bit="0" gate="0" /><gate type="IN" /><gate type="NOT">'
    '<input gate="0" /></gate><gate type="IN" /></gates>',
    # arity mismatch
    '<gates><output bit="0" gate="1" /><gate type="IN" /><gate type="AND">'
    '<input gate="0" /></gate></gates>',
    # function on non-LUT
    '<gates><output bit="0" gate="1" /><gate type="IN" /><gate type="NOT" '
    'function="12"><input gate="0" /></gate></gates>',
    # duplicate output bit
    '<gates><output bit="0" gate="0" /><output bit="0" gate="0" />'
    '<gate type="IN" /></gates>',
    # output references missing gate
    '<gates><output bit="0" gate="7" /><gate type="IN" /></gates>',
    # >8 IN gates
    '<gates><output bit="0" gate="0" />' + '<gate type="IN" />' * 9 + '</gates>',
    # malformed XML
    '<gates><gate type="IN" />',
])
def test_invalid_documents_rejected(bad):
    with pytest.raises(RuntimeError):
        _core.State.from_xml(bad)


def test_tables_recomputed_on_load():
    st = small_state()
    st2 = _core.State.from_xml(st.to_xml())
    for i in range(st.num_gates):
        assert st.gate(i)["table"] == st2.gate(i)["table"]


def test_file_name_format():
    st = small_state()
    name = st.file_name()
    # O-GGG-MMMM-NNNN-FFFFFFFF.xml; NNNN lists output bits in order of gate
    # inclusion: output 2 (the AND gate) was added before output 0 (the LUT).
    assert re.fullmatch(r"2-003-00\d\d-20-[0-9a-f]{8}\.xml", name), name


def test_save_and_load(tmp_path):
    st = small_state()
    path = st.save(str(tmp_path))
    assert os.path.exists(path)
    st2 = _core.State.load(path)
    assert st2.to_xml() == st.to_xml()


def test_sat_metric_accumulates():
    st = _core.State(2)
    st.add_gate(6, 0, 1)   # XOR: 12
    st.add_gate(1, 0, 1)   # AND: 7
    assert st.sat_metric == 19


def test_resume_checkpoint_search(tmp_path):
    """Search -> save -> load -> resume parity (the reference's -g flow)."""
    sbox, n = models.load("des_s1")
    eng = make_engine(seed=5, gpu="off", save_states=True, output_dir=str(tmp_path),
                      oneoutput=0)
    eng.set_sbox(sbox, n)
    st = eng.initial_state()
    eng.generate_graph_one_output(st)
    files = eng.saved_files()
    assert files
    st2 = _core.State.load(files[0])
    assert st2.outputs[0] >= 0
    # Loaded state evaluates correctly.
    for x in range(64):
        assert (st2.eval(x) & 1) == (sbox[x] & 1)
