"""Miscellaneous robustness: example script, XML-parser fuzzing."""

import os
import random
import subprocess
import sys

import pytest

from sboxgates_amd import _core

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_example_script_runs(tmp_path):
    r = subprocess.run(
        [sys.executable, os.path.join(REPO, "examples", "find_aes_circuit.py"), "0"],
        capture_output=True, text=True, timeout=240, cwd=str(tmp_path))
    assert r.returncode == 0, r.stderr
    assert "gates" in r.stdout


def test_xml_parser_fuzz_no_crash():
    """Random garbage and mutated documents must raise clean errors (or
    parse), never crash."""
    rng = random.Random(7)
    base = _core.State(3)
    base.add_gate(6, 0, 1)
    base.set_output(0, 3)
    good = base.to_xml()
    for trial in range(300):
        kind = rng.randrange(3)
        if kind == 0:
            doc = bytes(rng.getrandbits(8) for _ in range(rng.randrange(200)))
            doc = doc.decode("latin1")
        elif kind == 1:
            # Mutate the good document.
            pos = rng.randrange(len(good))
            doc = good[:pos] + rng.choice("<>&\"'/x0") + good[pos + 1:]
        else:
            # Truncate.
            doc = good[:rng.randrange(len(good))]
        try:
            st = _core.State.from_xml(doc)
            # If it parsed, it must be a consistent state.
            assert 0 <= st.num_gates <= 500
        except RuntimeError:
            pass


def test_xml_parser_entities_and_comments():
    xml = """<?xml version="1.0"?>
<!-- comment -->
<gates>
  <output bit="0" gate="1" />
  <gate type="IN" /><!-- inline -->
  <gate type='NOT'><input gate='0'/></gate>
</gates>"""
    st = _core.State.from_xml(xml)
    assert st.num_gates == 2
    assert st.gate(1)["type_name"] == "NOT"


def test_load_sbox_rejects_bad_sizes(tmp_path):
    p = tmp_path / "bad.txt"
    p.write_text("1 2 3")  # not a power of two
    with pytest.raises(RuntimeError):
        _core.load_sbox_file(str(p))
