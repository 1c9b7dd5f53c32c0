"""Artifact CI: every tracked XML in results/ must validate under exactly the
(sbox, permute, output bits, gate count) that results/MANIFEST.json claims
for it.

This is the integrity check for published claims: a mislabeled artifact (the
round-1 permute-sweep table attributed files to the wrong --permute values)
fails here. Validation is by ground-truth DAG evaluation on every input
pattern — independent of the cached truth tables the engine maintains.
"""

import glob
import json
import os

import pytest

from sboxgates_amd import _core

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
RESULTS = os.path.join(REPO, "results")
SBOXES = os.path.join(REPO, "sboxgates_amd", "sboxes")


def _load_manifest():
    with open(os.path.join(RESULTS, "MANIFEST.json")) as f:
        return json.load(f)


def _load_sbox(name):
    with open(os.path.join(SBOXES, name + ".txt")) as f:
        vals = [int(t, 16) for t in f.read().split()]
    return vals, len(vals).bit_length() - 1


def test_every_tracked_xml_is_manifested():
    # Hunt scratch dirs are not published claims; everything else is.
    tracked = {
        os.path.relpath(p, RESULTS)
        for p in glob.glob(os.path.join(RESULTS, "**", "*.xml"), recursive=True)
        if "des_hunt" not in p and "scratch" not in p
    }
    manifested = {e["file"] for e in _load_manifest()}
    assert tracked == manifested, (
        "untracked-in-manifest: %s; manifest-missing-file: %s"
        % (sorted(tracked - manifested), sorted(manifested - tracked))
    )


@pytest.mark.parametrize("entry", _load_manifest(), ids=lambda e: e["file"])
def test_artifact_validates_under_claimed_config(entry):
    path = os.path.join(RESULTS, entry["file"])
    st = _core.State.load(path)
    vals, num_inputs = _load_sbox(entry["sbox"])
    assert st.num_inputs == num_inputs
    assert st.num_gates - num_inputs == entry["gates"], "gate count mismatch"

    claimed = set(entry["outputs"])
    present = {b for b in range(8) if st.outputs[b] >= 0}
    assert present == claimed, "output slots mismatch"

    p = entry["permute"]
    for x in range(len(vals)):
        got = st.eval(x)
        want = vals[x ^ p]
        for b in claimed:
            assert ((got >> b) & 1) == ((want >> b) & 1), (
                "artifact %s wrong at input %d bit %d under permute %d"
                % (entry["file"], x, b, p)
            )
