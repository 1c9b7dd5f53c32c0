"""High-level search API.

    from sboxgates_amd.search import find_circuit
    st = find_circuit("des_s1", bit=0)          # gate-mode, one output bit
    st = find_circuit("rijndael", bit=0, lut=True, gpu="force")
    st = find_circuit(my_table, lut=True)       # all outputs, beam search

Returns a State whose outputs are wired; every circuit is validated by
DAG evaluation against the table before being returned.
"""

from typing import Optional, Sequence, Union

from . import _core, models
from .ops import make_engine, mask_for_inputs
from .utils import validate_circuit


def find_circuit(sbox: Union[str, Sequence[int]], bit: Optional[int] = None,
                 lut: bool = False, seed: Optional[int] = None,
                 gpu: str = "auto", iterations: int = 1, metric: str = "gates",
                 try_nots: bool = False, gate_bitfield: Optional[int] = None,
                 permute: int = 0, save_dir: Optional[str] = None,
                 verbosity: int = -1):
    """Finds a gate/LUT circuit for an S-box.

    sbox: bundled name, file path, or a sequence of 2^n ints.
    bit:  output bit to realize (None = all outputs via beam search).
    """
    if isinstance(sbox, str):
        table, n = models.load(sbox, permute)
    else:
        table, n = models.load_table(sbox, permute)

    if bit is not None:
        eng = make_engine(lut_graph=lut, seed=seed, gpu=gpu, oneoutput=bit,
                          iterations=iterations, metric=metric,
                          try_nots=try_nots, gate_bitfield=gate_bitfield,
                          save_states=save_dir is not None,
                          output_dir=save_dir or "", verbosity=verbosity)
        eng.set_sbox(table, n)
        st = eng.initial_state()
        best = None
        for _ in range(iterations):
            trial = st.copy()
            out = eng.create_circuit(trial, eng.target(bit), mask_for_inputs(n))
            if out < 0:
                continue
            trial.set_output(bit, out)
            if best is None or trial.num_gates < best.num_gates:
                best = trial
                st.max_gates = trial.num_gates
        if best is None:
            raise RuntimeError("no circuit found within bounds")
        assert validate_circuit(best, table, n, bit=bit)
        return best

    # All outputs: beam search; the best checkpoint is the result.
    eng2 = make_engine(lut_graph=lut, seed=seed, gpu=gpu,
                       iterations=iterations, metric=metric, try_nots=try_nots,
                       gate_bitfield=gate_bitfield, save_states=True,
                       output_dir=save_dir or ".", verbosity=verbosity)
    eng2.set_sbox(table, n)
    eng2.generate_graph(eng2.initial_state())
    files = eng2.saved_files()
    if not files:
        raise RuntimeError("no circuit found within bounds")
    best = None
    for f in files:
        st = _core.State.load(f)
        wired = sum(1 for b in range(8) if st.outputs[b] >= 0)
        if wired < eng2.num_outputs:
            continue
        if best is None or st.num_gates < best.num_gates:
            best = st
    assert best is not None
    assert validate_circuit(best, table, n)
    return best
