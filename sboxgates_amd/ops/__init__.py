"""Op-level access to the native engine primitives.

Everything here dispatches into the in-tree native extension; the HIP
kernels are used automatically when a GPU is visible (Options.gpu="auto"),
and `Options.gpu="force"` makes any silent CPU fallback an error — GPU
tests and the bench run with "force" so a missing/broken kernel fails
loudly instead of quietly passing on the CPU path.
"""

from .. import _core
from .._core import (  # noqa: F401
    combination_rank,
    decode_pair,
    function_lists,
    gen_lut_ttable,
    gen_ttable_2,
    generate_target,
    graph_to_dot,
    graph_to_source,
    lut5_solve,
    lut7_ordering,
    lut7_solve,
    make_2_input_fun,
    mask_for_inputs,
    n_choose_k,
    naive_check_n_lut_possible,
    naive_get_lut_function,
    nth_combination,
    splits5,
    tt_eq_mask,
    ttable_to_string,
)


def make_engine(lut_graph=False, seed=None, gpu="auto", oneoutput=-1,
                iterations=1, metric="gates", try_nots=False,
                save_states=False, output_dir="", verbosity=-1,
                gate_bitfield=None, ctx=None, jobs=1):
    """Builds an Engine with the given search options."""
    o = _core.Options()
    o.lut_graph = lut_graph
    if seed is not None:
        o.seeded = True
        o.seed = seed
    o.gpu = gpu
    o.oneoutput = oneoutput
    o.iterations = iterations
    o.jobs = jobs
    o.metric = metric
    o.try_nots = try_nots
    o.save_states = save_states
    o.output_dir = output_dir
    o.verbosity = verbosity
    if gate_bitfield is not None:
        o.set_avail_gates(gate_bitfield)
    o.derive_function_lists()
    return _core.Engine(o, ctx)


def scan(engine, k, state, target, mask, begin, end, seed=0, count_all=False):
    """Runs a 3/5/7-LUT combination scan over [begin, end).

    Returns (found, res[10], evaluated). res layout matches the
    reference's wire format (see sbg/scan.hpp).
    """
    return engine.scan_pool(k, state, target, mask, begin, end, seed, count_all)
