"""End-to-end search engine tests (CPU path). Every produced circuit is
validated by evaluating the DAG on all input patterns — a stronger oracle
than the reference's truth-table asserts."""

import pytest

from sboxgates_amd import _core, models
from sboxgates_amd.ops import make_engine, mask_for_inputs, n_choose_k
from sboxgates_amd.utils import validate_circuit


def run_one_bit(name, bit, lut=False, seed=1, **kw):
    sbox, n = models.load(name)
    eng = make_engine(lut_graph=lut, seed=seed, gpu="off", save_states=False, **kw)
    eng.set_sbox(sbox, n)
    st = eng.initial_state()
    out = eng.create_circuit(st, eng.target(bit), mask_for_inputs(n))
    assert out >= 0
    st.set_output(bit, out)
    assert validate_circuit(st, sbox, n, bit=bit)
    return st


def test_des_s1_bit0_gate_mode():
    st = run_one_bit("des_s1", 0)
    gates = st.num_gates - st.num_inputs
    # The bundled reference artifact for this bit is 19 gates
    # (des_s1_bit0.svg); randomized greedy search lands nearby.
    assert gates <= 30


def test_des_s1_bit0_best_of_iterations():
    """Iterative deepening across seeds reaches the reference's 19-gate
    quality bar."""
    best = min(run_one_bit("des_s1", 0, seed=s).num_gates - 6 for s in range(8))
    assert best <= 21


def test_des_s1_all_bits():
    for bit in range(4):
        run_one_bit("des_s1", bit, seed=3 + bit)


def test_des_s1_lut_mode():
    st = run_one_bit("des_s1", 0, lut=True)
    gates = st.num_gates - st.num_inputs
    assert gates <= 15  # LUT graphs are much smaller
    assert any(st.gate(i)["type_name"] == "LUT" for i in range(6, st.num_gates))


def test_crypto1_functions():
    for name in ("crypto1_fa", "crypto1_fb", "crypto1_fc"):
        run_one_bit(name, 0, seed=9)


def test_sat_metric_mode():
    sbox, n = models.load("des_s1")
    eng = make_engine(seed=4, gpu="off", save_states=False, metric="sat",
                      try_nots=True)
    eng.set_sbox(sbox, n)
    st = eng.initial_state()
    out = eng.create_circuit(st, eng.target(0), mask_for_inputs(n))
    assert out >= 0
    st.set_output(0, out)
    assert validate_circuit(st, sbox, n, bit=0)
    assert st.sat_metric > 0


def test_restricted_gate_set():
    # The CI set 10694 = NAND/NOR-ish subset (.travis.yml:43).
    st = run_one_bit("des_s1", 0, seed=6, gate_bitfield=10694)
    assert validate_circuit  # circuit already validated in run_one_bit


def test_generate_graph_full_sbox(tmp_path):
    """Full multi-output beam search on a small S-box writes checkpoints
    whose circuits evaluate correctly on every output."""
    sbox, n = models.load("des_s1")
    eng = make_engine(seed=8, gpu="off", save_states=True,
                      output_dir=str(tmp_path))
    eng.set_sbox(sbox, n)
    eng.generate_graph(eng.initial_state())
    files = eng.saved_files()
    assert files
    final = _core.State.load(files[-1])
    wired = [b for b in range(8) if final.outputs[b] >= 0]
    assert len(wired) == 4  # des_s1 has 4 output bits
    assert validate_circuit(final, sbox, n)


def test_generate_graph_one_output(tmp_path):
    sbox, n = models.load("des_s1")
    eng = make_engine(seed=10, gpu="off", save_states=True,
                      output_dir=str(tmp_path), oneoutput=2, iterations=2)
    eng.set_sbox(sbox, n)
    eng.generate_graph_one_output(eng.initial_state())
    files = eng.saved_files()
    assert len(files) >= 1
    for f in files:
        st = _core.State.load(f)
        assert validate_circuit(st, sbox, n, bit=2)


def test_permuted_search():
    sbox, n = models.load("des_s1", permute=63)
    eng = make_engine(seed=12, gpu="off", save_states=False)
    eng.set_sbox(sbox, n)
    st = eng.initial_state()
    out = eng.create_circuit(st, eng.target(0), mask_for_inputs(n))
    assert out >= 0
    st.set_output(0, out)
    assert validate_circuit(st, sbox, n, bit=0)


# NOTE: rijndael (8-input) searches are GPU-tier workloads — the reference
# needs an MPI cluster for them — and live in test_gpu.py.


def test_max_gates_bound_respected():
    sbox, n = models.load("des_s1")
    eng = make_engine(seed=13, gpu="off", save_states=False)
    eng.set_sbox(sbox, n)
    st = eng.initial_state()
    st.max_gates = 8  # 6 inputs + 2: essentially impossible
    out = eng.create_circuit(st, eng.target(0), mask_for_inputs(n))
    assert out == -1


def test_seeded_determinism():
    a = run_one_bit("des_s1", 0, seed=42).to_xml()
    b = run_one_bit("des_s1", 0, seed=42).to_xml()
    assert a == b


class TestScans:
    """CPU scan primitives: counting, range-splitting, solution validity."""

    def make(self, pool=40, seed=0xBEEF):
        sbox, n = models.load("rijndael")
        eng = make_engine(lut_graph=True, seed=1, gpu="off", save_states=False)
        eng.set_sbox(sbox, n)
        st = eng.initial_state()
        st.grow_pool_random(pool, seed)
        return eng, st, eng.target(0), mask_for_inputs(n)

    def test_scan_counts_full_range(self):
        eng, st, target, mask = self.make()
        for k in (3, 5):
            total = n_choose_k(st.num_gates, k)
            found, res, ev = eng.scan_pool(k, st, target, mask, 0, total,
                                           count_all=True)
            assert ev == total

    def test_scan_range_split_counts(self):
        eng, st, target, mask = self.make()
        total = n_choose_k(st.num_gates, 5)
        parts = [0, total // 3, total // 2, 2 * total // 3, total]
        ev_sum = 0
        for a, b in zip(parts, parts[1:]):
            _, _, ev = eng.scan_pool(5, st, target, mask, a, b, count_all=True)
            ev_sum += ev
        assert ev_sum == total

    def test_scan3_solution_valid(self):
        # Target realized by a 3-LUT of pool gates: plant one.
        eng, st, target, mask = self.make(pool=20)
        from sboxgates_amd.ops import gen_lut_ttable, tt_eq_mask
        t = gen_lut_ttable(0x96, st.gate(3)["table"], st.gate(9)["table"],
                           st.gate(15)["table"])
        found, res, ev = eng.scan_pool(3, st, t, mask, 0,
                                       n_choose_k(st.num_gates, 3))
        assert found
        got = gen_lut_ttable(res[0], st.gate(res[1])["table"],
                             st.gate(res[2])["table"], st.gate(res[3])["table"])
        assert tt_eq_mask(t, got, mask)

    def test_scan5_solution_valid(self):
        eng, st, target, mask = self.make(pool=18)
        from sboxgates_amd.ops import gen_lut_ttable, tt_eq_mask
        t_outer = gen_lut_ttable(0xE8, st.gate(2)["table"], st.gate(5)["table"],
                                 st.gate(11)["table"])
        t = gen_lut_ttable(0x4A, t_outer, st.gate(7)["table"], st.gate(13)["table"])
        found, res, ev = eng.scan_pool(5, st, t, mask, 0,
                                       n_choose_k(st.num_gates, 5))
        assert found
        t_o = gen_lut_ttable(res[0], st.gate(res[2])["table"],
                             st.gate(res[3])["table"], st.gate(res[4])["table"])
        t_i = gen_lut_ttable(res[1], t_o, st.gate(res[5])["table"],
                             st.gate(res[6])["table"])
        assert tt_eq_mask(t, t_i, mask)

    def test_scan7_solution_valid(self):
        eng, st, target, mask = self.make(pool=14)
        from sboxgates_amd.ops import gen_lut_ttable, tt_eq_mask
        t_o = gen_lut_ttable(0x35, st.gate(1)["table"], st.gate(4)["table"],
                             st.gate(8)["table"])
        t_m = gen_lut_ttable(0xC9, st.gate(2)["table"], st.gate(6)["table"],
                             st.gate(10)["table"])
        t = gen_lut_ttable(0x7B, t_o, t_m, st.gate(12)["table"])
        found, res, ev = eng.scan_pool(7, st, t, mask, 0,
                                       n_choose_k(st.num_gates, 7))
        assert found
        g_o = gen_lut_ttable(res[0], st.gate(res[3])["table"],
                             st.gate(res[4])["table"], st.gate(res[5])["table"])
        g_m = gen_lut_ttable(res[1], st.gate(res[6])["table"],
                             st.gate(res[7])["table"], st.gate(res[8])["table"])
        g_i = gen_lut_ttable(res[2], g_o, g_m, st.gate(res[9])["table"])
        assert tt_eq_mask(t, g_i, mask)


class TestScan4:
    """Gate-mode step-4 triple scan (matcher-based)."""

    def make(self, pool=30, bitfield=2 + 64 + 128, try_nots=False):
        sbox, n = models.load("rijndael")
        eng = make_engine(seed=1, gpu="off", save_states=False,
                          gate_bitfield=bitfield, try_nots=try_nots)
        eng.set_sbox(sbox, n)
        st = eng.initial_state()
        st.grow_pool_random(pool, 0xD00D)
        return eng, st, mask_for_inputs(8)

    def test_counts(self):
        eng, st, mask = self.make()
        total = n_choose_k(st.num_gates, 3)
        found, res, ev = eng.scan_pool(4, st, eng.target(0), mask, 0, total,
                                       count_all=True)
        assert ev == total

    def test_planted_composed_function_found(self):
        from sboxgates_amd.ops import function_lists, gen_lut_ttable, tt_eq_mask
        import random
        rng = random.Random(5)
        eng, st, mask = self.make(try_nots=True)
        _, _, threes = function_lists(2 + 64 + 128, True)
        for trial in range(10):
            f = rng.choice(threes)
            ids = sorted(rng.sample(range(st.num_gates), 3))
            target = gen_lut_ttable(f["fun"], st.gate(ids[0])["table"],
                                    st.gate(ids[1])["table"],
                                    st.gate(ids[2])["table"])
            found, res, ev = eng.scan_pool(4, st, target, mask, 0,
                                           n_choose_k(st.num_gates, 3))
            assert found, trial
            # Verify: avail fun at res[0] applied in order res[1] to the
            # found triple matches the target under the mask.
            perms = [(0, 1, 2), (0, 2, 1), (1, 0, 2), (1, 2, 0), (2, 0, 1),
                     (2, 1, 0)]
            sel = perms[res[1]]
            gids = [res[2], res[3], res[4]]
            fun = threes[res[0]]["fun"]
            got = gen_lut_ttable(fun, st.gate(gids[sel[0]])["table"],
                                 st.gate(gids[sel[1]])["table"],
                                 st.gate(gids[sel[2]])["table"])
            assert tt_eq_mask(target, got, mask)

    def test_scan4_matches_gate_search(self):
        """The scan-based step 4b must still let gate-mode searches finish
        (covered end-to-end by des tests); here: a target with no matching
        triple must not be 'found'."""
        eng, st, mask = self.make(pool=12, bitfield=2)  # AND only: tiny closure
        target = eng.target(3)
        found, res, ev = eng.scan_pool(4, st, target, mask, 0,
                                       n_choose_k(st.num_gates, 3))
        # With only AND compositions over a random pool, realizing an AES
        # output bit exactly is essentially impossible.
        assert not found


def test_find_circuit_api(tmp_path):
    from sboxgates_amd.search import find_circuit
    st = find_circuit("des_s1", bit=0, seed=3, gpu="off")
    assert st.outputs[0] >= 0
    st2 = find_circuit("crypto1_fa", bit=0, lut=True, seed=4, gpu="off")
    assert st2.outputs[0] >= 0
    st3 = find_circuit([1, 0, 2, 3], bit=1, seed=5, gpu="off")
    assert st3.outputs[1] >= 0


def test_find_circuit_all_outputs(tmp_path, monkeypatch):
    """find_circuit with bit=None runs the beam search and returns a fully
    wired, validated state."""
    monkeypatch.chdir(tmp_path)
    from sboxgates_amd.search import find_circuit
    st = find_circuit("des_s1", lut=False, seed=6, gpu="off",
                      save_dir=str(tmp_path))
    wired = [b for b in range(8) if st.outputs[b] >= 0]
    assert len(wired) == 4


def test_parallel_jobs_one_output(tmp_path):
    """--jobs: iterations as parallel independent engines; every produced
    checkpoint is a valid circuit."""
    sbox, n = models.load("des_s1")
    eng = make_engine(lut_graph=True, seed=33, gpu="off", save_states=True,
                      output_dir=str(tmp_path), oneoutput=0, iterations=6,
                      jobs=3)
    eng.set_sbox(sbox, n)
    eng.generate_graph_one_output(eng.initial_state())
    files = eng.saved_files()
    assert len(files) >= 2
    for f in files:
        st = _core.State.load(f)
        assert validate_circuit(st, sbox, n, bit=0)


def test_parallel_jobs_beam_search(tmp_path):
    """generate_graph with jobs: full multi-output beam with parallel
    (state x output) tasks; final checkpoint covers all outputs."""
    sbox, n = models.load("des_s1")
    eng = make_engine(seed=44, gpu="off", save_states=True,
                      output_dir=str(tmp_path), jobs=3)
    eng.set_sbox(sbox, n)
    eng.generate_graph(eng.initial_state())
    files = eng.saved_files()
    assert files
    final = _core.State.load(files[-1])
    wired = [b for b in range(8) if final.outputs[b] >= 0]
    assert len(wired) == 4
    assert validate_circuit(final, sbox, n)


def test_gpu_force_without_gpu_raises():
    if _core.gpu_available():
        pytest.skip("GPU present")
    with pytest.raises(RuntimeError):
        make_engine(gpu="force")


@pytest.mark.parametrize("bitfield", [0x8001, 8, 0x6, 1])
def test_degenerate_gate_sets_no_crash(bitfield):
    """TRUE/FALSE/pass-through-only vocabularies must terminate cleanly
    (found or not) rather than loop or crash."""
    sbox, n = models.load("crypto1_fa")
    eng = make_engine(seed=2, gpu="off", save_states=False,
                      gate_bitfield=bitfield)
    eng.set_sbox(sbox, n)
    st = eng.initial_state()
    st.max_gates = 14
    out = eng.create_circuit(st, eng.target(0), mask_for_inputs(n))
    if out >= 0:
        st.set_output(0, out)
        assert validate_circuit(st, sbox, n, bit=0)


def test_scan_windows_at_max_pool():
    """Boundary math at the MAX_GATES pool size (n=500): window counts are
    exact at the far end of the combination space."""
    sbox, n = models.load("rijndael")
    eng = make_engine(lut_graph=True, seed=1, gpu="off", save_states=False)
    eng.set_sbox(sbox, n)
    st = eng.initial_state()
    st.grow_pool_random(500, 99)
    assert st.num_gates == 500
    mask = mask_for_inputs(8)
    for k in (3, 5, 7):
        total = n_choose_k(500, k)
        for begin in (0, total // 2, total - 100_000):
            end = min(total, begin + 100_000)
            _, _, ev = eng.scan_pool(k, st, eng.target(0), mask, begin, end,
                                     count_all=True)
            assert ev == end - begin, (k, begin)


def test_degenerate_mux_no_abort():
    """A target whose both mux half-solutions are the selector bit made
    the reference abort (XOR(x,x) assert); this engine must complete.
    (Config found by randomized soak testing.)"""
    sbox, n = models.load_table([4, 6, 4, 0, 0, 2, 4, 2])
    eng = make_engine(seed=2108833752, gpu="off", save_states=False,
                      metric="sat", gate_bitfield=214)
    eng.set_sbox(sbox, n)
    st = eng.initial_state()
    out = eng.create_circuit(st, eng.target(0), mask_for_inputs(n))
    if out >= 0:
        st.set_output(0, out)
        assert validate_circuit(st, sbox, n, bit=0)
