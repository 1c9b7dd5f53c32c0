"""GPU kernel tests (gfx950) — run on an MI355X box.

Every test forces the GPU path (Options.gpu="force": silent CPU fallback is
impossible — the engine throws if the kernels are unavailable) and checks
the kernels against the CPU implementation of the same scan.
"""

import random

import pytest

from sboxgates_amd import _core, models
from sboxgates_amd.ops import (gen_lut_ttable, make_engine, mask_for_inputs,
                               n_choose_k, tt_eq_mask)
from sboxgates_amd.utils import validate_circuit

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def engines():
    if not _core.gpu_available():
        pytest.skip("no GPU")
    gpu = make_engine(lut_graph=True, seed=1, gpu="force", save_states=False)
    cpu = make_engine(lut_graph=True, seed=1, gpu="off", save_states=False)
    sbox, n = models.load("rijndael")
    gpu.set_sbox(sbox, n)
    cpu.set_sbox(sbox, n)
    assert gpu.gpu_active
    return gpu, cpu


def make_pool(engine, pool, seed):
    st = engine.initial_state()
    st.grow_pool_random(pool, seed)
    return st


@pytest.mark.parametrize("k,pool", [(3, 60), (3, 301), (5, 40), (5, 90), (7, 26)])
def test_scan_counts_match_cpu(engines, k, pool):
    gpu, cpu = engines
    st = make_pool(gpu, pool, 0xA0 + pool)
    target = gpu.target(0)
    mask = mask_for_inputs(8)
    total = n_choose_k(pool, k)
    end = min(total, 3_000_000)
    f_g, r_g, ev_g = gpu.scan_pool(k, st, target, mask, 0, end, seed=3,
                                   count_all=True)
    f_c, r_c, ev_c = cpu.scan_pool(k, st, target, mask, 0, end, seed=3,
                                   count_all=True)
    assert ev_g == ev_c == end


@pytest.mark.parametrize("k", [3, 5, 7])
def test_planted_solutions_found_and_valid(engines, k):
    """Plant a k-LUT-decomposable target; the kernel must find a valid
    (not necessarily identical) solution."""
    gpu, cpu = engines
    rng = random.Random(77 + k)
    pool = {3: 80, 5: 40, 7: 22}[k]
    st = make_pool(gpu, pool, 0xB0 + k)
    mask = mask_for_inputs(8)
    ids = rng.sample(range(pool), k)
    tabs = [st.gate(i)["table"] for i in ids]
    if k == 3:
        target = gen_lut_ttable(rng.randrange(256), *tabs)
    elif k == 5:
        t_o = gen_lut_ttable(rng.randrange(256), tabs[0], tabs[1], tabs[2])
        target = gen_lut_ttable(rng.randrange(256), t_o, tabs[3], tabs[4])
    else:
        t_o = gen_lut_ttable(rng.randrange(256), tabs[0], tabs[1], tabs[2])
        t_m = gen_lut_ttable(rng.randrange(256), tabs[3], tabs[4], tabs[5])
        target = gen_lut_ttable(rng.randrange(256), t_o, t_m, tabs[6])

    found, res, ev = gpu.scan_pool(k, st, target, mask, 0, n_choose_k(pool, k),
                                   seed=5)
    assert found
    if k == 3:
        got = gen_lut_ttable(res[0], st.gate(res[1])["table"],
                             st.gate(res[2])["table"], st.gate(res[3])["table"])
    elif k == 5:
        t_o = gen_lut_ttable(res[0], st.gate(res[2])["table"],
                             st.gate(res[3])["table"], st.gate(res[4])["table"])
        got = gen_lut_ttable(res[1], t_o, st.gate(res[5])["table"],
                             st.gate(res[6])["table"])
    else:
        t_o = gen_lut_ttable(res[0], st.gate(res[3])["table"],
                             st.gate(res[4])["table"], st.gate(res[5])["table"])
        t_m = gen_lut_ttable(res[1], st.gate(res[6])["table"],
                             st.gate(res[7])["table"], st.gate(res[8])["table"])
        got = gen_lut_ttable(res[2], t_o, t_m, st.gate(res[9])["table"])
    assert tt_eq_mask(target, got, mask)


def test_found_parity_sparse_masks(engines):
    """GPU and CPU must agree on found/not-found for sparse-mask scans
    (deep-recursion-like conditions) across many random instances.

    Half the instances plant a 5-LUT-decomposable target (realizable under
    ANY mask), so a hit is guaranteed and a regression that kills hit
    detection cannot hide behind not-found agreement; the other half use
    raw S-box output-bit targets for unbiased found/not-found parity."""
    gpu, cpu = engines
    import struct
    rng = random.Random(31)
    found_planted = 0
    for trial in range(20):
        pool = rng.choice([20, 30, 40])
        st = make_pool(gpu, pool, rng.getrandbits(32))
        # Sparse mask: search deep-recursion-like conditions.
        words = [0, 0, 0, 0]
        for _ in range(rng.choice([6, 10, 16])):
            i = rng.randrange(256)
            words[i // 64] |= 1 << (i % 64)
        mask = struct.pack("<4Q", *words)
        planted = trial % 2 == 0
        if planted:
            ids = rng.sample(range(pool), 5)
            tabs = [st.gate(i)["table"] for i in ids]
            t_o = gen_lut_ttable(rng.randrange(256), tabs[0], tabs[1], tabs[2])
            target = gen_lut_ttable(rng.randrange(256), t_o, tabs[3], tabs[4])
        else:
            target = gpu.target(rng.randrange(8))
        total = n_choose_k(pool, 5)
        f_g, r_g, _ = gpu.scan_pool(5, st, target, mask, 0, total, seed=trial)
        f_c, r_c, _ = cpu.scan_pool(5, st, target, mask, 0, total, seed=trial)
        assert f_g == f_c, trial
        if planted:
            assert f_g, f"planted 5-LUT not found in trial {trial}"
            found_planted += 1
        if f_g:
            t_o = gen_lut_ttable(r_g[0], st.gate(r_g[2])["table"],
                                 st.gate(r_g[3])["table"], st.gate(r_g[4])["table"])
            got = gen_lut_ttable(r_g[1], t_o, st.gate(r_g[5])["table"],
                                 st.gate(r_g[6])["table"])
            assert tt_eq_mask(target, got, mask)
    assert found_planted == 10


def test_range_split_counts(engines):
    gpu, _ = engines
    st = make_pool(gpu, 80, 7)
    target = gpu.target(0)
    mask = mask_for_inputs(8)
    total = n_choose_k(80, 5)
    parts = [0, total // 4, total // 2, total]
    ev = 0
    for a, b in zip(parts, parts[1:]):
        _, _, e = gpu.scan_pool(5, st, target, mask, a, b, count_all=True)
        ev += e
    assert ev == total


def test_des_lut_search_end_to_end_gpu():
    """Full DES S1 bit 0 LUT search with forced GPU kernels; circuit
    validated by evaluation."""
    eng = make_engine(lut_graph=True, seed=9, gpu="force", save_states=False)
    sbox, n = models.load("des_s1")
    eng.set_sbox(sbox, n)
    st = eng.initial_state()
    out = eng.create_circuit(st, eng.target(0), mask_for_inputs(n))
    assert out >= 0
    st.set_output(0, out)
    assert validate_circuit(st, sbox, n, bit=0)
    assert eng.stats()["gpu_scans"] > 0


def test_scan5_throughput_sane(engines):
    """The 5-LUT kernel must beat the CPU path by a wide margin on a
    large range (guards against silently serialized kernels)."""
    import time
    gpu, cpu = engines
    st = make_pool(gpu, 150, 3)
    target = gpu.target(0)
    mask = mask_for_inputs(8)
    n_range = 50_000_000
    t0 = time.perf_counter()
    _, _, ev = gpu.scan_pool(5, st, target, mask, 0, n_range, count_all=True)
    gpu_time = time.perf_counter() - t0
    assert ev == n_range
    t0 = time.perf_counter()
    _, _, ev_c = cpu.scan_pool(5, st, target, mask, 0, 2_000_000, count_all=True)
    cpu_time = time.perf_counter() - t0
    gpu_rate = n_range / gpu_time
    cpu_rate = 2_000_000 / cpu_time
    print(f"gpu {gpu_rate:.3g} cand/s, cpu {cpu_rate:.3g} cand/s, "
          f"speedup {gpu_rate / cpu_rate:.1f}x")
    assert gpu_rate > 20 * cpu_rate


def test_scan4_gpu_matches_cpu(engines):
    """Gate-mode step-4 triple kernel vs CPU path: counts + planted hits."""
    import random
    from sboxgates_amd.ops import function_lists, gen_lut_ttable, tt_eq_mask
    gpu = make_engine(seed=1, gpu="force", save_states=False, try_nots=True)
    cpu = make_engine(seed=1, gpu="off", save_states=False, try_nots=True)
    sbox, n = models.load("rijndael")
    gpu.set_sbox(sbox, n)
    cpu.set_sbox(sbox, n)
    st = gpu.initial_state()
    st.grow_pool_random(120, 0xFEED)
    mask = mask_for_inputs(8)
    total = n_choose_k(120, 3)
    # Count parity.
    _, _, ev_g = gpu.scan_pool(4, st, gpu.target(0), mask, 0, total,
                               count_all=True)
    _, _, ev_c = cpu.scan_pool(4, st, cpu.target(0), mask, 0, total,
                               count_all=True)
    assert ev_g == ev_c == total
    # Planted composed-function targets must be found and verify.
    _, _, threes = function_lists(2 + 64 + 128, True)
    rng = random.Random(3)
    perms = [(0, 1, 2), (0, 2, 1), (1, 0, 2), (1, 2, 0), (2, 0, 1), (2, 1, 0)]
    for trial in range(5):
        f = rng.choice(threes)
        ids = sorted(rng.sample(range(120), 3))
        target = gen_lut_ttable(f["fun"], st.gate(ids[0])["table"],
                                st.gate(ids[1])["table"], st.gate(ids[2])["table"])
        found, res, _ = gpu.scan_pool(4, st, target, mask, 0, total)
        assert found, trial
        sel = perms[res[1]]
        gids = [res[2], res[3], res[4]]
        got = gen_lut_ttable(threes[res[0]]["fun"],
                             st.gate(gids[sel[0]])["table"],
                             st.gate(gids[sel[1]])["table"],
                             st.gate(gids[sel[2]])["table"])
        assert tt_eq_mask(target, got, mask)


def test_rijndael_gate_mode_bit0_gpu():
    """AES S-box output bit 0, 2-input {AND,OR,XOR} gate set, single GPU —
    BASELINE config 2's single-output core, with the step-4 triple scan on
    the k_scan4 kernel."""
    eng = make_engine(seed=11, gpu="force", save_states=False)
    sbox, n = models.load("rijndael")
    eng.set_sbox(sbox, n)
    st = eng.initial_state()
    out = eng.create_circuit(st, eng.target(0), mask_for_inputs(n))
    assert out >= 0
    st.set_output(0, out)
    assert validate_circuit(st, sbox, n, bit=0)
    assert eng.stats()["gpu_scans"] > 0


def test_scan7_hit_buffer_overflow_splitting(engines, monkeypatch):
    """A tiny hit cap must trigger range splitting, not truncation: the
    planted solution is still found."""
    import os
    import subprocess
    import sys
    # The cap is read at GpuEngine creation; run in a subprocess with the
    # env var set.
    code = r'''
import sys
sys.path.insert(0, ".")
import random
from sboxgates_amd import models
from sboxgates_amd.ops import (gen_lut_ttable, make_engine, mask_for_inputs,
                               n_choose_k, tt_eq_mask)
eng = make_engine(lut_graph=True, seed=1, gpu="force", save_states=False)
sbox, n = models.load("rijndael")
eng.set_sbox(sbox, n)
st = eng.initial_state()
st.grow_pool_random(20, 77)
mask = mask_for_inputs(8)
rng = random.Random(9)
ids = sorted(rng.sample(range(20), 7))
tabs = [st.gate(i)["table"] for i in ids]
t_o = gen_lut_ttable(0x3C, tabs[0], tabs[1], tabs[2])
t_m = gen_lut_ttable(0xA5, tabs[3], tabs[4], tabs[5])
target = gen_lut_ttable(0x96, t_o, t_m, tabs[6])
found, res, ev = eng.scan_pool(7, st, target, mask, 0, n_choose_k(20, 7), seed=3)
assert found, "planted 7-LUT not found under tiny hit cap"
g_o = gen_lut_ttable(res[0], st.gate(res[3])["table"], st.gate(res[4])["table"],
                     st.gate(res[5])["table"])
g_m = gen_lut_ttable(res[1], st.gate(res[6])["table"], st.gate(res[7])["table"],
                     st.gate(res[8])["table"])
g_i = gen_lut_ttable(res[2], g_o, g_m, st.gate(res[9])["table"])
assert tt_eq_mask(target, g_i, mask)
print("overflow-split ok")
'''
    env = dict(os.environ)
    env["SBOXGATES_HIT_CAP"] = "64"
    r = subprocess.run([sys.executable, "-c", code], env=env,
                       capture_output=True, text=True, timeout=300,
                       cwd=os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    assert r.returncode == 0, r.stderr
    assert "overflow-split ok" in r.stdout


def test_distributed_gpu_search_two_ranks():
    """World-2 distributed LUT search with BOTH ranks driving GPU kernels
    (gloo control plane, both ranks on device 0 — the protocol and kernel
    interaction exactly match the multi-GPU RCCL deployment)."""
    import json
    import os
    import subprocess
    import sys
    import tempfile
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    worker = os.path.join(repo, "tests", "_dist_worker.py")
    with tempfile.TemporaryDirectory() as d:
        out = os.path.join(d, "result")
        procs = []
        for rank in range(2):
            env = dict(os.environ)
            env.update({
                "RANK": str(rank), "WORLD_SIZE": "2", "LOCAL_RANK": str(rank),
                "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": "29871",
                "SBOXGATES_TEST_GPU": "1",
                "SBOXGATES_CHUNK5": "20000", "SBOXGATES_CHUNK7": "100000",
            })
            procs.append(subprocess.Popen(
                [sys.executable, worker, "one_output_search", out], env=env,
                cwd=d, stdout=subprocess.PIPE, stderr=subprocess.PIPE,
                text=True))
        for rank, p in enumerate(procs):
            stdout, stderr = p.communicate(timeout=300)
            assert p.returncode == 0, f"rank {rank}:\n{stderr}"
        with open(out + ".rank0") as f:
            r0 = json.load(f)
        assert r0["ok"]
        assert r0["stats"]["gpu_scans"] > 0, "GPU kernels must be used"


def test_random_window_count_parity(engines):
    """Counts over random [begin, end) windows must match the CPU scans
    exactly for every k (exercises prefix/slice boundary math)."""
    import random
    gpu, cpu = engines
    rng = random.Random(123)
    st = make_pool(gpu, 60, 0xFACE)
    target = gpu.target(0)
    mask = mask_for_inputs(8)
    for k in (3, 4, 5, 7):
        total = n_choose_k(60, k if k != 4 else 3)
        for _ in range(6):
            a = rng.randrange(total)
            b = min(total, a + rng.choice([1, 7, 1000, 50_000, 2_000_000]))
            f_g, _, ev_g = gpu.scan_pool(k, st, target, mask, a, b,
                                         count_all=True)
            f_c, _, ev_c = cpu.scan_pool(k, st, target, mask, a, b,
                                         count_all=True)
            assert ev_g == ev_c == b - a, (k, a, b, ev_g, ev_c)


def test_gpu_window_fuzz_soak(engines):
    """Randomized GPU-vs-CPU fuzz over (k, pool, window, mask sparsity):
    evaluated counts must match exactly, found verdicts must agree, and any
    hit must verify under the mask. This is the GPU-path analog of the CPU
    fuzz soaks (round-1 soaks exercised only the CPU path)."""
    import struct
    gpu, cpu = engines
    rng = random.Random(0xF022)
    for trial in range(60):
        k = rng.choice([3, 4, 5, 5, 7, 7])
        pool = rng.choice([15, 25, 40, 70])
        if k == 7 and pool > 40:
            pool = 40
        st = make_pool(gpu, pool, rng.getrandbits(32))
        # Mask: dense, sparse, or full.
        kind = rng.randrange(3)
        if kind == 0:
            mask = mask_for_inputs(8)
        else:
            words = [0, 0, 0, 0]
            nbits = rng.choice([4, 12, 40, 150])
            for _ in range(nbits):
                i = rng.randrange(256)
                words[i // 64] |= 1 << (i % 64)
            mask = struct.pack("<4Q", *words)
        target = gpu.target(rng.randrange(8))
        total = n_choose_k(pool, 3 if k == 4 else k)
        a = rng.randrange(total)
        b = min(total, a + rng.choice([3, 500, 30_000, 200_000]))
        f_g, r_g, ev_g = gpu.scan_pool(k, st, target, mask, a, b,
                                       seed=trial, count_all=True)
        f_c, r_c, ev_c = cpu.scan_pool(k, st, target, mask, a, b,
                                       seed=trial, count_all=True)
        assert ev_g == ev_c == b - a, (trial, k, pool, a, b, ev_g, ev_c)
        # Early-exit mode: verdicts agree; hits verify.
        f_g, r_g, _ = gpu.scan_pool(k, st, target, mask, a, b, seed=trial)
        f_c, _, _ = cpu.scan_pool(k, st, target, mask, a, b, seed=trial)
        assert f_g == f_c, (trial, k, pool, a, b)
        if f_g and k in (3, 5, 7):
            if k == 3:
                got = gen_lut_ttable(r_g[0], st.gate(r_g[1])["table"],
                                     st.gate(r_g[2])["table"],
                                     st.gate(r_g[3])["table"])
            elif k == 5:
                t_o = gen_lut_ttable(r_g[0], st.gate(r_g[2])["table"],
                                     st.gate(r_g[3])["table"],
                                     st.gate(r_g[4])["table"])
                got = gen_lut_ttable(r_g[1], t_o, st.gate(r_g[5])["table"],
                                     st.gate(r_g[6])["table"])
            else:
                t_o = gen_lut_ttable(r_g[0], st.gate(r_g[3])["table"],
                                     st.gate(r_g[4])["table"],
                                     st.gate(r_g[5])["table"])
                t_m = gen_lut_ttable(r_g[1], st.gate(r_g[6])["table"],
                                     st.gate(r_g[7])["table"],
                                     st.gate(r_g[8])["table"])
                got = gen_lut_ttable(r_g[2], t_o, t_m,
                                     st.gate(r_g[9])["table"])
            assert tt_eq_mask(target, got, mask)


def test_parallel_jobs_on_gpu():
    """--jobs: two concurrent engines on one device, each running GPU
    kernels."""
    import subprocess
    import os
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    cli = os.path.join(repo, "bin", "sboxgates")
    if not os.path.exists(cli):
        pytest.skip("CLI not built")
    import tempfile
    with tempfile.TemporaryDirectory() as d:
        r = subprocess.run(
            [cli, "-l", "-o", "0", "-i", "4", "--jobs", "2", "--gpu",
             "--seed", "5", os.path.join(repo, "sboxgates_amd", "sboxes",
                                         "des_s1.txt")],
            cwd=d, capture_output=True, text=True, timeout=300)
        assert r.returncode == 0, r.stderr
        import glob as g
        assert g.glob(os.path.join(d, "1-*.xml"))


def test_concurrent_jobs_gate_mode_gpu():
    """Gate-mode --jobs: several engines sharing the one persistent scan
    service on one device (the config-2 full-graph deployment shape)."""
    import subprocess
    import os
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    cli = os.path.join(repo, "bin", "sboxgates")
    if not os.path.exists(cli):
        pytest.skip("CLI not built")
    import tempfile
    with tempfile.TemporaryDirectory() as d:
        r = subprocess.run(
            [cli, "-o", "0", "-i", "8", "--jobs", "4", "--gpu", "--seed", "7",
             os.path.join(repo, "sboxgates_amd", "sboxes", "des_s1.txt")],
            cwd=d, capture_output=True, text=True, timeout=420)
        assert r.returncode == 0, r.stderr
        import glob as g
        assert g.glob(os.path.join(d, "1-*.xml"))
