"""CLI contract tests — the reference CI's command list (.travis.yml:26-51)
reproduced against the native binary, minus the MPI launcher (the CLI is a
single process; multi-GPU goes through sboxgates_amd.parallel)."""

import glob
import os
import shutil
import subprocess

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
CLI = os.path.join(REPO, "bin", "sboxgates")
SBOX = os.path.join(REPO, "sboxgates_amd", "sboxes")


def run(args, cwd=None, timeout=120):
    return subprocess.run([CLI] + args, cwd=cwd, capture_output=True, text=True,
                          timeout=timeout)


@pytest.fixture(scope="module", autouse=True)
def cli_exists():
    if not os.path.exists(CLI):
        pytest.skip("CLI not built (run make)")


def des(cwd=None):
    return os.path.join(SBOX, "des_s1.txt")


def test_help():
    assert run(["--help"]).returncode == 0


def test_no_args_fails():
    assert run([]).returncode != 0


@pytest.mark.parametrize("args", [
    ["-a", "-123"],
    ["-a", "65536"],
    ["-i", "0"],
    ["-i", "-123"],
    ["-o", "-123"],
    ["-o", "8"],
    ["-p", "-123"],
    ["-p", "256"],
])
def test_bad_argument_values(args):
    assert run(args + [des()]).returncode != 0


def test_conflicting_conversions():
    assert run(["-c", "-d", "test.xml"]).returncode != 0


def test_lut_sat_conflict():
    assert run(["-l", "-s", des()]).returncode != 0


def test_nonexistent_input():
    assert run(["nonexisting.txt"]).returncode != 0


def test_output_bit_beyond_sbox():
    # des_s1 has 4 outputs; -o 7 must fail (sboxgates.c:1128-1134).
    assert run(["-o", "7", des()]).returncode != 0


def test_search_sat_not_resume(tmp_path):
    # mpirun -N 4 ./sboxgates -vv -i 3 -o 0 -s -n des_s1.txt; then resume.
    r = run(["-vv", "-i", "3", "-o", "0", "-s", "-n", "--seed", "5", "--cpu", des()],
            cwd=str(tmp_path))
    assert r.returncode == 0, r.stderr
    produced = glob.glob(os.path.join(str(tmp_path), "1*.xml"))
    assert produced
    r2 = run(["-vv", "-i", "1", "-s", "-n", "-o", "0", "--seed", "6", "--cpu",
              "-g", produced[0], des()], cwd=str(tmp_path))
    assert r2.returncode == 0, r2.stderr


def test_full_graph_restricted_gates_permute(tmp_path):
    # mpirun -N 4 ./sboxgates -vv -a 10694 -i 3 -p 63 des_s1.txt
    r = run(["-vv", "-a", "10694", "-i", "1", "-p", "63", "--seed", "7", "--cpu",
             des()], cwd=str(tmp_path), timeout=300)
    assert r.returncode == 0, r.stderr
    produced = glob.glob(os.path.join(str(tmp_path), "4-*.xml"))
    assert produced
    # Convert to DOT; pipe through graphviz when available.
    r3 = run(["-d", produced[0]])
    assert r3.returncode == 0 and r3.stdout.startswith("digraph sbox {")
    dot = shutil.which("dot")
    if dot:
        p = subprocess.run([dot, "-Tpng"], input=r3.stdout, capture_output=True,
                           text=False if False else True)
        assert p.returncode == 0
    # Convert to C; compile it.
    r4 = run(["-c", produced[0]])
    assert r4.returncode == 0
    gcc = shutil.which("gcc")
    if gcc:
        cfile = os.path.join(str(tmp_path), "test.c")
        open(cfile, "w").write(r4.stdout)
        p = subprocess.run([gcc, "-c", "-Wall", "-Wpedantic", "-Werror", cfile,
                            "-o", os.path.join(str(tmp_path), "test.o")])
        assert p.returncode == 0


def test_lut_search_and_cuda_output(tmp_path):
    # mpirun -N 10 ./sboxgates -vv -a 10694 -l -o 0 des_s1.txt
    r = run(["-vv", "-a", "10694", "-l", "-o", "0", "--seed", "8", "--cpu", des()],
            cwd=str(tmp_path), timeout=300)
    assert r.returncode == 0, r.stderr
    produced = glob.glob(os.path.join(str(tmp_path), "1-*.xml"))
    assert produced
    r2 = run(["-c", produced[0]])
    assert r2.returncode == 0
    assert "lop3.b32" in r2.stdout  # CUDA emission for LUT graphs
    # HIP backend on the same graph.
    r3 = run(["--convert-hip", produced[0]])
    assert r3.returncode == 0
    assert "lop3" not in r3.stdout


def test_version():
    r = run(["--version"])
    assert r.returncode == 0 and "sboxgates" in r.stdout


def test_identity_sbox(tmp_path):
    # The identity test vector: searches must find trivial pass-through.
    r = run(["-o", "0", "--seed", "3", "--cpu",
             os.path.join(SBOX, "identity.txt")], cwd=str(tmp_path))
    assert r.returncode == 0


def test_multi_device_threads(tmp_path):
    """--gpus N: in-process SPMD over worker threads (CPU path here; on a
    GPU node each thread drives its own device)."""
    r = run(["-l", "-o", "0", "--seed", "4", "--cpu", "--gpus", "3", des()],
            cwd=str(tmp_path), timeout=240)
    assert r.returncode == 0, r.stderr
    assert glob.glob(os.path.join(str(tmp_path), "1-*.xml"))


def test_bad_gpus_value():
    assert run(["--gpus", "0", des()]).returncode != 0


def test_beam_option(tmp_path):
    r = run(["--beam", "1", "--seed", "9", "--cpu", des()], cwd=str(tmp_path),
            timeout=300)
    assert r.returncode == 0, r.stderr
    assert run(["--beam", "0", des()]).returncode != 0
    assert run(["--beam", "21", des()]).returncode != 0


def test_jobs_option(tmp_path):
    r = run(["-l", "-o", "0", "-i", "4", "--jobs", "2", "--seed", "3", "--cpu",
             des()], cwd=str(tmp_path), timeout=300)
    assert r.returncode == 0, r.stderr
    assert glob.glob(os.path.join(str(tmp_path), "1-*.xml"))
    assert run(["--jobs", "0", des()]).returncode != 0


def test_max_gates_bound(tmp_path):
    # 6 inputs + 19 gates: usually fails for des bit0 (the bound prunes);
    # a generous bound succeeds.
    r = run(["-o", "0", "--seed", "2", "--cpu", "--max-gates", "60", des()],
            cwd=str(tmp_path))
    assert r.returncode == 0
    assert glob.glob(os.path.join(str(tmp_path), "1-*.xml"))
    assert run(["--max-gates", "0", des()]).returncode != 0


def test_resume_dir(tmp_path):
    """--resume-dir: a second invocation picks up the most advanced state
    from the first (checkpoint/restart across budget windows)."""
    d = os.path.join(str(tmp_path), "ckpt")
    r = run(["-a", "10694", "-p", "63", "--seed", "5", "--cpu",
             "--resume-dir", d, des()], cwd=str(tmp_path), timeout=300)
    assert r.returncode == 0, r.stderr
    first = set(glob.glob(os.path.join(d, "*.xml")))
    assert first, "first run produced no checkpoints"
    r2 = run(["-a", "10694", "-p", "63", "--seed", "6", "--cpu",
              "--resume-dir", d, des()], cwd=str(tmp_path), timeout=300)
    assert r2.returncode == 0, r2.stderr
    assert "Resuming from" in r2.stdout
    # It resumed from a full 4-output state, so the second run starts (and
    # ends) complete without redoing outputs.
    import re
    m = re.search(r"Resuming from \S+ \((\d+) outputs?", r2.stdout)
    assert m and int(m.group(1)) >= 1
