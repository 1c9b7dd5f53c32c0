"""Multi-process distributed-protocol tests (gloo backend, world size 2,
CPU). The same chunked symmetric protocol runs over RCCL on GPU nodes —
these tests pin down its correctness by construction (fixed collective
cadence; see sbg/search.cpp dist_scan_chunked)."""

import json
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
WORKER = os.path.join(REPO, "tests", "_dist_worker.py")


def launch(mode, tmp_path, world=2, extra_env=None, timeout=240):
    out = os.path.join(str(tmp_path), "result")
    procs = []
    for rank in range(world):
        env = dict(os.environ)
        env.update({
            "RANK": str(rank),
            "WORLD_SIZE": str(world),
            "LOCAL_RANK": str(rank),
            "MASTER_ADDR": "127.0.0.1",
            "MASTER_PORT": str(29650 + (os.getpid() % 1000)),
        })
        if extra_env:
            env.update(extra_env)
        procs.append(subprocess.Popen([sys.executable, WORKER, mode, out],
                                      env=env, cwd=str(tmp_path),
                                      stdout=subprocess.PIPE,
                                      stderr=subprocess.PIPE, text=True))
    results = {}
    for rank, p in enumerate(procs):
        try:
            stdout, stderr = p.communicate(timeout=timeout)
        except subprocess.TimeoutExpired:
            for q in procs:
                q.kill()
            pytest.fail(f"rank {rank} timed out (deadlock?)")
        assert p.returncode == 0, f"rank {rank} failed:\n{stderr}\n{stdout}"
        with open(out + f".rank{rank}") as f:
            results[rank] = json.load(f)
    return results


def test_distributed_lut_search(tmp_path):
    """World-2 LUT search: rank 0 drives, rank 1 serves; result valid."""
    res = launch("one_output_search", tmp_path)
    assert res[0]["ok"] and res[1]["ok"]
    assert res[0]["gates"] <= 15


def test_distributed_multi_chunk(tmp_path):
    """Tiny chunks force many allreduce rounds through the protocol."""
    res = launch("one_output_search", tmp_path,
                 extra_env={"SBOXGATES_CHUNK5": "2000", "SBOXGATES_CHUNK7": "20000"})
    assert res[0]["ok"] and res[1]["ok"]


def test_run_search_driver(tmp_path):
    """The parallel.run_search SPMD helper completes on both ranks."""
    res = launch("run_search_driver", tmp_path)
    assert res[0]["ok"] and res[1]["ok"]


def test_distributed_world3_small_chunks(tmp_path):
    """Three ranks, uneven ranges, forced multi-chunk cadence."""
    res = launch("one_output_search", tmp_path, world=3,
                 extra_env={"SBOXGATES_CHUNK5": "1500",
                            "SBOXGATES_CHUNK7": "15000"})
    assert all(res[r]["ok"] for r in range(3))


def test_distributed_world4_small_chunks(tmp_path):
    """World-4 multi-process search (the driver's 4-GPU shape on the gloo
    transport), with chunk sizes forcing many allreduce rounds."""
    res = launch("one_output_search", tmp_path, world=4,
                 extra_env={"SBOXGATES_CHUNK5": "5000",
                            "SBOXGATES_CHUNK7": "40000"})
    assert res[0]["ok"]
    assert res[0]["gates"] > 0
