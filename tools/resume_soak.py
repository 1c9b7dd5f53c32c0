"""Interrupt/resume soak: random S-boxes, full-graph searches repeatedly
KILLED mid-run and resumed from --resume-dir checkpoints until complete;
the final artifact is validated by DAG evaluation against the table.

Exercises the round-2 checkpoint/restart path (and the live-prefix state
copies + pair-loop changes underneath) the way the GPU budget windows use
it: SIGKILL at arbitrary points, then resumption from the best state.

    python tools/resume_soak.py --trials 200 --seed 1
"""

import argparse
import os
import random
import shutil
import subprocess
import sys
import tempfile
import time

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)
CLI = os.path.join(REPO, "bin", "sboxgates")


def random_sbox(rng, n_in, n_out):
    return [rng.randrange(1 << n_out) for _ in range(1 << n_in)]


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--trials", type=int, default=100)
    ap.add_argument("--seed", type=int, default=1)
    ap.add_argument("--report-every", type=int, default=10)
    args = ap.parse_args()

    from sboxgates_amd import _core

    rng = random.Random(args.seed)
    t0 = time.time()
    kills = 0
    for trial in range(args.trials):
        n_in = rng.choice([4, 5, 5, 6])
        n_out = rng.choice([2, 3, 4])
        table = random_sbox(rng, n_in, n_out)
        # Ensure every output bit is non-constant so all get wired.
        for b in range(n_out):
            vals = {(v >> b) & 1 for v in table}
            if len(vals) == 1:
                table[0] ^= 1 << b
        vocab = rng.choice(["194", "10694", "214", "65535"])
        d = tempfile.mkdtemp(prefix="soak")
        sfile = os.path.join(d, "sbox.txt")
        with open(sfile, "w") as f:
            f.write(" ".join("%02x" % v for v in table))
        ck = os.path.join(d, "ck")
        cmd = [CLI, "--cpu", "-a", vocab, "--beam", str(rng.choice([1, 2])),
               "--seed", str(rng.getrandbits(30)), "--resume-dir", ck, sfile]
        mode = rng.randrange(3)
        if mode == 1:
            cmd.insert(1, "-l")          # LUT-graph mode
        elif mode == 2 and vocab != "65535":
            cmd.insert(1, "-s")          # SAT metric
        if rng.randrange(4) == 0:
            cmd[1:1] = ["-n"]            # NOT-augmented step 4a
        # Run with random kill windows; the last attempts run to
        # completion (slow modes — LUT/SAT on 6 inputs — can legitimately
        # outlive every short window).
        done = False
        for attempt in range(14):
            p = subprocess.Popen(cmd, cwd=d, stdout=subprocess.DEVNULL,
                                 stderr=subprocess.PIPE, text=True)
            kill_after = rng.uniform(0.02, 0.6) if attempt < 12 else 180.0
            try:
                p.wait(timeout=kill_after)
                assert p.returncode == 0, p.stderr.read()
                done = True
                break
            except subprocess.TimeoutExpired:
                p.kill()
                p.wait()
                kills += 1
        assert done, f"trial {trial}: never completed"
        # Validate the most advanced artifact.
        best, best_outs = None, -1
        for name in os.listdir(ck):
            if not name.endswith(".xml"):
                continue
            try:
                st = _core.State.load(os.path.join(ck, name))
            except RuntimeError:
                # save_state() writes atomically (temp + rename), so a
                # parse failure here is a bug, not an interruption scar.
                raise AssertionError(
                    f"trial {trial}: truncated/corrupt checkpoint {name}")
            outs = [b for b in range(8) if st.outputs[b] >= 0]
            if len(outs) > best_outs:
                best, best_outs = st, len(outs)
        assert best is not None and best_outs == n_out, (trial, best_outs, n_out)
        for x in range(1 << n_in):
            got = best.eval(x)
            for b in range(n_out):
                assert ((got >> b) & 1) == ((table[x] >> b) & 1), (
                    f"trial {trial}: mismatch input {x} bit {b}")
        shutil.rmtree(d)
        if (trial + 1) % args.report_every == 0:
            print(f"[{time.time()-t0:7.1f}s] {trial+1}/{args.trials} ok "
                  f"({kills} mid-run kills so far)", flush=True)
    print(f"SOAK OK: {args.trials} interrupted-and-resumed searches, "
          f"{kills} kills, all final circuits valid")


if __name__ == "__main__":
    main()
