"""Subprocess worker for the multi-process (gloo) distributed tests.
Launched by test_parallel.py with torchrun-style env vars."""

import json
import os
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)


def main():
    import torch.distributed as dist

    from sboxgates_amd import _core, models
    from sboxgates_amd.ops import make_engine, mask_for_inputs
    from sboxgates_amd.parallel import init_from_env, make_ctx, run_search
    from sboxgates_amd.utils import validate_circuit

    mode = sys.argv[1]
    out_path = sys.argv[2]

    rank, world, _ = init_from_env(backend="gloo")
    ctx = make_ctx()

    sbox, n = models.load("des_s1")
    gpu_mode = "force" if os.environ.get("SBOXGATES_TEST_GPU") else "off"
    eng = make_engine(lut_graph=True, seed=21, gpu=gpu_mode, save_states=False,
                      oneoutput=0, ctx=ctx)
    eng.set_sbox(sbox, n)

    if mode == "one_output_search":
        if rank != 0:
            eng.worker_loop()
            result = {"rank": rank, "ok": True}
        else:
            st = eng.initial_state()
            out = eng.create_circuit(st, eng.target(0), mask_for_inputs(n))
            eng.stop_workers()
            assert out >= 0
            st.set_output(0, out)
            ok = validate_circuit(st, sbox, n, bit=0)
            result = {"rank": rank, "ok": bool(ok),
                      "gates": st.num_gates - st.num_inputs,
                      "stats": {k: int(v) for k, v in eng.stats().items()}}
    elif mode == "run_search_driver":
        files = run_search(eng, "one_output")
        result = {"rank": rank, "ok": True,
                  "files": files if files is not None else []}
    else:
        raise SystemExit(f"unknown mode {mode}")

    with open(out_path + f".rank{rank}", "w") as f:
        json.dump(result, f)
    dist.barrier()
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
