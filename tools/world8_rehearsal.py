"""World-N distributed-search rehearsal on however many GPUs are visible.

Runs the REAL distributed LUT search (rank 0 drives create_circuit, other
ranks serve scan work; fixed-cadence chunked allreduce protocol of
sbg/search.cpp dist_scan_chunked) under torchrun, with every collective
individually timed. On an 8-GPU node this is the BASELINE config-3
deployment; on a single-GPU box it is the deployment REHEARSAL the round-1
verdict asked for: all ranks pin to device 0 (local_rank % device_count),
so the protocol, kernel interleaving and collective cadence are exactly
those of the multi-GPU run — only the transport differs if RCCL refuses
co-located ranks (we then record the gloo numbers and say so).

    torchrun --nproc-per-node 8 --master-addr 127.0.0.1 \
        tools/world8_rehearsal.py --backend nccl --out gpurun_out/w8

Writes one JSON per rank: wall, per-collective latency stats, search
result.
"""

import argparse
import json
import os
import statistics
import sys
import time

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--backend", default="auto", choices=["auto", "nccl", "gloo"])
    ap.add_argument("--out", default="gpurun_out/w8")
    ap.add_argument("--sbox", default="rijndael")
    ap.add_argument("--bit", type=int, default=0)
    ap.add_argument("--chunk5", type=int, default=2_000_000)
    ap.add_argument("--chunk7", type=int, default=20_000_000)
    args = ap.parse_args()

    os.environ.setdefault("SBOXGATES_CHUNK5", str(args.chunk5))
    os.environ.setdefault("SBOXGATES_CHUNK7", str(args.chunk7))

    import torch
    import torch.distributed as dist

    from sboxgates_amd import _core, models
    from sboxgates_amd.ops import make_engine, mask_for_inputs
    from sboxgates_amd.utils import validate_circuit

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))

    use_gpu = torch.cuda.is_available()
    ndev = torch.cuda.device_count() if use_gpu else 0
    backend = args.backend
    if backend == "auto":
        backend = "nccl" if use_gpu else "gloo"
    if use_gpu:
        torch.cuda.set_device(local_rank % ndev)
    dist.init_process_group(backend=backend)

    device = torch.device("cuda") if (use_gpu and backend == "nccl") else torch.device("cpu")

    # Probe the backend with one tiny allreduce (this is where RCCL rejects
    # co-located ranks); surface the failure instead of hanging the run.
    t = torch.zeros(1, dtype=torch.int32, device=device)
    dist.all_reduce(t)

    # Timed collective wrappers around the native engine's DistCtx.
    lat_bcast = []
    lat_allreduce = []

    def bcast(data: bytes, root: int) -> bytes:
        t0 = time.perf_counter()
        buf = torch.frombuffer(bytearray(data), dtype=torch.uint8).to(device)
        dist.broadcast(buf, src=root)
        out = bytes(buf.cpu().numpy().tobytes())
        lat_bcast.append(time.perf_counter() - t0)
        return out

    def allreduce_min(v: int) -> int:
        t0 = time.perf_counter()
        buf = torch.tensor([v], dtype=torch.int32, device=device)
        dist.all_reduce(buf, op=dist.ReduceOp.MIN)
        r = int(buf.item())
        lat_allreduce.append(time.perf_counter() - t0)
        return r

    ctx = _core.PyDistCtx(rank, world, bcast, allreduce_min)

    sbox, n = models.load(args.sbox)
    eng = make_engine(lut_graph=True, seed=11, gpu="force" if use_gpu else "off",
                      save_states=False, oneoutput=args.bit, ctx=ctx)
    eng.set_sbox(sbox, n)

    t0 = time.perf_counter()
    result = {"rank": rank, "world": world, "backend": backend,
              "device": str(device), "ndev": ndev}
    if rank != 0:
        eng.worker_loop()
        result["role"] = "worker"
    else:
        st = eng.initial_state()
        out = eng.create_circuit(st, eng.target(args.bit), mask_for_inputs(n))
        eng.stop_workers()
        assert out >= 0, "no circuit found"
        st.set_output(args.bit, out)
        assert validate_circuit(st, sbox, n, bit=args.bit)
        result["role"] = "driver"
        result["gates"] = st.num_gates - st.num_inputs
        result["stats"] = {k: int(v) for k, v in eng.stats().items()}
    result["wall_s"] = time.perf_counter() - t0

    def stats(xs):
        if not xs:
            return None
        xs_us = sorted(x * 1e6 for x in xs)
        return {
            "count": len(xs_us),
            "mean_us": statistics.fmean(xs_us),
            "p50_us": xs_us[len(xs_us) // 2],
            "p95_us": xs_us[int(len(xs_us) * 0.95)],
            "max_us": xs_us[-1],
        }

    result["allreduce_min"] = stats(lat_allreduce)
    result["bcast"] = stats(lat_bcast)

    os.makedirs(os.path.dirname(args.out) or ".", exist_ok=True)
    with open(f"{args.out}.rank{rank}.json", "w") as f:
        json.dump(result, f, indent=1)
    if rank == 0:
        print(json.dumps(result))
    dist.barrier()
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
