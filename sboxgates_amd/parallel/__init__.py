"""Distributed (multi-GPU) search layer: one process per GPU, coordinated
with torch.distributed — backend "nccl" IS RCCL on ROCm, carrying the tiny
control-plane collectives over xGMI; "gloo" covers CPU-only runs and tests.

This replaces the reference's MPI SPMD layer (sboxgates.c:618-642,
lut.c:665-740). The protocol is the chunked symmetric scan implemented in
the native engine (sbg/search.cpp dist_scan_chunked): a ~33 KB work
broadcast per search round, then one 4-byte min-allreduce per chunk and a
20-byte winner broadcast — all latency-bound single-shot collectives
(every payload is far below xGMI's bandwidth regime).
"""

from .. import _core
from ..utils import env_rank


def _torch_dist():
    import torch  # noqa: F401
    import torch.distributed as dist
    return dist


def make_ctx():
    """Builds a PyDistCtx bound to the current torch.distributed process
    group (must already be initialized). Returns None when world size is 1
    (the native engine then uses its LocalCtx)."""
    import torch
    dist = _torch_dist()
    if not dist.is_initialized() or dist.get_world_size() == 1:
        return None
    rank = dist.get_rank()
    world = dist.get_world_size()
    backend = dist.get_backend()
    use_cuda = "nccl" in str(backend) and torch.cuda.is_available()
    device = torch.device("cuda") if use_cuda else torch.device("cpu")

    def bcast(data: bytes, root: int) -> bytes:
        t = torch.frombuffer(bytearray(data), dtype=torch.uint8).to(device)
        dist.broadcast(t, src=root)
        return bytes(t.cpu().numpy().tobytes())

    def allreduce_min(v: int) -> int:
        import torch as _t
        t = _t.tensor([v], dtype=_t.int32, device=device)
        dist.all_reduce(t, op=dist.ReduceOp.MIN)
        return int(t.item())

    return _core.PyDistCtx(rank, world, bcast, allreduce_min)


def init_from_env(backend=None):
    """Initializes torch.distributed from torchrun environment variables.
    Chooses nccl (RCCL) when CUDA/HIP devices are visible, else gloo.
    Returns (rank, world, local_rank). Safe to call with world size 1
    (no-op)."""
    import torch
    rank, world, local = env_rank()
    if world == 1:
        return rank, world, local
    dist = _torch_dist()
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    if backend == "nccl":
        torch.cuda.set_device(local)
    if not dist.is_initialized():
        dist.init_process_group(backend=backend)
    return rank, world, local


def run_search(engine, mode, state=None):
    """SPMD entry: rank 0 drives the search, other ranks serve scan work.

    mode: "graph" (all outputs), "one_output". Rank 0 returns after the
    search and releases the workers; other ranks return when released.
    """
    dist = _torch_dist()
    rank = dist.get_rank() if dist.is_initialized() else 0
    if rank != 0:
        engine.worker_loop()
        return None
    try:
        if state is None:
            state = engine.initial_state()
        if mode == "one_output":
            engine.generate_graph_one_output(state)
        else:
            engine.generate_graph(state)
    finally:
        engine.stop_workers()
    return engine.saved_files()
