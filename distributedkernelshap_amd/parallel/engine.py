"""Collective data-parallel engine: one process per MI355X GPU over RCCL/xGMI.

Replaces the reference's ray control/data plane (Redis + plasma object store,
SURVEY.md §2.3) with ``torch.distributed``:

* communicator setup   <- ``ray.init``                 (distributed.py:107-109)
* ``broadcast_array``  <- actor-ctor payload shipping  (distributed.py:125-128)
* static ``shard_bounds`` <- ``ActorPool.map_unordered`` scatter (:150-152)
* ``allgather_rows``   <- unordered gather + reorder   (:152-179)

Static contiguous sharding + the per-instance counter RNG make the gathered
result bitwise-identical to the single-rank result (tested in
``tests/test_distributed.py``). Payloads here are KB-to-MB scale, so the xGMI
topology concern (7 p2p links/GPU, ring collectives per-link bound) only
matters for the 1M-instance configs; `all_gather_into_tensor` handles both.
"""
from __future__ import annotations

import datetime
import os
from typing import List, Optional, Tuple

import numpy as np

__all__ = [
    "init_distributed",
    "is_distributed",
    "shard_bounds",
    "broadcast_array",
    "allgather_rows",
    "explain_sharded",
]


def is_distributed() -> bool:
    import torch.distributed as dist

    return dist.is_available() and dist.is_initialized()


def init_distributed(backend: Optional[str] = None) -> Tuple[int, int]:
    """Initialise the process group from torchrun env vars; returns
    (rank, world_size). Backend 'nccl' IS RCCL on ROCm; falls back to gloo
    when no GPU is visible (CPU CI)."""
    import torch
    import torch.distributed as dist

    if dist.is_initialized():
        return dist.get_rank(), dist.get_world_size()
    if "RANK" not in os.environ:
        return 0, 1
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29500")
    dist.init_process_group(backend=backend, timeout=datetime.timedelta(seconds=600))
    if backend == "nccl":
        torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", 0)))
    return dist.get_rank(), dist.get_world_size()


def shard_bounds(n: int, rank: int, world: int) -> Tuple[int, int]:
    """Static contiguous shard [lo, hi) for this rank (deterministic — no
    scheduler needed at 8 ranks)."""
    base, rem = divmod(n, world)
    lo = rank * base + min(rank, rem)
    hi = lo + base + (1 if rank < rem else 0)
    return lo, hi


def broadcast_array(arr: Optional[np.ndarray], src: int = 0) -> np.ndarray:
    """Broadcast a numpy array from src to all ranks (fit-time model weights /
    background set ship — RCCL broadcast over xGMI on the GPU path)."""
    import torch
    import torch.distributed as dist

    if not is_distributed():
        return arr
    device = "cuda" if dist.get_backend() == "nccl" else "cpu"
    rank = dist.get_rank()
    if rank == src:
        meta = torch.tensor(
            [arr.ndim] + list(arr.shape) + [0] * (8 - arr.ndim), dtype=torch.int64
        ).to(device)
    else:
        meta = torch.zeros(9, dtype=torch.int64, device=device)
    dist.broadcast(meta, src=src)
    ndim = int(meta[0].item())
    shape = tuple(int(x) for x in meta[1 : 1 + ndim])
    if rank == src:
        t = torch.from_numpy(np.ascontiguousarray(arr, dtype=np.float64)).to(device)
    else:
        t = torch.empty(shape, dtype=torch.float64, device=device)
    dist.broadcast(t, src=src)
    return t.cpu().numpy()


def allgather_rows(local, counts: List[int]):
    """All-gather variable-row-count blocks; returns the concatenation in
    rank order (the result gather of SURVEY.md §2.3 — per-instance shap rows
    over xGMI).

    A torch-tensor input stays on its device end to end: fp32, padded by at
    most ``max(counts) - count`` rows (static sharding makes that 0 or 1),
    one RCCL ``all_gather_into_tensor`` over xGMI, sliced and concatenated
    device-side. Numpy input keeps the legacy fp64 host staging (used by the
    CPU/gloo paths and old callers)."""
    import torch
    import torch.distributed as dist

    if not is_distributed():
        return local
    world = dist.get_world_size()
    maxc = max(counts)
    if torch.is_tensor(local):
        t = local  # dtype preserved: fp32 from the GPU engine, fp64 CPU oracle
        src_device = t.device
        nccl = dist.get_backend() == "nccl"
        if nccl and not t.is_cuda:
            t = t.cuda()
        elif not nccl and t.is_cuda:
            t = t.cpu()  # gloo transport (e.g. 2 ranks sharing one GPU)
        if t.shape[0] < maxc:  # pad ≤1 row (shard_bounds remainder)
            padrow = t.new_zeros((maxc - t.shape[0],) + tuple(t.shape[1:]))
            t = torch.cat([t, padrow], dim=0)
        t = t.contiguous()
        out = t.new_empty((world * maxc,) + tuple(t.shape[1:]))
        dist.all_gather_into_tensor(out, t)
        if not all(c == maxc for c in counts):
            blocks = [out[r * maxc : r * maxc + counts[r]] for r in range(world)]
            out = torch.cat(blocks, dim=0)
        return out.to(src_device)
    device = "cuda" if dist.get_backend() == "nccl" else "cpu"
    tail = local.shape[1:]
    pad = np.zeros((maxc,) + tail, dtype=np.float64)
    pad[: local.shape[0]] = local
    t = torch.from_numpy(pad).to(device)
    out = torch.empty((world * maxc,) + tail, dtype=torch.float64, device=device)
    dist.all_gather_into_tensor(out, t)
    out = out.cpu().numpy()
    blocks = [out[r * maxc : r * maxc + counts[r]] for r in range(world)]
    return np.concatenate(blocks, axis=0)


def explain_sharded(engine, X, as_tensor: bool = False, **kwargs):
    """Each rank explains its static shard; results all-gathered so every rank
    returns the full per-class shap matrices. Single-process: plain explain.

    The local shard result is taken as a ``(b, n_groups, n_out)`` fp32 device
    tensor (``engine.shap_values(as_tensor=True)``) and gathered with ONE
    device-side RCCL all-gather — no fp64 host bounce, no per-class gathers
    (VERDICT r01 item 1). ``as_tensor=True`` skips the final host conversion
    and returns the gathered ``(n, n_groups, n_out)`` tensor."""
    import torch.distributed as dist

    if not is_distributed():
        return engine.shap_values(X, as_tensor=as_tensor, **kwargs)
    rank, world = dist.get_rank(), dist.get_world_size()
    n = X.shape[0]
    lo, hi = shard_bounds(n, rank, world)
    counts = [shard_bounds(n, r, world)[1] - shard_bounds(n, r, world)[0] for r in range(world)]
    local = engine.shap_values(X[lo:hi], instance_offset=lo, as_tensor=True, **kwargs)
    gathered = allgather_rows(local, counts)
    if as_tensor:
        return gathered
    g = gathered.double().cpu().numpy()
    return [np.ascontiguousarray(g[:, :, o]) for o in range(g.shape[2])]
