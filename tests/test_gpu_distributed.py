"""On-hardware distributed tests on a single MI355X.

RCCL (like NCCL) supports one rank per GPU — two ranks sharing a device
deadlock in the collective — so the multi-process tests here run world=2
over gloo while BOTH ranks do their compute on cuda:0 (the full GPU sharded
pipeline, transport swapped), plus a world=1 RCCL group that exercises the
real nccl-backend code path of ``broadcast_array``/``allgather_rows`` on
device tensors. The true N-rank RCCL transport is exactly what the driver's
round-end SCALE run measures on an 8-GPU node (one rank per GPU)."""
import multiprocessing as mp
import os

import numpy as np
import pytest

pytestmark = pytest.mark.gpu


def _env(rank, world, port):
    os.environ.update(
        RANK=str(rank),
        WORLD_SIZE=str(world),
        MASTER_ADDR="127.0.0.1",
        MASTER_PORT=str(port),
        LOCAL_RANK="0",          # both ranks compute on the single leased GPU
    )


def _sharded_worker(rank, world, port, ret):
    _env(rank, world, port)
    import torch
    import torch.distributed as dist

    from distributedkernelshap_amd.core.engine import KernelShapEngine
    from distributedkernelshap_amd.models import LinearPredictor, make_adult_like
    from distributedkernelshap_amd.parallel import explain_sharded, init_distributed

    init_distributed(backend="gloo")   # transport only; compute is cuda:0
    torch.cuda.set_device(0)
    data = make_adult_like(n_instances=9, n_background=20, seed=5)
    pred = LinearPredictor.random(data.X.shape[1], 2, seed=5)
    eng = KernelShapEngine(
        pred, data.background, groups=data.groups, link="logit", seed=0,
        device="cuda",
    )
    sv = explain_sharded(eng, data.X)
    if rank == 0:
        single = eng.shap_values(data.X)  # same GPU, full batch, no sharding
        ret.put((
            [s.copy() for s in sv],
            [s.copy() for s in single],
        ))
    dist.barrier()
    dist.destroy_process_group()


def _spawn2(target, port):
    ctx = mp.get_context("spawn")
    ret = ctx.Queue()
    procs = [ctx.Process(target=target, args=(r, 2, port, ret)) for r in range(2)]
    for p in procs:
        p.start()
    out = ret.get(timeout=420)
    for p in procs:
        p.join(timeout=120)
        assert p.exitcode == 0
    return out


def test_sharded_gather_gpu_matches_single_rank():
    """2-rank explain_sharded with cuda:0 engines must equal the 1-rank GPU
    result bitwise: static sharding + the per-instance counter RNG make each
    shard's pipeline identical, and the gather is a pure concatenation."""
    sv, single = _spawn2(_sharded_worker, 29621)
    for o in range(2):
        assert np.array_equal(sv[o], single[o])


def _sample_sharded_worker(rank, world, port, ret):
    _env(rank, world, port)
    import torch
    import torch.distributed as dist

    from distributedkernelshap_amd.core.engine import KernelShapEngine
    from distributedkernelshap_amd.models import LinearPredictor, make_adult_like
    from distributedkernelshap_amd.parallel import init_distributed
    from distributedkernelshap_amd.parallel.sample_sharded import (
        explain_sample_sharded,
    )

    init_distributed(backend="gloo")
    torch.cuda.set_device(0)
    data = make_adult_like(n_instances=4, n_background=20, seed=6)
    pred = LinearPredictor.random(data.X.shape[1], 2, seed=6)
    eng = KernelShapEngine(
        pred, data.background, groups=data.groups, link="logit", seed=0,
        device="cuda",
    )
    sv = explain_sample_sharded(eng, data.X)
    if rank == 0:
        # single-GPU reference on the SAME device masks (counter RNG keyed by
        # instance id) — a CPU oracle would draw different random coalitions
        # and differ by sampling noise, not pipeline error
        ref = eng.shap_values(data.X)
        ret.put(([s.copy() for s in sv], [s.copy() for s in ref]))
    dist.barrier()
    dist.destroy_process_group()


def test_sample_sharded_gpu_matches_single_gpu():
    """GPU sample-sharded mode (fused predict per nsamples slice + all-reduce
    of the WLS normal equations) matches the unsharded single-GPU result on
    the same masks to fp32 pipeline tolerance."""
    sv, ref = _spawn2(_sample_sharded_worker, 29623)
    for o in range(2):
        err = np.abs(sv[o] - ref[o]).max()
        assert err < 2e-3, f"class {o} max err {err}"


def test_rccl_world1_device_collectives():
    """A real RCCL (nccl-backend) process group on the MI355X: the device
    broadcast/all-gather helpers run on the actual collective path the
    driver's multi-GPU SCALE run uses (world=1 is the one topology a single
    GPU supports — RCCL is one rank per device)."""
    import torch
    import torch.distributed as dist

    _env(0, 1, 29625)
    from distributedkernelshap_amd.parallel import (
        allgather_rows,
        broadcast_array,
        init_distributed,
    )

    rank, world = init_distributed(backend="nccl")
    assert (rank, world) == (0, 1)
    assert dist.get_backend() == "nccl"
    arr = np.arange(12.0).reshape(3, 4)
    out = broadcast_array(arr)
    assert np.array_equal(out, arr)
    t = torch.arange(24.0, device="cuda").reshape(4, 3, 2)
    g = allgather_rows(t, [4])
    assert g.is_cuda and torch.equal(g, t)
    dist.destroy_process_group()
