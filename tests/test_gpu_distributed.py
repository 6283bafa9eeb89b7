"""On-hardware RCCL tests: 2 ranks sharing one MI355X (RCCL supports multiple
ranks per device), exercising the real nccl(=RCCL) backend that the driver's
multi-GPU SCALE run uses — the device-side fp32 result gather
(``explain_sharded``/``allgather_rows``) and the sample-sharded Gram
all-reduce."""
import multiprocessing as mp
import os

import numpy as np
import pytest

pytestmark = pytest.mark.gpu


def _rccl_env(rank, world, port):
    os.environ.update(
        RANK=str(rank),
        WORLD_SIZE=str(world),
        MASTER_ADDR="127.0.0.1",
        MASTER_PORT=str(port),
        LOCAL_RANK="0",          # both ranks on the single leased GPU
    )


def _sharded_worker(rank, world, port, ret):
    _rccl_env(rank, world, port)
    import torch
    import torch.distributed as dist

    from distributedkernelshap_amd.core.engine import KernelShapEngine
    from distributedkernelshap_amd.models import LinearPredictor, make_adult_like
    from distributedkernelshap_amd.parallel import explain_sharded, init_distributed

    init_distributed(backend="nccl")
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
    out = ret.get(timeout=600)
    for p in procs:
        p.join(timeout=120)
        assert p.exitcode == 0
    return out


def test_rccl_sharded_gather_matches_single_rank():
    """2-rank RCCL explain_sharded (device fp32 all_gather_into_tensor) must
    equal the 1-rank GPU result bitwise: static sharding + the per-instance
    counter RNG make each shard's pipeline identical, and the gather is a
    pure concatenation."""
    sv, single = _spawn2(_sharded_worker, 29621)
    for o in range(2):
        # both paths round the same fp32 phi through fp64; equality is exact
        assert np.array_equal(sv[o], single[o])


def _sample_sharded_worker(rank, world, port, ret):
    _rccl_env(rank, world, port)
    import torch
    import torch.distributed as dist

    from distributedkernelshap_amd.core.engine import KernelShapEngine
    from distributedkernelshap_amd.models import LinearPredictor, make_adult_like
    from distributedkernelshap_amd.parallel import init_distributed
    from distributedkernelshap_amd.parallel.sample_sharded import (
        explain_sample_sharded,
    )

    init_distributed(backend="nccl")
    torch.cuda.set_device(0)
    data = make_adult_like(n_instances=4, n_background=20, seed=6)
    pred = LinearPredictor.random(data.X.shape[1], 2, seed=6)
    eng = KernelShapEngine(
        pred, data.background, groups=data.groups, link="logit", seed=0,
        device="cuda",
    )
    sv = explain_sample_sharded(eng, data.X)
    if rank == 0:
        # CPU fp64 oracle of the same problem for the accuracy bound
        cpu = KernelShapEngine(
            pred, data.background, groups=data.groups, link="logit", seed=0,
            device="cpu",
        )
        ref = cpu.shap_values(data.X)
        ret.put(([s.copy() for s in sv], [s.copy() for s in ref]))
    dist.barrier()
    dist.destroy_process_group()


def test_rccl_sample_sharded_gpu_matches_oracle():
    """GPU sample-sharded mode (fused predict per nsamples slice + RCCL
    all-reduce of the WLS normal equations) matches the CPU fp64 oracle to
    fp32 pipeline tolerance, and satisfies local accuracy by construction."""
    sv, ref = _spawn2(_sample_sharded_worker, 29623)
    for o in range(2):
        err = np.abs(sv[o] - ref[o]).max()
        assert err < 2e-3, f"class {o} max err {err}"
