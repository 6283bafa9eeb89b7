"""Distribution engine tests: pool mode (process replicas) and the collective
torch.distributed path (gloo, world_size=2) — both must reproduce the
sequential result exactly (SURVEY.md §4 rebuild strategy (d))."""
import multiprocessing as mp
import os

import numpy as np
import pytest

from distributedkernelshap_amd import KernelShap
from distributedkernelshap_amd.explainers.distributed import invert_permutation
from distributedkernelshap_amd.models import LinearPredictor, make_adult_like


def test_invert_permutation():
    p = [3, 0, 2, 1]
    s = invert_permutation(p)
    assert list(s[p]) == [0, 1, 2, 3]


@pytest.fixture(scope="module")
def problem():
    data = make_adult_like(n_instances=8, n_background=20, seed=2)
    pred = LinearPredictor.random(data.X.shape[1], 2, seed=2)
    return data, pred


def _sequential(problem, **kw):
    data, pred = problem
    ks = KernelShap(pred, link="logit", device="cpu")
    ks.fit(data.background, groups=data.groups, group_names=data.group_names)
    return ks.explain(data.X, **kw)


def test_pool_equals_sequential(problem):
    data, pred = problem
    seq = _sequential(problem)
    ks = KernelShap(
        pred,
        link="logit",
        device="cpu",
        distributed_opts={"n_workers": 2, "batch_size": 3},
    )
    ks.fit(data.background, groups=data.groups, group_names=data.group_names)
    par = ks.explain(data.X)
    # masks/weights are bitwise identical (counter RNG); the only allowed
    # divergence is BLAS summation order (forked workers run single-threaded)
    for o in range(2):
        assert np.allclose(par.shap_values[o], seq.shap_values[o], rtol=0, atol=1e-10)
    assert np.allclose(par.expected_value, seq.expected_value)
    ks._explainer.shutdown()


def test_pool_attribute_proxy(problem):
    data, pred = problem
    ks = KernelShap(
        pred, link="logit", device="cpu", distributed_opts={"n_workers": 2}
    )
    ks.fit(data.background, groups=data.groups)
    assert ks._explainer.vector_out is True
    assert ks._explainer.expected_value.shape == (2,)
    ks._explainer.shutdown()


# --------------------------------------------------------------------- #
# collective (torch.distributed gloo) path

def _collective_worker(rank, world, port, ret):
    os.environ.update(
        RANK=str(rank),
        WORLD_SIZE=str(world),
        MASTER_ADDR="127.0.0.1",
        MASTER_PORT=str(port),
        LOCAL_RANK=str(rank),
    )
    import torch.distributed as dist

    from distributedkernelshap_amd.core.engine import KernelShapEngine
    from distributedkernelshap_amd.parallel import explain_sharded, init_distributed

    init_distributed(backend="gloo")
    # 9 instances over 2 ranks: unequal shard counts exercise the
    # allgather pad/trim branch
    data = make_adult_like(n_instances=9, n_background=20, seed=2)
    pred = LinearPredictor.random(data.X.shape[1], 2, seed=2)
    eng = KernelShapEngine(
        pred, data.background, groups=data.groups, link="logit", seed=0, device="cpu"
    )
    sv = explain_sharded(eng, data.X)
    if rank == 0:
        ret.put([s.copy() for s in sv])
    dist.destroy_process_group()


def test_collective_gloo_world2():
    # sequential reference on the SAME 9-instance problem the workers build
    from distributedkernelshap_amd.core.engine import KernelShapEngine

    data = make_adult_like(n_instances=9, n_background=20, seed=2)
    pred = LinearPredictor.random(data.X.shape[1], 2, seed=2)
    eng = KernelShapEngine(
        pred, data.background, groups=data.groups, link="logit", seed=0,
        device="cpu",
    )
    seq = eng.shap_values(data.X)
    ctx = mp.get_context("spawn")
    ret = ctx.Queue()
    port = 29611
    procs = [
        ctx.Process(target=_collective_worker, args=(r, 2, port, ret))
        for r in range(2)
    ]
    for p in procs:
        p.start()
    sv = ret.get(timeout=180)
    for p in procs:
        p.join(timeout=60)
    for o in range(2):
        assert np.allclose(sv[o], seq[o], rtol=0, atol=1e-10)


# --------------------------------------------------------------------- #
# sample-axis sharding (sequence-parallel analogue)

def _sample_sharded_worker(rank, world, port, ret):
    os.environ.update(
        RANK=str(rank),
        WORLD_SIZE=str(world),
        MASTER_ADDR="127.0.0.1",
        MASTER_PORT=str(port),
        LOCAL_RANK=str(rank),
    )
    import torch.distributed as dist

    from distributedkernelshap_amd.core.engine import KernelShapEngine
    from distributedkernelshap_amd.parallel import init_distributed
    from distributedkernelshap_amd.parallel.sample_sharded import (
        explain_sample_sharded,
    )

    init_distributed(backend="gloo")
    data = make_adult_like(n_instances=3, n_background=20, seed=2)
    pred = LinearPredictor.random(data.X.shape[1], 2, seed=2)
    eng = KernelShapEngine(
        pred, data.background, groups=data.groups, link="logit", seed=0,
        device="cpu",
    )
    sv = explain_sample_sharded(eng, data.X)
    if rank == 0:
        ret.put([s.copy() for s in sv])
    dist.destroy_process_group()


def test_sample_sharded_equals_sequential(problem):
    """Sharding the coalition-sample axis + all-reducing the normal
    equations reproduces the single-process solve."""
    data = make_adult_like(n_instances=3, n_background=20, seed=2)
    pred = LinearPredictor.random(data.X.shape[1], 2, seed=2)
    from distributedkernelshap_amd.core.engine import KernelShapEngine

    eng = KernelShapEngine(
        pred, data.background, groups=data.groups, link="logit", seed=0,
        device="cpu",
    )
    seq = eng.shap_values(data.X)

    ctx = mp.get_context("spawn")
    ret = ctx.Queue()
    procs = [
        ctx.Process(target=_sample_sharded_worker, args=(r, 2, 29613, ret))
        for r in range(2)
    ]
    for p in procs:
        p.start()
    sv = ret.get(timeout=180)
    for p in procs:
        p.join(timeout=60)
    for o in range(2):
        assert np.allclose(sv[o], seq[o], rtol=0, atol=1e-8)


def test_pool_dead_worker_tolerated(problem):
    """With the shared task queue (map_unordered-style dynamic dispatch), a
    worker that dies while idle is harmless: the survivors drain the queue
    and the result is still complete and correct."""
    data, pred = problem
    ks = KernelShap(
        pred, link="logit", device="cpu",
        distributed_opts={"n_workers": 2, "batch_size": 2},
    )
    ks.fit(data.background, groups=data.groups)
    seq = KernelShap(pred, link="logit", device="cpu")
    seq.fit(data.background, groups=data.groups)
    expected = seq._explainer._engine.shap_values(data.X)
    # kill one worker behind the pool's back
    ks._explainer._procs[0].terminate()
    ks._explainer._procs[0].join()
    sv = ks._explainer.get_explanation(data.X)
    for o in range(2):
        assert np.allclose(sv[o], expected[o], rtol=0, atol=1e-8)


def test_pool_all_workers_dead_fails_fast(problem):
    """When no worker can make progress the pool must raise, not hang
    (reference quirk SURVEY.md §2.8 / §5.3: map_unordered blocked forever)."""
    data, pred = problem
    ks = KernelShap(
        pred, link="logit", device="cpu",
        distributed_opts={"n_workers": 2, "batch_size": 2},
    )
    ks.fit(data.background, groups=data.groups)
    for p in ks._explainer._procs:
        p.terminate()
        p.join()
    with pytest.raises(RuntimeError, match="died"):
        ks._explainer.get_explanation(data.X)


def test_explain_checkpointed_resume(tmp_path, problem):
    """Interrupted checkpointed run resumes without recomputing and matches
    the uninterrupted result exactly."""
    from distributedkernelshap_amd.core.engine import KernelShapEngine
    from distributedkernelshap_amd.parallel.checkpointed import (
        explain_checkpointed,
    )

    data, pred = problem
    eng = KernelShapEngine(
        pred, data.background, groups=data.groups, link="logit", seed=0,
        device="cpu",
    )
    full = eng.shap_values(data.X)
    out = str(tmp_path / "ckpt")
    # simulate interruption: run only the first chunk, then 'crash'
    calls = {"n": 0}
    orig = eng.shap_values

    def crashing(X, **kw):
        calls["n"] += 1
        if calls["n"] > 1:
            raise KeyboardInterrupt
        return orig(X, **kw)

    eng.shap_values = crashing
    with pytest.raises(KeyboardInterrupt):
        explain_checkpointed(eng, data.X, out, chunk_instances=3)
    eng.shap_values = orig
    # resume: only the remaining chunks run
    before = calls["n"]
    sv = explain_checkpointed(eng, data.X, out, chunk_instances=3)
    for o in range(2):
        assert np.allclose(sv[o], full[o], rtol=0, atol=1e-10)
    # first chunk was not recomputed
    import json as _json

    with open(out + "/manifest.json") as f:
        assert len(_json.load(f)["done"]) == 3


def test_shard_bounds_partition():
    from distributedkernelshap_amd.parallel import shard_bounds

    for n in (0, 1, 5, 7, 8, 2560, 1000003):
        for world in (1, 2, 3, 7, 8):
            spans = [shard_bounds(n, r, world) for r in range(world)]
            # contiguous exact partition, balanced within 1
            assert spans[0][0] == 0 and spans[-1][1] == n
            for (a, b), (c, d) in zip(spans, spans[1:]):
                assert b == c
            sizes = [b - a for a, b in spans]
            assert max(sizes) - min(sizes) <= 1
