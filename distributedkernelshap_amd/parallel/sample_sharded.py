"""Sample-axis sharded explanation — the sequence-parallel analogue for
KernelSHAP.

When a *single* instance's perturbation set (nsamples x background x
features) exceeds one GPU's capacity/latency budget, the coalition-sample
axis is sharded across ranks: every rank evaluates the model on its
contiguous slice of the nsamples rows, builds partial WLS normal equations
(the Gram matrix and rhs are plain sums over samples), and one small
all-reduce of ``(M-1)^2 + (M-1)*n_out`` floats recovers the exact
single-device solve — communication is independent of nsamples and
background size.

Masks are generated identically on every rank (counter-based RNG), so no
mask traffic is needed either.
"""
from __future__ import annotations

from typing import List, Optional

import numpy as np

from ..core.sampler import sample_masks
from .engine import is_distributed, shard_bounds

__all__ = ["explain_sample_sharded"]


def explain_sample_sharded(
    engine,
    X: np.ndarray,
    nsamples: Optional[int] = None,
) -> List[np.ndarray]:
    """Explain X with the nsamples axis sharded over the process group.

    Every rank returns the full per-class shap matrices (bitwise identical
    across ranks up to all-reduce summation order). Falls back to a plain
    ``shap_values`` call when torch.distributed is not initialised.
    """
    import torch
    import torch.distributed as dist

    if not is_distributed():
        return engine.shap_values(X, nsamples=nsamples)
    if getattr(engine, "_gpu", None) is not None:
        return _explain_sample_sharded_gpu(engine, X, nsamples)
    rank, world = dist.get_rank(), dist.get_world_size()

    X = np.atleast_2d(np.asarray(X, dtype=np.float64))
    b = X.shape[0]
    lfnull = engine.link(engine.fnull)
    fx = np.atleast_2d(np.asarray(engine.predictor(X)))
    total_all = engine.link(fx) - lfnull[None, :]

    phi = np.zeros((b, engine.n_groups, engine.n_out), dtype=np.float64)
    device = "cuda" if dist.get_backend() == "nccl" else "cpu"

    for i in range(b):
        x = X[i]
        varying = engine.varying_groups(x)
        m = len(varying)
        if m == 0:
            continue
        if m == 1:
            phi[i, varying[0]] = total_all[i]
            continue
        plan = engine._plan(m, nsamples)
        masks, kw = sample_masks(plan, engine.seed, i)
        lo, hi = shard_bounds(plan.nsamples, rank, world)
        # local model evaluation on this rank's coalition slice (the O(S*N*D)
        # part); CPU oracle path — the GPU engine's fused kernels cover the
        # single-GPU regime
        ey = engine._ey(x, masks[lo:hi], varying)
        ey_adj = engine.link(ey) - lfnull[None, :]
        # partial normal equations over the local sample slice
        z = masks[lo:hi].astype(np.float64)
        last = z[:, -1]
        ey2 = ey_adj - last[:, None] * total_all[i][None, :]
        etmp = z[:, :-1] - last[:, None]
        wz = etmp * kw[lo:hi, None]
        a_p = wz.T @ etmp                      # (m-1, m-1)
        r_p = wz.T @ ey2                       # (m-1, n_out)
        buf = torch.from_numpy(
            np.concatenate([a_p.ravel(), r_p.ravel()])
        ).to(device)
        dist.all_reduce(buf)                   # sum over sample shards
        flat = buf.cpu().numpy()
        mm = m - 1
        a = flat[: mm * mm].reshape(mm, mm)
        r = flat[mm * mm :].reshape(mm, engine.n_out)
        try:
            w = np.linalg.solve(a, r)
        except np.linalg.LinAlgError:
            w, *_ = np.linalg.lstsq(a, r, rcond=None)
        phi[i, varying[:-1]] = w
        phi[i, varying[-1]] = total_all[i] - w.sum(axis=0)
    return [np.ascontiguousarray(phi[:, :, o]) for o in range(engine.n_out)]


def _explain_sample_sharded_gpu(
    engine,
    X: np.ndarray,
    nsamples: Optional[int] = None,
) -> List[np.ndarray]:
    """GPU sample-sharded path: every rank generates the
    FULL coalition mask set on device (counter RNG — identical across ranks,
    zero mask traffic), runs the fused MFMA predict on its contiguous
    ``nsamples`` slice, builds the partial WLS normal equations device-side in
    fp64 and all-reduces the tiny ``(M-1)^2 + (M-1)*n_out`` buffer over RCCL.
    Communication is independent of nsamples/background size."""
    import torch as t
    import torch.distributed as dist

    gpu = engine._gpu
    rank, world = dist.get_rank(), dist.get_world_size()
    nccl = dist.get_backend() == "nccl"

    X = np.atleast_2d(np.asarray(X, dtype=np.float64))
    b = X.shape[0]
    X_dev = t.tensor(X, dtype=t.float32, device=gpu.device)
    fx = gpu._predict_rows_f64(X_dev)
    lfnull64 = gpu._link(gpu.fnull.double())
    total_all = (gpu._link(fx) - lfnull64[None, :]).float()   # (B, n_out)
    lfnull = lfnull64.float()
    eps = 1e-7

    phi = t.zeros(b, engine.n_groups, engine.n_out, dtype=t.float64,
                  device=gpu.device)
    vmat = gpu._varying_matrix_dev(X_dev).cpu().numpy()
    for i in range(b):
        varying = np.nonzero(vmat[i])[0]
        m = len(varying)
        if m == 0:
            continue
        if m == 1:
            phi[i, int(varying[0])] = total_all[i].double()
            continue
        plan = engine._plan(m, nsamples)
        masks, kw = gpu._device_masks(plan, np.array([i]))    # (1, S, m)
        lo, hi = shard_bounds(plan.nsamples, rank, world)
        sub = masks[:, lo:hi]                                 # view, contiguous
        kws = kw[:, lo:hi]
        sub_X = X_dev[i : i + 1]
        mpad = max(4, (m + 3) // 4 * 4)
        npad = (gpu.N + 15) // 16 * 16
        if gpu.linear is None:
            ey = gpu._ey_torch_module(sub, sub_X, varying)
        elif mpad <= 64 and npad <= 128 and gpu.n_out in (1, 2, 4):
            ey = gpu._ey_fused_linear(sub.contiguous(), sub_X, varying)
        else:
            ey = gpu._ey_linear_torch(sub.contiguous(), sub_X, varying)
        if gpu.link_name == "identity":
            ey_adj = ey - lfnull[None, None, :]
        else:
            p = ey.clamp(eps, 1.0 - eps)
            ey_adj = t.log(p / (1.0 - p)) - lfnull[None, None, :]
        # partial normal equations over the local sample slice (fp64)
        z = sub[0].double()                                   # (s_loc, m)
        last = z[:, -1:]
        etmp = z[:, :-1] - last
        ey2 = ey_adj[0].double() - last * total_all[i].double()[None, :]
        wz = etmp * kws[0].double()[:, None]
        a_p = wz.T @ etmp                                     # (m-1, m-1)
        r_p = wz.T @ ey2                                      # (m-1, n_out)
        buf = t.cat([a_p.reshape(-1), r_p.reshape(-1)])
        if not nccl:
            buf = buf.cpu()
        dist.all_reduce(buf)                                  # sum over shards
        buf = buf.to(gpu.device)
        mm = m - 1
        a = buf[: mm * mm].view(mm, mm)
        r = buf[mm * mm :].view(mm, engine.n_out)
        w = t.linalg.solve(a, r)
        vidx = t.tensor(varying, dtype=t.int64, device=gpu.device)
        phi[i, vidx[:-1]] = w
        phi[i, vidx[-1]] = total_all[i].double() - w.sum(dim=0)
    out = phi.cpu().numpy()
    return [np.ascontiguousarray(out[:, :, o]) for o in range(engine.n_out)]
