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
