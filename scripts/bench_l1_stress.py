"""Measure the stress config with default kwargs (l1_reg='auto' -> batched
device LARS) vs l1_reg=False — VERDICT r01 item 4's acceptance measurement."""
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import numpy as np

from distributedkernelshap_amd.core.engine import KernelShapEngine
from distributedkernelshap_amd.models import LinearPredictor, make_tabular

data = make_tabular(n_features=256, n_instances=64, n_background=1000, seed=1)
pred = LinearPredictor.random(256, 2, seed=0)
eng = KernelShapEngine(
    pred, data.background, groups=data.groups, link="logit", seed=0,
    device="cuda",
)


def timed(label, **kw):
    import torch

    eng.shap_values(data.X, nsamples=2 ** 14, **kw)   # warm
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    reps = 3
    for _ in range(reps):
        sv = eng.shap_values(data.X, nsamples=2 ** 14, **kw)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / reps
    nz = (np.abs(sv[0]) > 1e-12).sum(axis=1)
    print(f"{label}: {dt*1e3:.1f} ms/batch(64) = {64/dt:,.0f} expl/s; "
          f"support min/mean/max = {nz.min()}/{nz.mean():.0f}/{nz.max()}")
    return dt


t_off = timed("l1_reg=False  ", l1_reg=False)
t_auto = timed("l1_reg='auto' ", l1_reg="auto")
print(f"auto/off ratio: {t_auto/t_off:.2f}x")
