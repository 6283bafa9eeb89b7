"""Stage-level numerics diagnosis for the stress shapes (m=200 config of
test_engine_gpu_stress_paths): prints per-stage GPU-vs-fp64 error so a
failure can be attributed to the tiled predict, the Gram build, or the
solve conditioning."""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import numpy as np
import torch as t

from distributedkernelshap_amd.core.engine import KernelShapEngine
from distributedkernelshap_amd.core.links import logit
from distributedkernelshap_amd.models import LinearPredictor, make_tabular

data = make_tabular(n_features=200, n_instances=4, n_background=150, seed=1)
pred = LinearPredictor.random(200, 2, seed=1)
eng = KernelShapEngine(
    pred, data.background, groups=data.groups, link="logit", seed=0,
    device="cuda",
)
gpu = eng._gpu
X = data.X
sv = eng.shap_values(X, nsamples=2048, l1_reg=False)
fx = logit(pred(X))
for o in range(2):
    total = sv[o].sum(axis=1) + eng.expected_value[o]
    print("local acc class", o, np.abs(total - fx[:, o]).max())
cpu = KernelShapEngine(
    pred, data.background, groups=data.groups, link="logit", seed=0,
    device="cpu",
)
sv_c = cpu.shap_values(X, nsamples=2048, l1_reg=False)
for o in range(2):
    print("oracle err class", o, np.abs(sv[o] - sv_c[o]).max())

X_dev = t.tensor(X, dtype=t.float32, device="cuda")
varying = np.arange(200)
plan = eng._plan(200, 2048)
masks, kw = gpu._device_masks(plan, np.arange(4))
ey_t = gpu._ey_fused_tiled(masks, X_dev, varying).clone()
ey_r = gpu._ey_linear_torch(masks, X_dev, varying).clone()
ey_64 = gpu._ey_linear_f64(masks, X_dev, varying)
print("tiled vs torch ey:", (ey_t - ey_r).abs().max().item())
print("tiled vs fp64 ey:", (ey_t - ey_64.float()).abs().max().item())
print("torch vs fp64 ey:", (ey_r - ey_64.float()).abs().max().item())

fx64 = gpu._predict_rows_f64(X_dev)
lfx = gpu._link(fx64)
lfnull64 = gpu._link(gpu.fnull.double())
total_all = (lfx - lfnull64[None]).float()
lfnull = lfnull64.float()
# the engine's saturation-safe pairwise link (the clamp-based transform
# differs by O(1) on saturated rows — that was the round-2 accuracy bug)
ey_adj = gpu._link_ey(ey_t.clone(), lfnull, True)
totalc = total_all.contiguous()
phi_gram = gpu._solve_gram(masks, kw, ey_adj, totalc).clone()
phi_torch = gpu._solve_torch(masks, kw, ey_adj, totalc)
print("gram-solve vs torch-solve phi:", (phi_gram - phi_torch).abs().max().item())

b, s, m = masks.shape
W = (m + 63) // 64
packedw = t.empty(b, s, W, dtype=t.int64, device="cuda")
gpu.ext.pack_masks_words(masks, packedw)
mm = m - 1
a64 = t.empty(b, mm, mm, dtype=t.float64, device="cuda")
r64 = t.empty(b, mm, 2, dtype=t.float64, device="cuda")
gpu.ext.wls_gram(packedw, kw, ey_adj, totalc, a64, r64)
z = masks.double()
last = z[:, :, -1:]
etmp = z[:, :, :-1] - last
ey2 = ey_adj.double() - last * totalc.double()[:, None, :]
wz = etmp * kw.double()[:, :, None]
a_ref = t.bmm(wz.transpose(1, 2), etmp)
r_ref = t.bmm(wz.transpose(1, 2), ey2)
print("gram A err:", (a64 - a_ref).abs().max().item(),
      "scale", a_ref.abs().max().item())
print("gram r err:", (r64 - r_ref).abs().max().item(),
      "scale", r_ref.abs().max().item())
print("cond(A):", np.linalg.cond(a_ref[0].cpu().numpy()))
print("kw range:", kw.min().item(), kw.max().item())
# oracle phi via fp64 solve on fp64 ey (same pairwise link) for isolation
lr = (t.log(ey_64[..., 1].clamp_min(1e-300))
      - t.log(ey_64[..., 0].clamp_min(1e-300)))
ey_adj64 = t.stack([-lr, lr], dim=-1) - lfnull64[None, None, :]
phi_64 = gpu._solve_torch(masks, kw, ey_adj64.float(), totalc,
                          out_dtype=t.float64)
print("gram phi vs fp64-ey phi:", (phi_gram.double() - phi_64).abs().max().item())
