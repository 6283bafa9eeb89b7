"""GPU KernelSHAP pipeline: orchestrates the CDNA4 HIP kernels.

Per instance-bucket (instances sharing a varying-group pattern):

  enum masks (host, cached per plan)  ──┐
  fill_random_masks  (HIP K2, Philox)  ─┴─> masks (B,S,M) u8 on device
  fused_predict_linear (HIP K3-K6, MFMA)  -> ey (B,S,n_out)   [linear predictor]
     — or — synth_chunk (HIP K3') + torch module + weighted mean  [torch predictor]
  link transform (torch, fp32)           -> eyAdj
  wls_solve (HIP K7)                     -> phi (B,M,n_out)

The reference's equivalent work is the per-instance shap 0.35.0 loop invoked
from ``explainers/kernel_shap.py:250``: sampling -> synth -> predict_proba ->
mean -> WLS, one instance at a time on one CPU core. Here the whole
2,560-instance batch is a handful of kernel launches.
"""
from __future__ import annotations

import logging
from typing import List, Optional

import numpy as np

from . import load_extension
from ..core.sampler import CoalitionPlan

logger = logging.getLogger(__name__)

import os as _os
import time as _time

_TIMING = _os.environ.get("KSHAP_TIMING", "") == "1"


class _StageTimer:
    """Per-stage wall timing: prints when KSHAP_TIMING=1, records into a
    sink dict when tracing is enabled (SURVEY.md §5.1 — the reference only
    had whole-run timeit wall clocks; this gives per-phase numbers). Each
    mark syncs the device, so enable only for diagnosis."""

    def __init__(self, torch_mod, enabled, sink=None):
        self.t = torch_mod
        self.enabled = enabled or sink is not None
        self.sink = sink
        self.last = _time.perf_counter() if self.enabled else 0.0

    def mark(self, name):
        if not self.enabled:
            return
        self.t.cuda.synchronize()
        now = _time.perf_counter()
        ms = (now - self.last) * 1e3
        if _TIMING:
            print(f"[kshap-timing] {name}: {ms:.2f} ms", flush=True)
        if self.sink is not None:
            self.sink.setdefault(name, []).append(ms)
        self.last = now

_EPS = 1e-7  # fp32 logit clamp


class GpuKernelShap:
    """Device-side state + pipeline for one fitted engine."""

    def __init__(self, engine):
        import torch

        self.torch = torch
        self.engine = engine
        self.ext = load_extension()
        self.device = torch.device("cuda")
        t = torch

        self.bg = t.tensor(engine.background, dtype=t.float32, device=self.device)
        self.bg_w = t.tensor(engine.bg_weights, dtype=t.float32, device=self.device)
        self.col_group = t.tensor(engine._col_group, dtype=t.int64, device=self.device)
        self.n_groups = engine.n_groups
        self.n_out = engine.n_out
        self.N, self.D = self.bg.shape
        self.link_name = engine.link_name

        self.linear = None
        self.module = None
        lp = getattr(engine.predictor, "linear_params", None)
        if lp is not None:
            W, b, act = lp()
            self.linear = {
                "W": t.tensor(np.asarray(W), dtype=t.float32, device=self.device),
                "b": t.tensor(np.asarray(b), dtype=t.float32, device=self.device),
                "act": {"none": 0, "sigmoid": 1, "softmax": 2}[act],
            }
            # bg_part[n, g, o] = sum_{j in g} bg[n, j] * W[o, j]
            contrib = self.bg[:, :, None] * self.linear["W"].T[None, :, :]  # (N,D,o)
            bg_part = t.zeros(self.N, self.n_groups, self.n_out, device=self.device)
            bg_part.index_add_(1, self.col_group, contrib)
            self.bg_part = bg_part
            self.baseN = self.bg @ self.linear["W"].T + self.linear["b"]  # (N, o)
        else:
            tm = getattr(engine.predictor, "torch_module", None)
            if tm is not None:
                self.module = tm().to(self.device)
                if (engine.kernels.module_channels_last
                        and engine.kernels.module_autocast == "bf16"):
                    # conv nets: NHWC weights select MIOpen's fast bf16
                    # paths (measured: resnet bf16 12.4 -> 14.4 expl/s, but
                    # fp32 NCHW is FASTER than NHWC — 5.4 vs 3.2 — so the
                    # layout switch is tied to autocast)
                    try:
                        self.module = self.module.to(
                            memory_format=t.channels_last
                        )
                    except Exception:  # pragma: no cover
                        pass
            else:
                raise TypeError(
                    "GPU engine needs a predictor exposing linear_params() or "
                    "torch_module(); got a plain callable. Wrap it in "
                    "TorchPredictor or use device='cpu'."
                )

        self.fnull = t.tensor(engine.fnull, dtype=t.float32, device=self.device)
        # varying-group check via column extrema: x differs from ALL
        # background rows iff it differs from both extremes (O(B*D) instead
        # of O(B*N*D) per call)
        self.bg_min = self.bg.min(dim=0).values
        self.bg_max = self.bg.max(dim=0).values
        self._enum_cache: dict = {}
        # persistent workspaces: identical shapes every call, so reusing the
        # same device blocks avoids caching-allocator churn (sporadic ~90 ms
        # hipMalloc stalls measured at B=2560)
        self._ws: dict = {}
        self.trace: dict = None  # per-stage ms lists when tracing enabled
        # hipGraph capture of the fused pipeline (~60 fixed-shape launches):
        # keyed by (B, nsamples, varying pattern, offset); disable with
        # KSHAP_GRAPH=0 or automatically on capture failure
        self._graphs: dict = {}
        self._graph_hits: dict = {}
        # speculative replay state: (probe_key0, graph key) of the last
        # successful single-bucket graph call
        self._spec = None
        self._graphs_enabled = _os.environ.get("KSHAP_GRAPH", "1") == "1"
        # capture costs ~100 ms: only worth it for recurring shapes (a
        # serving workload with varying batch sizes must stay eager)
        self._graph_min_hits = int(_os.environ.get("KSHAP_GRAPH_MIN_HITS", "2"))

    def enable_tracing(self, on: bool = True) -> None:
        self.trace = {} if on else None

    def _buf(self, name, shape, dtype=None, zeroed=False):
        t = self.torch
        dtype = dtype or t.float32
        key = (name, tuple(shape), dtype)   # per-shape: captured hipGraphs
        buf = self._ws.get(key)             # hold raw pointers to these
        if buf is None:
            buf = (t.zeros if zeroed else t.empty)(
                *shape, dtype=dtype, device=self.device
            )
            self._ws[key] = buf
        return buf

    # ------------------------------------------------------------------ #

    def _link(self, p):
        t = self.torch
        if self.link_name == "identity":
            return p
        eps = 1e-15 if p.dtype == t.float64 else _EPS
        p = t.clamp(p, eps, 1.0 - eps)
        return t.log(p / (1.0 - p))

    def _predict_rows_f64(self, rows):
        """fp64 predict for the per-instance totals: link('logit') amplifies
        probability saturation (p ~ 1-1e-8 clips to 1-1e-7 in fp32, shifting
        logit by >2), so fx/fnull go through fp64 — B rows only, negligible."""
        t = self.torch
        if self.linear is not None:
            z = rows.double() @ self.linear["W"].double().T + self.linear["b"].double()
            a = self.linear["act"]
            if a == 1:
                return t.sigmoid(z)
            if a == 2:
                return t.softmax(z, dim=-1)
            return z
        with t.no_grad():
            return self.module(rows).double()

    def _predict_rows(self, rows):
        """Run the predictor on a device tensor of rows -> (n, n_out) fp32.
        ``module_autocast='bf16'`` wraps the module forward in bf16 autocast
        (MI355X bf16 matrix cores are 16x the f32 MFMA rate) — the lever for
        the predict-bound mlp/resnet configs."""
        t = self.torch
        if self.linear is not None:
            z = rows @ self.linear["W"].T + self.linear["b"]
            a = self.linear["act"]
            if a == 1:
                return t.sigmoid(z)
            if a == 2:
                return t.softmax(z, dim=-1)
            return z
        with t.no_grad():
            if self.engine.kernels.module_autocast == "bf16":
                with t.autocast(device_type="cuda", dtype=t.bfloat16):
                    out = self.module(rows)
            else:
                out = self.module(rows)
        return out.float()

    def _x_part(self, X_dev):
        """x_part[b, g, o] = sum_{j in g} x[b, j] * W[o, j]."""
        t = self.torch
        contrib = X_dev[:, :, None] * self.linear["W"].T[None, :, :]
        x_part = t.zeros(
            X_dev.shape[0], self.n_groups, self.n_out, device=self.device
        )
        x_part.index_add_(1, self.col_group, contrib)
        return x_part

    # ------------------------------------------------------------------ #

    def _device_masks(self, plan: CoalitionPlan, inst_ids: np.ndarray,
                      ids_dev=None):
        """masks (B,S,M) u8 + kernel weights (B,S) f32 on device.
        ``ids_dev`` overrides the RNG-key tensor (hipGraph static input)."""
        t = self.torch
        b = len(inst_ids)
        s, m = plan.nsamples, plan.m
        ne = plan.enum_masks.shape[0]
        key = (m, s)
        if key not in self._enum_cache:
            enum = t.tensor(plan.enum_masks, dtype=t.uint8, device=self.device)
            ew = t.tensor(plan.enum_weights, dtype=t.float32, device=self.device)
            cdf = t.tensor(
                np.cumsum(plan.random_size_probs).astype(np.float32),
                device=self.device,
            )
            szs = t.tensor(plan.random_sizes.astype(np.int32), device=self.device)
            self._enum_cache[key] = (enum, ew, cdf, szs)
        enum, ew, cdf, szs = self._enum_cache[key]

        masks = self._buf("masks", (b, s, m), t.uint8)
        masks[:, :ne] = enum[None]
        kw = t.empty(s, dtype=t.float32, device=self.device)
        kw[:ne] = ew
        if plan.n_random > 0:
            num_paired = int(np.floor((m - 1) / 2.0))
            if ids_dev is not None:
                ids = ids_dev
            else:
                base0 = int(inst_ids[0])
                if np.array_equal(inst_ids, np.arange(base0, base0 + b)):
                    # contiguous ids (common bucket): device arange, no H2D
                    ids = t.arange(base0, base0 + b, dtype=t.int32,
                                   device=self.device)
                else:
                    ids = t.tensor(inst_ids.astype(np.int32), device=self.device)
            self.ext.fill_random_masks(
                masks, ne, plan.n_random, cdf, szs, num_paired,
                int(self.engine.seed), ids,
            )
            kw[ne:] = plan.weight_left / plan.n_random
        kwb = self._buf("kwb", (b, s))
        kwb.copy_(kw[None, :].expand(b, s))
        return masks, kwb

    # ------------------------------------------------------------------ #

    def _diff_tensor(self, X_dev, varying, vidx_t=None):
        """diff[b, o, k, n] = x_part[b, k, o] - bg_part[n, k, o] over varying
        groups k (the algebraically folded masked-background blend)."""
        t = self.torch
        vidx = (vidx_t if vidx_t is not None
                else t.tensor(varying, dtype=t.int64, device=self.device))
        xv = self._x_part(X_dev)[:, vidx]       # (b, m, o)
        bgv = self.bg_part[:, vidx]             # (N, m, o)
        return xv.permute(0, 2, 1)[:, :, :, None] - bgv.permute(2, 1, 0)[None]

    def _ey_fused_linear(self, masks, X_dev, varying, vidx_t=None):
        """K3-K6 fused MFMA path (linear predictor, Mpad<=64, Npad<=128).
        The A operand is converted in-register from the raw u8 masks."""
        t = self.torch
        b, s, m = masks.shape
        mpad = max(4, (m + 3) // 4 * 4)
        npad = (self.N + 15) // 16 * 16
        if vidx_t is None:
            vidx_t = t.tensor(varying, dtype=t.int64, device=self.device)
        act, oimg = self._act_oimg()
        # pad slots are zeroed once at allocation and never written after
        # (buffer name keyed by m so a different varying count cannot see
        # stale values)
        diff = self._buf(f"diff{m}", (b, oimg, mpad, npad), zeroed=True)
        self.ext.build_diff_f32(
            self._x_part_img(X_dev, act), self._bg_part_img(act), vidx_t, diff
        )
        base = t.zeros(oimg, npad, device=self.device)
        base[:, : self.N] = self._base_img(act)
        wbg = t.zeros(npad, device=self.device)
        wbg[: self.N] = self.bg_w
        ey = self._buf("ey", (b, s, self.n_out))
        self.ext.fused_predict_linear(masks, diff, base, wbg, ey, act)
        return ey

    def _act_oimg(self):
        """Binary softmax runs on the logit DIFFERENCE (one operand image,
        half the MFMAs): act code 3."""
        act = self.linear["act"]
        if act == 2 and self.n_out == 2:
            return 3, 1
        return act, self.n_out

    def _x_part_img(self, X_dev, act):
        xp = self._x_part(X_dev)                       # (b, G, o)
        if act == 3:
            return (xp[:, :, 1:2] - xp[:, :, 0:1]).contiguous()
        return xp

    def _bg_part_img(self, act):
        if act == 3:
            if not hasattr(self, "_bg_part_d"):
                self._bg_part_d = (
                    self.bg_part[:, :, 1:2] - self.bg_part[:, :, 0:1]
                ).contiguous()
            return self._bg_part_d
        return self.bg_part

    def _base_img(self, act):
        if act == 3:
            return (self.baseN[:, 1] - self.baseN[:, 0])[None, :]
        return self.baseN.T

    def _ey_fused_bf16(self, masks, X_dev, varying, vidx_t=None, packed=None):
        """bf16 matrix-core predict (predict_dtype 'bf16x2'/'bf16'): one
        v_mfma_f32_16x16x32_bf16 spans the whole K dim; 'bf16x2' adds a lo
        correction image (B = hi + lo) for fp32-grade accuracy."""
        t = self.torch
        b, s, m = masks.shape
        npad = (self.N + 15) // 16 * 16
        split = 2 if self.engine.kernels.predict_dtype == "bf16x2" else 1
        if packed is not None:
            self.ext.pack_masks(masks, packed)   # graph-path convenience
        if vidx_t is None:
            vidx_t = t.tensor(varying, dtype=t.int64, device=self.device)
        act, oimg = self._act_oimg()
        diffB = self._buf(
            f"diffB{m}", (b, split, oimg, npad, 40), t.bfloat16, zeroed=True
        )
        self.ext.build_diff_bf16(
            self._x_part_img(X_dev, act), self._bg_part_img(act), vidx_t, diffB
        )
        base = t.zeros(oimg, npad, device=self.device)
        base[:, : self.N] = self._base_img(act)
        wbg = t.zeros(npad, device=self.device)
        wbg[: self.N] = self.bg_w
        ey = self._buf("ey", (b, s, self.n_out))
        self.ext.fused_predict_bf16(masks, diffB, base, wbg, ey, act)
        return ey

    def _ey_fused_tiled(self, masks, X_dev, varying, vidx_t=None):
        """Tiled MFMA fused predict for the stress shapes (Mpad>64 or
        Npad>128): diff streamed through LDS in
        (k-chunk x 128-col) tiles, per-column-tile partial reductions summed
        by a deterministic second kernel (no float atomics)."""
        t = self.torch
        b, s, m = masks.shape
        # Mpad rounds to the kernel's 16-deep LDS k-chunks so the inner MFMA
        # loop is fully unrolled (pad k-slices are zeros in the diff image)
        mpad = max(16, (m + 15) // 16 * 16)
        npad = (self.N + 15) // 16 * 16
        if vidx_t is None:
            vidx_t = t.tensor(varying, dtype=t.int64, device=self.device)
        act, oimg = self._act_oimg()
        diff = self._buf(f"difft{m}", (b, oimg, mpad, npad), zeroed=True)
        self.ext.build_diff_f32(
            self._x_part_img(X_dev, act), self._bg_part_img(act), vidx_t, diff
        )
        base = t.zeros(oimg, npad, device=self.device)
        base[:, : self.N] = self._base_img(act)
        wbg = t.zeros(npad, device=self.device)
        wbg[: self.N] = self.bg_w
        n_ntiles = (npad + 127) // 128
        partial = self._buf("ftpart", (b, n_ntiles, s, self.n_out))
        ey = self._buf("ey", (b, s, self.n_out))
        self.ext.fused_predict_tiled(masks, diff, base, wbg, partial, ey, act)
        return ey

    def _ey_linear_f64(self, masks, X_dev, varying, s_chunk=2048):
        """fp64 verification path (predict_dtype='fp64'):
        the SAME device coalition masks evaluated end-to-end in fp64 (weights,
        diff, GEMM, activation, background reduction all double). Perf is
        irrelevant — this mode exists so bench.py can print a measured
        ``max_phi_err_vs_fp64`` for the fp32 pipeline, making every headline
        number self-certifying (the reference computes in numpy fp64
        throughout, shap 0.35.0 via ``explainers/kernel_shap.py:250``)."""
        t = self.torch
        b, s, m = masks.shape
        W = self.linear["W"].double()
        bias = self.linear["b"].double()
        bg = self.bg.double()
        X64 = X_dev.double()
        contrib_x = X64[:, :, None] * W.T[None, :, :]
        x_part = t.zeros(b, self.n_groups, self.n_out, dtype=t.float64,
                         device=self.device)
        x_part.index_add_(1, self.col_group, contrib_x)
        contrib_b = bg[:, :, None] * W.T[None, :, :]
        bg_part = t.zeros(self.N, self.n_groups, self.n_out, dtype=t.float64,
                          device=self.device)
        bg_part.index_add_(1, self.col_group, contrib_b)
        vidx = t.tensor(varying, dtype=t.int64, device=self.device)
        diff = (x_part[:, vidx].permute(0, 2, 1)[:, :, :, None]
                - bg_part[:, vidx].permute(2, 1, 0)[None])       # (b,o,m,N)
        diff = diff.permute(0, 2, 3, 1).reshape(b, m, self.N * self.n_out)
        base = bg @ W.T + bias                                    # (N, o)
        wbg64 = self.bg_w.double()
        ey = t.empty(b, s, self.n_out, dtype=t.float64, device=self.device)
        for lo in range(0, s, s_chunk):
            hi = min(lo + s_chunk, s)
            logits = t.bmm(masks[:, lo:hi].double(), diff)
            logits = logits.view(b, hi - lo, self.N, self.n_out) + base[None, None]
            a = self.linear["act"]
            if a == 1:
                p = t.sigmoid(logits)
            elif a == 2:
                p = t.softmax(logits, dim=-1)
            else:
                p = logits
            ey[:, lo:hi] = t.einsum("bsno,n->bso", p, wbg64)
        return ey

    def _ey_fused_tiled_bf16(self, masks, X_dev, varying, vidx_t=None):
        """bf16 matrix-core tiled predict (opt-in predict_dtype for the
        stress shapes): one v_mfma_f32_16x16x32_bf16 per 32-deep k block —
        8x the f32 MFMA rate — with the hi+lo split B operand for
        fp32-grade results ('bf16x2')."""
        t = self.torch
        b, s, m = masks.shape
        mpad = max(32, (m + 31) // 32 * 32)     # FTB_KC contract
        npad = (self.N + 15) // 16 * 16
        if vidx_t is None:
            vidx_t = t.tensor(varying, dtype=t.int64, device=self.device)
        act, oimg = self._act_oimg()
        split = 2 if self.engine.kernels.predict_dtype == "bf16x2" else 1
        diffb = self._buf(
            f"difftb{m}", (b, split, oimg, npad, mpad), t.bfloat16,
            zeroed=True,
        )
        self.ext.build_diff_bf16_tiled(
            self._x_part_img(X_dev, act), self._bg_part_img(act), vidx_t,
            diffb,
        )
        base = t.zeros(oimg, npad, device=self.device)
        base[:, : self.N] = self._base_img(act)
        wbg = t.zeros(npad, device=self.device)
        wbg[: self.N] = self.bg_w
        n_ntiles = (npad + 127) // 128
        partial = self._buf("ftpart", (b, n_ntiles, s, self.n_out))
        ey = self._buf("ey", (b, s, self.n_out))
        self.ext.fused_predict_tiled_bf16(
            masks, diffb, base, wbg, partial, ey, act
        )
        return ey

    def _ey_linear_torch(self, masks, X_dev, varying, s_chunk=4096):
        """Library-GEMM fallback for shapes beyond the fused kernel's limits
        (stress configs: M>64 or N>128). ey = reduce(act(mask @ diff + base))."""
        t = self.torch
        b, s, m = masks.shape
        diff = self._diff_tensor(X_dev, varying)          # (b, o, m, N)
        diff = diff.permute(0, 2, 3, 1).reshape(b, m, self.N * self.n_out)
        ey = t.empty(b, s, self.n_out, device=self.device)
        for lo in range(0, s, s_chunk):
            hi = min(lo + s_chunk, s)
            logits = t.bmm(masks[:, lo:hi].float(), diff)
            logits = logits.view(b, hi - lo, self.N, self.n_out) + self.baseN[None, None]
            a = self.linear["act"]
            if a == 1:
                p = t.sigmoid(logits)
            elif a == 2:
                p = t.softmax(logits, dim=-1)
            else:
                p = logits
            ey[:, lo:hi] = t.einsum("bsno,n->bso", p, self.bg_w)
        return ey

    def _ey_torch_module(self, masks, X_dev, varying, chunk_rows=None):
        """K3' synth + torch predictor + weighted mean (arbitrary-predictor
        path). Synth tiles never leave the device; multiple
        instances are packed per predictor call so the python/launch overhead
        amortises over the 1M-instance MLP config."""
        t = self.torch
        b, s, m = masks.shape
        if chunk_rows is None:
            chunk_rows = max(self.N, self.engine.kernels.synth_chunk_rows or 1 << 19)
        # hard cap: the synth buffer must stay ~2 GB regardless of config
        chunk_rows = min(chunk_rows, max(self.N, (1 << 29) // max(1, self.D)))
        # never allocate more than the whole batch needs
        chunk_rows = min(chunk_rows, b * s * self.N)
        ey = t.empty(b, s, self.n_out, device=self.device)
        # masks cover varying groups only; map non-varying columns to a
        # sentinel always-zero mask column m
        vmap = np.full(self.n_groups, m, dtype=np.int64)
        for i, g in enumerate(varying):
            vmap[g] = i
        mfull = self._buf("mfull", (b, s, m + 1), t.uint8)
        mfull.zero_()
        mfull[:, :, :m] = masks
        colg = t.tensor(
            vmap[self.engine._col_group].astype(np.int32), device=self.device
        )
        # under bf16 autocast the synth kernel emits bf16 directly: halves
        # the perturbation-tensor write traffic and removes the whole-tensor
        # fp32->bf16 cast pass (mlp profile: 12.7 + ~6 ms per step)
        sdt = (t.bfloat16 if self.engine.kernels.module_autocast == "bf16"
               else t.float32)
        rows_per_inst = s * self.N
        if rows_per_inst <= chunk_rows:
            # pack g instances per call
            g_inst = max(1, min(b, chunk_rows // rows_per_inst))
            buf = self._buf("synth", (g_inst * rows_per_inst, self.D), sdt)
            for lo in range(0, b, g_inst):
                hi = min(lo + g_inst, b)
                rows = (hi - lo) * rows_per_inst
                out = buf[:rows]
                self.ext.synth_chunk(mfull, X_dev, self.bg, colg, out, lo, hi, 0, s)
                y = self._predict_rows(out)
                ey[lo:hi] = t.einsum(
                    "csno,n->cso",
                    y.view(hi - lo, s, self.N, self.n_out),
                    self.bg_w,
                )
        else:
            s_chunk = max(1, chunk_rows // self.N)
            buf = self._buf("synth", (s_chunk * self.N, self.D), sdt)
            for bi in range(b):
                for lo in range(0, s, s_chunk):
                    hi = min(lo + s_chunk, s)
                    rows = (hi - lo) * self.N
                    out = buf[:rows]
                    self.ext.synth_chunk(
                        mfull, X_dev, self.bg, colg, out, bi, bi + 1, lo, hi
                    )
                    y = self._predict_rows(out)
                    y = y.view(hi - lo, self.N, self.n_out)
                    ey[bi, lo:hi] = t.einsum("cno,n->co", y, self.bg_w)
        return ey

    # ------------------------------------------------------------------ #
    # hipGraph fast path

    def _fused_body(self, X_dev, plan, varying, vidx_t, ids_dev, phi_full):
        """The captureable single-bucket fused pipeline: X_dev -> phi_full.
        Pure device ops (no pageable H2D, no host syncs); instance RNG keys
        come from the static ``ids_dev`` buffer."""
        t = self.torch
        b = X_dev.shape[0]
        m = len(varying)
        fx = self._predict_rows_f64(X_dev)
        lfx = self._link(fx)
        lfnull64 = self._link(self.fnull.double())
        total = (lfx - lfnull64[None, :]).float().contiguous()
        lfnull = lfnull64.float()
        masks, kw = self._device_masks(plan, np.arange(b), ids_dev=ids_dev)
        kc = self.engine.kernels
        packed = self._buf("packed", (b, plan.nsamples), t.int64)
        self.ext.pack_masks(masks, packed)
        if (kc.predict_dtype in ("bf16", "bf16x2") and m <= 32
                and self.n_out in (1, 2, 4)):
            ey = self._ey_fused_bf16(masks, X_dev, varying, vidx_t, packed=None)
        else:
            ey = self._ey_fused_linear(masks, X_dev, varying, vidx_t)
        if self.link_name == "identity":
            ey_adj = ey.sub_(lfnull[None, None, :])
        else:
            ey.clamp_(_EPS, 1.0 - _EPS)
            ey.log_().sub_(t.log1p(-t.exp(ey)))
            ey_adj = ey.sub_(lfnull[None, None, :])
        phi = self._buf("phi", (b, m, self.n_out))
        self.ext.wls_solve(masks, kw, ey_adj, total, phi, packed)
        phi_full.zero_()
        phi_full[:, vidx_t] = phi

    def _graph_explain(self, X_dev, plan, varying, instance_offset,
                       as_tensor=False):
        """Capture-or-replay the fused pipeline; returns shap values or None
        when graphs are unavailable."""
        t = self.torch
        b = X_dev.shape[0]
        # instance ids are a graph INPUT (static buffer), so the key is
        # offset-independent: chunked 1M-instance sweeps replay one graph
        key = (b, plan.nsamples, varying.tobytes())
        entry = self._graphs.get(key)
        if entry is None:
            hits = self._graph_hits.get(key, 0) + 1
            self._graph_hits[key] = hits
            if hits < self._graph_min_hits or len(self._graphs) >= 8:
                return None        # eager until the shape proves recurring
            try:
                vidx_t = t.tensor(varying, dtype=t.int64, device=self.device)
                x_static = self._buf("Xg", (b, self.D))
                ids_static = self._buf("idsg", (b,), t.int32)
                phi_full = t.zeros(
                    b, self.n_groups, self.n_out, device=self.device
                )
                x_static.copy_(X_dev)
                ids_static.copy_(
                    t.arange(instance_offset, instance_offset + b,
                             dtype=t.int32, device=self.device)
                )
                # warmup on a side stream (required before capture)
                side = t.cuda.Stream()
                side.wait_stream(t.cuda.current_stream())
                with t.cuda.stream(side):
                    self._fused_body(
                        x_static, plan, varying, vidx_t, ids_static, phi_full
                    )
                t.cuda.current_stream().wait_stream(side)
                graph = t.cuda.CUDAGraph()
                with t.cuda.graph(graph):
                    self._fused_body(
                        x_static, plan, varying, vidx_t, ids_static, phi_full
                    )
                entry = (graph, x_static, ids_static, phi_full, vidx_t)
                self._graphs[key] = entry
            except Exception as e:  # pragma: no cover - capture support varies
                logger.warning(
                    "hipGraph capture failed (%r); continuing eagerly", e
                )
                self._graphs_enabled = False
                return None
        graph, x_static, ids_static, phi_full, _ = entry
        if x_static.data_ptr() != X_dev.data_ptr():
            x_static.copy_(X_dev)
        ids_static.copy_(
            t.arange(instance_offset, instance_offset + b, dtype=t.int32,
                     device=self.device)
        )
        graph.replay()
        if as_tensor:
            return phi_full.clone()  # static graph output buffer
        out = phi_full.double().cpu().numpy()
        return [np.ascontiguousarray(out[:, :, o]) for o in range(self.n_out)]

    # device-memory budget for per-call mask/workspace tensors; batches
    # whose projected footprint exceeds it are processed in instance chunks
    _CHUNK_BYTES = 2 << 30

    def _instances_per_chunk(self, nsamples: Optional[int]) -> int:
        from ..core.sampler import default_nsamples

        s = nsamples or default_nsamples(max(2, self.n_groups))
        g = self.n_groups
        # masks u8 + packed u64 + kwb f32 + ey f32 (+ column-tile partials
        # on the tiled stress path) per instance
        npad = (self.N + 15) // 16 * 16
        n_ntiles = (npad + 127) // 128
        per_inst = s * (g + 8 + 4 + 4 * self.n_out) + (1 << 14)
        if g > 64 or npad > 128:
            per_inst += s * 4 * self.n_out * n_ntiles        # ftpart
            mpad = max(4, (g + 3) // 4 * 4)
            per_inst += 4 * mpad * npad * max(1, self.n_out) # diff image
            per_inst += 8 * g * (g + self.n_out) * 2         # Gram + LU work
        return max(1, self._CHUNK_BYTES // per_inst)

    def shap_values(
        self,
        X: np.ndarray,
        nsamples: Optional[int] = None,
        l1_reg="auto",
        instance_offset: int = 0,
        as_tensor: bool = False,
    ) -> List[np.ndarray]:
        """``as_tensor=True`` returns the raw ``(B, n_groups, n_out)`` fp32
        device tensor (caller-owned) instead of per-class numpy arrays — the
        distributed gather path all-gathers it over RCCL without any host
        bounce."""
        t = self.torch
        b_total = X.shape[0]
        chunk = self._instances_per_chunk(nsamples)
        if b_total > chunk:
            # large batches stream through in fixed-size chunks (1M-instance
            # configs: the whole-batch mask tensor would be TB-scale)
            if as_tensor:
                out = t.empty(
                    b_total, self.n_groups, self.n_out, device=self.device,
                    dtype=(t.float64
                           if self.engine.kernels.predict_dtype == "fp64"
                           else t.float32),
                )
                for lo in range(0, b_total, chunk):
                    hi = min(lo + chunk, b_total)
                    out[lo:hi] = self.shap_values(
                        X[lo:hi], nsamples=nsamples, l1_reg=l1_reg,
                        instance_offset=instance_offset + lo, as_tensor=True,
                    )
                return out
            parts = [
                self.shap_values(
                    X[lo : lo + chunk], nsamples=nsamples, l1_reg=l1_reg,
                    instance_offset=instance_offset + lo,
                )
                for lo in range(0, b_total, chunk)
            ]
            return [
                np.concatenate([p[o] for p in parts], axis=0)
                for o in range(self.n_out)
            ]
        timer = _StageTimer(t, _TIMING, sink=self.trace)
        if t.is_tensor(X):
            b = X.shape[0]
            if X.is_cuda:
                X_dev = X.float()
            else:
                X_dev = self._buf("X", (b, X.shape[1]))
                X_dev.copy_(X, non_blocking=True)   # pinned source = async DMA
        else:
            X = np.ascontiguousarray(X, dtype=np.float64)
            b = X.shape[0]
            X_dev = t.tensor(X, dtype=t.float32, device=self.device)
        timer.mark("h2d")

        fp64 = self.engine.kernels.predict_dtype == "fp64"
        fx = self._predict_rows_f64(X_dev)              # (B, n_out) fp64
        lfx = self._link(fx)
        lfnull64 = self._link(self.fnull.double())
        total_all = (lfx - lfnull64[None, :])           # (B, n_out) fp64
        if not fp64:
            total_all = total_all.float()
        lfnull = lfnull64.float()

        phi_full = t.zeros(b, self.n_groups, self.n_out, device=self.device,
                           dtype=t.float64 if fp64 else t.float32)

        # speculative fast path: when the last call used a captured graph,
        # assume the varying pattern repeats, replay immediately and verify
        # the device-computed pattern probe in the SAME sync as the result
        # (one host sync per call instead of two); mismatch falls through to
        # the eager path below
        if (
            self._graphs_enabled
            and not fp64
            and self._spec is not None
            and self.linear is not None
            and self.n_groups <= 62
        ):
            sk0, skey, sm = self._spec
            sb, sns, soff = skey[0], skey[1], skey[3]
            entry = self._graphs.get((skey[0], skey[1], skey[2]))
            # the replay is only valid if THIS call's plan (for the varying
            # count the probe will verify) resolves to the captured sample
            # count — nsamples=None must re-resolve through the default, not
            # wildcard-match the old graph — and its l1 mode stays off
            spec_plan = self.engine._plan(sm, nsamples) if entry is not None else None
            if (entry is not None and sb == b and spec_plan.nsamples == sns
                    and not self._l1_active(spec_plan, l1_reg)
                    and soff == int(instance_offset)):
                gbool = self._varying_matrix_dev(X_dev)
                keys = (gbool.long() << t.arange(
                    self.n_groups, device=self.device)).sum(dim=1)
                probe = t.stack([(keys == keys[0]).all().long(), keys[0]])
                graph, x_static, ids_static, gphi, _ = entry
                if x_static.data_ptr() != X_dev.data_ptr():
                    x_static.copy_(X_dev)
                graph.replay()
                # d2h through persistent pinned staging, enqueued BEFORE the
                # probe sync so ONE sync covers replay + result transfer (a
                # pageable .double().cpu() here cost ~0.2 ms of alloc+copy)
                staged = None if as_tensor else self._phi_stage_enqueue(gphi)
                pc = probe.cpu()                    # the single sync
                timer.mark("spec")
                if bool(pc[0]) and int(pc[1]) == sk0:
                    if as_tensor:
                        return gphi.clone()  # gphi is the graph's static buffer
                    timer.mark("d2h")
                    return [
                        np.ascontiguousarray(staged[:, :, o])
                        for o in range(self.n_out)
                    ]
                # pattern changed: discard the replayed result, run eagerly
                self._spec = None

        # bucket instances by varying-group pattern (benchmark case: 1 bucket)
        if self.n_groups <= 62:
            # fast path: pack each instance's pattern into an int64 key on
            # device and transfer only (all_same, key0) — 16 bytes instead of
            # the (B, G) matrix (the D2H sync dominated this stage)
            gbool = self._varying_matrix_dev(X_dev)     # (B, G) bool, device
            keys = (gbool.long() << t.arange(
                self.n_groups, device=self.device)).sum(dim=1)
            probe = t.stack([(keys == keys[0]).all().long(), keys[0]]).cpu()
            timer.mark("varying")
            if bool(probe[0]):
                key0 = int(probe[1])
                uniq = np.array(
                    [[(key0 >> g) & 1 for g in range(self.n_groups)]],
                    dtype=bool,
                )
                inverse = np.zeros(b, dtype=np.int64)
            else:
                vmat = gbool.cpu().numpy()
                ks = vmat @ (1 << np.arange(self.n_groups, dtype=np.uint64))
                _, first, inverse = np.unique(
                    ks, return_index=True, return_inverse=True
                )
                uniq = vmat[first]
        else:
            vmat = self._varying_matrix_dev(X_dev).cpu().numpy()
            timer.mark("varying")
            uniq, inverse = np.unique(vmat, axis=0, return_inverse=True)
        timer.mark("bucket")

        kc = self.engine.kernels
        if self._graphs_enabled and not fp64 and uniq.shape[0] == 1 \
                and self.linear is not None \
                and kc.fused_predict and kc.wls_mode in ("auto", "mfma"):
            varying0 = np.nonzero(uniq[0])[0]
            m0 = len(varying0)
            if m0 >= 2:
                plan0 = self.engine._plan(m0, nsamples)
                mpad0 = max(4, (m0 + 3) // 4 * 4)
                npad0 = (self.N + 15) // 16 * 16
                if (
                    mpad0 <= 64
                    and npad0 <= 128
                    and self.n_out in (1, 2, 4)
                    and not self._l1_active(plan0, l1_reg)
                ):
                    out = self._graph_explain(
                        X_dev, plan0, varying0, instance_offset,
                        as_tensor=as_tensor,
                    )
                    if out is not None:
                        key0 = 0
                        for g in varying0:
                            key0 |= 1 << int(g)
                        self._spec = (
                            key0,
                            (b, plan0.nsamples, varying0.tobytes(),
                             int(instance_offset)),
                            m0,
                        )
                        timer.mark("graph")
                        return out

        for u in range(uniq.shape[0]):
            varying = np.nonzero(uniq[u])[0]
            m = len(varying)
            ids = np.nonzero(inverse == u)[0]
            ids_t = t.tensor(ids, dtype=t.int64, device=self.device)
            if m == 0:
                continue
            if m == 1:
                phi_full[ids_t, int(varying[0])] = total_all[ids_t]
                continue
            plan = self.engine._plan(m, nsamples)
            gids = ids + instance_offset
            masks, kw = self._device_masks(plan, gids)
            timer.mark("masks")
            sub_X = X_dev[ids_t]
            if fp64:
                phi = self._bucket_fp64(
                    plan, masks, sub_X, varying, ids_t, total_all, lfnull64,
                    l1_reg,
                )
                timer.mark("wls")
                vidx_t = t.tensor(varying, dtype=t.int64, device=self.device)
                phi_full[ids_t[:, None], vidx_t[None, :]] = phi
                continue
            packed = None
            if m <= 64:
                # packed u64 masks feed the MFMA WLS Gram build
                packed = self._buf("packed", (len(ids), plan.nsamples), t.int64)
            pairwise = False
            if self.linear is not None:
                act3 = self._act_oimg()[0] == 3
                mpad = max(4, (m + 3) // 4 * 4)
                npad = (self.N + 15) // 16 * 16
                use_bf16 = (
                    kc.predict_dtype in ("bf16", "bf16x2")
                    and m <= 32 and npad <= 128 and self.n_out in (1, 2, 4)
                )
                if packed is not None:
                    self.ext.pack_masks(masks, packed)
                split = 2 if kc.predict_dtype == "bf16x2" else 1
                bf16_tiled_ok = (
                    kc.predict_dtype in ("bf16", "bf16x2")
                    and split * (1 if act3 else self.n_out) * 128 * 40 * 2
                        <= 64 * 1024
                )
                if kc.fused_predict and use_bf16:
                    ey = self._ey_fused_bf16(masks, sub_X, varying, packed=None)
                elif (kc.fused_predict and kc.predict_dtype == "fp32"
                        and mpad <= 64 and npad <= 128
                        and self.n_out in (1, 2, 4)):
                    ey = self._ey_fused_linear(masks, sub_X, varying)
                elif (kc.fused_predict and bf16_tiled_ok
                        and self.n_out in (1, 2, 4)):
                    # bf16 matrix-core tiled path (opt-in dtype), any shape
                    ey = self._ey_fused_tiled_bf16(masks, sub_X, varying)
                    pairwise = act3
                elif kc.fused_predict and self.n_out in (1, 2, 4):
                    # stress shapes stay on the hand-written MFMA path; the
                    # tiled kernel dual-accumulates (p0, p1) so the pairwise
                    # logit survives saturated probabilities
                    ey = self._ey_fused_tiled(masks, sub_X, varying)
                    pairwise = act3
                else:
                    ey = self._ey_linear_torch(masks, sub_X, varying)
                    pairwise = act3  # torch softmax is relatively accurate
            else:
                if packed is not None:
                    self.ext.pack_masks(masks, packed)
                ey = self._ey_torch_module(masks, sub_X, varying)
            timer.mark("predict")
            ey_adj = self._link_ey(ey, lfnull, pairwise)
            total = total_all[ids_t].contiguous()
            if self._l1_active(plan, l1_reg):
                if kc.l1_device:
                    support, g64, r64 = self._l1_select_batched(
                        masks, kw, ey_adj, l1_reg
                    )
                    phi = self._solve_selected(
                        masks, kw, ey_adj, total, support, g64, r64
                    )
                else:
                    phi = self._solve_host_l1(masks, kw, ey_adj, total, l1_reg)
            else:
                phi = self._buf("phi", (len(ids), m, self.n_out))
                # fp32 Gram conditioning degrades with M (Shapley kernel
                # weights span orders of magnitude): MFMA path covers
                # (M-1)+n_out <= 16; the scalar fp32 kernel up to M=24;
                # larger M goes to the fp64 batched torch solve
                use_kernel = (
                    kc.wls_mode != "torch"
                    and 2 <= m
                    and self.n_out <= 8
                    and (m - 1 + self.n_out <= 16 or m <= 24)
                )
                if use_kernel:
                    # wls_mode 'generic' disables the MFMA Gram build by
                    # withholding the packed masks
                    pk = None if kc.wls_mode == "generic" else packed
                    self.ext.wls_solve(masks, kw, ey_adj, total, phi, pk)
                elif (kc.wls_mode in ("auto", "mfma") and m <= 513
                        and self.n_out <= 8):
                    # stress shapes: MFMA-tiled Gram build + fp64 solve
                    phi = self._solve_gram(masks, kw, ey_adj, total)
                else:
                    phi = self._solve_torch(masks, kw, ey_adj, total)
            timer.mark("wls")
            vidx_t = t.tensor(varying, dtype=t.int64, device=self.device)
            phi_full[ids_t[:, None], vidx_t[None, :]] = phi

        if as_tensor:
            return phi_full
        out = phi_full.double().cpu().numpy()
        timer.mark("d2h")
        return [np.ascontiguousarray(out[:, :, o]) for o in range(self.n_out)]

    # ------------------------------------------------------------------ #

    def _varying_matrix_dev(self, X_dev):
        """K1: per-instance varying-group booleans (device tensor).

        A column varies iff some background value differs from x there; all
        values lie in [bg_min, bg_max], so it suffices to test the extremes
        (tolerance mirrors the CPU oracle's isclose)."""
        t = self.torch
        b = X_dev.shape[0]
        close_min = (self.bg_min[None] - X_dev).abs() <= (
            1e-5 * self.bg_min[None].abs() + 1e-8
        )
        close_max = (self.bg_max[None] - X_dev).abs() <= (
            1e-5 * self.bg_max[None].abs() + 1e-8
        )
        coldiff = (~(close_min & close_max)).to(t.int32)  # (B, D)
        gcount = t.zeros(b, self.n_groups, dtype=t.int32, device=self.device)
        gcount.index_add_(1, self.col_group, coldiff)
        return gcount > 0

    def _l1_active(self, plan, l1_reg) -> bool:
        m = plan.m
        max_samples = 2 ** 30 if m > 30 else 2 ** m - 2
        frac = plan.nsamples / max_samples
        if l1_reg == "auto":
            return frac < 0.2
        return l1_reg not in (None, False, 0)

    def _phi_stage_enqueue(self, gphi):
        """Enqueue device fp32->fp64 convert + async D2H into a persistent
        pinned buffer; returns the numpy view (valid after the caller's next
        sync). Callers copy out per class immediately, so the buffer can be
        reused next call."""
        t = self.torch
        shape = tuple(gphi.shape)
        d64 = self._buf("phi_d64", shape, t.float64)
        d64.copy_(gphi)
        key = ("phi_pin", shape)
        pin = self._ws.get(key)
        if pin is None:
            pin = t.empty(*shape, dtype=t.float64)
            try:
                pin = pin.pin_memory()
            except RuntimeError:  # pragma: no cover
                pass
            self._ws[key] = pin
        pin.copy_(d64, non_blocking=True)
        return pin.numpy()

    def _link_ey(self, ey, lfnull, pairwise):
        """In-place link transform (ey is a workspace). ``pairwise``: ey
        carries relatively-accurate (p0_sum, p1_sum) class sums, so the
        logit is computed as log(p1) - log(p0) — the reference-faithful
        fp64-numpy formula, with no clamp cliff at saturation."""
        t = self.torch
        if self.link_name == "identity":
            return ey.sub_(lfnull[None, None, :])
        if pairwise and self.n_out == 2:
            tiny = 1e-300 if ey.dtype == t.float64 else 1e-38
            lratio = (t.log(ey[..., 1].clamp_min(tiny))
                      - t.log(ey[..., 0].clamp_min(tiny)))
            ey[..., 0] = -lratio
            ey[..., 1] = lratio
            return ey.sub_(lfnull[None, None, :])
        eps = 1e-15 if ey.dtype == t.float64 else _EPS
        ey.clamp_(eps, 1.0 - eps)
        ey.log_().sub_(t.log1p(-t.exp(ey)))  # log(p/(1-p)) in place
        return ey.sub_(lfnull[None, None, :])

    def _bucket_fp64(self, plan, masks, sub_X, varying, ids_t, total_all,
                     lfnull64, l1_reg):
        """One bucket of the fp64 verification pipeline: fp64 ey on the same
        device masks, exact fp64 kernel weights (rebuilt from the plan's
        fp64 arrays, not the fp32 device copy), fp64 link and WLS."""
        t = self.torch
        if self.linear is None:
            raise TypeError(
                "predict_dtype='fp64' is the linear-predictor verification "
                "mode; torch-module predictors have no fp64 reference "
                "(convert the module to double and use device='cpu' instead)."
            )
        b = sub_X.shape[0]
        ey = self._ey_linear_f64(masks, sub_X, varying)
        kw64 = t.from_numpy(
            np.concatenate([
                plan.enum_weights,
                np.full(plan.n_random,
                        plan.weight_left / max(plan.n_random, 1)),
            ])
        ).to(self.device)
        kw = kw64[None, :].expand(b, plan.nsamples).contiguous()
        act3 = self._act_oimg()[0] == 3
        if self.link_name == "identity":
            ey_adj = ey - lfnull64[None, None, :]
        elif act3 and self.n_out == 2:
            # pairwise logit on the fp64 softmax pair — same formula as the
            # fp32 tiled path, so the self-check isolates arithmetic only
            lratio = (t.log(ey[..., 1].clamp_min(1e-300))
                      - t.log(ey[..., 0].clamp_min(1e-300)))
            ey_adj = (t.stack([-lratio, lratio], dim=-1)
                      - lfnull64[None, None, :])
        else:
            p = ey.clamp(1e-15, 1.0 - 1e-15)
            ey_adj = t.log(p / (1.0 - p)) - lfnull64[None, None, :]
        total = total_all[ids_t].contiguous()
        if self._l1_active(plan, l1_reg):
            return self._solve_host_l1(masks, kw, ey_adj, total, l1_reg).double()
        return self._solve_torch(masks, kw, ey_adj, total, out_dtype=t.float64)

    def _solve_gram(self, masks, kw, ey_adj, total):
        """Stress-shape WLS: the S-dependent
        O(S*(M-1)^2) normal-equation build runs as the hand-written
        ``wls_gram_kernel`` (16x16 MFMA tiles over packed mask bits, fp64
        via 256-sample chunked promotion); only the S-independent
        (M-1)x(M-1) solve uses library fp64. Replaces the fp64 torch bmm of
        ``_solve_torch`` — which materialised multiple (b, S, M) fp64
        tensors (~GBs of HBM traffic at the stress config) — with a kernel
        that reads just the packed bits."""
        t = self.torch
        b, s, m = masks.shape
        w_words = (m + 63) // 64
        packedw = self._buf(f"packedW{w_words}", (b, s, w_words), t.int64)
        self.ext.pack_masks_words(masks, packedw)
        mm = m - 1
        a64 = self._buf(f"gramA{mm}", (b, mm, mm), t.float64)
        r64 = self._buf(f"gramR{mm}", (b, mm, self.n_out), t.float64)
        self.ext.wls_gram(packedw, kw, ey_adj, total, a64, r64)
        try:
            w = t.linalg.solve(a64, r64)
        except Exception:
            w = t.linalg.lstsq(a64, r64).solution
        phi_last = total.double()[:, None, :] - w.sum(dim=1, keepdim=True)
        return t.cat([w, phi_last], dim=1).float()

    def _solve_torch(self, masks, kw, ey_adj, total, out_dtype=None):
        """Batched torch WLS fallback: normal equations via bmm +
        torch.linalg.solve, in fp64 — the Shapley kernel weights span
        several orders of magnitude, so the Gram matrix at M~200 is too
        ill-conditioned for fp32 normal equations (MI355X fp64 is cheap
        relative to this cold path). Also the fp64 verification solver."""
        t = self.torch
        z = masks.double()
        last = z[:, :, -1:]
        etmp = z[:, :, :-1] - last                       # (b, s, m-1)
        ey2 = ey_adj.double() - last * total.double()[:, None, :]
        wz = etmp * kw.double()[:, :, None]
        a = t.bmm(wz.transpose(1, 2), etmp)              # (b, m-1, m-1)
        r = t.bmm(wz.transpose(1, 2), ey2)               # (b, m-1, o)
        w = t.linalg.solve(a, r)
        phi_last = total.double()[:, None, :] - w.sum(dim=1, keepdim=True)
        return t.cat([w, phi_last], dim=1).to(out_dtype or t.float32)

    def _l1_select_batched(self, masks, kw, ey_adj, l1_reg):
        """K8 on device: whole-batch L1 pre-selection.

        The weighted normal equations (Gram + correlations) are built by the
        MFMA ``wls_gram`` kernel called with a phantom last feature (bit m is
        always 0, so the constraint elimination is a no-op and the kernel
        returns the RAW m x m Gram); the LARS/lasso path then runs as ONE
        batched torch fp64 iteration over all instances
        (``core.lars.batched_lars_select``, selection-identical to the
        sklearn calls of the CPU oracle) — no per-instance host loop.
        Returns a (b, m) bool support tensor."""
        from ..core.lars import batched_lars_select

        t = self.torch
        b, s, m = masks.shape
        if l1_reg == "auto":
            if not getattr(self, "_warned_l1_auto", False):
                logger.warning(
                    "l1_reg='auto' engaged on a GPU batch (sampled fraction "
                    "of the 2^%d coalitions < 0.2): running batched device "
                    "LARS (AIC) pre-selection over all %d instances. For "
                    "dense problems pass l1_reg=False to skip selection "
                    "entirely.", m, b,
                )
                self._warned_l1_auto = True
            mode, nf, alpha = "aic", None, None
        elif l1_reg in ("aic", "bic"):
            mode, nf, alpha = l1_reg, None, None
        elif isinstance(l1_reg, str) and l1_reg.startswith("num_features("):
            mode, nf, alpha = ("num_features",
                               int(l1_reg[len("num_features("):-1]), None)
        elif isinstance(l1_reg, (int, float)) and not isinstance(l1_reg, bool):
            mode, nf, alpha = "alpha", None, float(l1_reg)
        else:
            raise ValueError(f"Unsupported l1_reg: {l1_reg!r}")

        w_words = (m + 64) // 64          # phantom last bit m -> W*64 >= m+1
        packedw = self._buf(f"packedL{w_words}", (b, s, w_words), t.int64)
        self.ext.pack_masks_words(masks, packedw)
        g64 = self._buf(f"l1G{m}", (b, m, m), t.float64)
        r64 = self._buf(f"l1R{m}", (b, m, self.n_out), t.float64)
        zero_tot = self._buf("l1tot", (b, self.n_out), zeroed=True)
        self.ext.wls_gram(packedw, kw, ey_adj, zero_tot, g64, r64)
        c64 = r64[:, :, 0].contiguous()
        y = ey_adj[:, :, 0]
        yty = (kw.double() * y.double() * y.double()).sum(1)
        zbar = ybar = None
        if mode != "num_features":
            # sklearn Lasso/LassoLarsIC centre the (weighted) design
            sqw = kw.clamp_min(0).sqrt()
            zsum = t.zeros(b, m, dtype=t.float64, device=self.device)
            for lo in range(0, s, 8192):
                hi = min(lo + 8192, s)
                zsum += t.bmm(
                    sqw[:, lo:hi].unsqueeze(1).double(),
                    masks[:, lo:hi].double(),
                ).squeeze(1)
            zbar = zsum / s
            ybar = (sqw.double() * y.double()).sum(1) / s
        support = batched_lars_select(
            g64, c64, yty, n_samples=s, mode=mode, num_features=nf,
            alpha=alpha, zbar=zbar, ybar=ybar,
        )
        return support, g64, r64

    def _solve_selected(self, masks, kw, ey_adj, total, support, g64, r64):
        """Constrained WLS over per-instance selected supports, derived
        ALGEBRAICALLY from the full raw normal equations the selection
        already built (no second pass over the S samples): with s = selected
        features, l = last of them,

          A_el[i,j] = G[si,sj] - G[si,l] - G[l,sj] + G[l,l]
          r_el[i,o] = (c_o[si] - c_o[l]) - (G[si,l] - G[l,l]) * total_o

        Instances are grouped by identical support and each group solved as
        one batched fp64 system."""
        t = self.torch
        b, s, m = masks.shape
        phi = t.zeros(b, m, self.n_out, device=self.device)
        sup_np = support.cpu().numpy().astype(bool)
        groups: dict = {}
        for i in range(b):
            groups.setdefault(sup_np[i].tobytes(), []).append(i)
        for idxs in groups.values():
            sel = np.nonzero(sup_np[idxs[0]])[0]
            if len(sel) == 0:
                sel = np.arange(m)       # empty selection: solve over all
            ids_t = t.tensor(idxs, dtype=t.int64, device=self.device)
            tot_g = total[ids_t].double()                  # (bg, n_out)
            if len(sel) == 1:
                phi[ids_t, int(sel[0])] = tot_g.float()
                continue
            sel_t = t.tensor(sel, dtype=t.int64, device=self.device)
            st, l = sel_t[:-1], int(sel_t[-1])
            gg = g64[ids_t]                                # (bg, m, m)
            rr = r64[ids_t]                                # (bg, m, n_out)
            gss = gg[:, st][:, :, st]
            gcol = gg[:, st, l]                            # (bg, k-1)
            gll = gg[:, l, l]
            a_el = gss - gcol[:, :, None] - gcol[:, None, :] + gll[:, None, None]
            r_el = ((rr[:, st] - rr[:, l][:, None, :])
                    - (gcol - gll[:, None])[:, :, None] * tot_g[:, None, :])
            try:
                w = t.linalg.solve(a_el, r_el)
            except Exception:
                w = t.linalg.lstsq(a_el, r_el).solution
            phi_last = tot_g[:, None, :] - w.sum(dim=1, keepdim=True)
            phi_sub = t.cat([w, phi_last], dim=1).float()
            phi[ids_t[:, None], sel_t[None, :]] = phi_sub
        return phi

    def _solve_host_l1(self, masks, kw, ey_adj, total, l1_reg):
        """Cold path: l1 feature selection + solve on host, per instance."""
        t = self.torch
        masks_h = masks.cpu().numpy()
        kw_h = kw.double().cpu().numpy()
        ey_h = ey_adj.double().cpu().numpy()
        tot_h = total.double().cpu().numpy()
        from ..core.solver import solve_wls

        b, s, m = masks_h.shape
        phi = np.zeros((b, m, self.n_out))
        plan_stub = type("PlanStub", (), {"nsamples": s})()
        for i in range(b):
            nz = self.engine._l1_select(masks_h[i], kw_h[i], ey_h[i], plan_stub, l1_reg)
            phi[i] = solve_wls(masks_h[i], kw_h[i], ey_h[i], tot_h[i], nonzero_inds=nz)
        return t.tensor(phi, dtype=t.float32, device=self.device)
