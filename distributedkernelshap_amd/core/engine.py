"""Device-agnostic KernelSHAP engine.

This is the native replacement for the algorithm core the reference delegates
to ``shap.KernelExplainer`` (reference ``explainers/kernel_shap.py:14,229,250``).
The CPU path (numpy, vectorised, chunked) is the CI oracle; the GPU path
(``device='cuda'``) dispatches to the HIP extension in
``distributedkernelshap_amd.ops`` — hand-written CDNA4 kernels for mask
sampling, fused masked-background predict (MFMA), background reduction and the
batched constrained WLS solve.

Pipeline per instance:
  varying groups -> coalition masks + kernel weights -> masked-background
  perturbation synth -> batched predict -> weighted mean over background ->
  link transform -> constrained WLS -> phi.
"""
from __future__ import annotations

import logging
from typing import Callable, List, Optional, Sequence, Union

import numpy as np

from .links import convert_to_link
from .sampler import CoalitionPlan, plan_coalitions, sample_masks
from .solver import solve_wls

logger = logging.getLogger(__name__)

__all__ = ["KernelShapEngine"]


def _as_2d(x: np.ndarray) -> np.ndarray:
    x = np.asarray(x)
    if x.ndim == 1:
        x = x.reshape(1, -1)
    return x


class KernelShapEngine:
    """Batched KernelSHAP over a fixed background set.

    Parameters
    ----------
    predictor : callable (n, D) -> (n, n_out), e.g. predict_proba. May expose
        ``linear_params() -> (W, b, activation)`` to enable the fused GPU path.
    background : (N, D) background dataset (already summarised if desired).
    bg_weights : optional (N,) weights (e.g. kmeans cluster occupancies,
        reference ``explainers/kernel_shap.py:542``); default uniform.
    groups : list of column-index lists — one entry per explained feature
        (one-hot blocks grouped, reference §2.6); default singleton columns.
    link : 'identity' | 'logit'.
    seed : base seed for the per-instance counter-based RNG.
    device : 'cpu' | 'cuda' | 'auto'.
    """

    def __init__(
        self,
        predictor: Callable[[np.ndarray], np.ndarray],
        background: np.ndarray,
        bg_weights: Optional[np.ndarray] = None,
        groups: Optional[Sequence[Sequence[int]]] = None,
        link: str = "identity",
        seed: int = 0,
        device: str = "auto",
        chunk_rows: int = 1 << 20,
        kernels=None,
    ):
        from ..config import KernelConfig

        self.kernels = kernels if kernels is not None else KernelConfig()
        self.predictor = predictor
        self.background = np.ascontiguousarray(_as_2d(background), dtype=np.float64)
        n, d = self.background.shape
        if bg_weights is None:
            bg_weights = np.full(n, 1.0 / n)
        else:
            bg_weights = np.asarray(bg_weights, dtype=np.float64)
            bg_weights = bg_weights / bg_weights.sum()
        self.bg_weights = bg_weights
        if groups is None:
            groups = [[j] for j in range(d)]
        self.groups = [np.asarray(g, dtype=np.int64) for g in groups]
        cols = np.concatenate(self.groups) if self.groups else np.array([], dtype=np.int64)
        if sorted(cols.tolist()) != list(range(d)):
            raise ValueError("groups must partition the feature columns exactly")
        self.n_groups = len(self.groups)
        # group id per column, for fast mask expansion
        self._col_group = np.empty(d, dtype=np.int64)
        for gi, g in enumerate(self.groups):
            self._col_group[g] = gi
        self.link_name = link if isinstance(link, str) else "custom"
        self.link, self.link_inv = convert_to_link(link)
        self.seed = int(seed)
        self.chunk_rows = int(chunk_rows)
        self.device = self._resolve_device(device)

        # fit-time quantities (K9): fnull / expected_value
        fbg = np.asarray(self.predictor(self.background))
        if fbg.ndim == 1:
            fbg = fbg.reshape(-1, 1)
        self.vector_out = fbg.shape[1] > 1 or np.asarray(fbg).ndim > 1
        self.n_out = fbg.shape[1]
        self.fnull = bg_weights @ fbg  # (n_out,)
        self.expected_value = self.link(self.fnull)
        self.D = d
        self.N = n
        self._plan_cache: dict = {}
        # algebraic fast path for linear predictors (same folding as the GPU
        # kernel): logits(synth[s,n]) = base[n] + sum_k mask[s,k]*diff[k,n]
        self._lin = None
        lp = getattr(predictor, "linear_params", None)
        if lp is not None:
            W, bias, act = lp()
            W = np.asarray(W, dtype=np.float64)
            contrib = self.background[:, :, None] * W.T[None, :, :]  # (N,D,o)
            bg_part = np.zeros((n, self.n_groups, W.shape[0]))
            np.add.at(bg_part, (slice(None), self._col_group), contrib)
            self._lin = {
                "W": W,
                "b": np.asarray(bias, dtype=np.float64),
                "act": act,
                "bg_part": bg_part,
                "base": self.background @ W.T + np.asarray(bias),
            }
        self._gpu = None
        if self.device == "cuda":
            from ..ops import gpu_engine

            self._gpu = gpu_engine.GpuKernelShap(self)

    @staticmethod
    def _resolve_device(device: str) -> str:
        if device == "auto":
            try:
                import torch

                return "cuda" if torch.cuda.is_available() else "cpu"
            except Exception:
                return "cpu"
        return device

    # ------------------------------------------------------------------ #

    def enable_tracing(self, on: bool = True) -> None:
        """Record per-stage wall times of the GPU pipeline (see
        ``get_trace``). No-op on the CPU path."""
        if self._gpu is not None:
            self._gpu.enable_tracing(on)

    def get_trace(self) -> Optional[dict]:
        """Stage -> list of ms per shap_values call (None if not tracing)."""
        return self._gpu.trace if self._gpu is not None else None

    def _plan(self, m: int, nsamples: Optional[int]) -> CoalitionPlan:
        key = (m, nsamples)
        if key not in self._plan_cache:
            self._plan_cache[key] = plan_coalitions(m, nsamples)
        return self._plan_cache[key]

    def varying_groups(self, x: np.ndarray) -> np.ndarray:
        """K1: indices of groups where x differs from at least one background row."""
        diff = ~np.isclose(self.background, x[None, :], rtol=1e-5, atol=1e-8)
        anydiff = diff.any(axis=0)  # (D,)
        gdiff = np.zeros(self.n_groups, dtype=bool)
        np.logical_or.at(gdiff, self._col_group, anydiff)
        return np.nonzero(gdiff)[0]

    def shap_values(
        self,
        X: np.ndarray,
        nsamples: Optional[int] = None,
        l1_reg: Union[str, int, float] = "auto",
        instance_offset: int = 0,
        silent: bool = True,
        as_tensor: bool = False,
    ) -> List[np.ndarray]:
        """Shapley values for a batch of instances.

        Returns a list of ``n_out`` arrays of shape ``(B, n_groups)`` — the
        reference's ``shap_values`` output layout
        (``explainers/kernel_shap.py:250`` / ``build_explanation`` contract).
        With ``as_tensor=True`` the raw ``(B, n_groups, n_out)`` fp32 torch
        tensor is returned instead (device-resident on the GPU path) so the
        distributed result gather never bounces through the host.

        ``instance_offset`` keys the per-instance counter RNG so distributed
        shards reproduce the single-process result exactly.
        """
        if self._gpu is not None:
            # torch tensors (incl. pinned/device) pass through untouched
            return self._gpu.shap_values(
                X, nsamples=nsamples, l1_reg=l1_reg,
                instance_offset=instance_offset, as_tensor=as_tensor,
            )
        if isinstance(X, (list, tuple)) or type(X).__module__ == "numpy":
            pass
        elif hasattr(X, "cpu"):  # torch tensor on the CPU path
            X = X.cpu().numpy()
        X = _as_2d(X).astype(np.float64)
        b = X.shape[0]
        fx = np.asarray(self.predictor(X))
        if fx.ndim == 1:
            fx = fx.reshape(-1, 1)
        lfx = self.link(fx)  # (B, n_out)
        lfnull = self.link(self.fnull)  # (n_out,)
        phi = np.zeros((b, self.n_groups, self.n_out), dtype=np.float64)
        for i in range(b):
            phi[i] = self._explain_one(
                X[i], lfx[i] - lfnull, nsamples, l1_reg, instance_offset + i
            )
        if as_tensor:
            # CPU oracle keeps fp64 (bitwise parity with the numpy path);
            # the GPU engine's as_tensor path is fp32 device-resident
            import torch

            return torch.from_numpy(phi)
        return [np.ascontiguousarray(phi[:, :, o]) for o in range(self.n_out)]

    # ------------------------------------------------------------------ #

    def _explain_one(
        self,
        x: np.ndarray,
        total: np.ndarray,
        nsamples: Optional[int],
        l1_reg,
        instance_index: int,
    ) -> np.ndarray:
        varying = self.varying_groups(x)
        m = len(varying)
        phi = np.zeros((self.n_groups, self.n_out), dtype=np.float64)
        if m == 0:
            return phi
        if m == 1:
            phi[varying[0]] = total
            return phi
        plan = self._plan(m, nsamples)
        masks, kw = sample_masks(plan, self.seed, instance_index)
        ey = self._ey(x, masks, varying)  # (S, n_out) model-space
        ey_adj = self.link(ey) - self.link(self.fnull)[None, :]
        nonzero = self._l1_select(masks, kw, ey_adj, plan, l1_reg)
        phi[varying] = solve_wls(masks, kw, ey_adj, total, nonzero_inds=nonzero, m_full=m)
        return phi

    def _ey(self, x: np.ndarray, masks: np.ndarray, varying: np.ndarray) -> np.ndarray:
        """K3-K5: masked-background synth -> predict -> weighted mean over bg.

        Linear predictors take the algebraically folded path (no synth
        materialisation); arbitrary callables use the chunked synth path
        (the CI oracle for the GPU synth kernel).
        """
        if self._lin is not None:
            return self._ey_linear(x, masks, varying)
        s = masks.shape[0]
        n, d = self.background.shape
        # expand group mask (S, m_varying) to column mask (S, D)
        colmask = np.zeros((s, d), dtype=bool)
        for mi, gi in enumerate(varying):
            colmask[:, self.groups[gi]] = masks[:, mi, None].astype(bool)
        ey = np.empty((s, self.n_out), dtype=np.float64)
        chunk = max(1, self.chunk_rows // max(n, 1))
        for lo in range(0, s, chunk):
            hi = min(lo + chunk, s)
            cm = colmask[lo:hi]  # (c, D)
            synth = np.where(cm[:, None, :], x[None, None, :], self.background[None, :, :])
            y = np.asarray(self.predictor(synth.reshape(-1, d)))
            if y.ndim == 1:
                y = y.reshape(-1, 1)
            y = y.reshape(hi - lo, n, self.n_out)
            ey[lo:hi] = np.einsum("cno,n->co", y, self.bg_weights)
        return ey

    def _ey_linear(self, x: np.ndarray, masks: np.ndarray, varying: np.ndarray) -> np.ndarray:
        lin = self._lin
        w = lin["W"]
        x_contrib = x[:, None] * w.T                       # (D, o)
        x_part = np.zeros((self.n_groups, w.shape[0]))
        np.add.at(x_part, self._col_group, x_contrib)
        diff = x_part[None, varying] - lin["bg_part"][:, varying]   # (N, m, o)
        act = lin["act"]
        mf = masks.astype(np.float64)
        if act == "softmax" and diff.shape[2] == 2:
            # binary softmax = one sigmoid on the logit difference: a single
            # (S,m)@(m,N) BLAS GEMM + one exp pass
            dd = diff[:, :, 1] - diff[:, :, 0]              # (N, m)
            based = lin["base"][:, 1] - lin["base"][:, 0]   # (N,)
            zd = mf @ dd.T + based[None, :]
            p1 = 1.0 / (1.0 + np.exp(-zd))
            ey1 = p1 @ self.bg_weights
            return np.stack([1.0 - ey1, ey1], axis=1)
        # logits[s, n, o] = base[n, o] + masks[s] . diff[n]
        zz = np.einsum("sm,nmo->sno", mf, diff)
        zz += lin["base"][None, :, :]
        if act == "sigmoid":
            p = 1.0 / (1.0 + np.exp(-zz))
        elif act == "softmax":
            zz -= zz.max(axis=-1, keepdims=True)
            np.exp(zz, out=zz)
            p = zz / zz.sum(axis=-1, keepdims=True)
        else:
            p = zz
        return np.einsum("sno,n->so", p, self.bg_weights)

    # ------------------------------------------------------------------ #

    def _l1_select(
        self,
        masks: np.ndarray,
        kw: np.ndarray,
        ey_adj: np.ndarray,
        plan: CoalitionPlan,
        l1_reg,
    ) -> Optional[np.ndarray]:
        """K8: optional L1 feature pre-selection (shap 0.35.0 semantics).

        Returns column indices of the mask matrix to solve over, or None for
        all. Host-side sklearn (cold path). Uses the first
        output dimension for selection, like shap.
        """
        m = masks.shape[1]
        max_samples = 2 ** 30 if m > 30 else 2 ** m - 2
        fraction_evaluated = plan.nsamples / max_samples
        use_auto = l1_reg == "auto" and fraction_evaluated < 0.2
        if l1_reg in (None, False, 0, "auto") and not use_auto:
            return None
        z = masks.astype(np.float64)
        # shap's augmented-data trick: weighted sqrt-transform for lasso
        w = np.sqrt(kw)
        zw = z * w[:, None]
        yw = ey_adj[:, 0] * w
        from sklearn import linear_model

        if use_auto or l1_reg == "aic" or l1_reg == "bic":
            crit = "bic" if l1_reg == "bic" else "aic"
            model = linear_model.LassoLarsIC(criterion=crit)
            model.fit(zw, yw)
            nz = np.nonzero(model.coef_)[0]
        elif isinstance(l1_reg, str) and l1_reg.startswith("num_features("):
            k = int(l1_reg[len("num_features("):-1])
            coefs = linear_model.lars_path(zw, yw, max_iter=k)[2]
            nz = np.nonzero(coefs[:, -1])[0]
        elif isinstance(l1_reg, (int, float)):
            model = linear_model.Lasso(alpha=float(l1_reg))
            model.fit(zw, yw)
            nz = np.nonzero(model.coef_)[0]
        else:
            raise ValueError(f"Unsupported l1_reg: {l1_reg!r}")
        return nz if len(nz) > 0 else None
