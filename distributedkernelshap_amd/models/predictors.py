"""Predictor registry.

The reference's "model" is a pickled sklearn LogisticRegression invoked via
``predict_proba`` (``benchmarks/ray_pool.py:34``, ``scripts/fit_adult_model.py:27-32``).
Here predictors are first-class objects:

* ``LinearPredictor`` — native logistic-regression path; exposes
  ``linear_params()`` so the GPU engine can run the fused masked-GEMM +
  activation + background-reduction HIP kernel without materialising the
  perturbation matrix.
* ``TorchPredictor`` — wraps any torch ``nn.Module`` (MLP / ResNet configs);
  the GPU engine materialises synth tiles on-device and calls the module
  stream-ordered.
"""
from __future__ import annotations

from typing import Optional

import numpy as np

__all__ = ["LinearPredictor", "TorchPredictor", "make_predictor"]


def _softmax(z: np.ndarray) -> np.ndarray:
    z = z - z.max(axis=-1, keepdims=True)
    e = np.exp(z)
    return e / e.sum(axis=-1, keepdims=True)


class LinearPredictor:
    """Multinomial logistic regression: softmax(X @ W.T + b).

    Mirrors sklearn ``LogisticRegression(multi_class='multinomial').predict_proba``
    (reference ``scripts/fit_adult_model.py:27-32``) with explicit weights.
    """

    def __init__(self, weights: np.ndarray, bias: np.ndarray, activation: str = "softmax"):
        self.weights = np.asarray(weights, dtype=np.float64)  # (n_out, D)
        self.bias = np.asarray(bias, dtype=np.float64)  # (n_out,)
        if activation not in ("softmax", "sigmoid", "none"):
            raise ValueError(f"unknown activation {activation}")
        self.activation = activation

    @property
    def n_out(self) -> int:
        return self.weights.shape[0]

    def __call__(self, X: np.ndarray) -> np.ndarray:
        z = np.asarray(X, dtype=np.float64) @ self.weights.T + self.bias
        if self.activation == "softmax":
            return _softmax(z)
        if self.activation == "sigmoid":
            return 1.0 / (1.0 + np.exp(-z))
        return z

    # protocol hook for the fused GPU path
    def linear_params(self):
        return self.weights, self.bias, self.activation

    @classmethod
    def random(cls, d: int, n_out: int = 2, seed: int = 0, scale: float = 0.5,
               activation: str = "softmax"):
        rng = np.random.Generator(np.random.Philox(key=[seed, 0x11EA7]))
        w = rng.normal(0.0, scale, size=(n_out, d))
        b = rng.normal(0.0, 0.1, size=(n_out,))
        return cls(w, b, activation)


class TorchPredictor:
    """Wraps a torch module as a numpy-callable predictor.

    The module maps (n, D) float32 -> (n, n_out) probabilities. The GPU engine
    detects this class and keeps the perturbation tiles on-device
    (``torch_module()`` / ``device`` accessors), avoiding host round-trips
    (SURVEY.md §7.3 "arbitrary-predictor path").
    """

    def __init__(self, module, device: Optional[str] = None, batch_rows: int = 1 << 16):
        import torch

        self.module = module.eval()
        if device is None:
            device = "cuda" if torch.cuda.is_available() else "cpu"
        self.device = device
        self.module.to(device)
        self.batch_rows = batch_rows

    def torch_module(self):
        return self.module

    def __call__(self, X: np.ndarray) -> np.ndarray:
        import torch

        X = np.asarray(X, dtype=np.float32)
        outs = []
        with torch.no_grad():
            for lo in range(0, X.shape[0], self.batch_rows):
                xb = torch.from_numpy(X[lo : lo + self.batch_rows]).to(self.device)
                outs.append(self.module(xb).float().cpu().numpy())
        return np.concatenate(outs, axis=0)


def make_predictor(kind: str, d: int, n_out: int = 2, seed: int = 0, **kw):
    """Registry entry point used by benchmarks and serve config."""
    if kind == "linear":
        return LinearPredictor.random(d, n_out, seed)
    if kind == "mlp":
        import torch
        from torch import nn

        torch.manual_seed(seed)
        hidden = kw.get("hidden", 256)
        layers = kw.get("layers", 2)
        mods = []
        prev = d
        for _ in range(layers):
            mods += [nn.Linear(prev, hidden), nn.ReLU()]
            prev = hidden
        mods += [nn.Linear(prev, n_out), nn.Softmax(dim=-1)]
        return TorchPredictor(nn.Sequential(*mods), device=kw.get("device"))
    raise ValueError(f"unknown predictor kind {kind!r}")
