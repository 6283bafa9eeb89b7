"""Coalition (mask) sampling for KernelSHAP.

Reimplements the sampling semantics the reference delegates to
``shap.KernelExplainer`` (driven from reference ``explainers/kernel_shap.py:250``):

* paired subset-size enumeration with Shapley-kernel weights
  ``(M-1)/(s*(M-s))`` while the sample budget allows full enumeration of a
  size (and its complement size),
* a random phase drawing leftover masks from the residual size
  distribution, in complement pairs, each with unit weight,
* final renormalisation of the random-phase weights so they carry exactly
  the residual kernel-weight mass.

Differences from shap 0.35.0 (intentional, documented in SURVEY.md §2.4):
duplicate random masks are kept as separate unit-weight rows instead of
incrementing a dict-tracked weight — the WLS normal equations are identical
either way — and the RNG is a counter-based generator keyed on
``(seed, instance_index)`` instead of the global numpy stream.
"""
from __future__ import annotations

from dataclasses import dataclass
from math import comb
from typing import Optional

import numpy as np

__all__ = [
    "CoalitionPlan",
    "plan_coalitions",
    "sample_masks",
    "default_nsamples",
]


def default_nsamples(m: int) -> int:
    """shap 0.35.0 default budget: ``2*M + 2**11``, capped at full enumeration."""
    ns = 2 * m + 2 ** 11
    if m <= 30:
        ns = min(ns, 2 ** m - 2)
    return ns


@dataclass
class CoalitionPlan:
    """Instance-independent part of the sampling decision for a given M.

    Attributes
    ----------
    m: number of varying groups.
    nsamples: total mask rows that will be emitted.
    enum_masks: (n_enum, m) uint8 — fully enumerated masks (deterministic).
    enum_weights: (n_enum,) float64 — their normalized kernel weights.
    num_full_subsets: number of fully enumerated subset sizes.
    n_random: rows to fill with random masks (nsamples - n_enum).
    random_size_probs: (n_sizes_left,) probabilities over remaining subset
        sizes (the *pair-collapsed* distribution; complements drawn implicitly).
    random_sizes: the actual subset sizes corresponding to random_size_probs.
    weight_left: residual normalized weight mass carried by the random rows.
    """

    m: int
    nsamples: int
    enum_masks: np.ndarray
    enum_weights: np.ndarray
    num_full_subsets: int
    n_random: int
    random_size_probs: np.ndarray
    random_sizes: np.ndarray
    weight_left: float


def _all_subsets(m: int, s: int) -> np.ndarray:
    """All C(m, s) binary masks of weight s over m items, lexicographic."""
    from itertools import combinations

    out = np.zeros((comb(m, s), m), dtype=np.uint8)
    for i, c in enumerate(combinations(range(m), s)):
        out[i, list(c)] = 1
    return out


def plan_coalitions(m: int, nsamples: Optional[int] = None) -> CoalitionPlan:
    """Build the deterministic enumeration + the residual random-size plan.

    Mirrors the subset-size loop of shap 0.35.0 ``KernelExplainer.explain``
    (reference call chain: ``explainers/kernel_shap.py:250`` -> shap ``explain``).
    """
    if m < 2:
        raise ValueError("plan_coalitions requires m >= 2 (m<2 is special-cased)")
    if nsamples is None:
        nsamples = default_nsamples(m)
    if m <= 30:
        nsamples = min(nsamples, 2 ** m - 2)

    num_subset_sizes = int(np.ceil((m - 1) / 2.0))
    num_paired_subset_sizes = int(np.floor((m - 1) / 2.0))

    weight_vector = np.array(
        [(m - 1.0) / (i * (m - i)) for i in range(1, num_subset_sizes + 1)],
        dtype=np.float64,
    )
    weight_vector[:num_paired_subset_sizes] *= 2.0
    weight_vector /= weight_vector.sum()

    remaining_weight_vector = weight_vector.copy()
    num_full_subsets = 0
    samples_left = nsamples

    enum_blocks = []
    enum_weight_blocks = []

    for i, subset_size in enumerate(range(1, num_subset_sizes + 1)):
        nsubsets = comb(m, subset_size)
        if subset_size <= num_paired_subset_sizes:
            nsubsets *= 2
        # can we fully enumerate this size (and its complement)?
        if samples_left * remaining_weight_vector[i] / nsubsets >= 1.0 - 1e-8:
            num_full_subsets += 1
            samples_left -= nsubsets
            if remaining_weight_vector[i] < 1.0:
                remaining_weight_vector[i:] = remaining_weight_vector[i:] / (
                    1.0 - remaining_weight_vector[i]
                )
                remaining_weight_vector[i] = 0.0  # consumed
            w = weight_vector[i] / comb(m, subset_size)
            if subset_size <= num_paired_subset_sizes:
                w /= 2.0
            masks = _all_subsets(m, subset_size)
            enum_blocks.append(masks)
            enum_weight_blocks.append(np.full(masks.shape[0], w))
            if subset_size <= num_paired_subset_sizes:
                enum_blocks.append(1 - masks)
                enum_weight_blocks.append(np.full(masks.shape[0], w))
        else:
            break

    if enum_blocks:
        enum_masks = np.concatenate(enum_blocks, axis=0)
        enum_weights = np.concatenate(enum_weight_blocks, axis=0)
    else:
        enum_masks = np.zeros((0, m), dtype=np.uint8)
        enum_weights = np.zeros((0,), dtype=np.float64)

    n_random = samples_left
    sizes_left = np.arange(num_full_subsets + 1, num_subset_sizes + 1)
    if num_full_subsets < num_subset_sizes:
        probs = weight_vector[num_full_subsets:].copy()
        # paired sizes are drawn once and their complement added implicitly;
        # the unpaired middle size (odd M-1 case) has no distinct complement.
        probs /= probs.sum()
        weight_left = float(weight_vector[num_full_subsets:].sum())
    else:
        probs = np.zeros((0,))
        weight_left = 0.0

    return CoalitionPlan(
        m=m,
        nsamples=int(nsamples),
        enum_masks=enum_masks,
        enum_weights=enum_weights,
        num_full_subsets=num_full_subsets,
        n_random=int(n_random),
        random_size_probs=probs,
        random_sizes=sizes_left,
        weight_left=weight_left,
    )


def sample_masks(
    plan: CoalitionPlan, seed: int, instance_index: int
) -> tuple[np.ndarray, np.ndarray]:
    """Emit the full (nsamples, m) mask matrix + kernel weights for one instance.

    The random phase uses a Philox counter-based generator keyed on
    ``(seed, instance_index)`` so results are deterministic and independent of
    sharding/order — replacing the reference's per-actor global-numpy reseed
    (``explainers/kernel_shap.py:226-229``).
    """
    m = plan.m
    masks = np.zeros((plan.nsamples, m), dtype=np.uint8)
    weights = np.zeros((plan.nsamples,), dtype=np.float64)
    ne = plan.enum_masks.shape[0]
    masks[:ne] = plan.enum_masks
    weights[:ne] = plan.enum_weights

    if plan.n_random > 0:
        rng = np.random.Generator(np.random.Philox(key=[seed, instance_index]))
        num_paired = int(np.floor((m - 1) / 2.0))
        n_rand = plan.n_random
        # vectorised random phase: draw all subset sizes at once, allocate
        # rows by prefix sum (paired draws consume 2 rows), build subsets via
        # the rank-threshold trick
        sizes_draw = rng.choice(
            plan.random_sizes, size=n_rand, p=plan.random_size_probs
        )
        paired = sizes_draw <= num_paired
        rows_per = 1 + paired.astype(np.int64)
        cum = np.cumsum(rows_per)
        jcut = int(np.searchsorted(cum, n_rand))  # first draw reaching budget
        sizes_draw = sizes_draw[: jcut + 1]
        paired = paired[: jcut + 1]
        starts = cum[: jcut + 1] - rows_per[: jcut + 1]
        # random subset of size s: the s smallest ranks of a random row
        rand = rng.random((jcut + 1, m))
        ranks = rand.argsort(axis=1).argsort(axis=1)
        draw_masks = (ranks < sizes_draw[:, None]).astype(np.uint8)
        masks[ne + starts] = draw_masks
        comp_ok = paired & (starts + 1 < n_rand)
        masks[ne + starts[comp_ok] + 1] = 1 - draw_masks[comp_ok]
        weights[ne:] = plan.weight_left / n_rand

    return masks, weights


def sampler_chunks(n_random: int) -> int:
    """Chunk count of the GPU sampler's deterministic per-wave partitioning
    (mirrors ``kshap_sampler_chunks`` in kshap_kernels.hip): ~256 rows per
    chunk, a multiple of 8 in [8, 64]. Plans with n_random <= 2048 keep the
    historic 8 chunks, so their device masks are bit-for-bit stable."""
    c = (n_random + 255) // 256
    c = (c + 7) // 8 * 8
    return max(8, min(64, c))
