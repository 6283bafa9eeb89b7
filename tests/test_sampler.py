"""Unit tests for coalition sampling (CPU oracle of HIP kernel K2)."""
from math import comb

import numpy as np
import pytest

from distributedkernelshap_amd.core.sampler import (
    default_nsamples,
    plan_coalitions,
    sample_masks,
)


def test_default_nsamples():
    assert default_nsamples(12) == 2 * 12 + 2048
    # full-enumeration cap for small M
    assert default_nsamples(4) == 2 ** 4 - 2
    assert default_nsamples(10) == 2 ** 10 - 2


def test_full_enumeration_small_m():
    m = 6
    plan = plan_coalitions(m)  # budget covers 2^6-2 = 62 masks
    assert plan.nsamples == 62
    assert plan.n_random == 0
    masks, w = sample_masks(plan, seed=0, instance_index=0)
    # all non-trivial subsets present exactly once
    keys = {tuple(row) for row in masks}
    assert len(keys) == 62
    sizes = masks.sum(axis=1)
    assert sizes.min() == 1 and sizes.max() == m - 1
    # weights sum to 1 (normalized shapley kernel over enumerated sizes)
    assert np.isclose(w.sum(), 1.0)
    # per-mask weight matches the shapley kernel (M-1)/(C(M,s)*s*(M-s)) ratio
    for s in range(1, m):
        ws = w[sizes == s]
        assert np.allclose(ws, ws[0])
    r12 = w[sizes == 1][0] / w[sizes == 2][0]
    expect = ((m - 1) / (1 * (m - 1)) / comb(m, 1)) / ((m - 1) / (2 * (m - 2)) / comb(m, 2))
    assert np.isclose(r12, expect)


def test_adult_config_plan():
    # M=12 benchmark config: nsamples = 2072; the weighted budget check
    # (samples_left * residual_weight / nsubsets >= 1) admits sizes 1-3
    # fully (596 masks incl. complements), size 4 fails (1476*0.424/990 < 1)
    plan = plan_coalitions(12)
    assert plan.nsamples == 2072
    assert plan.num_full_subsets == 3
    n_enum = sum(2 * comb(12, s) for s in (1, 2, 3))
    assert plan.enum_masks.shape == (n_enum, 12)
    assert plan.n_random == 2072 - n_enum
    masks, w = sample_masks(plan, seed=0, instance_index=3)
    assert masks.shape == (2072, 12)
    assert np.isclose(w.sum(), 1.0)
    # random rows carry exactly the residual kernel-weight mass
    assert np.isclose(w[n_enum:].sum(), plan.weight_left)
    # random sizes only from non-enumerated range (draws 4-6, complements 6-8)
    rs = masks[n_enum:].sum(axis=1)
    assert rs.min() >= 4 and rs.max() <= 8


def test_complement_pairing():
    plan = plan_coalitions(12)
    masks, _ = sample_masks(plan, seed=0, instance_index=0)
    ne = plan.enum_masks.shape[0]
    # enumerated block contains each mask's complement
    keys = {tuple(r) for r in masks[:ne]}
    for r in masks[:ne]:
        assert tuple(1 - r) in keys
    # random phase: paired draws are adjacent complements (size<=5 paired for M=12)
    rnd = masks[ne:]
    sizes = rnd.sum(axis=1)
    i = 0
    while i < len(rnd) - 1:
        if sizes[i] <= 5:
            assert np.array_equal(rnd[i + 1], 1 - rnd[i])
            i += 2
        else:
            i += 1


def test_determinism_and_instance_keying():
    plan = plan_coalitions(12)
    a1, w1 = sample_masks(plan, seed=0, instance_index=5)
    a2, w2 = sample_masks(plan, seed=0, instance_index=5)
    b, _ = sample_masks(plan, seed=0, instance_index=6)
    assert np.array_equal(a1, a2) and np.array_equal(w1, w2)
    assert not np.array_equal(a1, b)


def test_size_distribution_matches_kernel():
    """Random-phase subset-size frequencies follow the residual Shapley
    kernel distribution (property test): tested on M=24
    with a 15k-draw random phase."""
    m = 24
    plan = plan_coalitions(m, nsamples=20000)
    masks, w = sample_masks(plan, seed=0, instance_index=0)
    ne = plan.enum_masks.shape[0]
    assert plan.num_full_subsets == 2  # sizes 1-2 enumerable within 20000
    sizes = masks[ne:].sum(axis=1)
    counts = np.bincount(sizes, minlength=m + 1).astype(float)
    # paired draws: size s and complement m-s appear equally often
    for s in range(3, 12):
        assert abs(counts[s] - counts[m - s]) / counts[s] < 0.08
    # draw frequency proportional to the doubled shapley kernel weight
    w3 = 2 * (m - 1) / (3 * (m - 3))
    w4 = 2 * (m - 1) / (4 * (m - 4))
    got = counts[3] / counts[4]
    assert abs(got - w3 / w4) / (w3 / w4) < 0.12
    assert np.isclose(w.sum(), 1.0)
