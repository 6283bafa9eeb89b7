"""Checkpointed explanation for very large instance sets.

The reference's only resume surface was the incremental timing pickle
(SURVEY.md §5.4); for the 1M-instance configs an interrupted job should not
recompute finished work. ``explain_checkpointed`` streams X through the
engine in fixed-size chunks, appends per-class shap blocks to .npy memmaps
and records completed chunks in a JSON manifest — a restart skips them.
The per-instance counter RNG guarantees the resumed result is identical to
an uninterrupted run.
"""
from __future__ import annotations

import json
import os
from typing import List, Optional

import numpy as np

__all__ = ["explain_checkpointed"]


def explain_checkpointed(
    engine,
    X: np.ndarray,
    out_dir: str,
    chunk_instances: int = 4096,
    nsamples: Optional[int] = None,
    l1_reg="auto",
    instance_offset: int = 0,
) -> List[np.ndarray]:
    """Explain X in chunks with on-disk checkpointing; returns the full
    per-class shap matrices (memmap-backed)."""
    X = np.atleast_2d(np.asarray(X, dtype=np.float64))
    b = X.shape[0]
    os.makedirs(out_dir, exist_ok=True)
    manifest_path = os.path.join(out_dir, "manifest.json")
    n_out, g = engine.n_out, engine.n_groups

    manifest = {"done": [], "b": b, "n_groups": g, "n_out": n_out,
                "chunk_instances": chunk_instances, "seed": engine.seed,
                "nsamples": nsamples, "l1_reg": repr(l1_reg),
                "instance_offset": instance_offset}
    if os.path.exists(manifest_path):
        with open(manifest_path) as f:
            old = json.load(f)
        # a restart with different sampling/regularisation parameters must
        # not mix chunks computed under the old ones
        if all(old.get(k) == manifest[k]
               for k in ("b", "n_groups", "chunk_instances", "seed",
                         "nsamples", "l1_reg", "instance_offset")):
            manifest = old
        # else: incompatible checkpoint -> start over (files rewritten below)

    mode = "r+" if manifest["done"] and all(
        os.path.exists(os.path.join(out_dir, f"shap_class{o}.npy"))
        for o in range(n_out)
    ) else "w+"
    mms = [
        np.lib.format.open_memmap(
            os.path.join(out_dir, f"shap_class{o}.npy"),
            mode=mode, dtype=np.float64, shape=(b, g),
        )
        if mode == "w+"
        else np.lib.format.open_memmap(
            os.path.join(out_dir, f"shap_class{o}.npy"), mode="r+"
        )
        for o in range(n_out)
    ]
    if mode == "w+":
        manifest["done"] = []

    done = set(manifest["done"])
    for lo in range(0, b, chunk_instances):
        if lo in done:
            continue
        hi = min(lo + chunk_instances, b)
        sv = engine.shap_values(
            X[lo:hi], nsamples=nsamples, l1_reg=l1_reg,
            instance_offset=instance_offset + lo,
        )
        for o in range(n_out):
            mms[o][lo:hi] = sv[o]
            mms[o].flush()
        manifest["done"] = sorted(done | {lo})
        done.add(lo)
        with open(manifest_path, "w") as f:
            json.dump(manifest, f)
    return [np.asarray(m) for m in mms]
