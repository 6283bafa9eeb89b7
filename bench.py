"""Flagship benchmark: explanations/sec on the Adult-shaped KernelSHAP config.

BASELINE.json metric: "explanations/sec (2560 inst, 100-sample background) at
1/2/4/8 MI355X" — 12 feature groups, nsamples = 2*12 + 2048 = 2072, logit
link, logistic-regression predictor, synthetic Adult-shaped data with
random-init weights (no network). Reference floors: 1.47 expl/s
sequential, 20.5 expl/s 32-worker node, 44.9 expl/s 56-worker k8s cluster.

One *step* = explaining 2,560 instances per GPU (weak scaling: per-GPU work
fixed as N grows; whole-job value = N * 2560 / step_time). Launched by the
driver as  torchrun --nnodes=1 --nproc-per-node N bench.py --gpus N ...
(one rank per GPU over RCCL).
"""
from __future__ import annotations

import argparse
import json
import os
import time

import numpy as np

BASELINE_EXPL_PER_S = 20.5  # reference best single-node (125.05 s / 2560)


def main() -> None:
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=5)
    p.add_argument("--warmup", type=int, default=2)
    p.add_argument("--instances", type=int, default=2560, help="instances per GPU")
    p.add_argument("--background", type=int, default=100)
    p.add_argument("--device", default="auto", choices=["auto", "cuda", "cpu"])
    args = p.parse_args()

    import torch

    from distributedkernelshap_amd.core.engine import KernelShapEngine
    from distributedkernelshap_amd.models import LinearPredictor, make_adult_like
    from distributedkernelshap_amd.parallel import (
        allgather_rows,
        broadcast_array,
        init_distributed,
        is_distributed,
    )

    rank, world = init_distributed()
    use_cuda = (
        torch.cuda.is_available() if args.device == "auto" else args.device == "cuda"
    )
    device = "cuda" if use_cuda else "cpu"

    # per-rank synthetic shard (weak scaling: fixed per-GPU work); model
    # weights + background are created on rank 0 and broadcast (RCCL/xGMI)
    data = make_adult_like(
        n_instances=args.instances, n_background=args.background, seed=1000 + rank
    )
    pred0 = LinearPredictor.random(data.X.shape[1], 2, seed=0)
    W = broadcast_array(pred0.weights)
    bias = broadcast_array(pred0.bias)
    background = broadcast_array(data.background)
    pred = LinearPredictor(W, bias)

    engine = KernelShapEngine(
        pred,
        background,
        groups=data.groups,
        link="logit",
        seed=0,
        device=device,
    )

    if use_cuda:
        X_in = torch.from_numpy(data.X.astype(np.float32))
        try:
            X_in = X_in.pin_memory()
        except RuntimeError:
            pass
    else:
        X_in = data.X

    def step() -> np.ndarray:
        sv = engine.shap_values(X=X_in, instance_offset=rank * args.instances)
        # gather per-instance shap rows (class 0) to every rank, reference
        # order_result parity
        if is_distributed():
            counts = [args.instances] * world
            return allgather_rows(sv[0], counts)
        return sv[0]

    def sync():
        if use_cuda:
            torch.cuda.synchronize()
        if is_distributed():
            import torch.distributed as dist

            dist.barrier()

    for _ in range(args.warmup):
        step()
    sync()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        out = step()
    sync()
    elapsed = time.perf_counter() - t0

    # max over ranks
    if is_distributed():
        import torch.distributed as dist

        t = torch.tensor([elapsed], dtype=torch.float64,
                         device="cuda" if dist.get_backend() == "nccl" else "cpu")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    n_gpus = world
    ms_per_step = elapsed / args.steps * 1000.0
    value = n_gpus * args.instances * args.steps / elapsed
    if rank == 0:
        result = {
            "metric": "explanations/sec (2560 inst, 100-sample background)",
            "value": value,
            "unit": "explanations/s",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_per_step,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": value / BASELINE_EXPL_PER_S,
            "dtype": "fp32",
            "data": "synthetic",
            "config": {
                "model": "adult-logreg (12 groups, D=50, n_out=2, logit link)",
                "global_batch": n_gpus * args.instances,
                "background": args.background,
                "nsamples": 2072,
                "parallelism": f"dp{n_gpus}",
                "device": device,
                "out_shape": list(out.shape),
            },
        }
        print(json.dumps(result))


if __name__ == "__main__":
    main()
