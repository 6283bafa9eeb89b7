"""Flagship benchmark: explanations/sec for distributed KernelSHAP on MI355X.

Default (--config adult) is the BASELINE.json headline metric:
"explanations/sec (2560 inst, 100-sample background) at 1/2/4/8 MI355X" —
12 feature groups, nsamples = 2*12 + 2048 = 2072, logit link, logistic-
regression predictor, synthetic Adult-shaped data with random-init weights.
Reference floors: 1.47 expl/s sequential, 20.5 expl/s
32-worker node, 44.9 expl/s 56-worker k8s cluster.

Extra configs (BASELINE.json configs 3-5):
  --config stress   256 features x 1000-sample background, nsamples=2^14
                    (library-GEMM + batched torch WLS paths, HBM streaming)
  --config mlp      64 features, MLP predictor (synth kernel + torch module)
  --config resnet   ResNet-18 on 224x224 superpixel masks (large
                    perturbation-batch predict path)

One *step* = explaining ``--instances`` instances per GPU (weak scaling:
per-GPU work fixed as N grows; whole-job value = N * instances / step_time).
Launched by the driver as  torchrun --nnodes=1 --nproc-per-node N bench.py
--gpus N ...  (one rank per GPU over RCCL).
"""
from __future__ import annotations

import argparse
import json
import time

import numpy as np

BASELINE_EXPL_PER_S = 20.5  # reference best single-node (125.05 s / 2560)

CONFIG_DEFAULTS = {
    "adult": {"instances": 2560, "background": 100},
    "stress": {"instances": 64, "background": 1000, "nsamples": 2 ** 14},
    "mlp": {"instances": 256, "background": 100},
    "resnet": {"instances": 2, "background": 1},
}


def build_problem(cfg: str, args, rank: int):
    """Returns (X, background, groups, group_names, predictor, nsamples,
    model_desc)."""
    from distributedkernelshap_amd.models import (
        LinearPredictor,
        TorchPredictor,
        make_adult_like,
        make_tabular,
    )

    if cfg == "adult":
        data = make_adult_like(
            n_instances=args.instances, n_background=args.background,
            seed=1000 + rank,
        )
        pred = LinearPredictor.random(data.X.shape[1], 2, seed=0)
        return (data.X, data.background, data.groups, data.group_names, pred,
                None, "adult-logreg (12 groups, D=50, n_out=2, logit link)")
    if cfg == "stress":
        data = make_tabular(
            n_features=256, n_instances=args.instances,
            n_background=args.background, seed=1000 + rank,
        )
        pred = LinearPredictor.random(256, 2, seed=0)
        return (data.X, data.background, data.groups, data.group_names, pred,
                args.nsamples or 2 ** 14,
                "stress-linear (256 groups, bg=1000, nsamples=2^14)")
    if cfg == "mlp":
        from distributedkernelshap_amd.models import make_predictor

        data = make_tabular(
            n_features=64, n_instances=args.instances,
            n_background=args.background, seed=1000 + rank,
        )
        pred = make_predictor("mlp", 64, 2, seed=0, hidden=256, layers=2)
        return (data.X, data.background, data.groups, data.group_names, pred,
                None, "mlp-256x2 (64 groups, torch predictor)")
    if cfg == "resnet":
        import torch

        from distributedkernelshap_amd.models.resnet import (
            make_superpixel_problem,
            resnet18,
        )

        X, bg, groups, names = make_superpixel_problem(
            n_instances=args.instances, hw=224, patch=32, seed=1000 + rank
        )
        module = resnet18(num_classes=10, seed=0)
        pred = TorchPredictor(module)
        return (X, bg, groups, names, pred, args.nsamples,
                "resnet18-superpixel (49 patches, 224x224)")
    raise ValueError(cfg)


def main() -> None:
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=5)
    p.add_argument("--warmup", type=int, default=2)
    p.add_argument("--config", default="adult",
                   choices=["adult", "stress", "mlp", "resnet"])
    p.add_argument("--instances", type=int, default=None,
                   help="instances per GPU (default per config)")
    p.add_argument("--background", type=int, default=None)
    p.add_argument("--nsamples", type=int, default=None)
    p.add_argument("--device", default="auto", choices=["auto", "cuda", "cpu"])
    p.add_argument("--dtype", default="fp32",
                   choices=["fp32", "bf16x2", "bf16", "fp64"],
                   help="GPU predict compute mode: fp32 MFMA (default), "
                        "bf16 matrix cores with hi+lo split (fp32-grade), "
                        "plain bf16 (fastest), or fp64 (full-double "
                        "verification mode, linear predictors)")
    p.add_argument("--selfcheck", type=int, default=64, metavar="K",
                   help="after timing, re-explain the first K instances in "
                        "the fp64 device mode (same masks) and report "
                        "max_phi_err_vs_fp64 in the JSON; 0 disables "
                        "(linear-predictor configs only)")
    args = p.parse_args()
    defaults = CONFIG_DEFAULTS[args.config]
    if args.instances is None:
        args.instances = defaults["instances"]
    if args.background is None:
        args.background = defaults["background"]
    if args.nsamples is None:
        args.nsamples = defaults.get("nsamples")

    import torch

    from distributedkernelshap_amd.core.engine import KernelShapEngine
    from distributedkernelshap_amd.parallel import (
        allgather_rows,
        broadcast_array,
        init_distributed,
        is_distributed,
    )

    rank, world = init_distributed()
    if args.gpus != world:
        raise SystemExit(
            f"--gpus {args.gpus} does not match the launched world size "
            f"{world}: for N>1 launch as `python -m torch.distributed.run "
            f"--nnodes=1 --nproc-per-node N --master-addr 127.0.0.1 bench.py "
            f"--gpus N ...` (one rank per GPU over RCCL)"
        )
    use_cuda = (
        torch.cuda.is_available() if args.device == "auto" else args.device == "cuda"
    )
    device = "cuda" if use_cuda else "cpu"

    X, background, groups, group_names, pred, nsamples, model_desc = build_problem(
        args.config, args, rank
    )
    # weights + background ship from rank 0 (RCCL broadcast over xGMI) — the
    # reference's actor-constructor broadcast
    from distributedkernelshap_amd.models import LinearPredictor

    if isinstance(pred, LinearPredictor):
        W = broadcast_array(pred.weights)
        b_ = broadcast_array(pred.bias)
        pred = LinearPredictor(W, b_, pred.activation)
    background = broadcast_array(background)

    from distributedkernelshap_amd.config import KernelConfig

    engine = KernelShapEngine(
        pred, background, groups=groups, link="logit", seed=0, device=device,
        kernels=KernelConfig(
            predict_dtype=args.dtype,
            # torch-module configs (mlp/resnet): bf16 dtype selects autocast
            # around the module forward
            module_autocast=("bf16" if args.dtype in ("bf16", "bf16x2")
                             else "off"),
        ),
    )

    if use_cuda:
        X_in = torch.from_numpy(X.astype(np.float32))
        try:
            X_in = X_in.pin_memory()
        except RuntimeError:
            pass
    else:
        X_in = X

    ekw = {}
    if nsamples is not None:
        ekw["nsamples"] = nsamples
    if args.config != "adult":
        # high-dim configs sample a tiny fraction of 2^M subsets, which would
        # trip shap's l1_reg='auto' LARS pre-selection (host-side, per
        # instance); the benchmark measures the un-regularised WLS path
        ekw["l1_reg"] = False

    def step():
        # gather per-instance shap rows to every rank — the reference's
        # order_result gather, done as ONE device-side fp32
        # RCCL all_gather_into_tensor over xGMI (no host bounce)
        if is_distributed():
            sv = engine.shap_values(
                X=X_in, instance_offset=rank * args.instances,
                as_tensor=True, **ekw
            )
            counts = [args.instances] * world
            return allgather_rows(sv, counts)
        sv = engine.shap_values(
            X=X_in, instance_offset=rank * args.instances, **ekw
        )
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

    if is_distributed():
        import torch.distributed as dist

        t = torch.tensor([elapsed], dtype=torch.float64,
                         device="cuda" if dist.get_backend() == "nccl" else "cpu")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    n_gpus = world
    ms_per_step = elapsed / args.steps * 1000.0
    value = n_gpus * args.instances * args.steps / elapsed

    # precision self-certification: re-explain a prefix
    # of this rank's shard in the fp64 device mode — identical masks via the
    # counter RNG — and report the measured fp32-pipeline error bound
    max_phi_err = None
    if (args.selfcheck and use_cuda and args.dtype != "fp64"
            and isinstance(pred, LinearPredictor)):
        k = min(args.selfcheck, args.instances)
        off = rank * args.instances
        sv_fast = engine.shap_values(X[:k], instance_offset=off, **ekw)
        eng64 = KernelShapEngine(
            pred, background, groups=groups, link="logit", seed=0,
            device=device, kernels=KernelConfig(predict_dtype="fp64"),
        )
        sv_64 = eng64.shap_values(X[:k], instance_offset=off, **ekw)
        max_phi_err = float(
            max(np.abs(sv_fast[o] - sv_64[o]).max() for o in range(len(sv_64)))
        )
        if is_distributed():
            import torch.distributed as dist

            e = torch.tensor(
                [max_phi_err], dtype=torch.float64,
                device="cuda" if dist.get_backend() == "nccl" else "cpu",
            )
            dist.all_reduce(e, op=dist.ReduceOp.MAX)
            max_phi_err = float(e.item())

    if rank == 0:
        headline = (args.config == "adult" and args.instances == 2560
                    and args.background == 100)
        metric = (
            "explanations/sec (2560 inst, 100-sample background)"
            if headline
            else f"explanations/sec ({args.config} config, "
                 f"{args.instances} inst/GPU)"
        )
        result = {
            "metric": metric,
            "value": value,
            "unit": "explanations/s",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_per_step,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": (value / BASELINE_EXPL_PER_S
                            if headline else None),
            "dtype": ("fp32" if args.dtype == "fp32" or device == "cpu"
                      else ("bf16x2 (hi+lo split, fp32-grade)"
                            if args.dtype == "bf16x2"
                            else args.dtype)),
            "max_phi_err_vs_fp64": max_phi_err,
            "data": "synthetic",
            "config": {
                "model": model_desc,
                "global_batch": n_gpus * args.instances,
                "background": args.background,
                "nsamples": args.nsamples or "default(2M+2048)",
                "parallelism": f"dp{n_gpus}",
                "device": device,
                "out_shape": list(out.shape),
            },
        }
        print(json.dumps(result))


if __name__ == "__main__":
    main()
