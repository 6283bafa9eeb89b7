"""Worker-pool benchmark CLI (reference ``benchmarks/ray_pool.py`` C7 parity).

Explains the 2,560-instance test set with a pool of worker-process replicas
(``--workers -1`` = sequential in-process baseline, like the reference), for
each minibatch size in ``--batch``, ``--nruns`` times; appends wall-clock
timings to a results pickle named with the reference's filename scheme so the
analysis tooling parses either framework's results.

On a GPU box a single worker saturates the MI355X — sweep ``--workers`` for
API parity, sweep GPUs via ``torchrun bench.py`` for the scaling curve.
"""
import argparse
import logging
import os
import pickle
import sys
import timeit

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import numpy as np  # noqa: E402

logging.basicConfig(level=logging.INFO)
logger = logging.getLogger(__name__)


def fit_kernel_shap_explainer(clf, background, groups, group_names, distributed_opts,
                              device="auto"):
    """Reference ``fit_kernel_shap_explainer`` (ray_pool.py:18-38): logit
    link, seed 0, grouped categorical columns."""
    from distributedkernelshap_amd import KernelShap

    explainer = KernelShap(
        clf, link="logit", feature_names=group_names, seed=0,
        distributed_opts=distributed_opts, device=device,
    )
    explainer.fit(background, group_names=group_names, groups=groups)
    return explainer


def run_explainer(explainer, X, batch_size, nruns, result_path):
    """Timer loop with incremental pickling (ray_pool.py:41-79: each run
    re-writes the file so a killed sweep keeps completed runs)."""
    result = {"t_elapsed": []}
    for run in range(nruns):
        logger.info("run %d/%d", run + 1, nruns)
        t_start = timeit.default_timer()
        explanation = explainer.explain(X)
        t_elapsed = timeit.default_timer() - t_start
        logger.info("Time elapsed: %.4f s", t_elapsed)
        result["t_elapsed"].append(t_elapsed)
        with open(result_path, "wb") as f:
            pickle.dump(result, f)
    return explanation


def main():
    from distributedkernelshap_amd.utils import get_filename, load_data, load_model

    parser = argparse.ArgumentParser()
    parser.add_argument("--workers", type=int, default=-1,
                        help="-1 = sequential baseline (no pool)")
    parser.add_argument("-b", "--batch", nargs="+", type=int, default=[1, 5, 10])
    parser.add_argument("--benchmark", type=int, default=1, choices=[0, 1],
                        help="0 = single run of the first batch size "
                             "(reference ray_pool.py flag parity)")
    parser.add_argument("--nruns", type=int, default=5)
    parser.add_argument("--instances", type=int, default=2560)
    parser.add_argument("--actor-cpu-fraction", type=float, default=1.0)
    parser.add_argument("--device", default="auto")
    parser.add_argument("--assets-dir", default="assets")
    parser.add_argument("--results-dir", default="results")
    parser.add_argument("--config-file", default=None,
                        help="TOML config (distributedkernelshap_amd.config) "
                             "overriding bench/engine defaults")
    args = parser.parse_args()
    if not args.benchmark:
        args.nruns = 1
        args.batch = args.batch[:1]
    if args.config_file:
        from distributedkernelshap_amd.config import Config

        cfg = Config.from_toml(args.config_file)
        args.instances = cfg.bench.instances
        args.nruns = cfg.bench.nruns
        args.batch = cfg.bench.batch_sizes
        args.results_dir = cfg.bench.results_dir
        args.assets_dir = cfg.bench.assets_dir
        if cfg.distributed.n_workers:
            args.workers = cfg.distributed.n_workers

    data = load_data(args.assets_dir)
    model_path = os.path.join(args.assets_dir, "predictor.pkl")
    if not os.path.exists(model_path):
        logger.info("no fitted model found; running scripts/fit_model.py")
        import subprocess

        subprocess.run(
            [sys.executable, os.path.join(os.path.dirname(__file__), "..",
                                          "scripts", "fit_model.py"),
             "--assets-dir", args.assets_dir],
            check=True,
        )
    clf = load_model(model_path)
    acc = float(np.mean(np.argmax(clf(data.X_test), axis=1) == data.y_test))
    logger.info("Model test accuracy: %.4f", acc)
    X = data.X_test[: args.instances]

    if args.workers == -1:
        logger.info("sequential baseline (no pool)")
        explainer = fit_kernel_shap_explainer(
            clf, data.background, data.groups, data.group_names, None, args.device
        )
        path = get_filename(-1, 0, results_dir=args.results_dir)
        run_explainer(explainer, X, None, args.nruns, path)
        return

    for batch_size in args.batch:
        logger.info("workers=%d batch_size=%d", args.workers, batch_size)
        opts = {
            "n_workers": args.workers,
            "batch_size": batch_size,
            "actor_cpu_fraction": args.actor_cpu_fraction,
        }
        explainer = fit_kernel_shap_explainer(
            clf, data.background, data.groups, data.group_names, opts, args.device
        )
        path = get_filename(
            args.workers, batch_size, args.actor_cpu_fraction,
            results_dir=args.results_dir,
        )
        run_explainer(explainer, X, batch_size, args.nruns, path)
        if hasattr(explainer._explainer, "shutdown"):
            explainer._explainer.shutdown()


if __name__ == "__main__":
    main()
