"""Results analysis (reference Analysis.ipynb C16 as a script): parse the
results pickles, print a table, optionally plot mean±std bars."""
import argparse
import glob
import os
import pickle
import re
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import numpy as np

PAT_POOL = re.compile(r"ray_workers_(-?\d+)_bsize_(\d+)_actorfr_([\d.]+)\.pkl")
PAT_SERVE = re.compile(r"ray_replicas_(\d+)_maxbatch_(\d+)\.pkl")


def collect(results_dir):
    rows = []
    for path in sorted(glob.glob(os.path.join(results_dir, "*.pkl"))):
        name = os.path.basename(path)
        with open(path, "rb") as f:
            res = pickle.load(f)
        t = np.asarray(res["t_elapsed"])
        m = PAT_POOL.match(name)
        if m:
            rows.append(("pool", int(m.group(1)), int(m.group(2)),
                         t.mean(), t.std(), len(t)))
            continue
        m = PAT_SERVE.match(name)
        if m:
            rows.append(("serve", int(m.group(1)), int(m.group(2)),
                         t.mean(), t.std(), len(t)))
    return rows


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--results-dir", default="results")
    p.add_argument("--plot", action="store_true")
    args = p.parse_args()
    rows = collect(args.results_dir)
    if not rows:
        print("no results found in", args.results_dir)
        return
    print(f"{'mode':6s} {'workers':>8s} {'batch':>6s} {'mean_s':>10s} {'std':>8s} {'runs':>5s}")
    for mode, w, b, mean, std, n in rows:
        print(f"{mode:6s} {w:8d} {b:6d} {mean:10.4f} {std:8.4f} {n:5d}")
    if args.plot:
        import matplotlib

        matplotlib.use("Agg")
        import matplotlib.pyplot as plt

        for mode in ("pool", "serve"):
            sel = [r for r in rows if r[0] == mode]
            if not sel:
                continue
            ws = [r[1] for r in sel]
            means = [r[3] for r in sel]
            stds = [r[4] for r in sel]
            plt.figure(figsize=(8, 4))
            plt.bar(range(len(sel)), means, yerr=stds)
            plt.xticks(range(len(sel)), [f"w{w}b{b}" for _, w, b, *_ in sel],
                       rotation=45)
            plt.ylabel("seconds (2,560 explanations)")
            plt.title(f"{mode} benchmark")
            plt.tight_layout()
            out = os.path.join(args.results_dir, f"{mode}_results.png")
            plt.savefig(out)
            print("wrote", out)


if __name__ == "__main__":
    main()
