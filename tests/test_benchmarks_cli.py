"""Smoke tests for the benchmark CLIs (tiny sizes, CPU)."""
import json
import os
import pickle
import subprocess
import sys

import pytest

ROOT = os.path.join(os.path.dirname(__file__), "..")


def _run(args, timeout=600, env_extra=None):
    env = dict(os.environ)
    if env_extra:
        env.update(env_extra)
    return subprocess.run(
        [sys.executable] + args, cwd=ROOT, capture_output=True, text=True,
        timeout=timeout, env=env,
    )


def test_bench_json_contract(tmp_path):
    r = _run(["bench.py", "--instances", "8", "--steps", "1", "--warmup", "0",
              "--device", "cpu"])
    assert r.returncode == 0, r.stderr[-800:]
    line = [l for l in r.stdout.splitlines() if l.startswith("{")][-1]
    obj = json.loads(line)
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                "dtype", "data", "config"):
        assert key in obj, key
    assert obj["n_gpus"] == 1 and obj["scaling"] == "weak"
    assert obj["config"]["global_batch"] == 8
    # a reduced instance count must NOT claim the headline metric/baseline
    assert "2560 inst" not in obj["metric"]
    assert obj["vs_baseline"] is None
    assert "max_phi_err_vs_fp64" in obj


def test_pool_cli_writes_results(tmp_path):
    assets = str(tmp_path / "assets")
    results = str(tmp_path / "results")
    r = _run([
        "benchmarks/pool.py", "--workers", "2", "--batch", "4", "--nruns", "1",
        "--instances", "8", "--assets-dir", assets, "--results-dir", results,
    ])
    assert r.returncode == 0, r.stderr[-800:]
    path = os.path.join(results, "ray_workers_2_bsize_4_actorfr_1.0.pkl")
    assert os.path.exists(path)
    with open(path, "rb") as f:
        res = pickle.load(f)
    assert len(res["t_elapsed"]) == 1


def test_analysis_cli(tmp_path):
    results = str(tmp_path)
    with open(os.path.join(results, "ray_workers_4_bsize_1_actorfr_1.0.pkl"), "wb") as f:
        pickle.dump({"t_elapsed": [1.0, 1.2]}, f)
    r = _run(["benchmarks/analysis.py", "--results-dir", results])
    assert r.returncode == 0
    assert "pool" in r.stdout and "4" in r.stdout


def test_serve_cli_tiny(tmp_path):
    assets = str(tmp_path / "assets")
    results = str(tmp_path / "results")
    r = _run([
        "benchmarks/serve_explanations.py", "--instances", "6", "--nruns", "1",
        "--benchmark", "0", "--concurrency", "4", "--port", "8877",
        "--assets-dir", assets, "--results-dir", results,
    ], timeout=900)
    assert r.returncode == 0, (r.stderr[-800:], r.stdout[-300:])
    assert os.path.exists(os.path.join(results, "ray_replicas_1_maxbatch_64.pkl"))
