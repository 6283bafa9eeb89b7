"""Config system + utils tests (SURVEY.md §5.6 / C6 parity)."""
import os

import numpy as np
import pytest

from distributedkernelshap_amd.config import Config
from distributedkernelshap_amd.utils import (
    Bunch,
    batch,
    get_filename,
    load_data,
    load_model,
    methdispatch,
)


def test_config_roundtrip(tmp_path):
    toml = tmp_path / "cfg.toml"
    toml.write_text(
        """
[engine]
link = "logit"
seed = 7

[distributed]
n_workers = 4
batch_size = 10

[bench]
instances = 128
"""
    )
    cfg = Config.from_toml(str(toml))
    assert cfg.engine.link == "logit" and cfg.engine.seed == 7
    assert cfg.distributed.n_workers == 4
    assert cfg.distributed.to_opts()["batch_size"] == 10
    assert cfg.bench.instances == 128
    assert cfg.serve.port == 8800  # defaults preserved
    d = cfg.to_dict()
    assert d["kernels"]["wls_mode"] == "auto"


def test_config_unknown_key():
    with pytest.raises(KeyError):
        Config.from_dict({"engine": {"nope": 1}})
    with pytest.raises(KeyError):
        Config.from_dict({"nope": {}})


def test_bunch():
    b = Bunch(a=1, b=2)
    assert b.a == 1 and b["b"] == 2
    b.c = 3
    assert b["c"] == 3
    with pytest.raises(AttributeError):
        b.nope


def test_batch_splitting():
    X = np.arange(20).reshape(10, 2)
    bs = batch(X, batch_size=3)
    assert [len(x) for x in bs] == [3, 3, 3, 1]
    bs = batch(X, n_batches=4)
    assert sum(len(x) for x in bs) == 10
    assert np.concatenate(bs).tolist() == X.tolist()


def test_get_filename(tmp_path):
    p = get_filename(8, 5, 0.5, results_dir=str(tmp_path))
    assert p.endswith("ray_workers_8_bsize_5_actorfr_0.5.pkl")
    p = get_filename(4, 0, serve=True, max_batch_size=10, results_dir=str(tmp_path))
    assert p.endswith("ray_replicas_4_maxbatch_10.pkl")


def test_methdispatch():
    class C:
        @methdispatch
        def f(self, x):
            return "default"

        @f.register(int)
        def _(self, x):
            return "int"

    c = C()
    assert c.f(1) == "int"
    assert c.f("s") == "default"


def test_load_data_and_model_cache(tmp_path):
    d = str(tmp_path)
    data = load_data(d)
    assert data.X_test.shape == (2560, 50)
    assert data.background.shape == (100, 50)
    assert len(data.groups) == 12
    # cached second load is identical
    data2 = load_data(d)
    assert np.array_equal(data.X_test, data2.X_test)
    # fit + load model round trip (sklearn -> native LinearPredictor)
    from sklearn.linear_model import LogisticRegression

    import pickle

    clf = LogisticRegression(max_iter=200).fit(
        data.X_train[:2000], data.y_train[:2000]
    )
    with open(os.path.join(d, "predictor.pkl"), "wb") as f:
        pickle.dump(clf, f)
    pred = load_model(os.path.join(d, "predictor.pkl"))
    probs = pred(data.X_test[:50])
    ref = clf.predict_proba(data.X_test[:50])
    assert np.allclose(probs, ref, atol=1e-10)
    assert hasattr(pred, "linear_params")


def test_pool_cli_config_file(tmp_path):
    import subprocess
    import sys

    cfgf = tmp_path / "c.toml"
    cfgf.write_text(
        f"""
[bench]
instances = 6
nruns = 1
batch_sizes = [3]
results_dir = "{tmp_path}/res"
assets_dir = "{tmp_path}/assets"

[distributed]
n_workers = 2
"""
    )
    root = os.path.join(os.path.dirname(__file__), "..")
    r = subprocess.run(
        [sys.executable, "benchmarks/pool.py", "--config-file", str(cfgf)],
        cwd=root, capture_output=True, text=True, timeout=600,
    )
    assert r.returncode == 0, r.stderr[-500:]
    assert os.path.exists(
        os.path.join(str(tmp_path), "res", "ray_workers_2_bsize_3_actorfr_1.0.pkl")
    )


def test_kernel_config_on_cpu_engine():
    """predict_dtype/wls_mode are GPU dispatch knobs; a CPU engine accepts
    them without effect."""
    from distributedkernelshap_amd.config import KernelConfig
    from distributedkernelshap_amd.core.engine import KernelShapEngine
    from distributedkernelshap_amd.models import LinearPredictor

    rng = np.random.default_rng(0)
    pred = LinearPredictor.random(6, 2)
    eng = KernelShapEngine(
        pred, rng.normal(size=(10, 6)), link="logit", device="cpu",
        kernels=KernelConfig(predict_dtype="bf16x2", wls_mode="torch"),
    )
    sv = eng.shap_values(rng.normal(size=(2, 6)))
    assert sv[0].shape == (2, 6)
