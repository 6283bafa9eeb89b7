"""Serving façade tests (C5/C8 parity): replicas, JSON contract, dynamic
batching equivalence."""
import json

import numpy as np
import pytest

from distributedkernelshap_amd.models import LinearPredictor, make_adult_like
from distributedkernelshap_amd.serve import (
    BatchKernelShapModel,
    KernelShapModel,
    create_app,
)


@pytest.fixture(scope="module")
def served():
    data = make_adult_like(n_instances=6, n_background=20, seed=3)
    pred = LinearPredictor.random(data.X.shape[1], 2, seed=3)
    ckw = {"link": "logit", "device": "cpu"}
    fkw = {"groups": data.groups, "group_names": data.group_names}
    return data, pred, ckw, fkw


def test_model_replica_json(served):
    data, pred, ckw, fkw = served
    model = KernelShapModel(pred, data.background, ckw, fkw)
    body = model({"array": data.X[:1].tolist()})
    obj = json.loads(body)
    assert len(obj["data"]["shap_values"]) == 2
    assert np.asarray(obj["data"]["shap_values"][0]).shape == (1, 12)
    # round-trips through the Explanation contract
    from distributedkernelshap_amd.interface import Explanation

    exp = Explanation.from_json(body)
    assert exp.data["link"] == "logit"


def test_batch_model_single_call_equals_stacked(served):
    """Coalesced batch == one explain over the stacked array (identical RNG
    keys); per-request singles differ only by sampling noise, so local
    accuracy is the invariant there."""
    data, pred, ckw, fkw = served
    model = BatchKernelShapModel(pred, data.background, ckw, fkw)
    reqs = [{"array": data.X[i : i + 1].tolist()} for i in range(4)]
    batched = model.batch(reqs)
    stacked = model({"array": data.X[:4].tolist()})
    sv_stacked = np.asarray(json.loads(stacked)["data"]["shap_values"][0])
    for i, bs in enumerate(batched):
        b = np.asarray(json.loads(bs)["data"]["shap_values"][0])
        assert np.allclose(b, sv_stacked[i : i + 1], atol=1e-12)
    # per-request singles: same local-accuracy total despite different masks
    from distributedkernelshap_amd.core.links import logit

    ev = np.asarray(model.explainer.expected_value)
    for i, r in enumerate(reqs):
        s = np.asarray(json.loads(model(r))["data"]["shap_values"][0])
        total = s.sum(axis=1)[0] + ev[0]
        assert np.isclose(total, logit(pred(data.X[i : i + 1]))[0, 0], atol=1e-9)


def test_http_app_dynamic_batching(served):
    from fastapi.testclient import TestClient

    data, pred, ckw, fkw = served
    model = BatchKernelShapModel(pred, data.background, ckw, fkw)
    app = create_app(model, max_batch_size=8, max_wait_ms=5)
    with TestClient(app) as client:
        assert client.get("/healthz").json()["status"] == "ok"
        ev = client.get("/expected_value").json()["expected_value"]
        assert len(ev) == 2
        r = client.post("/explain", json={"array": data.X[:2].tolist()})
        assert r.status_code == 200
        obj = r.json()
        assert np.asarray(obj["data"]["shap_values"][0]).shape == (2, 12)


def test_http_app_concurrent_requests(served):
    """Concurrent posts are coalesced by the batcher and all answered."""
    import threading

    from fastapi.testclient import TestClient

    data, pred, ckw, fkw = served
    model = BatchKernelShapModel(pred, data.background, ckw, fkw)
    app = create_app(model, max_batch_size=8, max_wait_ms=20)
    results = {}
    with TestClient(app) as client:
        def post(i):
            r = client.post("/explain", json={"array": data.X[i : i + 1].tolist()})
            results[i] = r

        threads = [threading.Thread(target=post, args=(i,)) for i in range(6)]
        for th in threads:
            th.start()
        for th in threads:
            th.join()
    assert len(results) == 6
    for i, r in results.items():
        assert r.status_code == 200
        sv = np.asarray(r.json()["data"]["shap_values"][0])
        assert sv.shape == (1, 12)


def test_metrics_endpoint(served):
    from fastapi.testclient import TestClient

    data, pred, ckw, fkw = served
    model = BatchKernelShapModel(pred, data.background, ckw, fkw)
    app = create_app(model, max_batch_size=4, max_wait_ms=5)
    with TestClient(app) as client:
        client.post("/explain", json={"array": data.X[:1].tolist()})
        r = client.get("/metrics")
        assert r.status_code == 200
        assert b"kshap_requests_total" in r.content


def test_bad_requests_return_400(served):
    from fastapi.testclient import TestClient

    data, pred, ckw, fkw = served
    model = BatchKernelShapModel(pred, data.background, ckw, fkw)
    app = create_app(model, max_batch_size=4, max_wait_ms=5)
    with TestClient(app) as client:
        assert client.post("/explain", content=b"not json").status_code == 400
        assert client.post("/explain", json={"nope": 1}).status_code == 400
        # wrong feature width -> engine error surfaced as 400, server stays up
        r = client.post("/explain", json={"array": [[1.0, 2.0]]})
        assert r.status_code == 400
        # ragged (non-numeric-matrix) array rejected before batching
        assert client.post(
            "/explain", json={"array": [[1.0, 2.0], [1.0]]}
        ).status_code == 400
        # and a good request still works afterwards
        ok = client.post("/explain", json={"array": data.X[:1].tolist()})
        assert ok.status_code == 200


def test_malformed_request_does_not_poison_cobatched(served):
    """A malformed request coalesced with good ones must 400 alone: good
    requests in the same dynamic batch still return 200 (pre-enqueue
    validation + per-request fallback in the batcher)."""
    import concurrent.futures

    from fastapi.testclient import TestClient

    data, pred, ckw, fkw = served
    model = BatchKernelShapModel(pred, data.background, ckw, fkw)
    app = create_app(model, max_batch_size=8, max_wait_ms=50)
    with TestClient(app) as client:
        def post(payload):
            return client.post("/explain", json=payload)

        good = {"array": data.X[:1].tolist()}
        bad = {"array": [[1.0, 2.0]]}          # wrong width
        with concurrent.futures.ThreadPoolExecutor(4) as ex:
            futs = [ex.submit(post, good), ex.submit(post, bad),
                    ex.submit(post, good), ex.submit(post, good)]
            codes = [f.result().status_code for f in futs]
        assert codes[1] == 400
        assert codes[0] == codes[2] == codes[3] == 200
