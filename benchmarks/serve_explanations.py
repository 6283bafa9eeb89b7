"""HTTP serving benchmark (reference ``benchmarks/serve_explanations.py`` C8
parity): start a uvicorn server with the dynamic-batching app, fan the 2,560
test instances out as concurrent single-instance requests, time the whole
run, pickle ``{'t_elapsed': [...]}`` with the reference's filename scheme.
"""
import argparse
import concurrent.futures
import json
import logging
import multiprocessing as mp
import os
import pickle
import sys
import time
import timeit

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import numpy as np  # noqa: E402

logging.basicConfig(level=logging.INFO)
logger = logging.getLogger(__name__)


def _server_main(host, port, max_batch_size, assets_dir, device, replica_id=0):
    # pin each replica to a GPU round-robin (before torch import)
    if device in ("auto", "cuda"):
        n = os.environ.get("KSHAP_POOL_NGPUS")
        if n and int(n) > 0:
            os.environ["HIP_VISIBLE_DEVICES"] = str(replica_id % int(n))
    import uvicorn

    from distributedkernelshap_amd.serve import BatchKernelShapModel, create_app
    from distributedkernelshap_amd.utils import load_data, load_model

    data = load_data(assets_dir)
    clf = load_model(os.path.join(assets_dir, "predictor.pkl"))
    model = BatchKernelShapModel(
        clf,
        data.background,
        {"link": "logit", "seed": 0, "device": device},
        {"groups": data.groups, "group_names": data.group_names},
    )
    app = create_app(model, max_batch_size=max_batch_size)
    uvicorn.run(app, host=host, port=port, log_level="warning")


def distribute_requests(X, urls, max_workers, batch_mode="ray",
                        max_batch_size=64):
    """Fan requests out over replicas. 'ray' mode sends one request per
    instance (reference ``distribute_request``/``explain``,
    serve_explanations.py:96-139); 'default' pre-splits into client-side
    minibatches (reference k8s variant batch_mode)."""
    import http.client
    import itertools
    import threading
    import urllib.parse

    if batch_mode == "default":
        n_batches = (X.shape[0] + max_batch_size - 1) // max_batch_size
        instances = np.array_split(X, n_batches)
    else:
        instances = np.split(X, X.shape[0])
    tls = threading.local()
    thread_seq = itertools.count()
    parsed = [urllib.parse.urlsplit(u) for u in urls]

    def post(item):
        # persistent keep-alive http.client connection per thread, each
        # thread pinned to one replica (httpx cost ~1 ms/request dominated
        # the client side)
        _i, x = item
        conn = getattr(tls, "conn", None)
        if conn is None:
            tls.target = parsed[next(thread_seq) % len(parsed)]
            conn = tls.conn = http.client.HTTPConnection(
                tls.target.hostname, tls.target.port, timeout=120.0
            )
        body = json.dumps({"array": x.tolist()})
        headers = {"Content-Type": "application/json"}
        try:
            conn.request("POST", tls.target.path, body=body, headers=headers)
            resp = conn.getresponse()
            payload = resp.read()
        except (http.client.HTTPException, OSError):
            conn.close()
            conn = tls.conn = http.client.HTTPConnection(
                tls.target.hostname, tls.target.port, timeout=120.0
            )
            conn.request("POST", tls.target.path, body=body, headers=headers)
            resp = conn.getresponse()
            payload = resp.read()
        if resp.status != 200:
            raise RuntimeError(f"HTTP {resp.status}: {payload[:200]!r}")
        return payload.decode()

    with concurrent.futures.ThreadPoolExecutor(max_workers=max_workers) as pool:
        return list(pool.map(post, enumerate(instances)))


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--replicas", type=int, default=1,
                        help="server replica processes (round-robin clients); "
                             "one GPU replica already saturates a device")
    parser.add_argument("--max-batch-size", type=int, default=64)
    parser.add_argument("--instances", type=int, default=2560)
    parser.add_argument("--benchmark", type=int, default=1, choices=[0, 1],
                        help="0 = single run (reference flag parity)")
    parser.add_argument("--nruns", type=int, default=3)
    parser.add_argument("--concurrency", type=int, default=32)
    parser.add_argument("--batch-mode", default="ray", choices=["ray", "default"],
                        help="'ray': one request per instance (server-side "
                             "coalescing); 'default': client-side minibatches "
                             "of --max-batch-size (reference "
                             "k8s_serve_explanations.py:180-185 parity)")
    parser.add_argument("--host", default="127.0.0.1")
    parser.add_argument("--port", type=int, default=8800)
    parser.add_argument("--device", default="auto")
    parser.add_argument("--assets-dir", default="assets")
    parser.add_argument("--results-dir", default="results")
    args = parser.parse_args()

    if not args.benchmark:
        args.nruns = 1

    from distributedkernelshap_amd.utils import get_filename, load_data

    data = load_data(args.assets_dir)
    X = data.X_test[: args.instances]
    assert X.shape[0] == args.instances
    if not os.path.exists(os.path.join(args.assets_dir, "predictor.pkl")):
        logger.info("no fitted model found; running scripts/fit_model.py")
        import subprocess

        subprocess.run(
            [sys.executable,
             os.path.join(os.path.dirname(__file__), "..", "scripts", "fit_model.py"),
             "--assets-dir", args.assets_dir],
            check=True,
        )

    ctx = mp.get_context("spawn")
    servers = []
    urls = []
    for r in range(args.replicas):
        port = args.port + r
        srv = ctx.Process(
            target=_server_main,
            args=(args.host, port, args.max_batch_size, args.assets_dir,
                  args.device, r),
            daemon=True,
        )
        srv.start()
        servers.append(srv)
        urls.append(f"http://{args.host}:{port}/explain")
    # wait for readiness of every replica
    import httpx

    for r in range(args.replicas):
        for _ in range(600):
            try:
                if httpx.get(
                    f"http://{args.host}:{args.port + r}/healthz", timeout=2.0
                ).status_code == 200:
                    break
            except Exception:
                time.sleep(0.5)
        else:
            raise RuntimeError(f"server replica {r} did not become ready")

    path = get_filename(
        args.replicas, 0, serve=True, max_batch_size=args.max_batch_size,
        results_dir=args.results_dir,
    )
    result = {"t_elapsed": []}
    try:
        for run in range(args.nruns):
            logger.info("run %d/%d", run + 1, args.nruns)
            t_start = timeit.default_timer()
            responses = distribute_requests(
                X, urls, args.concurrency, args.batch_mode, args.max_batch_size
            )
            t_elapsed = timeit.default_timer() - t_start
            logger.info("Time elapsed: %.4f s (%d responses)",
                        t_elapsed, len(responses))
            sv = np.asarray(json.loads(responses[0])["data"]["shap_values"][0])
            assert sv.shape[1] == len(data.groups)
            result["t_elapsed"].append(t_elapsed)
            with open(path, "wb") as f:
                pickle.dump(result, f)
    finally:
        for srv in servers:
            srv.terminate()
        for srv in servers:
            srv.join(timeout=10)


if __name__ == "__main__":
    main()
