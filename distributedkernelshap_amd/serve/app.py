"""HTTP serving façade with dynamic batching.

Native replacement for the reference's ray-serve deployment
(``explainers/wrappers.py`` C5 + ``benchmarks/serve_explanations.py`` C8):

* :class:`KernelShapModel` / :class:`BatchKernelShapModel` — request-callable
  replicas constructing + fitting a ``KernelShap`` at build time (the
  reference fitted one per serve replica, ``wrappers.py:41``). Batch requests
  are explained in ONE engine call — the reference looped instances
  one-by-one under ``@serve.accept_batch`` (``wrappers.py:81-88``, a known
  quirk SURVEY.md §2.8).
* :func:`create_app` — FastAPI app whose ``/explain`` endpoint feeds an
  asyncio dynamic batcher: requests are coalesced up to ``max_batch_size`` or
  ``max_wait_ms`` and dispatched to the engine in a worker thread, so GPU
  batching replaces ray's request router.
"""
# NOTE: no `from __future__ import annotations` here — FastAPI resolves the
# /explain endpoint's `Request` annotation via get_type_hints, which fails on
# stringified annotations when the type is imported inside create_app.
import asyncio
import json
import logging
import os
import time
from typing import Any, Dict, List, Optional

import numpy as np

logger = logging.getLogger(__name__)

__all__ = ["KernelShapModel", "BatchKernelShapModel", "create_app"]

_METRICS = None


def _metrics():
    """Process-wide metric singletons (SURVEY.md §5.5 observability);
    prometheus registries reject duplicate names, so create_app may run many
    times but metrics register once."""
    global _METRICS
    if _METRICS is None:
        try:
            from prometheus_client import (
                CONTENT_TYPE_LATEST,
                Counter,
                Histogram,
                generate_latest,
            )

            _METRICS = (
                Counter("kshap_requests_total", "explain requests"),
                Histogram("kshap_request_seconds", "explain latency"),
                Histogram(
                    "kshap_batch_size", "coalesced batch sizes",
                    buckets=[1, 2, 4, 8, 16, 32, 64, 128],
                ),
                generate_latest,
                CONTENT_TYPE_LATEST,
            )
        except ImportError:  # pragma: no cover
            _METRICS = (None, None, None, None, None)
    return _METRICS


class KernelShapModel:
    """Serving replica: fit at construction, explain per request
    (reference ``explainers/wrappers.py:10-59``)."""

    def __init__(
        self,
        predictor,
        background_data: np.ndarray,
        constructor_kwargs: Optional[Dict[str, Any]] = None,
        fit_kwargs: Optional[Dict[str, Any]] = None,
    ):
        from ..explainers.kernel_shap import KernelShap

        constructor_kwargs = dict(constructor_kwargs or {})
        fit_kwargs = dict(fit_kwargs or {})
        if not callable(predictor):
            raise TypeError("predictor must be callable (e.g. predict_proba)")
        self.explainer = KernelShap(predictor, **constructor_kwargs)
        self.explainer.fit(background_data, **fit_kwargs)

    def __call__(self, request_json: Dict[str, Any]) -> str:
        arr = np.asarray(request_json["array"], dtype=np.float64)
        explanation = self.explainer.explain(arr, silent=True)
        return explanation.to_json()


class BatchKernelShapModel(KernelShapModel):
    """Batch replica: one engine call for N coalesced requests
    (reference ``explainers/wrappers.py:62-88``, minus the per-instance loop)."""

    def batch(self, requests_json: List[Dict[str, Any]]) -> List[str]:
        arrays = [np.atleast_2d(np.asarray(r["array"], dtype=np.float64))
                  for r in requests_json]
        sizes = [a.shape[0] for a in arrays]
        stacked = np.concatenate(arrays, axis=0)
        explanation = self.explainer.explain(stacked, silent=True)
        sv = explanation.data["shap_values"]
        out = []
        lo = 0
        for size in sizes:
            piece = {
                "meta": explanation.meta,
                "data": {
                    **{k: v for k, v in explanation.data.items()
                       if k not in ("shap_values", "raw")},
                    "shap_values": [s[lo : lo + size] for s in sv],
                },
            }
            from ..interface import NumpyEncoder

            out.append(json.dumps(piece, cls=NumpyEncoder))
            lo += size
        return out


def create_app(
    model: KernelShapModel,
    max_batch_size: int = 64,
    max_wait_ms: float = 2.0,
):
    """Build the FastAPI app with an asyncio dynamic batcher in front of the
    replica (replaces ray-serve's router + ``@serve.accept_batch``)."""
    from fastapi import FastAPI, Request

    app = FastAPI(title="distributedkernelshap-amd")
    queue: asyncio.Queue = asyncio.Queue()

    req_count, req_latency, batch_hist, generate_latest, content_type = _metrics()

    async def _batcher():
        while True:
            item = await queue.get()
            batch = [item]
            try:
                deadline = asyncio.get_event_loop().time() + max_wait_ms / 1e3
                while len(batch) < max_batch_size:
                    timeout = deadline - asyncio.get_event_loop().time()
                    if timeout <= 0:
                        break
                    batch.append(await asyncio.wait_for(queue.get(), timeout))
            except asyncio.TimeoutError:
                pass
            payloads = [b[0] for b in batch]
            futures = [b[1] for b in batch]
            if batch_hist is not None:
                batch_hist.observe(len(batch))
            loop = asyncio.get_event_loop()
            _dbg = os.environ.get("KSHAP_TIMING") == "1"
            _t0 = time.perf_counter() if _dbg else 0.0
            try:
                if isinstance(model, BatchKernelShapModel) and len(payloads) > 1:
                    try:
                        results = await loop.run_in_executor(
                            None, model.batch, payloads
                        )
                    except Exception:
                        # one malformed co-batched request must not 400 the
                        # well-formed ones: retry each request individually so
                        # only the offending ones fail
                        results = []
                        for p in payloads:
                            try:
                                results.append(
                                    await loop.run_in_executor(None, model, p)
                                )
                            except Exception as e:
                                results.append(e)
                elif isinstance(model, BatchKernelShapModel):
                    results = await loop.run_in_executor(None, model.batch, payloads)
                else:
                    results = []
                    for p in payloads:
                        try:
                            results.append(await loop.run_in_executor(None, model, p))
                        except Exception as e:
                            results.append(e)
                if _dbg:
                    print(
                        f"[kshap-serve] batch={len(batch)} "
                        f"explain_ms={(time.perf_counter() - _t0) * 1e3:.1f}",
                        flush=True,
                    )
                for fut, res in zip(futures, results):
                    if fut.done():
                        continue
                    if isinstance(res, Exception):
                        fut.set_exception(res)
                    else:
                        fut.set_result(res)
            except Exception as e:  # propagate to every waiter
                for fut in futures:
                    if not fut.done():
                        fut.set_exception(e)

    @app.on_event("startup")
    async def _start():
        app.state.batcher = asyncio.create_task(_batcher())

    @app.on_event("shutdown")
    async def _stop():
        app.state.batcher.cancel()

    @app.get("/healthz")
    async def healthz():
        return {"status": "ok"}

    @app.get("/expected_value")
    async def expected_value():
        return {"expected_value": np.asarray(model.explainer.expected_value).tolist()}

    @app.api_route("/explain", methods=["GET", "POST"])
    async def explain(request: Request):
        from fastapi.responses import JSONResponse

        t0 = time.perf_counter()
        try:
            payload = await request.json()
        except Exception:
            return JSONResponse({"error": "body must be JSON"}, status_code=400)
        if not isinstance(payload, dict) or "array" not in payload:
            return JSONResponse(
                {"error": "payload must be a JSON object with an 'array' key"},
                status_code=400,
            )
        # validate shape/dtype BEFORE enqueueing so a malformed request can
        # never poison the batch it would have been coalesced into
        try:
            arr = np.atleast_2d(np.asarray(payload["array"], dtype=np.float64))
            expected_d = getattr(
                getattr(model.explainer, "_engine", None), "D", None
            )
            if expected_d is not None and arr.shape[1] != expected_d:
                return JSONResponse(
                    {"error": f"'array' must have {expected_d} columns, "
                              f"got {arr.shape[1]}"},
                    status_code=400,
                )
        except (ValueError, TypeError) as e:
            return JSONResponse(
                {"error": f"'array' is not a numeric matrix: {e}"},
                status_code=400,
            )
        fut: asyncio.Future = asyncio.get_event_loop().create_future()
        await queue.put((payload, fut))
        try:
            body = await fut
        except Exception as e:
            return JSONResponse(
                {"error": f"explanation failed: {e!r}"}, status_code=400
            )
        if req_count is not None:
            req_count.inc()
            req_latency.observe(time.perf_counter() - t0)
        from fastapi.responses import Response

        return Response(content=body, media_type="application/json")

    @app.get("/metrics")
    async def metrics():
        from fastapi.responses import Response

        if req_count is None:
            return Response(status_code=404)
        return Response(content=generate_latest(), media_type=content_type)

    return app
