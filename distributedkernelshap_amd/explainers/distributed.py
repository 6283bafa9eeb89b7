"""Distribution engine: data-parallel explanation over instances.

Native re-design of the reference's ray actor pool
(``explainers/distributed.py``, C4 in SURVEY.md §2.1). Two modes:

* **pool** (default; API parity with the reference): N worker *processes*
  spawned with ``multiprocessing``, each holding a fitted explainer replica
  (construction == the reference's actor-ctor broadcast). Minibatches are
  dispatched with ``(batch_idx, batch)`` tagging, results collected unordered,
  the completion permutation inverted and per-class results concatenated —
  the exact semantics of ``DistributedExplainer.get_explanation`` /
  ``order_result`` (reference ``explainers/distributed.py:130-179``).

* **collective**: when ``torch.distributed`` is already initialised (one rank
  per MI355X GPU over RCCL/xGMI, launched by torchrun), every rank explains a
  static contiguous shard and results are all-gathered. This is the bench.py
  path — see ``distributedkernelshap_amd.parallel``.

The per-instance counter-based RNG (``core.sampler``) makes results identical
across worker counts and modes — unlike the reference, where per-actor numpy
reseeding made distributed results depend on the dispatch schedule.
"""
from __future__ import annotations

import logging
import multiprocessing as mp
from typing import Any, Callable, Dict, List, Sequence, Tuple

import numpy as np

from ..utils import batch as batch_split

logger = logging.getLogger(__name__)

__all__ = [
    "DistributedExplainer",
    "invert_permutation",
    "kernel_shap_target_fn",
    "kernel_shap_postprocess_fn",
]


def invert_permutation(p: Sequence[int]) -> np.ndarray:
    """Return s with s[p[i]] = i (reference ``explainers/distributed.py:65-82``)."""
    p = np.asarray(p)
    s = np.empty_like(p)
    s[p] = np.arange(p.size)
    return s


def kernel_shap_target_fn(worker, item: Tuple[int, np.ndarray, int], kwargs=None):
    """Dispatch one tagged minibatch to a worker replica
    (reference ``explainers/distributed.py:11-34``)."""
    kwargs = kwargs or {}
    batch_idx, batch, offset = item
    return worker.get_explanation((batch_idx, batch), instance_offset=offset, **kwargs)


def kernel_shap_postprocess_fn(
    ordered_result: List[List[np.ndarray]],
) -> List[np.ndarray]:
    """Concatenate per-class shap value blocks
    (reference ``explainers/distributed.py:37-62``)."""
    n_classes = len(ordered_result[0])
    return [
        np.concatenate([r[c] for r in ordered_result], axis=0)
        for c in range(n_classes)
    ]


def _worker_main(worker_id, explainer_type, init_args, init_kwargs, task_q,
                 result_q, attr_q=None):
    """Worker process: build an explainer replica, serve explain/attr requests.

    CUDA replicas are pinned round-robin to the node's GPUs via
    HIP_VISIBLE_DEVICES — set before torch is imported in this process
    (spawn context; torch is only imported lazily by the replica ctor)."""
    try:
        if init_kwargs.get("device") == "cuda":
            import os

            n_gpus = int(os.environ.get("KSHAP_POOL_NGPUS", "0"))
            if not n_gpus:
                try:
                    import subprocess

                    out = subprocess.run(
                        ["rocm-smi", "--showid", "--csv"],
                        capture_output=True, text=True, timeout=10,
                    ).stdout
                    n_gpus = max(1, out.count("card"))
                except Exception:
                    n_gpus = 1
            os.environ["HIP_VISIBLE_DEVICES"] = str(worker_id % n_gpus)
        replica = explainer_type(*init_args, **init_kwargs)
    except Exception as e:  # construction failure must not hang the pool
        result_q.put(("fatal", worker_id, repr(e)))
        return
    result_q.put(("ready", worker_id, None))
    while True:
        msg = task_q.get()
        if msg is None:
            return
        kind = msg[0]
        try:
            if kind == "explain":
                _, batch_idx, batch, offset, kwargs = msg
                out = replica.get_explanation(
                    (batch_idx, batch), instance_offset=offset, **kwargs
                )
                result_q.put(("result", out[0], out[1]))
            elif kind == "attr":
                # attribute replies go to a dedicated queue so a fetch while
                # explain results are in flight can never consume (and lose)
                # a 'result' message
                _, name = msg
                (attr_q or result_q).put(
                    ("attr", name, replica.return_attribute(name))
                )
        except Exception as e:
            q = (attr_q or result_q) if kind == "attr" else result_q
            q.put(("error", msg[1] if len(msg) > 1 else None, repr(e)))


class DistributedExplainer:
    """Generic worker-pool orchestration
    (reference ``explainers/distributed.py:85-179``).

    Parameters mirror the reference: ``distributed_opts`` (``n_workers``,
    ``batch_size``), the explainer type to replicate, and its init
    args/kwargs (the broadcast payload).
    """

    def __init__(
        self,
        distributed_opts: Dict[str, Any],
        explainer_type: Callable,
        explainer_init_args: Sequence,
        explainer_init_kwargs: Dict[str, Any],
        concatenate_results: bool = True,
    ):
        self.n_workers = int(distributed_opts.get("n_workers") or 1)
        self.batch_size = distributed_opts.get("batch_size", 1)
        self.algorithm = distributed_opts.get("algorithm", "kernel_shap")
        self.concatenate_results = concatenate_results
        self._explainer_type = explainer_type
        self._init_args = tuple(explainer_init_args)
        self._init_kwargs = dict(explainer_init_kwargs)
        self._attr_cache: Dict[str, Any] = {}
        # fork is robust for CPU replicas (no __main__ re-import); CUDA
        # replicas need spawn (HIP contexts don't survive fork).
        method = distributed_opts.get(
            "mp_context",
            "spawn" if explainer_init_kwargs.get("device") == "cuda" else "fork",
        )
        self._ctx = mp.get_context(method)
        self._task_q = None
        self._procs: List[Any] = []
        self._result_q = None
        self._attr_q = None
        self.create_parallel_pool()

    # ------------------------------------------------------------------ #

    def create_parallel_pool(self) -> None:
        """Spawn N replica processes (reference :120-128; each ctor ships the
        pickled predictor + background — the 'broadcast'). All workers pull
        from ONE shared task queue: dynamic greedy dispatch, the same load
        balancing as the reference's ``ActorPool.map_unordered``
        (``explainers/distributed.py:150-152``) — a slow batch delays only
        itself, not a statically assigned queue behind it."""
        self._result_q = self._ctx.Queue()
        self._attr_q = self._ctx.Queue()
        self._task_q = self._ctx.Queue()
        for wid in range(self.n_workers):
            p = self._ctx.Process(
                target=_worker_main,
                args=(
                    wid,
                    self._explainer_type,
                    self._init_args,
                    self._init_kwargs,
                    self._task_q,
                    self._result_q,
                    self._attr_q,
                ),
                daemon=True,
            )
            p.start()
            self._procs.append(p)
        ready = 0
        while ready < self.n_workers:
            kind, wid, payload = self._result_q.get()
            if kind == "fatal":
                self.shutdown()
                raise RuntimeError(f"worker {wid} failed to construct: {payload}")
            if kind == "ready":
                ready += 1

    def __getattr__(self, item):
        """Proxy shared replica state (expected_value, vector_out) from worker 0
        (reference ``explainers/distributed.py:113-118`` did this via a racy
        idle-actor peek; here it is an explicit request/reply)."""
        if item.startswith("_") or item in self.__dict__:
            raise AttributeError(item)
        cache = self.__dict__.get("_attr_cache", {})
        if item in cache:
            return cache[item]
        self._task_q.put(("attr", item))
        while True:
            msg = self._attr_q.get()
            if msg[0] == "attr" and msg[1] == item:
                cache[item] = msg[2]
                return msg[2]
            if msg[0] == "error":
                raise AttributeError(f"{item}: {msg[2]}")

    # ------------------------------------------------------------------ #

    def get_explanation(self, X: np.ndarray, **kwargs):
        """Shard X into minibatches, scatter round-robin, gather unordered,
        re-order and concatenate (reference :130-179)."""
        batches = batch_split(X, batch_size=self.batch_size)
        offsets = np.concatenate([[0], np.cumsum([b.shape[0] for b in batches])])[:-1]
        base = int(kwargs.pop("instance_offset", 0))
        n_batches = len(batches)
        for i, b in enumerate(batches):
            self._task_q.put(("explain", i, b, base + int(offsets[i]), kwargs))
        import queue as _queue

        unordered: List[Tuple[int, Any]] = []
        while len(unordered) < n_batches:
            try:
                msg = self._result_q.get(timeout=5)
            except _queue.Empty:
                # failure detection (the reference hung
                # map_unordered forever on a dead actor) — fail fast if any
                # worker process died
                dead = [i for i, p in enumerate(self._procs) if not p.is_alive()]
                if dead:
                    self.shutdown()
                    raise RuntimeError(
                        f"worker process(es) {dead} died; "
                        f"{len(unordered)}/{n_batches} batches completed"
                    )
                continue
            if msg[0] == "result":
                unordered.append((msg[1], msg[2]))
            elif msg[0] == "error":
                self.shutdown()
                raise RuntimeError(f"worker failed on batch {msg[1]}: {msg[2]}")
        return self.order_result(unordered)

    def order_result(self, unordered: List[Tuple[int, Any]]):
        """Invert the completion permutation, then post-process
        (reference :156-179)."""
        indices = [u[0] for u in unordered]
        results = [u[1] for u in unordered]
        perm = invert_permutation(indices)
        ordered = [results[perm[i]] for i in range(len(results))]
        if self.concatenate_results:
            return kernel_shap_postprocess_fn(ordered)
        return ordered

    # ------------------------------------------------------------------ #

    def shutdown(self) -> None:
        if self._task_q is not None:
            for _ in self._procs:
                try:
                    self._task_q.put(None)
                except Exception:
                    pass
        for p in self._procs:
            p.join(timeout=5)
            if p.is_alive():
                p.terminate()
        self._task_q = None
        self._procs = []

    def __del__(self):  # best-effort cleanup
        try:
            self.shutdown()
        except Exception:
            pass
