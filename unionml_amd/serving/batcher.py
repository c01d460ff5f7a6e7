"""Dynamic request batcher for online serving.

Coalesces concurrent ``/predict`` requests into one model forward
(SURVEY.md §2c row 'hipGraph inference step'): requests queue up to
``max_delay_ms``; the worker drains up to ``max_batch_size`` of them,
runs ONE prediction over the concatenated features and scatters results
back to each caller's future.

On an MI355X the per-bucket forward is a hipGraph replay: the model's
predictor advertises graph capture via
:class:`unionml_amd.serving.graph_runner.GraphedPredictor` (batch sizes
are padded up to the nearest bucket {1,2,4,...,max_batch_size} so every
replayed graph is shape-static). On CPU the same batcher runs the plain
predictor — one code path, two execution substrates.
"""

import asyncio
import threading
import time
from collections import deque
from dataclasses import dataclass
from typing import Any, Deque, List, Optional

from unionml_amd._logging import logger


@dataclass
class _Request:
    features_raw: Any
    n_rows: int
    future: "asyncio.Future"
    loop: "asyncio.AbstractEventLoop"
    t_submit: float = 0.0


def bucket_for(n: int, max_batch: int) -> int:
    """Smallest power-of-two bucket >= n (capped at max_batch)."""
    b = 1
    while b < n and b < max_batch:
        b <<= 1
    return min(b, max_batch)


class DynamicBatcher:
    """Thread-backed micro-batcher feeding a single predictor."""

    def __init__(self, model, max_batch_size: int = 64, max_delay_ms: float = 0.0):
        self.model = model
        self.max_batch_size = max_batch_size
        self.max_delay_s = max_delay_ms / 1000.0
        self._queue: Deque[_Request] = deque()
        self._cv = threading.Condition()
        self._stop = False
        self._thread: Optional[threading.Thread] = None
        self._graphed = None  # set lazily on first GPU batch
        self._n_batches = 0
        self._n_requests = 0
        # bounded reservoirs: a long-running server must not leak one
        # entry per request (stats() quantiles come from the most recent
        # window, which is also the operationally interesting one)
        self._batch_rows: Deque[int] = deque(maxlen=4096)
        self._done_latency_ms: Deque[float] = deque(maxlen=16384)

    def start(self):
        # build the hipGraph runner (and capture all buckets) up front so
        # no request pays the capture cost
        predictor = self.model._predictor
        factory = getattr(predictor, "__unionml_graphed__", None)
        if factory is not None and self.model.artifact is not None:
            try:
                self._graphed = factory(self.model.artifact.model_object, self.max_batch_size)
            except Exception:
                logger.exception("graph runner init failed; serving eagerly")
        self._thread = threading.Thread(target=self._worker, daemon=True, name="unionml-batcher")
        self._thread.start()

    def stop(self):
        with self._cv:
            self._stop = True
            self._cv.notify_all()
        if self._thread is not None:
            self._thread.join(timeout=5)

    async def submit(self, features_raw: Any):
        loop = asyncio.get_running_loop()
        future = loop.create_future()
        n_rows = len(features_raw) if hasattr(features_raw, "__len__") else 1
        req = _Request(
            features_raw=features_raw,
            n_rows=n_rows,
            future=future,
            loop=loop,
            t_submit=time.monotonic(),
        )
        with self._cv:
            self._queue.append(req)
            self._cv.notify()
        return await future

    def stats(self) -> dict:
        """Server-side batching stats (for bench/diagnostics)."""
        lat = sorted(self._done_latency_ms)
        pick = lambda q: lat[min(len(lat) - 1, int(q * len(lat)))] if lat else None  # noqa: E731
        return {
            "batches": self._n_batches,
            "requests": self._n_requests,
            "rows_per_batch": (sum(self._batch_rows) / max(1, len(self._batch_rows))),
            "server_p50_ms": pick(0.50),
            "server_p99_ms": pick(0.99),
        }

    # ------------------------------------------------------------------

    def _drain(self) -> List[_Request]:
        """Collect requests for one batch, waiting up to max_delay for
        more once the first arrives."""
        with self._cv:
            while not self._queue and not self._stop:
                self._cv.wait(timeout=0.1)
            if self._stop and not self._queue:
                return []
            # max_delay 0 = adaptive batching: take whatever is queued
            # NOW — under load, requests naturally accumulate while the
            # previous batch is in flight, so coalescing emerges without
            # adding latency (and without the GIL-thrashing micro-wait
            # loop that starves the server's event loop)
            if self.max_delay_s > 0:
                deadline = time.monotonic() + self.max_delay_s
                while (
                    sum(r.n_rows for r in self._queue) < self.max_batch_size
                    and time.monotonic() < deadline
                    and not self._stop
                ):
                    remaining = deadline - time.monotonic()
                    if remaining > 0:
                        self._cv.wait(timeout=remaining)
            batch, rows = [], 0
            while self._queue and rows + self._queue[0].n_rows <= self.max_batch_size:
                req = self._queue.popleft()
                rows += req.n_rows
                batch.append(req)
            if not batch and self._queue:  # single oversized request
                batch.append(self._queue.popleft())
            return batch

    def _numeric_fast_path_ok(self) -> bool:
        """True when requests can bypass the pandas feature pipeline:
        the forward is a graphed runner (consumes a float32 matrix) and
        the dataset uses the default feature loader/transformer with
        declared feature columns — the per-request DataFrame build is
        pure overhead on the /predict hot path then."""
        ds = self.model._dataset
        return (
            self._graphed is not None
            and ds._feature_loader == ds._default_feature_loader
            and ds._feature_transformer == ds._default_feature_transformer
            and bool(ds._features)
        )

    def _predict_batch(self, features_list: List[Any]):
        """One forward over the concatenated features of the batch."""
        import numpy as np
        import pandas as pd

        model = self.model
        ds = model._dataset

        if self._numeric_fast_path_ok():
            names = ds._features
            rows: List[Any] = []
            counts = []
            try:
                for f in features_list:
                    if not (isinstance(f, list) and f and isinstance(f[0], dict)):
                        raise TypeError  # non-records request: general path
                    counts.append(len(f))
                    rows.extend([rec[c] for c in names] for rec in f)
                merged = np.asarray(rows, dtype=np.float32)
            except (TypeError, KeyError, ValueError):
                pass
            else:
                predictions = self._forward(merged)
                out, offset = [], 0
                for n in counts:
                    out.append(predictions[offset : offset + n])
                    offset += n
                return out
        loaded = [ds.get_features(f) for f in features_list]
        first = loaded[0]
        if isinstance(first, pd.DataFrame):
            merged = pd.concat(loaded, ignore_index=True)
        elif isinstance(first, np.ndarray):
            merged = np.concatenate(loaded, axis=0)
        elif hasattr(first, "shape") and hasattr(first, "device"):  # torch tensor
            import torch

            merged = torch.cat(loaded, dim=0)
        else:
            merged = [row for f in loaded for row in f]

        predictions = self._forward(merged)
        # scatter back by row counts
        out = []
        offset = 0
        for f in loaded:
            n = len(f)
            out.append(predictions[offset : offset + n])
            offset += n
        return out

    def _forward(self, merged_features):
        """The single fused forward. Uses the hipGraph-captured runner
        when the predictor provides one, else the plain predictor."""
        model = self.model
        predictor = model._predictor
        factory = getattr(predictor, "__unionml_graphed__", None)
        if factory is not None:
            if self._graphed is None:
                self._graphed = factory(model.artifact.model_object, self.max_batch_size)
            return self._graphed(merged_features)
        return model._run_predictor(model.artifact.model_object, merged_features)

    def _worker(self):
        from unionml_amd.fastapi import _jsonable

        while True:
            batch = self._drain()
            if not batch:
                if self._stop:
                    return
                continue
            try:
                results = self._predict_batch([r.features_raw for r in batch])
                now = time.monotonic()
                self._n_batches += 1
                self._batch_rows.append(sum(r.n_rows for r in batch))
                for req, res in zip(batch, results):
                    self._n_requests += 1
                    self._done_latency_ms.append((now - req.t_submit) * 1000.0)
                    req.loop.call_soon_threadsafe(req.future.set_result, _jsonable(res))
            except Exception as exc:
                logger.exception("batched prediction failed")
                for req in batch:
                    if not req.future.done():
                        req.loop.call_soon_threadsafe(req.future.set_exception, exc)
