"""Prometheus serving metrics (net-new vs the reference, whose only
observability is /health + prediction callbacks — SURVEY.md §5).

Exposed on ``GET /metrics`` of the serving app: request and row
counters, error counter, and a request-latency histogram bucketed for
sub-millisecond GPU serving.
"""

import uuid

from prometheus_client import (
    CollectorRegistry,
    Counter,
    Histogram,
    generate_latest,
    CONTENT_TYPE_LATEST,
)

LATENCY_BUCKETS = (
    0.0001, 0.00025, 0.0005, 0.001, 0.0025, 0.005, 0.01, 0.025, 0.05,
    0.1, 0.25, 0.5, 1.0, 2.5,
)


class ServingMetrics:
    """Per-app registry (own CollectorRegistry so multiple serving apps
    in one process never collide)."""

    def __init__(self, model_name: str):
        self.registry = CollectorRegistry()
        labels = {"model": model_name, "instance_id": uuid.uuid4().hex[:8]}
        self._requests = Counter(
            "unionml_predict_requests_total",
            "Completed /predict requests",
            list(labels),
            registry=self.registry,
        ).labels(**labels)
        self._rows = Counter(
            "unionml_predict_rows_total",
            "Feature rows predicted",
            list(labels),
            registry=self.registry,
        ).labels(**labels)
        self._errors = Counter(
            "unionml_predict_errors_total",
            "Failed /predict requests",
            list(labels),
            registry=self.registry,
        ).labels(**labels)
        self._latency = Histogram(
            "unionml_predict_latency_seconds",
            "End-to-end /predict handler latency",
            list(labels),
            registry=self.registry,
            buckets=LATENCY_BUCKETS,
        ).labels(**labels)

    def observe(self, n_rows: int, seconds: float) -> None:
        self._requests.inc()
        self._rows.inc(n_rows)
        self._latency.observe(seconds)

    def error(self) -> None:
        self._errors.inc()

    def render(self):
        return generate_latest(self.registry), CONTENT_TYPE_LATEST
