"""Prometheus metrics registry — name/label parity with the reference.

Reference series (full table /root/reference/docs/prometheus.md:29-47;
definitions core.py:24-61 and engine.py:14-54). All series are labeled
``(component_type, component_id)``. Registration helpers are idempotent so
re-created Services in one process reuse the existing collector
(reference core.py:45-52).

Batched-engine semantics (SURVEY.md §7 hard part 3): when the engine
processes a batch of N lines in one call, ``processing_duration_seconds``
observes the per-line amortized latency N times' worth via
``observe_batch`` (total_seconds recorded as N observations of
total/N) so ``histogram_quantile`` keeps its per-line meaning.
"""
from __future__ import annotations

from typing import Dict, Tuple

from prometheus_client import Counter, Enum, Gauge, Histogram, REGISTRY

LABELS = ("component_type", "component_id")

#: Buckets match the reference exactly (core.py:41).
DURATION_BUCKETS = (0.001, 0.005, 0.01, 0.025, 0.05, 0.1, 0.25, 0.5, 1.0, 2.5, 5.0, 10.0)

_registered: Dict[Tuple[str, type], object] = {}


def _get(metric_cls, name: str, doc: str, **kwargs):
    key = (name, metric_cls)
    if key in _registered:
        return _registered[key]
    # Another instance may have registered it under a previous import path.
    existing = getattr(REGISTRY, "_names_to_collectors", {}).get(name)
    if existing is not None:
        _registered[key] = existing
        return existing
    metric = metric_cls(name, doc, LABELS, **kwargs)
    _registered[key] = metric
    return metric


def get_counter(name: str, doc: str) -> Counter:
    return _get(Counter, name, doc)


def get_histogram(name: str, doc: str, buckets=DURATION_BUCKETS) -> Histogram:
    return _get(Histogram, name, doc, buckets=buckets)


def get_enum(name: str, doc: str, states) -> Enum:
    key = (name, Enum)
    if key in _registered:
        return _registered[key]
    existing = getattr(REGISTRY, "_names_to_collectors", {}).get(name)
    if existing is not None:
        _registered[key] = existing
        return existing
    metric = Enum(name, doc, LABELS, states=states)
    _registered[key] = metric
    return metric


def get_gauge(name: str, doc: str) -> Gauge:
    return _get(Gauge, name, doc)


class ServiceMetrics:
    """Per-service handle over the shared collectors, pre-bound to labels."""

    def __init__(self, component_type: str, component_id: str) -> None:
        labels = dict(component_type=component_type, component_id=component_id)

        self.engine_running = get_enum(
            "engine_running", "Engine state", states=["running", "stopped"]
        ).labels(**labels)
        self.engine_starts_total = get_counter(
            "engine_starts_total", "Number of engine starts"
        ).labels(**labels)
        self.engine_dist_degraded = get_counter(
            "engine_dist_degraded_total",
            "Collective-path failures degraded to the socket loop",
        ).labels(**labels)
        self.processing_duration_seconds = get_histogram(
            "processing_duration_seconds", "Per-line processing latency (s)"
        ).labels(**labels)
        self.data_processed_bytes_total = get_counter(
            "data_processed_bytes_total", "Bytes handed to the component"
        ).labels(**labels)
        self.data_processed_lines_total = get_counter(
            "data_processed_lines_total", "Lines handed to the component"
        ).labels(**labels)
        self.data_read_bytes_total = get_counter(
            "data_read_bytes_total", "Bytes received on the engine socket"
        ).labels(**labels)
        self.data_read_lines_total = get_counter(
            "data_read_lines_total", "Frames received on the engine socket"
        ).labels(**labels)
        self.data_written_bytes_total = get_counter(
            "data_written_bytes_total", "Bytes sent to outputs"
        ).labels(**labels)
        self.data_written_lines_total = get_counter(
            "data_written_lines_total", "Frames sent to outputs"
        ).labels(**labels)
        self.data_dropped_bytes_total = get_counter(
            "data_dropped_bytes_total", "Bytes dropped on failed sends"
        ).labels(**labels)
        self.data_dropped_lines_total = get_counter(
            "data_dropped_lines_total", "Frames dropped on failed sends"
        ).labels(**labels)
        self.processing_errors_total = get_counter(
            "processing_errors_total", "Component process() exceptions"
        ).labels(**labels)
        # MI355X-native additions
        self.engine_batch_size = get_histogram(
            "engine_batch_frames",
            "Frames per engine batch",
            buckets=(1, 2, 4, 8, 16, 32, 64, 128, 256, 512, 1024, 4096),
        ).labels(**labels)

    def observe_batch(self, total_seconds: float, n_lines: int) -> None:
        """Record per-line amortized latency for a batch of ``n_lines``."""
        if n_lines <= 0:
            return
        per_line = total_seconds / n_lines
        # One observation per line keeps rate() and quantiles per-line;
        # cap the loop for very large batches by bulk-observing.
        if n_lines <= 64:
            for _ in range(n_lines):
                self.processing_duration_seconds.observe(per_line)
        else:
            # Approximate: observe 64 samples carrying the same quantile
            # information, then fix up _count/_sum via direct observe calls.
            for _ in range(64):
                self.processing_duration_seconds.observe(per_line)
            # remaining lines: account sum/count cheaply
            remaining = n_lines - 64
            h = self.processing_duration_seconds
            # prometheus_client Histogram child: _sum and _buckets
            try:
                h._sum.inc(per_line * remaining)
                for i, bound in enumerate(h._upper_bounds):
                    if per_line <= bound:
                        h._buckets[i].inc(remaining)
                        break
            except AttributeError:  # fall back to the slow path
                for _ in range(remaining):
                    h.observe(per_line)
