"""Request metrics registry with Prometheus text exposition.

Analogue of the reference server's observability surface (SURVEY §5:
prometheus_exporter.h:29-39, request_logger.h:33, servables/tensorflow/
util.cc:36-66): request counters, latency quantiles, bytes packed, and a
``render_prometheus()`` text page. Used by both the loopback server and the
benchmark harness.
"""
from __future__ import annotations

import threading
from bisect import insort
from typing import Dict, List


class _Latency:
    __slots__ = ("samples", "total", "count")

    def __init__(self):
        self.samples: List[float] = []  # sorted, capped reservoir
        self.total = 0.0
        self.count = 0

    def observe(self, seconds: float, cap: int = 65536):
        self.total += seconds
        self.count += 1
        if len(self.samples) < cap:
            insort(self.samples, seconds)

    def quantile(self, q: float) -> float:
        if not self.samples:
            return 0.0
        idx = min(len(self.samples) - 1, int(q * len(self.samples)))
        return self.samples[idx]


class MetricsRegistry:
    def __init__(self):
        self._lock = threading.Lock()
        self._counters: Dict[str, float] = {}
        self._latency: Dict[str, _Latency] = {}

    # -- write ----------------------------------------------------------
    def inc(self, name: str, value: float = 1.0):
        with self._lock:
            self._counters[name] = self._counters.get(name, 0.0) + value

    def observe_request(self, method: str, seconds: float):
        with self._lock:
            self._counters[f"request_count{{method={method!r}}}"] = (
                self._counters.get(f"request_count{{method={method!r}}}", 0.0)
                + 1)
            lat = self._latency.setdefault(method, _Latency())
            lat.observe(seconds)

    def observe_bytes(self, direction: str, nbytes: int):
        self.inc(f"bytes_total{{direction={direction!r}}}", nbytes)

    # -- read -----------------------------------------------------------
    def latency_quantiles(self, method: str):
        with self._lock:
            lat = self._latency.get(method)
            if lat is None or lat.count == 0:
                return {}
            return {
                "p50": lat.quantile(0.50),
                "p90": lat.quantile(0.90),
                "p99": lat.quantile(0.99),
                "mean": lat.total / lat.count,
                "count": lat.count,
            }

    def counters(self):
        with self._lock:
            return dict(self._counters)

    def render_prometheus(self) -> str:
        """Prometheus text exposition format (the reference serves an
        equivalent scrape page behind MonitoringConfig —
        monitoring_config.proto:7-19)."""
        lines = []
        with self._lock:
            for name, value in sorted(self._counters.items()):
                lines.append(f":tensorflow:serving:{name} {value:g}")
            for method, lat in sorted(self._latency.items()):
                if lat.count:
                    for q in (0.5, 0.9, 0.99):
                        lines.append(
                            f':tensorflow:serving:request_latency_seconds'
                            f'{{method="{method}",quantile="{q}"}} '
                            f'{lat.quantile(q):.6f}')
                    lines.append(
                        f':tensorflow:serving:request_latency_seconds_count'
                        f'{{method="{method}"}} {lat.count}')
        return "\n".join(lines) + "\n"
