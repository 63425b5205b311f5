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
        # read-time merge sources for requests accounted outside python
        # (the C++ echo fast path): callables returning
        # {method: {"count", "total_s", "samples_s", "bytes_rx",
        #           "bytes_tx"}}
        self._sources = []

    def attach_source(self, fn) -> None:
        with self._lock:
            self._sources.append(fn)

    def _merged_external(self):
        merged: Dict[str, dict] = {}
        for fn in list(self._sources):
            try:
                for method, d in fn().items():
                    m = merged.setdefault(method, {
                        "count": 0, "total_s": 0.0, "samples_s": [],
                        "bytes_rx": 0, "bytes_tx": 0})
                    m["count"] += int(d.get("count", 0))
                    m["total_s"] += float(d.get("total_s", 0.0))
                    m["samples_s"].extend(d.get("samples_s", []))
                    m["bytes_rx"] += int(d.get("bytes_rx", 0))
                    m["bytes_tx"] += int(d.get("bytes_tx", 0))
            except Exception:  # noqa: BLE001 - observers must not break
                pass
        return merged

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
        ext = self._merged_external().get(method)
        with self._lock:
            lat = self._latency.get(method)
            count = (lat.count if lat else 0) + (ext["count"] if ext else 0)
            if count == 0:
                return {}
            samples = list(lat.samples) if lat else []
            total = lat.total if lat else 0.0
            if ext:
                samples.extend(ext["samples_s"])
                samples.sort()
                total += ext["total_s"]
            if not samples:
                return {"count": count, "mean": total / count}

            def q(p):
                return samples[min(len(samples) - 1, int(p * len(samples)))]

            return {
                "p50": q(0.50),
                "p90": q(0.90),
                "p99": q(0.99),
                "mean": total / count,
                "count": count,
            }

    def counters(self):
        out = {}
        with self._lock:
            out.update(self._counters)
        for method, d in self._merged_external().items():
            key = f"request_count{{method={method!r}}}"
            out[key] = out.get(key, 0.0) + d["count"]
            for direction, nbytes in (("rx", d["bytes_rx"]),
                                      ("tx", d["bytes_tx"])):
                bkey = f"bytes_total{{direction={direction!r}}}"
                out[bkey] = out.get(bkey, 0.0) + nbytes
        return out

    def render_prometheus(self) -> str:
        """Prometheus text exposition format (the reference serves an
        equivalent scrape page behind MonitoringConfig —
        monitoring_config.proto:7-19)."""
        lines = []
        ext = self._merged_external()
        for name, value in sorted(self.counters().items()):
            lines.append(f":tensorflow:serving:{name} {value:g}")
        with self._lock:
            methods = set(self._latency) | set(ext)
            for method in sorted(methods):
                lat = self._latency.get(method)
                samples = sorted(
                    (list(lat.samples) if lat else []) +
                    (ext.get(method, {}).get("samples_s", [])))
                count = (lat.count if lat else 0) + \
                    ext.get(method, {}).get("count", 0)
                if count and samples:
                    for q in (0.5, 0.9, 0.99):
                        v = samples[min(len(samples) - 1,
                                        int(q * len(samples)))]
                        lines.append(
                            f':tensorflow:serving:request_latency_seconds'
                            f'{{method="{method}",quantile="{q}"}} '
                            f'{v:.6f}')
                    lines.append(
                        f':tensorflow:serving:request_latency_seconds_count'
                        f'{{method="{method}"}} {count}')
        return "\n".join(lines) + "\n"
