"""Metrics registry: counters, quantiles, Prometheus text format."""
from min_tfs_client_amd.utils.metrics import MetricsRegistry


def test_counters_and_latency():
    m = MetricsRegistry()
    for i in range(100):
        m.observe_request("predict", (i + 1) / 1000.0)
    q = m.latency_quantiles("predict")
    assert q["count"] == 100
    assert 0.045 <= q["p50"] <= 0.055
    assert 0.095 <= q["p99"] <= 0.101
    assert abs(q["mean"] - 0.0505) < 1e-6


def test_bytes_counters():
    m = MetricsRegistry()
    m.observe_bytes("tx", 1000)
    m.observe_bytes("tx", 500)
    assert m.counters()["bytes_total{direction='tx'}"] == 1500


def test_prometheus_rendering():
    m = MetricsRegistry()
    m.observe_request("predict", 0.01)
    page = m.render_prometheus()
    assert ":tensorflow:serving:request_count" in page
    assert 'request_latency_seconds{method="predict",quantile="0.5"}' \
        in page
    assert page.endswith("\n")


def test_empty_quantiles():
    m = MetricsRegistry()
    assert m.latency_quantiles("nothing") == {}
