"""REST API, Prometheus endpoint, and the model_server CLI assembly."""
import json
import urllib.error
import urllib.request

import numpy as np
import pytest

from min_tfs_client_amd.rest import RestApiServer
from min_tfs_client_amd.server import ModelManager, Servable, identity_servable


@pytest.fixture()
def manager():
    mgr = ModelManager()
    mgr.load("default", identity_servable(), version=1)

    def double(inputs):
        return {k: np.asarray(v) * 2 for k, v in inputs.items()}

    mgr.load("double", Servable(double), version=3)
    return mgr


@pytest.fixture()
def rest(manager):
    with RestApiServer(manager, port=0) as srv:
        yield srv


def _get(rest, path):
    return json.loads(urllib.request.urlopen(
        f"http://127.0.0.1:{rest.port}{path}").read())


def _post(rest, path, payload):
    req = urllib.request.Request(
        f"http://127.0.0.1:{rest.port}{path}",
        data=json.dumps(payload).encode(),
        headers={"Content-Type": "application/json"})
    return json.loads(urllib.request.urlopen(req).read())


def test_rest_status(rest):
    body = _get(rest, "/v1/models/default")
    assert body == {"model_version_status": [
        {"version": "1", "state": "AVAILABLE", "status": {}}]}


def test_rest_status_not_found(rest):
    with pytest.raises(urllib.error.HTTPError) as err:
        _get(rest, "/v1/models/missing")
    assert err.value.code == 404


def test_rest_predict_instances_row_format(rest):
    body = _post(rest, "/v1/models/double:predict",
                 {"instances": [[1.0, 2.0], [3.0, 4.0]]})
    assert body == {"predictions": [[2.0, 4.0], [6.0, 8.0]]}


def test_rest_predict_columnar_format(rest):
    body = _post(rest, "/v1/models/double:predict",
                 {"inputs": {"x": [[1.0]], "y": [[10.0]]}})
    assert body["outputs"]["x"] == [[2.0]]
    assert body["outputs"]["y"] == [[20.0]]


def test_rest_predict_named_instances(rest):
    body = _post(rest, "/v1/models/double:predict",
                 {"instances": [{"a": [1.0], "b": [5.0]},
                                {"a": [2.0], "b": [6.0]}]})
    preds = body["predictions"]
    assert preds[0]["a"] == [2.0] and preds[1]["b"] == [12.0]


def test_rest_predict_versioned(rest):
    body = _post(rest, "/v1/models/double/versions/3:predict",
                 {"instances": [[1.0]]})
    assert body == {"predictions": [[2.0]]}
    with pytest.raises(urllib.error.HTTPError) as err:
        _post(rest, "/v1/models/double/versions/9:predict",
              {"instances": [[1.0]]})
    assert err.value.code == 404


def test_rest_predict_bad_payload(rest):
    with pytest.raises(urllib.error.HTTPError) as err:
        _post(rest, "/v1/models/double:predict", {"bogus": 1})
    assert err.value.code == 400


def test_rest_metadata(rest):
    body = _get(rest, "/v1/models/default/metadata")
    assert body["model_spec"]["name"] == "default"
    assert "signature_def" in body["metadata"]


def test_prometheus_page(rest, manager):
    _post(rest, "/v1/models/double:predict", {"instances": [[1.0]]})
    page = urllib.request.urlopen(
        f"http://127.0.0.1:{rest.port}/monitoring/prometheus/metrics"
    ).read().decode()
    assert ":tensorflow:serving:" in page
    assert "rest_predict" in page


# ---------------------------------------------------------------------------
# CLI assembly (make_server wiring, no subprocess)
# ---------------------------------------------------------------------------

def test_cli_make_server_single_model(tmp_path):
    from min_tfs_client_amd.model_server import build_arg_parser, make_server
    vdir = tmp_path / "mymodel" / "1"
    vdir.mkdir(parents=True)
    (vdir / "identity").touch()
    args = build_arg_parser().parse_args([
        "--port", "0", "--rest_api_port", "0",
        "--model_name", "mymodel",
        "--model_base_path", str(tmp_path / "mymodel"),
        "--file_system_poll_wait_seconds", "0",
    ])
    server, source, rest = make_server(args)
    try:
        server.start()
        source.start()
        from min_tfs_client_amd.turbo import TurboPredictClient
        import torch
        with TurboPredictClient(server.address) as c:
            out = c.predict("mymodel", {"x": torch.ones(2)})
            assert torch.equal(out["x"], torch.ones(2))
    finally:
        source.stop()
        server.stop(0)


def test_cli_model_config_file(tmp_path):
    from min_tfs_client_amd.model_server import build_arg_parser, make_server
    for name in ("a", "b"):
        vdir = tmp_path / name / "1"
        vdir.mkdir(parents=True)
        (vdir / "identity").touch()
    cfg = tmp_path / "models.config"
    cfg.write_text(f"""
model_config_list {{
  config {{ name: "a" base_path: "{tmp_path}/a" model_platform: "tensorflow" }}
  config {{
    name: "b" base_path: "{tmp_path}/b"
    model_version_policy {{ all {{ }} }}
  }}
}}
""")
    args = build_arg_parser().parse_args([
        "--port", "0", "--model_config_file", str(cfg),
        "--file_system_poll_wait_seconds", "0",
    ])
    server, source, rest = make_server(args)
    try:
        server.start()
        source.start()
        assert server.manager.version_statuses("a")[0][1] == 30
        assert server.manager.version_statuses("b")[0][1] == 30
    finally:
        source.stop()
        server.stop(0)


def test_cli_batching_flag(tmp_path):
    from min_tfs_client_amd.batching import BatchingServable
    from min_tfs_client_amd.model_server import build_arg_parser, make_server
    vdir = tmp_path / "m" / "1"
    vdir.mkdir(parents=True)
    (vdir / "identity").touch()
    args = build_arg_parser().parse_args([
        "--port", "0", "--model_base_path", str(tmp_path / "m"),
        "--model_name", "m", "--enable_batching",
        "--max_batch_size", "16",
        "--file_system_poll_wait_seconds", "0",
    ])
    server, source, _ = make_server(args)
    try:
        server.start()
        source.start()
        servable = server.manager.get("m")
        assert isinstance(servable, BatchingServable)
    finally:
        source.stop()
        server.stop(0)


def test_rest_tracing_endpoints(rest):
    _post(rest, "/v1/tracing:start", {})
    _post(rest, "/v1/models/double:predict", {"instances": [[1.0]]})
    _post(rest, "/v1/tracing:stop", {})
    body = _get(rest, "/v1/tracing/export")
    assert "traceEvents" in body


def test_rest_classify_and_regress(manager):
    from min_tfs_client_amd.examples_adapter import (
        ClassificationAdapter, RegressionAdapter)
    from min_tfs_client_amd.server import Servable

    def fn(features):
        s = features["x"].sum(axis=1)
        return {"scores": np.stack([s, -s], 1), "value": s}

    manager.load("clf", ClassificationAdapter(Servable(fn),
                                              labels=["a", "b"]), version=1)
    manager.load("reg", RegressionAdapter(Servable(fn)), version=1)
    with RestApiServer(manager, port=0) as rest_srv:
        body = _post(rest_srv, "/v1/models/clf:classify",
                     {"examples": [{"x": [1.0, 2.0]}]})
        assert body == {"results": [[["a", 3.0], ["b", -3.0]]]}
        body = _post(rest_srv, "/v1/models/reg:regress",
                     {"examples": [{"x": [2.0, 2.0]}, {"x": [1.0, 0.0]}]})
        assert body["results"][0] == 4.0
        with pytest.raises(urllib.error.HTTPError) as err:
            _post(rest_srv, "/v1/models/default:classify",
                  {"examples": [{"x": [1.0]}]})
        assert err.value.code == 400


def test_cli_monitoring_config(tmp_path):
    """--monitoring_config_file moves the Prometheus path
    (monitoring_config.proto:7-19 semantics)."""
    from min_tfs_client_amd.model_server import build_arg_parser, make_server
    vdir = tmp_path / "m" / "1"
    vdir.mkdir(parents=True)
    (vdir / "identity").touch()
    mon = tmp_path / "monitoring.config"
    mon.write_text(
        'prometheus_config { enable: true path: "/custom/metrics" }')
    import socket
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    rest_port = s.getsockname()[1]
    s.close()
    args = build_arg_parser().parse_args([
        "--port", "0", "--rest_api_port", str(rest_port),
        "--model_name", "m", "--model_base_path", str(tmp_path / "m"),
        "--monitoring_config_file", str(mon),
        "--file_system_poll_wait_seconds", "0",
    ])
    server, source, rest_srv = make_server(args)
    try:
        server.start()
        source.start()
        rest_srv.start()
        body = urllib.request.urlopen(
            f"http://127.0.0.1:{rest_srv.port}/custom/metrics").read()
        assert b":tensorflow:serving:" in body or body == b"\n"
        with pytest.raises(urllib.error.HTTPError):
            urllib.request.urlopen(
                f"http://127.0.0.1:{rest_srv.port}"
                f"/monitoring/prometheus/metrics")
    finally:
        source.stop()
        rest_srv.stop()
        server.stop(0)


def test_rest_label_route(manager):
    manager.set_version_label("double", "stable", 3)
    with RestApiServer(manager, port=0) as rest_srv:
        body = _post(rest_srv, "/v1/models/double/labels/stable:predict",
                     {"instances": [[2.0]]})
        assert body == {"predictions": [[4.0]]}
        with pytest.raises(urllib.error.HTTPError) as err:
            _post(rest_srv, "/v1/models/double/labels/nope:predict",
                  {"instances": [[2.0]]})
        assert err.value.code == 404
