"""HTTP/REST API + Prometheus scrape endpoint.

Analogue of the reference's evhttp REST server (SURVEY §2.4:
http_rest_api_handler.h:63-81, util/json_tensor.h:145-198) with the same
route shapes:

  POST /v1/models/{model}[/versions/{v}]:predict
  GET  /v1/models/{model}[/versions/{v}]            -> GetModelStatus JSON
  GET  /v1/models/{model}/metadata                  -> signature metadata
  GET  /monitoring/prometheus/metrics               -> Prometheus text page
       (path configurable via MonitoringConfig.prometheus_config.path,
        monitoring_config.proto:7-19)

JSON tensor codec (json_tensor.h semantics subset):
  request  {"instances": [row, ...]}  (row format)  or
           {"inputs": {name: nested-list, ...}}     (columnar format)
  response {"predictions": [...]} for row requests, {"outputs": {...}}
           for columnar — matching TF-Serving's format mirroring.
"""
from __future__ import annotations

import json
import re
import threading
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer
from typing import Optional

import numpy as np

from .server import ModelManager
from .utils.metrics import MetricsRegistry

_MODEL_RE = re.compile(
    r"^/v1/models/(?P<model>[^/:]+)"
    r"(?:/versions/(?P<version>\d+)|/labels/(?P<label>[^/:]+))?"
    r"(?P<rest>:predict|:classify|:regress|/metadata)?$")


def _json_to_inputs(payload: dict):
    """Returns (inputs dict, row_format flag)."""
    if "instances" in payload:
        instances = payload["instances"]
        if not isinstance(instances, list) or not instances:
            raise ValueError("instances must be a non-empty list")
        if isinstance(instances[0], dict):
            keys = instances[0].keys()
            inputs = {}
            for k in keys:
                inputs[k] = np.asarray([inst[k] for inst in instances])
            return inputs, True
        return {"inputs": np.asarray(instances)}, True
    if "inputs" in payload:
        inp = payload["inputs"]
        if isinstance(inp, dict):
            return {k: np.asarray(v) for k, v in inp.items()}, False
        return {"inputs": np.asarray(inp)}, False
    raise ValueError('request must carry "instances" or "inputs"')


def _to_jsonable(v):
    arr = np.asarray(v)
    if arr.dtype.kind in "OSU":
        return arr.astype(str).tolist()
    if arr.dtype == np.float16:
        arr = arr.astype(np.float32)
    return arr.tolist()


def _outputs_to_json(outputs: dict, row_format: bool) -> dict:
    outs = {}
    for k, v in outputs.items():
        try:
            import torch
            if isinstance(v, torch.Tensor):
                v = v.cpu().float().numpy() if v.dtype.is_floating_point \
                    else v.cpu().numpy()
        except ImportError:  # pragma: no cover
            pass
        outs[k] = v
    if row_format:
        if len(outs) == 1:
            return {"predictions": _to_jsonable(next(iter(outs.values())))}
        names = sorted(outs)
        n = len(_to_jsonable(outs[names[0]]))
        preds = []
        for i in range(n):
            preds.append({k: _to_jsonable(outs[k])[i] for k in names})
        return {"predictions": preds}
    return {"outputs": {k: _to_jsonable(v) for k, v in outs.items()}}


_STATE_NAMES = {0: "UNKNOWN", 10: "START", 20: "LOADING", 30: "AVAILABLE",
                40: "UNLOADING", 50: "END"}


class RestApiServer:
    """Threaded HTTP server over a ModelManager (shared with the gRPC
    server, like reference server.cc:368-377)."""

    def __init__(self, manager: ModelManager, port: int = 0,
                 metrics: Optional[MetricsRegistry] = None,
                 prometheus_path: str = "/monitoring/prometheus/metrics"):
        self.manager = manager
        self.metrics = metrics or MetricsRegistry()
        self.prometheus_path = prometheus_path
        outer = self

        class Handler(BaseHTTPRequestHandler):
            def log_message(self, *args):  # quiet
                pass

            def _send(self, code: int, body: dict | str,
                      content_type="application/json"):
                data = (json.dumps(body) if isinstance(body, dict)
                        else body).encode()
                self.send_response(code)
                self.send_header("Content-Type", content_type)
                self.send_header("Content-Length", str(len(data)))
                self.end_headers()
                self.wfile.write(data)

            def _error(self, code, msg):
                self._send(code, {"error": msg})

            # ---- GET: status / metadata / prometheus / tracing ------
            def do_GET(self):
                if self.path == outer.prometheus_path:
                    self._send(200, outer.metrics.render_prometheus(),
                               content_type="text/plain")
                    return
                if self.path == "/v1/tracing/export":
                    # remote trace capture (ProfilerService analogue,
                    # reference server.cc:324,339): returns the chrome-trace
                    # JSON accumulated since tracing:start
                    from .utils.tracing import Tracer
                    import tempfile
                    t = Tracer.get()
                    with tempfile.NamedTemporaryFile("r", suffix=".json",
                                                     delete=False) as f:
                        path = f.name
                    n = t.export(path)
                    body = open(path).read()
                    import os as _os
                    _os.unlink(path)
                    self._send(200, body, content_type="application/json")
                    return
                m = _MODEL_RE.match(self.path)
                if not m:
                    self._error(404, f"Malformed request: GET {self.path}")
                    return
                name = m.group("model")
                version = m.group("version")
                if m.group("rest") == "/metadata":
                    self._metadata(name, version)
                    return
                try:
                    statuses = outer.manager.version_statuses(name)
                except KeyError as e:
                    self._error(404, str(e))
                    return
                want = int(version) if version else None
                entries = []
                for ver, state, err in statuses:
                    if want is not None and ver != want:
                        continue
                    entry = {"version": str(ver),
                             "state": _STATE_NAMES.get(state, "UNKNOWN"),
                             "status": {}}
                    if err:
                        entry["status"] = {"error_code": err[0],
                                           "error_message": err[1]}
                    entries.append(entry)
                if not entries:
                    self._error(404, f"Could not find version {want} of "
                                     f"model {name}")
                    return
                self._send(200, {"model_version_status": entries})

            def _metadata(self, name, version):
                try:
                    servable = outer.manager.get(
                        name, int(version) if version else None)
                except KeyError as e:
                    self._error(404, str(e))
                    return
                sig = {"method_name": servable.signature.get(
                    "method_name", "tensorflow/serving/predict")}
                for io_key in ("inputs", "outputs"):
                    sig[io_key] = {
                        alias: {"dtype": dtype, "shape": shape}
                        for alias, (dtype, shape)
                        in servable.signature.get(io_key, {}).items()}
                self._send(200, {
                    "model_spec": {"name": name,
                                   "version": version or "latest"},
                    "metadata": {"signature_def": {
                        "signature_def": {
                            servable.signature_name: sig}}}})

            # ---- POST: predict / tracing control ---------------------
            def do_POST(self):
                import time as _t
                t0 = _t.perf_counter()
                if self.path in ("/v1/tracing:start", "/v1/tracing:stop"):
                    from .utils.tracing import Tracer
                    t = Tracer.get()
                    if self.path.endswith(":start"):
                        t.clear()
                        t.start()
                    else:
                        t.stop()
                    self._send(200, {"tracing": t.enabled})
                    return
                m = _MODEL_RE.match(self.path)
                if not m or m.group("rest") not in (":predict", ":classify",
                                                    ":regress"):
                    self._error(404, f"Malformed request: POST {self.path}")
                    return
                name = m.group("model")
                version = m.group("version")
                label = m.group("label")
                method = m.group("rest")
                length = int(self.headers.get("Content-Length", 0))
                try:
                    payload = json.loads(self.rfile.read(length) or b"{}")
                except json.JSONDecodeError as e:
                    self._error(400, str(e))
                    return
                try:
                    servable = outer.manager.get(
                        name, int(version) if version else None, label)
                except KeyError as e:
                    self._error(404, str(e))
                    return
                if method in (":classify", ":regress"):
                    self._classify_regress(servable, method, payload)
                    outer.metrics.observe_request(
                        "rest" + method.replace(":", "_"),
                        _t.perf_counter() - t0)
                    return
                try:
                    inputs, row_format = _json_to_inputs(payload)
                except ValueError as e:
                    self._error(400, str(e))
                    return
                try:
                    outputs = servable(inputs)
                except Exception as e:  # noqa: BLE001
                    self._error(500, str(e))
                    return
                self._send(200, _outputs_to_json(outputs, row_format))
                outer.metrics.observe_request("rest_predict",
                                              _t.perf_counter() - t0)

            def _classify_regress(self, servable, method, payload):
                """TF REST classify/regress format
                (http_rest_api_handler.h:63-81): request {"examples":
                [{feat: value}, ...]}; classify responds {"results":
                [[[label, score], ...], ...]}, regress {"results":
                [v, ...]}."""
                from .examples_adapter import examples_input
                examples = payload.get("examples")
                if not isinstance(examples, list) or not examples:
                    self._error(400, 'request must carry "examples"')
                    return
                try:
                    input_proto = examples_input(examples)
                except (TypeError, ValueError) as e:
                    self._error(400, str(e))
                    return
                fn_name = "classify" if method == ":classify" else "regress"
                fn = getattr(servable, fn_name, None)
                if fn is None:
                    self._error(400, f"Expected a {fn_name} signature "
                                     f"for this model")
                    return
                try:
                    result = fn(input_proto)
                except Exception as e:  # noqa: BLE001
                    self._error(500, str(e))
                    return
                if method == ":classify":
                    results = [[[c.label, c.score] for c in cls.classes]
                               for cls in result.classifications]
                else:
                    results = [r.value for r in result.regressions]
                self._send(200, {"results": results})

        self._httpd = ThreadingHTTPServer(("127.0.0.1", port), Handler)
        self.port = self._httpd.server_port
        self._thread = threading.Thread(target=self._httpd.serve_forever,
                                        daemon=True)

    def start(self):
        self._thread.start()
        return self

    def stop(self):
        self._httpd.shutdown()
        self._httpd.server_close()

    def __enter__(self):
        return self.start()

    def __exit__(self, *exc):
        self.stop()
