"""Native-transport serving backend: the C++ gRPC (HTTP/2) server.

Round 1 established that the python-grpcio stack caps the 19 MB Predict
loopback at ~9.5 GB/s of machine copy bandwidth (profiles/README.md,
"Throughput ceiling attribution"). This module mounts the same service
surface — all 5 PredictionService rpcs + both ModelService rpcs
(reference prediction_service.proto:15-31, model_service.proto:12-24) —
on the from-scratch C++ HTTP/2 transport (ops/csrc/grpc_transport.cpp):

* the Predict data plane never touches python-protobuf: request bytes are
  parsed by the C++ codec, responses are serialized by it, and the
  identity-echo case runs entirely in C++ without the GIL;
* send/receive are ~2 copies per hop (writev straight from the wire
  buffer; DATA read directly into the message buffer) — the MI355X-native
  realization of TF's two-slice zero-copy encode
  (grpc_tensor_coding.cc:140-248);
* the wire protocol is standard gRPC: grpcio clients interoperate
  (tests/integration/test_native_transport.py pins both directions).
"""
from __future__ import annotations

import time
from typing import Optional

import grpc
import numpy as np

try:
    import torch
except ImportError:  # pragma: no cover
    torch = None

from .utils.metrics import MetricsRegistry
from .wire import messages as pb

_PS = "/tensorflow.serving.PredictionService/"
_MS = "/tensorflow.serving.ModelService/"
PREDICT_PATH = _PS + "Predict"


class HandlerAbort(Exception):
    """Raised by handlers to return a gRPC status; the C++ server reads
    .grpc_code / .grpc_details."""

    def __init__(self, code: grpc.StatusCode, details: str):
        super().__init__(details)
        self.grpc_code = code.value[0]
        self.grpc_details = details


class _AbortContext:
    """Minimal grpc.ServicerContext lookalike for the shared service
    implementations (PredictionServiceImpl et al. only use .abort)."""

    def abort(self, code, details):
        raise HandlerAbort(code, details)

    def set_code(self, code):  # pragma: no cover - stub parity
        self._code = code

    def set_details(self, details):  # pragma: no cover - stub parity
        self._details = details


def _proto_handler(method, req_cls):
    """bytes-in/bytes-out adapter over a (request, context) -> response
    service method."""

    def handler(view):
        try:
            req = req_cls.FromString(bytes(view))
        except Exception as e:  # noqa: BLE001
            raise HandlerAbort(grpc.StatusCode.INVALID_ARGUMENT,
                               f"request parsing error: {e}")
        resp = method(req, _AbortContext())
        return resp.SerializeToString()

    return handler


def _serialize_response_py(spec, version, outputs,
                           use_content: bool = True) -> bytes:
    """python-protobuf PredictResponse builder (string tensors — per-
    element string_val — and typed-output mode, which reproduces
    TF-Serving's default AsProtoField representation)."""
    from .tensors import ndarray_to_tensor_proto, tensor_to_tensor_proto

    resp = pb.PredictResponse()
    resp.model_spec.name = spec["name"]
    if version is not None:
        resp.model_spec.version.value = version
    resp.model_spec.signature_name = (
        spec["signature_name"] or "serving_default")
    for k, v in outputs.items():
        if torch is not None and isinstance(v, torch.Tensor):
            proto = tensor_to_tensor_proto(v.cpu(), use_content)
        else:
            proto = ndarray_to_tensor_proto(np.asarray(v), use_content)
        resp.outputs[k].CopyFrom(proto)
    return resp.SerializeToString()


class StreamingReply:
    """Handler return value for the C++ server's overlapped send path:
    the wire skeleton plus (offset, nbytes, ptr, is_device) payload
    regions. The server sends skeleton spans from `buffer`, host regions
    zero-copy from tensor memory, and device regions through the pooled
    pinned-staging pipeline with the socket write as the chunk consumer —
    hipMemcpyAsync of chunk i+1 overlaps the DATA-frame send of chunk i.
    `keepalive` pins the region tensors until the send completes (the
    server holds this object through the send)."""

    __slots__ = ("buffer", "_wire_regions", "keepalive")

    def __init__(self, buffer, regions, keepalive):
        self.buffer = buffer
        self._wire_regions = regions
        self.keepalive = keepalive


def _raw_predict_bytes_handler(manager, device: str,
                               metrics: MetricsRegistry,
                               request_logger=None,
                               use_content: bool = True):
    """Raw-bytes Predict handler on the C++ codec (the python fallback
    behind the all-C++ echo fast path; runs real servables).
    ``use_content=False`` reproduces TF-Serving's default AsProtoField
    typed-field responses (predict_util.cc:222-226)."""
    from .ops import require_native

    def handler(view, pspec=None, pouts=None):
        t0 = time.perf_counter()
        native = require_native()
        data = memoryview(view)
        if pouts is not None:
            # request was prospected while it streamed in: inputs are
            # already device tensors (C++ reader H2D'd the spans)
            spec, inputs, _filter = pspec, pouts, pspec["output_filter"]
        else:
            try:
                spec, inputs, _filter = native.parse_predict_request(
                    data, device, 1)
            except Exception as e:  # noqa: BLE001
                raise HandlerAbort(grpc.StatusCode.INVALID_ARGUMENT,
                                   f"request parsing error: {e}")
        version = spec["version"] if spec["version"] >= 0 else None
        label = spec.get("version_label") or None
        try:
            servable = manager.get(spec["name"], version, label)
        except KeyError as e:
            raise HandlerAbort(grpc.StatusCode.NOT_FOUND, str(e))
        from .server import validate_inputs_against_signature
        err = validate_inputs_against_signature(servable, inputs)
        if err is not None:
            raise HandlerAbort(grpc.StatusCode.INVALID_ARGUMENT, err)
        try:
            outputs = servable(inputs)
        except ValueError as e:
            raise HandlerAbort(grpc.StatusCode.INVALID_ARGUMENT, str(e))
        except Exception as e:  # noqa: BLE001
            raise HandlerAbort(grpc.StatusCode.INTERNAL, str(e))
        from .server import validate_output_filter
        err = validate_output_filter(servable, outputs, _filter)
        if err is not None:
            raise HandlerAbort(grpc.StatusCode.INVALID_ARGUMENT, err)
        if _filter:
            outputs = {k: v for k, v in outputs.items() if k in _filter}
        names = list(outputs.keys())
        tensors = [] if use_content else None
        for k in (names if use_content else ()):
            v = outputs[k]
            if not isinstance(v, torch.Tensor):
                arr = np.asarray(v)
                if arr.dtype.kind in ("S", "U", "O"):
                    # string tensors can't ride the torch/tensor_content
                    # path; build the response with the python codec
                    tensors = None
                    break
                v = torch.as_tensor(arr)
            tensors.append(v)
        if tensors is None:
            blob = _serialize_response_py(spec, version, outputs,
                                          use_content)
        elif request_logger is None:
            # streaming reply: skeleton + payload regions; the C++ server
            # overlaps DMA (device) / sends zero-copy (host). Skipped when
            # request logging is on — the logger needs the full bytes.
            blob, regions, keepalive = native.serialize_predict_streaming(
                False, spec["name"], -1 if version is None else version,
                spec["signature_name"] or "serving_default", names,
                tensors)
            if regions and any(r[3] for r in regions):
                torch.cuda.current_stream().synchronize()
            metrics.observe_request("predict", time.perf_counter() - t0)
            return StreamingReply(blob, regions, keepalive)
        else:
            blob = native.serialize_predict_response(
                spec["name"], -1 if version is None else version,
                spec["signature_name"] or "serving_default", names,
                tensors, 1)
        metrics.observe_request("predict", time.perf_counter() - t0)
        if request_logger is not None:
            request_logger.log_predict(spec["name"], bytes(data),
                                       bytes(blob))
        return blob

    return handler


class NativeTransportServer:
    """Mounts the service implementations on the C++ HTTP/2 gRPC server.

    Used by ModelServer(transport="native"); owns the C++ server object
    and keeps the identity-echo fast-path table in sync with the model
    manager's state (via the manager's state event bus).
    """

    def __init__(self, manager, prediction_service, model_service,
                 address: str, device: str = "cpu",
                 metrics: Optional[MetricsRegistry] = None,
                 request_logger=None, max_workers: int = 16,
                 output_encoding: str = "tensor_content",
                 profiler_service=None):
        from . import _transport as T
        self._T = T
        self.manager = manager
        self._srv = T.GrpcServer(address, max_workers)
        self.metrics = metrics or MetricsRegistry()
        # requests served by the C++ echo fast path are accounted in C++;
        # merge them into the registry at read time
        self.metrics.attach_source(self._cxx_stats)
        self._use_content = output_encoding == "tensor_content"
        ps, ms = prediction_service, model_service
        pred_handler = _raw_predict_bytes_handler(
            manager, device, self.metrics, request_logger,
            use_content=self._use_content)
        import os as _os
        if (device.startswith("cuda")
                and _os.environ.get("MI355X_RX_PARSE", "1") != "0"):
            import torch as _torch
            dev_idx = _torch.device(device).index or 0
            # requests stream their tensor_content spans to the GPU while
            # still arriving (echo-table models are skipped in C++)
            self._srv.register_handler_parsed(PREDICT_PATH, pred_handler,
                                              dev_idx)
        else:
            self._srv.register_handler(PREDICT_PATH, pred_handler)
        self._srv.register_handler(
            _PS + "Classify",
            _proto_handler(ps.Classify, pb.ClassificationRequest))
        self._srv.register_handler(
            _PS + "Regress",
            _proto_handler(ps.Regress, pb.RegressionRequest))
        self._srv.register_handler(
            _PS + "MultiInference",
            _proto_handler(ps.MultiInference, pb.MultiInferenceRequest))
        self._srv.register_handler(
            _PS + "GetModelMetadata",
            _proto_handler(ps.GetModelMetadata, pb.GetModelMetadataRequest))
        self._srv.register_handler(
            _MS + "GetModelStatus",
            _proto_handler(ms.GetModelStatus, pb.GetModelStatusRequest))
        self._srv.register_handler(
            _MS + "HandleReloadConfigRequest",
            _proto_handler(ms.HandleReloadConfigRequest,
                           pb.ReloadConfigRequest))
        if profiler_service is not None:
            _PROF = "/tensorflow.ProfilerService/"
            self._srv.register_handler(
                _PROF + "Profile",
                _proto_handler(profiler_service.Profile, pb.ProfileRequest))
            self._srv.register_handler(
                _PROF + "Monitor",
                _proto_handler(profiler_service.Monitor, pb.MonitorRequest))
        self._request_logger = request_logger
        if request_logger is not None and hasattr(request_logger,
                                                  "subscribe"):
            request_logger.subscribe(self._refresh_echo_table)
        manager.subscribe(self._on_state_change)
        self._refresh_echo_table()
        self.address = None

    # -- echo fast path bookkeeping -------------------------------------
    def _on_state_change(self, name, version, state):
        self._refresh_echo_table()

    def _cxx_stats(self):
        stats = self._srv.stats()
        out = {}
        if PREDICT_PATH in stats:
            out["predict"] = stats[PREDICT_PATH]
        return out

    def _refresh_echo_table(self):
        """A model name enters the C++ echo table only when its LATEST
        available version is an identity servable; the version set lists
        exactly the identity versions (explicit requests for other
        versions fall through to the python handler). Disabled entirely in
        typed-output mode (the C++ echo answers with tensor_content)."""
        if not self._use_content:
            self._srv.set_echo_models(PREDICT_PATH, {})
            return
        logged = set()
        if self._request_logger is not None and hasattr(
                self._request_logger, "logged_models"):
            logged = self._request_logger.logged_models()
        models = {}
        by_name = {}
        for name, version, servable in self.manager.iter_available():
            by_name.setdefault(name, []).append((version, servable))
        for name, versions in by_name.items():
            if name in logged:  # logged requests must reach python
                continue
            versions.sort(key=lambda pair: pair[0])
            latest_servable = versions[-1][1]
            if getattr(latest_servable, "is_identity", False):
                idents = {v for v, s in versions
                          if getattr(s, "is_identity", False)}
                models[name] = idents
        self._srv.set_echo_models(PREDICT_PATH, models)

    # -- lifecycle -------------------------------------------------------
    def add_listener(self, address: str) -> None:
        """Extra listen address (e.g. --grpc_socket_path); before start."""
        self._srv.add_address(address)

    def start(self):
        self.address = self._srv.start()
        return self.address

    def stop(self):
        self._srv.stop()
