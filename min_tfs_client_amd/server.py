"""In-process PredictionService / ModelService server.

The reference repo has no loopback server — its integration tests require a
real out-of-process ``tensorflow_model_server`` (reference
requests_test.py:12-14, actions.yml:48). This module supplies the native
analogue the blueprint calls for: a grpcio server wired to our
own generated-style stubs, with

* a ``ModelManager`` whose version lifecycle mirrors TF-Serving's
  ``ModelVersionStatus.State`` machine (START→LOADING→AVAILABLE→UNLOADING→END,
  reference get_model_status.proto:20-68, core/loader_harness.h),
* pluggable ``Servable`` callables (identity echo, torch modules),
* Predict / Classify / Regress / MultiInference / GetModelMetadata and
  GetModelStatus / HandleReloadConfigRequest implementations matching the
  server-side semantics inventoried in SURVEY §2.4
  (predict_util.cc:89-226, model_service_impl.cc),
* optional request batching along dim 0 (batching_session.h:80-101 analogue)
  via ``min_tfs_client_amd.batching``,
* request metrics (req/s, latency) via ``min_tfs_client_amd.utils.metrics``.

Error messages follow predict_util.cc:65-85 shapes for drop-in UX.
"""
from __future__ import annotations

import os
import threading
import time
from concurrent import futures
from typing import Callable, Dict, Optional

import grpc
import numpy as np

try:
    import torch
except ImportError:  # pragma: no cover
    torch = None

from .tensors import (
    ndarray_to_tensor_proto,
    tensor_proto_to_ndarray,
    tensor_to_tensor_proto,
)
from .types import DataType
from .utils.metrics import MetricsRegistry
from .wire import messages as pb
from .wire.grpc_stubs import (
    ModelServiceServicer,
    PredictionServiceServicer,
    ProfilerServiceServicer,
    add_ModelServiceServicer_to_server,
    add_ProfilerServiceServicer_to_server,
    add_PredictionServiceServicer_to_server,
)

# ModelVersionStatus.State values (get_model_status.proto:27-45)
STATE_UNKNOWN = 0
STATE_START = 10
STATE_LOADING = 20
STATE_AVAILABLE = 30
STATE_UNLOADING = 40
STATE_END = 50


class Servable:
    """A loaded model version: maps {name: array} -> {name: array}.

    ``fn`` may return numpy arrays or torch tensors. ``signature`` describes
    inputs/outputs for GetModelMetadata: {"inputs": {alias: (dtype_enum,
    shape)}, "outputs": {...}, "method_name": str}.
    """

    def __init__(self, fn: Callable[[Dict[str, np.ndarray]], Dict[str, np.ndarray]],
                 signature: Optional[dict] = None,
                 signature_name: str = "serving_default"):
        self.fn = fn
        self.signature = signature or {}
        self.signature_name = signature_name

    def __call__(self, inputs: Dict[str, np.ndarray]) -> Dict[str, np.ndarray]:
        return self.fn(inputs)


def identity_servable() -> Servable:
    """Echo model: outputs == inputs with ``_output``-suffixed aliases when
    inputs use the reference fixture's ``*_input`` aliases, else same keys
    (mirrors the reference's identity SavedModel fixture,
    generate_tensorflow_model.py:12-57)."""

    def fn(inputs):
        out = {}
        for k, v in inputs.items():
            name = k[:-len("_input")] + "_output" if k.endswith("_input") else k
            out[name] = v
        return out

    s = Servable(fn, signature={"method_name": "tensorflow/serving/predict"})
    s.is_identity = True  # enables the all-C++ echo fast path in raw mode
    return s


class _Version:
    def __init__(self, servable: Optional[Servable]):
        self.servable = servable
        self.state = STATE_START
        self.status_error = None  # (code, message) on failed load


class ModelManager:
    """Servable registry with TF-Serving-style version states and a state
    event bus (ServableStateMonitor analogue, core/servable_state_monitor.h:
    subscribers get (model, version, state) on every transition)."""

    def __init__(self):
        self._lock = threading.RLock()
        self._models: Dict[str, Dict[int, _Version]] = {}
        # {model: {label: version}} — ModelSpec.version_label routing
        # (model.proto:9-33 oneof version_choice; ModelConfig.version_labels)
        self._labels: Dict[str, Dict[str, int]] = {}
        self._subscribers = []

    # -- state event bus ------------------------------------------------
    def subscribe(self, callback) -> None:
        """callback(model_name, version, state) on every transition."""
        with self._lock:
            self._subscribers.append(callback)

    def _notify(self, name: str, version: int, state: int) -> None:
        for cb in list(self._subscribers):
            try:
                cb(name, version, state)
            except Exception:  # noqa: BLE001 - observers must not break serving
                pass

    def wait_for_state(self, name: str, version: int, state: int,
                       timeout: float = 30.0) -> bool:
        """Blocks until (name, version) reaches `state` (monitor-style
        helper used by tests and warm-start orchestration)."""
        event = threading.Event()

        def cb(n, v, s):
            if n == name and v == version and s == state:
                event.set()

        self.subscribe(cb)
        with self._lock:
            v = self._models.get(name, {}).get(version)
            if v is not None and v.state == state:
                return True
        return event.wait(timeout)

    # -- lifecycle ------------------------------------------------------
    def load(self, name: str, servable: Servable, version: int = 1) -> None:
        with self._lock:
            versions = self._models.setdefault(name, {})
            v = _Version(servable)
            versions[version] = v
            v.state = STATE_LOADING
            self._notify(name, version, STATE_LOADING)
            v.state = STATE_AVAILABLE
        self._notify(name, version, STATE_AVAILABLE)

    def fail_load(self, name: str, version: int, code: int, msg: str) -> None:
        with self._lock:
            versions = self._models.setdefault(name, {})
            v = _Version(None)
            v.state = STATE_END
            v.status_error = (code, msg)
            versions[version] = v
        self._notify(name, version, STATE_END)

    def unload(self, name: str, version: Optional[int] = None) -> None:
        with self._lock:
            if name not in self._models:
                return
            versions = self._models[name]
            targets = [version] if version is not None else list(versions)
            for ver in targets:
                if ver in versions:
                    versions[ver].state = STATE_UNLOADING
                    self._notify(name, ver, STATE_UNLOADING)
                    servable = versions[ver].servable
                    close = getattr(servable, "close", None)
                    if callable(close):
                        close()  # e.g. BatchingServable's batcher thread
                    versions[ver].state = STATE_END
                    versions[ver].servable = None
                    self._notify(name, ver, STATE_END)

    def set_version_label(self, name: str, label: str,
                          version: int) -> None:
        """Labels may only point at AVAILABLE versions (TF semantics:
        --allow_version_labels_for_unavailable_models defaults false)."""
        with self._lock:
            v = self._models.get(name, {}).get(version)
            if v is None or v.state != STATE_AVAILABLE:
                raise KeyError(
                    f"Request to assign label to version {version} of model "
                    f"{name}, which is not currently available for "
                    f"inference")
            self._labels.setdefault(name, {})[label] = version

    # -- lookup ---------------------------------------------------------
    def get(self, name: str, version: Optional[int] = None,
            version_label: Optional[str] = None) -> Servable:
        """Resolve to an AVAILABLE servable; raises KeyError with a
        TF-Serving-shaped message."""
        with self._lock:
            if name not in self._models:
                raise KeyError(f"Servable not found for request: Latest({name})")
            if version is None and version_label:
                version = self._labels.get(name, {}).get(version_label)
                if version is None:
                    raise KeyError(
                        f"Servable not found for request: Specific({name}, "
                        f"label {version_label})")
            versions = self._models[name]
            if version is not None:
                v = versions.get(version)
                if v is None or v.state != STATE_AVAILABLE:
                    raise KeyError(
                        f"Servable not found for request: Specific({name}, "
                        f"{version})")
                return v.servable
            avail = [ver for ver, v in versions.items()
                     if v.state == STATE_AVAILABLE]
            if not avail:
                raise KeyError(f"Servable not found for request: Latest({name})")
            return versions[max(avail)].servable

    def version_statuses(self, name: str):
        with self._lock:
            if name not in self._models:
                raise KeyError(
                    f"Could not find any versions of model {name}")
            return [(ver, v.state, v.status_error)
                    for ver, v in sorted(self._models[name].items())]

    def model_names(self):
        with self._lock:
            return list(self._models)

    def iter_available(self):
        """Snapshot of (name, version, servable) for every AVAILABLE
        version (used by the native transport's echo-table refresh)."""
        out = []
        with self._lock:
            for name, versions in self._models.items():
                for ver, v in versions.items():
                    if v.state == STATE_AVAILABLE and v.servable is not None:
                        out.append((name, ver, v.servable))
        return out


# ---------------------------------------------------------------------------
# SignatureDef-driven request validation (predict_util.cc:66-146 semantics)
# ---------------------------------------------------------------------------

def validate_inputs_against_signature(servable, inputs) -> Optional[str]:
    """Returns a TF-Serving-shaped INVALID_ARGUMENT message when the
    request's input aliases do not match the servable's declared input
    signature, else None. Servables without a declared input signature
    accept any aliases (mirrors serving a SignatureDef-less model).

    Message shapes match reference predict_util.cc:66-111
    (VerifyRequestInputsSize / PreProcessPrediction).
    """
    sig_inputs = (servable.signature or {}).get("inputs")
    if not sig_inputs:
        return None
    req = set(inputs)
    sig = set(sig_inputs)
    if len(req) != len(sig):
        extra = sorted(req - sig)
        missing = sorted(sig - req)
        return (
            f"input size does not match signature: {len(req)}!={len(sig)} "
            f"len({{{','.join(sorted(req))}}}) != "
            f"len({{{','.join(sorted(sig))}}}). "
            f"Sent extra: {{{','.join(extra)}}}. "
            f"Missing but required: {{{','.join(missing)}}}.")
    for alias in sorted(req):
        if alias not in sig:
            return (
                f"input tensor alias not found in signature: {alias}. "
                f"Inputs expected to be in the set "
                f"{{{','.join(sorted(sig))}}}.")
    return None


def validate_output_filter(servable, outputs, output_filter) -> Optional[str]:
    """Validates Predict.output_filter entries against the declared output
    signature (when present) or the produced outputs; unknown aliases and
    duplicates are INVALID_ARGUMENT per predict_util.cc:119-136 (the
    reference never silently drops a filter entry)."""
    if not output_filter:
        return None
    sig_outputs = (servable.signature or {}).get("outputs")
    known = set(sig_outputs) if sig_outputs else set(outputs)
    seen = set()
    for alias in output_filter:
        if alias not in known:
            return (
                f"output tensor alias not found in signature: {alias} "
                f"Outputs expected to be in the set "
                f"{{{','.join(sorted(known))}}}.")
        if alias in seen:
            return f"duplicate output tensor alias: {alias}"
        seen.add(alias)
    return None


# ---------------------------------------------------------------------------
# Service implementations
# ---------------------------------------------------------------------------

def _decode_input(proto):
    """Decode a request TensorProto to the natural host type: torch tensor
    for DT_BFLOAT16 (numpy cannot carry bf16), numpy array otherwise."""
    if torch is not None and proto.dtype == 14:  # DT_BFLOAT16
        from .tensors import tensor_proto_to_tensor
        return tensor_proto_to_tensor(proto)
    return tensor_proto_to_ndarray(proto)


def _abort(context, code, msg):
    # context.abort raises; the fallback covers non-grpc contexts in tests
    context.abort(code, msg)
    raise grpc.RpcError(msg)  # pragma: no cover


class PredictionServiceImpl(PredictionServiceServicer):
    def __init__(self, manager: ModelManager,
                 output_encoding: str = "tensor_content",
                 metrics: Optional[MetricsRegistry] = None,
                 request_logger=None):
        assert output_encoding in ("tensor_content", "typed")
        self._manager = manager
        self._use_content = output_encoding == "tensor_content"
        self.metrics = metrics or MetricsRegistry()
        self.request_logger = request_logger

    # -- helpers --------------------------------------------------------
    def _resolve(self, model_spec, context):
        version = None
        label = None
        which = model_spec.WhichOneof("version_choice")
        if which == "version":
            version = model_spec.version.value
        elif which == "version_label":
            label = model_spec.version_label
        try:
            return self._manager.get(model_spec.name, version, label)
        except KeyError as e:
            _abort(context, grpc.StatusCode.NOT_FOUND, str(e))

    def _encode_outputs(self, response, outputs, output_filter=()):
        wanted = set(output_filter) if output_filter else None
        for k, v in outputs.items():
            if wanted is not None and k not in wanted:
                continue
            if torch is not None and isinstance(v, torch.Tensor):
                proto = tensor_to_tensor_proto(v.cpu(), self._use_content)
            else:
                proto = ndarray_to_tensor_proto(np.asarray(v),
                                                self._use_content)
            response.outputs[k].CopyFrom(proto)

    # -- rpcs -----------------------------------------------------------
    def Predict(self, request, context):
        t0 = time.perf_counter()
        servable = self._resolve(request.model_spec, context)
        try:
            inputs = {k: _decode_input(v) for k, v in request.inputs.items()}
        except Exception as e:
            _abort(context, grpc.StatusCode.INVALID_ARGUMENT,
                   f"tensor parsing error: {e}")
        err = validate_inputs_against_signature(servable, inputs)
        if err is not None:
            _abort(context, grpc.StatusCode.INVALID_ARGUMENT, err)
        try:
            outputs = servable(inputs)
        except ValueError as e:
            # e.g. batch larger than max_batch_size (BatchingServable):
            # TF-Serving surfaces these as INVALID_ARGUMENT, not INTERNAL
            _abort(context, grpc.StatusCode.INVALID_ARGUMENT, str(e))
        except Exception as e:
            _abort(context, grpc.StatusCode.INTERNAL, str(e))
        err = validate_output_filter(servable, outputs,
                                     request.output_filter)
        if err is not None:
            _abort(context, grpc.StatusCode.INVALID_ARGUMENT, err)
        response = pb.PredictResponse()
        response.model_spec.CopyFrom(request.model_spec)
        response.model_spec.signature_name = (
            request.model_spec.signature_name or "serving_default")
        self._encode_outputs(response, outputs, request.output_filter)
        self.metrics.observe_request("predict", time.perf_counter() - t0)
        if self.request_logger is not None:
            self.request_logger.log_predict(request.model_spec.name,
                                            request, response)
        return response

    def Classify(self, request, context):
        t0 = time.perf_counter()
        servable = self._resolve(request.model_spec, context)
        classify_fn = getattr(servable, "classify", None)
        if classify_fn is None:
            _abort(context, grpc.StatusCode.UNIMPLEMENTED,
                   f"Expected a classification signature for model "
                   f"{request.model_spec.name}")
        result = classify_fn(request.input)
        # example-count metric (servables/tensorflow/util.cc:36-66 analogue)
        n_examples = len(request.input.example_list.examples) or \
            len(request.input.example_list_with_context.examples)
        self.metrics.inc(
            f"request_example_counts{{model={request.model_spec.name!r}}}",
            n_examples)
        response = pb.ClassificationResponse()
        response.model_spec.CopyFrom(request.model_spec)
        response.result.CopyFrom(result)
        self.metrics.observe_request("classify", time.perf_counter() - t0)
        return response

    def Regress(self, request, context):
        t0 = time.perf_counter()
        servable = self._resolve(request.model_spec, context)
        regress_fn = getattr(servable, "regress", None)
        if regress_fn is None:
            _abort(context, grpc.StatusCode.UNIMPLEMENTED,
                   f"Expected a regression signature for model "
                   f"{request.model_spec.name}")
        result = regress_fn(request.input)
        n_examples = len(request.input.example_list.examples) or \
            len(request.input.example_list_with_context.examples)
        self.metrics.inc(
            f"request_example_counts{{model={request.model_spec.name!r}}}",
            n_examples)
        response = pb.RegressionResponse()
        response.model_spec.CopyFrom(request.model_spec)
        response.result.CopyFrom(result)
        self.metrics.observe_request("regress", time.perf_counter() - t0)
        return response

    def MultiInference(self, request, context):
        response = pb.MultiInferenceResponse()
        for task in request.tasks:
            servable = self._resolve(task.model_spec, context)
            r = response.results.add()
            r.model_spec.CopyFrom(task.model_spec)
            if task.method_name == "tensorflow/serving/classify":
                fn = getattr(servable, "classify", None)
                if fn is None:
                    _abort(context, grpc.StatusCode.UNIMPLEMENTED,
                           f"Expected a classification signature for model "
                           f"{task.model_spec.name}")
                r.classification_result.CopyFrom(fn(request.input))
            elif task.method_name == "tensorflow/serving/regress":
                fn = getattr(servable, "regress", None)
                if fn is None:
                    _abort(context, grpc.StatusCode.UNIMPLEMENTED,
                           f"Expected a regression signature for model "
                           f"{task.model_spec.name}")
                r.regression_result.CopyFrom(fn(request.input))
            else:
                _abort(context, grpc.StatusCode.INVALID_ARGUMENT,
                       f"Unsupported signature method_name: "
                       f"{task.method_name}")
        return response

    def GetModelMetadata(self, request, context):
        servable = self._resolve(request.model_spec, context)
        response = pb.GetModelMetadataResponse()
        response.model_spec.CopyFrom(request.model_spec)
        sdm = pb.SignatureDefMap()
        sig = sdm.signature_def[servable.signature_name]
        sig.method_name = servable.signature.get(
            "method_name", "tensorflow/serving/predict")
        for io_key in ("inputs", "outputs"):
            for alias, (dtype_enum, shape) in servable.signature.get(
                    io_key, {}).items():
                info = getattr(sig, io_key)[alias]
                info.name = f"{alias}:0"
                info.dtype = dtype_enum
                for d in shape:
                    info.tensor_shape.dim.add().size = d
        any_msg = response.metadata["signature_def"]
        any_msg.type_url = ("type.googleapis.com/"
                            "tensorflow.serving.SignatureDefMap")
        any_msg.value = sdm.SerializeToString()
        return response


class ModelServiceImpl(ModelServiceServicer):
    def __init__(self, manager: ModelManager,
                 servable_factory: Optional[Callable[[str, str], Servable]]
                 = None,
                 storage_source=None):
        self._manager = manager
        # used by HandleReloadConfigRequest to instantiate servables for
        # config entries: (name, version_dir) -> Servable
        self._servable_factory = servable_factory
        # preferred reload route: the FileSystemStoragePathSource, so the
        # poller's bookkeeping and the admin API stay consistent
        self._storage_source = storage_source

    def GetModelStatus(self, request, context):
        response = pb.GetModelStatusResponse()
        try:
            statuses = self._manager.version_statuses(request.model_spec.name)
        except KeyError as e:
            _abort(context, grpc.StatusCode.NOT_FOUND, str(e))
        want_version = None
        if request.model_spec.HasField("version"):
            want_version = request.model_spec.version.value
        for version, state, err in statuses:
            if want_version is not None and version != want_version:
                continue
            s = response.model_version_status.add()
            s.version = version
            s.state = state
            # status is always present (empty == OK), matching
            # tensorflow_model_server's JSON shape asserted by the
            # reference's own test (requests_test.py:43-50).
            s.status.SetInParent()
            if err is not None:
                s.status.error_code = err[0]
                s.status.error_message = err[1]
        if not response.model_version_status:
            _abort(context, grpc.StatusCode.NOT_FOUND,
                   f"Could not find version {want_version} of model "
                   f"{request.model_spec.name}")
        return response

    def HandleReloadConfigRequest(self, request, context):
        response = pb.ReloadConfigResponse()
        cfg = request.config
        if cfg.WhichOneof("config") != "model_config_list":
            response.status.error_code = pb.ErrorCode.INVALID_ARGUMENT
            response.status.error_message = (
                "ServerCore accepts only model_config_list")
            return response
        entries = list(cfg.model_config_list.config)
        if self._storage_source is not None:
            # route through the storage source: set_models unloads removed
            # models AND invalidates its _loaded bookkeeping, so polled
            # models stay consistent with admin reloads
            from .repository import VersionPolicy
            configs = {c.name: c.base_path for c in entries}
            policies = {
                c.name: VersionPolicy.from_proto(
                    c.model_version_policy
                    if c.HasField("model_version_policy") else None)
                for c in entries}
            try:
                self._storage_source.set_models(configs, policies)
                self._storage_source.poll_once()
                for c in entries:
                    for label, ver in dict(c.version_labels).items():
                        self._manager.set_version_label(c.name, label, ver)
            except Exception as e:  # noqa: BLE001
                response.status.error_code = pb.ErrorCode.UNKNOWN
                response.status.error_message = str(e)
                return response
            response.status.error_code = pb.ErrorCode.OK
            return response
        wanted = {c.name: c.base_path for c in entries}
        # unload models not in the new config; load new ones via factory
        for name in self._manager.model_names():
            if name not in wanted:
                self._manager.unload(name)
        if self._servable_factory is not None:
            for name, base_path in wanted.items():
                if name not in self._manager.model_names():
                    try:
                        self._load_via_factory(name, base_path)
                    except Exception as e:  # noqa: BLE001
                        response.status.error_code = pb.ErrorCode.UNKNOWN
                        response.status.error_message = str(e)
                        return response
        response.status.error_code = pb.ErrorCode.OK
        return response

    def _load_via_factory(self, name: str, base_path: str) -> None:
        """TF layout: base_path contains numeric version dirs; the factory
        receives a VERSION directory. A base_path with no numeric children
        is treated as a single version-1 directory (legacy layouts)."""
        import os
        versions = []
        if os.path.isdir(base_path):
            versions = sorted(
                int(e) for e in os.listdir(base_path)
                if e.isdigit() and os.path.isdir(os.path.join(base_path,
                                                              e)))
        if versions:
            latest = versions[-1]
            vdir = os.path.join(
                base_path,
                next(e for e in os.listdir(base_path)
                     if e.isdigit() and int(e) == latest))
            self._manager.load(name, self._servable_factory(name, vdir),
                               version=latest)
        else:
            self._manager.load(name,
                               self._servable_factory(name, base_path))


# ---------------------------------------------------------------------------
# Server wrapper
# ---------------------------------------------------------------------------

def _raw_predict_handler(manager: ModelManager, device: str,
                         metrics: MetricsRegistry, request_logger=None):
    """Raw-bytes Predict: request bytes in, response bytes out, all codec
    work in the C++ extension (turbo path — see turbo.py)."""
    from .ops import require_native

    def raw_predict(data, context):
        t0 = time.perf_counter()
        native = require_native()
        try:
            spec, inputs, _filter = native.parse_predict_request(
                data, device, 1)
        except Exception as e:  # noqa: BLE001
            _abort(context, grpc.StatusCode.INVALID_ARGUMENT,
                   f"request parsing error: {e}")
        version = spec["version"] if spec["version"] >= 0 else None
        label = spec.get("version_label") or None
        try:
            servable = manager.get(spec["name"], version, label)
        except KeyError as e:
            _abort(context, grpc.StatusCode.NOT_FOUND, str(e))
        if getattr(servable, "is_identity", False) and not _filter:
            out = native.echo_predict(data)
            metrics.observe_bytes("rx", len(data))
            metrics.observe_bytes("tx", len(out))
            metrics.observe_request("predict", time.perf_counter() - t0)
            if request_logger is not None:
                request_logger.log_predict(spec["name"], bytes(data),
                                           bytes(out))
            return out
        err = validate_inputs_against_signature(servable, inputs)
        if err is not None:
            _abort(context, grpc.StatusCode.INVALID_ARGUMENT, err)
        try:
            outputs = servable(inputs)
        except ValueError as e:
            _abort(context, grpc.StatusCode.INVALID_ARGUMENT, str(e))
        except Exception as e:  # noqa: BLE001
            _abort(context, grpc.StatusCode.INTERNAL, str(e))
        err = validate_output_filter(servable, outputs, _filter)
        if err is not None:
            _abort(context, grpc.StatusCode.INVALID_ARGUMENT, err)
        if _filter:
            outputs = {k: v for k, v in outputs.items() if k in _filter}
        names = list(outputs.keys())
        tensors = []
        for k in names:
            v = outputs[k]
            if not isinstance(v, torch.Tensor):
                v = torch.as_tensor(np.asarray(v))
            tensors.append(v)
        blob = native.serialize_predict_response(
            spec["name"], -1 if version is None else version,
            spec["signature_name"] or "serving_default", names, tensors, 1)
        metrics.observe_request("predict", time.perf_counter() - t0)
        if request_logger is not None:
            request_logger.log_predict(spec["name"], bytes(data),
                                       bytes(blob))
        return blob

    def identity(x):
        return x

    return grpc.unary_unary_rpc_method_handler(
        raw_predict, request_deserializer=identity,
        response_serializer=identity)


class ProfilerServiceImpl(ProfilerServiceServicer):
    """tensorflow.ProfilerService analogue — the reference model server
    registers this service alongside Model/Prediction
    (reference server.cc:324,339; profiler_service.proto:12-17).

    ``Profile`` captures ``duration_ms`` of in-process spans from the
    chrome-trace Tracer (utils/tracing.py) and returns them as a
    ``trace_viewer.json`` tool_data payload; ``Monitor`` returns a
    human-readable metrics snapshot (level 2 adds per-method latency
    quantiles), matching the proto's "properly formatted string data"
    contract."""

    def __init__(self, metrics: MetricsRegistry):
        self.metrics = metrics

    def Profile(self, request, context):
        import tempfile

        from .utils.tracing import Tracer
        from .wire import messages as pb
        tracer = Tracer.get()
        tracer.clear()
        tracer.start()
        dur_ms = request.duration_ms or 1000
        time.sleep(min(dur_ms, 60_000) / 1000.0)
        tracer.stop()
        fd, path = tempfile.mkstemp(suffix=".json")
        os.close(fd)
        try:
            n = tracer.export(path)
            with open(path, "rb") as f:
                data = f.read()
        finally:
            os.unlink(path)
        resp = pb.ProfileResponse()
        resp.empty_trace = n == 0
        td = resp.tool_data.add()
        td.name = "trace_viewer.json"
        td.data = data
        return resp

    def Monitor(self, request, context):
        from .wire import messages as pb
        if request.duration_ms:
            time.sleep(min(request.duration_ms, 10_000) / 1000.0)
        lines = []
        if request.timestamp:
            lines.append(f"timestamp: {time.time():.3f}")
        for name, value in sorted(self.metrics.counters().items()):
            lines.append(f"{name}: {value}")
        if request.monitoring_level >= 2:
            with self.metrics._lock:
                methods = sorted(self.metrics._latency)
            for method in methods:
                q = self.metrics.latency_quantiles(method)
                if not q:
                    continue
                lines.append(f"latency[{method}]: " + ", ".join(
                    f"{k}={v * 1e3:.3f}ms" if k != "count" else f"count={v}"
                    for k, v in q.items()))
        resp = pb.MonitorResponse()
        resp.data = "\n".join(lines) + "\n"
        return resp


class ModelServer:
    """Build-and-start wrapper (Server::BuildAndStart analogue,
    reference server.cc:291-339).

    ``raw_predict=True`` swaps the Predict method onto the C++ codec
    (identity (de)serializers; turbo clients and standard protobuf clients
    both interoperate — the wire bytes are identical). ``address`` may be a
    "unix:///path.sock" target for low-overhead loopback serving.
    """

    def __init__(self, port: int = 0, max_workers: int = 16,
                 output_encoding: str = "tensor_content",
                 manager: Optional[ModelManager] = None,
                 servable_factory=None,
                 raw_predict: bool = False,
                 device: str = "cpu",
                 address: Optional[str] = None,
                 shm_handshake_dir: Optional[str] = None,
                 transport: str = "native",
                 storage_source=None):
        from .utils.allocator import tune_malloc
        tune_malloc()
        from .request_logging import ServerRequestLogger
        self.manager = manager or ModelManager()
        self.metrics = MetricsRegistry()
        self.request_logger = ServerRequestLogger()
        self.prediction_service = None
        self.model_service = None
        self._server = None
        self._native = None
        if transport == "native":
            # C++ HTTP/2 gRPC server (native_transport.py): the Predict
            # data plane stays off python-grpcio entirely
            from .native_transport import NativeTransportServer
            self.prediction_service = PredictionServiceImpl(
                self.manager, output_encoding, self.metrics,
                self.request_logger)
            self.model_service = ModelServiceImpl(self.manager,
                                                  servable_factory,
                                                  storage_source)
            self.profiler_service = ProfilerServiceImpl(self.metrics)
            native_addr = address if address is not None \
                else f"127.0.0.1:{port}"
            self._native = NativeTransportServer(
                self.manager, self.prediction_service, self.model_service,
                native_addr, device=device, metrics=self.metrics,
                request_logger=self.request_logger,
                max_workers=max_workers,
                output_encoding=output_encoding,
                profiler_service=self.profiler_service)
            self.shm_listener = None
            if shm_handshake_dir:
                from .shm import ShmListener
                self.shm_listener = ShmListener(self.manager,
                                                shm_handshake_dir,
                                                device=device)
            self.address = native_addr
            self.port = port
            return
        if transport != "grpcio":
            raise ValueError(f"unknown transport {transport!r} "
                             "(expected 'native' or 'grpcio')")
        self._server = grpc.server(
            futures.ThreadPoolExecutor(max_workers=max_workers),
            options=[("grpc.max_send_message_length", 1 << 30),
                     ("grpc.max_receive_message_length", 1 << 30),
                     ("grpc.http2.max_frame_size", 16 * 1024 * 1024 - 1)])
        self.prediction_service = PredictionServiceImpl(
            self.manager, output_encoding, self.metrics,
            self.request_logger)
        self.model_service = ModelServiceImpl(self.manager, servable_factory,
                                              storage_source)
        self.profiler_service = ProfilerServiceImpl(self.metrics)
        add_ProfilerServiceServicer_to_server(self.profiler_service,
                                              self._server)
        if raw_predict:
            from .wire import messages as _pb
            from .wire.grpc_stubs import (
                add_ModelServiceServicer_to_server as _add_ms)
            handlers = {
                "Predict": _raw_predict_handler(self.manager, device,
                                                self.metrics,
                                                self.request_logger),
                "Classify": grpc.unary_unary_rpc_method_handler(
                    self.prediction_service.Classify,
                    request_deserializer=_pb.ClassificationRequest.FromString,
                    response_serializer=(
                        _pb.ClassificationResponse.SerializeToString)),
                "Regress": grpc.unary_unary_rpc_method_handler(
                    self.prediction_service.Regress,
                    request_deserializer=_pb.RegressionRequest.FromString,
                    response_serializer=(
                        _pb.RegressionResponse.SerializeToString)),
                "MultiInference": grpc.unary_unary_rpc_method_handler(
                    self.prediction_service.MultiInference,
                    request_deserializer=(
                        _pb.MultiInferenceRequest.FromString),
                    response_serializer=(
                        _pb.MultiInferenceResponse.SerializeToString)),
                "GetModelMetadata": grpc.unary_unary_rpc_method_handler(
                    self.prediction_service.GetModelMetadata,
                    request_deserializer=(
                        _pb.GetModelMetadataRequest.FromString),
                    response_serializer=(
                        _pb.GetModelMetadataResponse.SerializeToString)),
            }
            self._server.add_generic_rpc_handlers((
                grpc.method_handlers_generic_handler(
                    "tensorflow.serving.PredictionService", handlers),))
            _add_ms(self.model_service, self._server)
        else:
            add_PredictionServiceServicer_to_server(self.prediction_service,
                                                    self._server)
            add_ModelServiceServicer_to_server(self.model_service,
                                               self._server)
        self.shm_listener = None
        if shm_handshake_dir:
            from .shm import ShmListener
            self.shm_listener = ShmListener(self.manager,
                                            shm_handshake_dir,
                                            device=device)
        if address is not None:
            self.address = address
            self.port = self._server.add_insecure_port(address)
        else:
            self.port = self._server.add_insecure_port(f"127.0.0.1:{port}")
            self.address = f"127.0.0.1:{self.port}"

    def start(self) -> "ModelServer":
        if self._native is not None:
            self.address = self._native.start()
            if not self.address.startswith("unix:"):
                self.port = int(self.address.rsplit(":", 1)[1])
        else:
            self._server.start()
        if self.shm_listener is not None:
            self.shm_listener.start()
        return self

    def stop(self, grace: Optional[float] = None) -> None:
        if self.shm_listener is not None:
            self.shm_listener.stop()
        if self._native is not None:
            self._native.stop()
        else:
            self._server.stop(grace)

    def __enter__(self):
        return self.start()

    def __exit__(self, *exc):
        self.stop(0)
