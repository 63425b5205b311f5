"""Classify/Regress example adapters + request logging."""
import numpy as np
import pytest

from min_tfs_client_amd.examples_adapter import (
    ClassificationAdapter,
    RegressionAdapter,
    examples_input,
    examples_to_feature_arrays,
    make_example,
)
from min_tfs_client_amd.repository import read_tfrecord
from min_tfs_client_amd.request_logging import (
    FileLogCollector,
    RequestLogger,
    ServerRequestLogger,
)
from min_tfs_client_amd.server import Servable
from min_tfs_client_amd.tensors import ndarray_to_tensor_proto
from min_tfs_client_amd.wire import messages as pb


# -- example feature codec ---------------------------------------------------

def test_make_example_arms():
    ex = make_example({"age": 42, "score": 1.5, "name": "bob",
                       "vec": np.array([1.0, 2.0], np.float32)})
    assert list(ex.features.feature["age"].int64_list.value) == [42]
    assert list(ex.features.feature["score"].float_list.value) == [1.5]
    assert list(ex.features.feature["name"].bytes_list.value) == [b"bob"]
    assert list(ex.features.feature["vec"].float_list.value) == [1.0, 2.0]


def test_examples_to_feature_arrays():
    inp = examples_input([{"x": [1.0, 2.0], "label": 1},
                          {"x": [3.0, 4.0], "label": 0}])
    feats = examples_to_feature_arrays(inp)
    np.testing.assert_array_equal(
        feats["x"], np.array([[1.0, 2.0], [3.0, 4.0]], np.float32))
    np.testing.assert_array_equal(feats["label"], [[1], [0]])


def test_examples_with_context_merged():
    inp = pb.Input()
    ewc = inp.example_list_with_context
    ewc.context.CopyFrom(make_example({"common": 7}))
    ewc.examples.add().CopyFrom(make_example({"x": 1.0}))
    ewc.examples.add().CopyFrom(make_example({"x": 2.0}))
    feats = examples_to_feature_arrays(inp)
    np.testing.assert_array_equal(feats["common"], [[7], [7]])


def test_empty_input_raises():
    with pytest.raises(ValueError, match="empty"):
        examples_to_feature_arrays(pb.Input())


# -- adapters ----------------------------------------------------------------

def _scores_servable():
    def fn(features):
        x = features["x"]
        return {"scores": np.stack([x.sum(axis=1), -x.sum(axis=1)],
                                   axis=1)}
    return Servable(fn)


def test_classification_adapter():
    adapter = ClassificationAdapter(_scores_servable(),
                                    labels=["pos", "neg"])
    result = adapter.classify(examples_input(
        [{"x": [1.0, 2.0]}, {"x": [0.5, 0.5]}]))
    assert len(result.classifications) == 2
    first = result.classifications[0]
    assert [c.label for c in first.classes] == ["pos", "neg"]
    assert first.classes[0].score == pytest.approx(3.0)


def test_regression_adapter():
    def fn(features):
        return {"value": features["x"].sum(axis=1)}
    adapter = RegressionAdapter(Servable(fn))
    result = adapter.regress(examples_input(
        [{"x": [1.0, 2.0]}, {"x": [3.0, 3.0]}]))
    assert [r.value for r in result.regressions] == [3.0, 6.0]


def test_classify_rpc_through_server():
    from min_tfs_client_amd.client import TensorServingClient
    from min_tfs_client_amd.server import ModelServer
    with ModelServer(port=0) as srv:
        srv.manager.load("clf", ClassificationAdapter(
            _scores_servable(), labels=["a", "b"]), version=1)
        c = TensorServingClient("127.0.0.1", srv.port)
        try:
            resp = c.classify("clf", examples_input([{"x": [2.0, 2.0]}]))
            classes = resp.result.classifications[0].classes
            assert classes[0].label == "a"
            assert classes[0].score == pytest.approx(4.0)
        finally:
            c.close()


def test_regress_rpc_through_raw_server():
    """Classify/Regress still speak protobuf when Predict is raw."""
    from min_tfs_client_amd.client import TensorServingClient
    from min_tfs_client_amd.server import ModelServer

    def fn(features):
        return {"value": features["x"].sum(axis=1)}

    with ModelServer(port=0, raw_predict=True) as srv:
        srv.manager.load("reg", RegressionAdapter(Servable(fn)), version=1)
        c = TensorServingClient("127.0.0.1", srv.port)
        try:
            resp = c.regress("reg", examples_input([{"x": [1.0, 1.5]}]))
            assert resp.result.regressions[0].value == pytest.approx(2.5)
        finally:
            c.close()


def test_multi_inference_rpc():
    from min_tfs_client_amd.client import TensorServingClient
    from min_tfs_client_amd.server import ModelServer

    def fn(features):
        s = features["x"].sum(axis=1)
        return {"scores": np.stack([s, -s], 1), "value": s}

    with ModelServer(port=0) as srv:
        srv.manager.load("m", ClassificationAdapter(Servable(fn)),
                         version=1)
        reg = RegressionAdapter(Servable(fn))
        srv.manager.load("m2", reg, version=1)
        c = TensorServingClient("127.0.0.1", srv.port)
        try:
            resp = c.multi_inference_request(
                [("m", "tensorflow/serving/classify"),
                 ("m2", "tensorflow/serving/regress")],
                examples_input([{"x": [1.0, 1.0]}]))
            assert resp.results[0].classification_result.classifications
            assert resp.results[1].regression_result.regressions[0].value \
                == pytest.approx(2.0)
        finally:
            c.close()


# -- request logging ---------------------------------------------------------

def _predict_pair():
    req = pb.PredictRequest()
    req.model_spec.name = "m"
    req.inputs["x"].CopyFrom(ndarray_to_tensor_proto(
        np.ones(2, np.float32)))
    resp = pb.PredictResponse()
    resp.outputs["x"].CopyFrom(ndarray_to_tensor_proto(
        np.ones(2, np.float32)))
    return req, resp


def test_request_logger_roundtrip(tmp_path):
    collector = FileLogCollector(str(tmp_path / "log"))
    logger = RequestLogger(collector, sampling_rate=1.0)
    req, resp = _predict_pair()
    assert logger.log_predict(req, resp)
    collector.flush()
    records = read_tfrecord(str(tmp_path / "log.m.log"))
    log = pb.PredictionLog.FromString(records[0])
    assert log.WhichOneof("log_type") == "predict_log"
    assert log.predict_log.request.model_spec.name == "m"
    assert log.log_metadata.sampling_config.sampling_rate == 1.0


def test_request_logger_sampling(tmp_path):
    logger = RequestLogger(FileLogCollector(str(tmp_path / "l")),
                           sampling_rate=0.0)
    req, resp = _predict_pair()
    assert not logger.log_predict(req, resp)
    assert logger.seen == 1 and logger.logged == 0


def test_server_request_logger_config(tmp_path):
    srl = ServerRequestLogger()
    cfg = pb.LoggingConfig()
    cfg.log_collector_config.filename_prefix = str(tmp_path / "plog")
    cfg.sampling_config.sampling_rate = 1.0
    srl.configure("m", cfg)
    req, resp = _predict_pair()
    srl.log_predict("m", req, resp)
    srl.flush_all()
    assert read_tfrecord(str(tmp_path / "plog.m.log"))
    srl.configure("m", None)
    assert srl.get("m") is None


def test_logging_through_raw_server(tmp_path):
    """Raw-path logging captures byte-level request/response pairs that
    replay as warmup records."""
    import torch
    from min_tfs_client_amd.server import ModelServer, identity_servable
    from min_tfs_client_amd.turbo import TurboPredictClient
    pytest.importorskip("min_tfs_client_amd._native")

    with ModelServer(port=0, raw_predict=True) as srv:
        srv.manager.load("m", identity_servable(), version=1)
        cfg = pb.LoggingConfig()
        cfg.log_collector_config.filename_prefix = str(tmp_path / "raw")
        cfg.sampling_config.sampling_rate = 1.0
        srv.request_logger.configure("m", cfg)
        with TurboPredictClient(srv.address) as c:
            c.predict("m", {"x": torch.ones(3)})
        srv.request_logger.flush_all()
    records = read_tfrecord(str(tmp_path / "raw.m.log"))
    log = pb.PredictionLog.FromString(records[0])
    assert "x" in log.predict_log.request.inputs
    assert "x" in log.predict_log.response.outputs


# -- tracing -----------------------------------------------------------------

def test_tracer_spans_and_export(tmp_path):
    import torch
    from min_tfs_client_amd.utils.tracing import Tracer
    from min_tfs_client_amd.server import ModelServer, identity_servable
    from min_tfs_client_amd.turbo import TurboPredictClient

    tracer = Tracer.get()
    tracer.clear()
    tracer.start()
    try:
        with ModelServer(port=0, raw_predict=True) as srv:
            srv.manager.load("m", identity_servable(), version=1)
            with TurboPredictClient(srv.address) as c:
                c.predict("m", {"x": torch.ones(4)})
    finally:
        tracer.stop()
    out = str(tmp_path / "trace.json")
    n = tracer.export(out)
    assert n >= 3  # serialize, rpc, parse
    import json as _json
    events = _json.load(open(out))["traceEvents"]
    names = {e["name"] for e in events}
    assert {"turbo.serialize", "turbo.rpc", "turbo.parse"} <= names


def test_tracer_disabled_is_noop():
    from min_tfs_client_amd.utils.tracing import Tracer, trace_span
    t = Tracer.get()
    t.clear()
    assert not t.enabled
    with trace_span("nothing"):
        pass
    assert t.export("/dev/null") == 0


# -- shm native helpers ------------------------------------------------------

def test_shm_wait_and_store_helpers():
    import pytest as _pytest
    native = _pytest.importorskip("min_tfs_client_amd._native")
    import threading
    buf = bytearray(64)
    mv = memoryview(buf)
    # timeout returns sentinel
    assert native.shm_wait_value(mv, 0, 7, 0.05) == 0xFFFFFFFF
    # store+wait round trip across a thread
    def setter():
        import time
        time.sleep(0.05)
        native.shm_store_value(mv, 0, 7)
    t = threading.Thread(target=setter)
    t.start()
    assert native.shm_wait_value(mv, 0, 7, 5.0) == 7
    t.join()
    # immediate hit
    assert native.shm_wait_value(mv, 0, 7, 0.01) == 7


def test_log_collector_appends_across_flushes(tmp_path):
    collector = FileLogCollector(str(tmp_path / "log"))
    logger = RequestLogger(collector, sampling_rate=1.0)
    req, resp = _predict_pair()
    logger.log_predict(req, resp)
    collector.flush()
    logger.log_predict(req, resp)
    collector.flush()
    collector.flush()  # idempotent on empty buffer
    records = read_tfrecord(str(tmp_path / "log.m.log"))
    assert len(records) == 2


def test_log_collector_auto_flush(tmp_path):
    collector = FileLogCollector(str(tmp_path / "auto"))
    logger = RequestLogger(collector, sampling_rate=1.0)
    req, resp = _predict_pair()
    for _ in range(300):
        logger.log_predict(req, resp)
    # 256-record auto-flush happened without an explicit flush()
    records = read_tfrecord(str(tmp_path / "auto.m.log"))
    assert len(records) >= 256
    collector.flush()
    assert len(read_tfrecord(str(tmp_path / "auto.m.log"))) == 300
