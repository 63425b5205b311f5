"""Integration tests against the in-process loopback server — the mirror of
reference requests_test.py:12-50 (Predict echo of string/float/int64 +
GetModelStatus AVAILABLE), self-contained: no external
tensorflow_model_server needed (the loopback servicer is the component the
reference conspicuously lacks)."""
import json

import grpc
import numpy as np
import pytest
from google.protobuf.json_format import MessageToJson

from min_tfs_client_amd.client import TensorServingClient
from min_tfs_client_amd.server import ModelServer, Servable, identity_servable
from min_tfs_client_amd.tensors import tensor_proto_to_ndarray


@pytest.fixture(scope="module")
def server():
    with ModelServer(port=0) as srv:
        srv.manager.load("default", identity_servable(), version=1)
        yield srv


@pytest.fixture(scope="module")
def client(server):
    c = TensorServingClient("127.0.0.1", server.port)
    yield c
    c.close()


def test_predict_echo_all_three_dtypes(client):
    """Mirror of reference requests_test.py:17-36 (identity model echoes
    string/float/int64 inputs)."""
    inputs = {
        "string_input": np.array(["hello"]),
        "float_input": np.array([1.25], dtype=np.float32),
        "int_input": np.array([10], dtype=np.int64),
    }
    response = client.predict_request("default", inputs)
    assert tensor_proto_to_ndarray(
        response.outputs["string_output"])[0] == b"hello"
    np.testing.assert_array_equal(
        tensor_proto_to_ndarray(response.outputs["float_output"]),
        np.array([1.25], dtype=np.float32))
    np.testing.assert_array_equal(
        tensor_proto_to_ndarray(response.outputs["int_output"]),
        np.array([10], dtype=np.int64))


def test_predict_typed_field_encoding(client):
    x = np.array([3.0, 4.0], dtype=np.float32)
    response = client.predict_request("default", {"x": x},
                                      use_tensor_content=False)
    np.testing.assert_array_equal(
        tensor_proto_to_ndarray(response.outputs["x"]), x)


def test_predict_large_image_batch(client):
    """BASELINE config-2 shape on CPU: 32x3x224x224 fp32."""
    x = np.random.default_rng(1).random((4, 3, 16, 16)).astype(np.float32)
    response = client.predict_request("default", {"images": x})
    np.testing.assert_array_equal(
        tensor_proto_to_ndarray(response.outputs["images"]), x)


def test_model_status_available(client):
    """Mirror of reference requests_test.py:39-50 (state == AVAILABLE via
    JSON comparison)."""
    response = client.model_status_request("default")
    as_json = json.loads(MessageToJson(response))
    assert as_json == {
        "model_version_status": [
            {"version": "1", "state": "AVAILABLE", "status": {}}
        ]
    }


def test_model_status_not_found(client):
    with pytest.raises(grpc.RpcError) as err:
        client.model_status_request("nonexistent")
    assert err.value.code() == grpc.StatusCode.NOT_FOUND


def test_predict_model_not_found(client):
    with pytest.raises(grpc.RpcError) as err:
        client.predict_request("nonexistent", {"x": np.zeros(1,
                                                             np.float32)})
    assert err.value.code() == grpc.StatusCode.NOT_FOUND


def test_predict_specific_version(client):
    response = client.predict_request(
        "default", {"x": np.ones(2, np.float32)}, model_version=1)
    assert "x" in response.outputs


def test_predict_wrong_version_not_found(client):
    with pytest.raises(grpc.RpcError) as err:
        client.predict_request("default", {"x": np.ones(2, np.float32)},
                               model_version=99)
    assert err.value.code() == grpc.StatusCode.NOT_FOUND


def test_output_filter(client):
    response = client.predict_request(
        "default",
        {"a": np.ones(1, np.float32), "b": np.ones(1, np.float32)},
        output_filter=["a"])
    assert set(response.outputs) == {"a"}


def test_get_model_metadata(client):
    response = client.get_model_metadata_request("default")
    assert response.metadata["signature_def"].type_url.endswith(
        "tensorflow.serving.SignatureDefMap")


def test_reload_config_unloads_models(server):
    server.manager.load("temp_model", identity_servable(), version=1)
    c = TensorServingClient("127.0.0.1", server.port)
    try:
        resp = c.reload_config_request({"default": "/models/default"})
        assert resp.status.error_code == 0
        with pytest.raises(grpc.RpcError):
            c.predict_request("temp_model", {"x": np.zeros(1, np.float32)})
        # default still up
        c.predict_request("default", {"x": np.zeros(1, np.float32)})
    finally:
        c.close()


def test_version_lifecycle(server):
    server.manager.load("lifecycle", identity_servable(), version=1)
    server.manager.load("lifecycle", identity_servable(), version=2)
    c = TensorServingClient("127.0.0.1", server.port)
    try:
        st = c.model_status_request("lifecycle")
        states = {s.version: s.state for s in st.model_version_status}
        assert states == {1: 30, 2: 30}
        server.manager.unload("lifecycle", version=1)
        st = c.model_status_request("lifecycle")
        states = {s.version: s.state for s in st.model_version_status}
        assert states == {1: 50, 2: 30}  # END, AVAILABLE
        # latest-version resolution now picks v2
        r = c.predict_request("lifecycle", {"x": np.zeros(1, np.float32)})
        assert "x" in r.outputs
    finally:
        c.close()


def test_torch_cpu_tensor_input(client):
    import torch
    t = torch.arange(6, dtype=torch.float32).reshape(2, 3)
    response = client.predict_request("default", {"x": t})
    np.testing.assert_array_equal(
        tensor_proto_to_ndarray(response.outputs["x"]), t.numpy())


def test_bf16_tensor_input(client):
    import torch
    t = torch.tensor([1.5, -2.0], dtype=torch.bfloat16)
    response = client.predict_request("default", {"x": t})
    out_proto = response.outputs["x"]
    assert out_proto.dtype == 14
    from min_tfs_client_amd.tensors import tensor_proto_to_tensor
    assert torch.equal(tensor_proto_to_tensor(out_proto), t)


def test_enable_retries_option(server):
    """Retry-enabled channel still works (transparent retries on
    UNAVAILABLE; opt-in — SURVEY §5 notes the reference has none)."""
    c = TensorServingClient("127.0.0.1", server.port, enable_retries=True)
    try:
        x = np.ones(3, np.float32)
        resp = c.predict_request("default", {"x": x})
        np.testing.assert_array_equal(
            tensor_proto_to_ndarray(resp.outputs["x"]), x)
    finally:
        c.close()


def test_version_label_routing(server):
    """ModelSpec.version_label resolution (model.proto oneof
    version_choice; ModelConfig.version_labels semantics)."""
    server.manager.load("labeled", identity_servable(), version=1)
    server.manager.load("labeled", identity_servable(), version=2)
    server.manager.set_version_label("labeled", "stable", 1)
    server.manager.set_version_label("labeled", "canary", 2)
    c = TensorServingClient("127.0.0.1", server.port)
    try:
        x = np.ones(2, np.float32)
        r = c.predict_request("labeled", {"x": x}, version_label="stable")
        assert "x" in r.outputs
        with pytest.raises(grpc.RpcError) as err:
            c.predict_request("labeled", {"x": x}, version_label="nope")
        assert err.value.code() == grpc.StatusCode.NOT_FOUND
        # labels may only target AVAILABLE versions
        with pytest.raises(KeyError):
            server.manager.set_version_label("labeled", "bad", 99)
    finally:
        c.close()


def test_state_event_bus():
    """ServableStateMonitor analogue: subscribers see every transition,
    wait_for_state blocks until a target state."""
    from min_tfs_client_amd.server import ModelManager
    mgr = ModelManager()
    events = []
    mgr.subscribe(lambda n, v, s: events.append((n, v, s)))
    mgr.load("m", identity_servable(), version=1)
    assert (("m", 1, 20) in events) and (("m", 1, 30) in events)
    assert mgr.wait_for_state("m", 1, 30, timeout=1)
    mgr.unload("m", 1)
    assert events[-1] == ("m", 1, 50)
    # a broken subscriber must not break serving
    mgr.subscribe(lambda *a: 1 / 0)
    mgr.load("m2", identity_servable(), version=1)
    assert mgr.get("m2") is not None


def test_typed_output_encoding_mode():
    """output_encoding='typed' reproduces TF-Serving's default
    AsProtoField responses (predict_util.cc:222-226) — the representation
    the reference client's decoder requires."""
    with ModelServer(port=0, output_encoding="typed") as srv:
        srv.manager.load("m", identity_servable(), version=1)
        c = TensorServingClient("127.0.0.1", srv.port)
        try:
            x = np.array([1.5, 2.5], dtype=np.float32)
            resp = c.predict_request("m", {"x": x})
            proto = resp.outputs["x"]
            assert len(proto.tensor_content) == 0
            assert list(proto.float_val) == [1.5, 2.5]
            np.testing.assert_array_equal(tensor_proto_to_ndarray(proto), x)
        finally:
            c.close()


def test_metadata_signature_unpack():
    """GetModelMetadata's Any payload unpacks into a SignatureDefMap with
    the servable's declared inputs/outputs."""
    from min_tfs_client_amd.models import resnet50_servable
    from min_tfs_client_amd.wire import messages as pb
    with ModelServer(port=0) as srv:
        srv.manager.load("resnet50", resnet50_servable(), version=1)
        c = TensorServingClient("127.0.0.1", srv.port)
        try:
            resp = c.get_model_metadata_request("resnet50")
            sdm = pb.SignatureDefMap()
            assert resp.metadata["signature_def"].type_url.endswith(
                "tensorflow.serving.SignatureDefMap")
            sdm.MergeFromString(resp.metadata["signature_def"].value)
            sig = sdm.signature_def["serving_default"]
            assert sig.method_name == "tensorflow/serving/predict"
            assert sig.inputs["images"].dtype == 1
            assert [d.size for d in
                    sig.inputs["images"].tensor_shape.dim] == \
                [-1, 3, 224, 224]
            assert sig.outputs["logits"].name == "logits:0"
        finally:
            c.close()


def test_decode_predict_response_helper(client):
    import torch
    from min_tfs_client_amd.client import decode_predict_response
    x = np.arange(4, dtype=np.float32)
    resp = client.predict_request("default", {"x": x})
    outs = decode_predict_response(resp)
    np.testing.assert_array_equal(outs["x"], x)
    touts = decode_predict_response(resp, as_numpy=False)
    assert torch.equal(touts["x"], torch.arange(4, dtype=torch.float32))


def test_client_backend_selection(server):
    """TensorServingClient rides the C++ transport by default and falls
    back to grpcio whenever grpcio-specific features are requested."""
    host, port = "127.0.0.1", server.port
    with TensorServingClient(host=host, port=port) as c:
        assert c.backend == "native"
        r = c.predict_request("default", {
            "float_input": np.float32(np.random.rand(2, 3))})
        assert "float_output" in r.outputs
    with TensorServingClient(host=host, port=port,
                             backend="grpcio") as c:
        assert c.backend == "grpcio"
        r = c.predict_request("default", {
            "float_input": np.float32(np.random.rand(2, 3))})
        assert "float_output" in r.outputs
    with TensorServingClient(host=host, port=port,
                             enable_retries=True) as c:
        assert c.backend == "grpcio"  # retry policy is grpcio-specific
    import pytest as _pytest
    with _pytest.raises(ValueError):
        TensorServingClient(host=host, port=port, backend="native",
                            enable_retries=True)
