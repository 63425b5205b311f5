"""C++ wire codec cross-validation against the python protobuf layer.

Every byte the C++ serializer emits must parse identically in python
protobuf (and vice versa) — this pins the hand-rolled writer/parser
(ops/csrc/wire.h) to the generated-code wire format.
"""
import numpy as np
import pytest
import torch

from min_tfs_client_amd.wire import messages as pb

native = pytest.importorskip(
    "min_tfs_client_amd._native",
    reason="_native extension not built (python setup.py build_ext "
           "--inplace)")


def test_serialize_request_parses_in_python_protobuf():
    t1 = torch.arange(24, dtype=torch.float32).reshape(2, 3, 4)
    t2 = torch.tensor([1.5, -2.0], dtype=torch.bfloat16)
    blob = native.serialize_predict_request(
        "mymodel", 3, "serving_default", ["x", "y"], [t1, t2], 0)
    req = pb.PredictRequest.FromString(blob)
    assert req.model_spec.name == "mymodel"
    assert req.model_spec.version.value == 3
    assert req.model_spec.signature_name == "serving_default"
    assert sorted(req.inputs) == ["x", "y"]
    assert req.inputs["x"].dtype == 1
    assert [d.size for d in req.inputs["x"].tensor_shape.dim] == [2, 3, 4]
    assert req.inputs["x"].tensor_content == t1.numpy().tobytes()
    assert req.inputs["y"].dtype == 14  # DT_BFLOAT16


def test_serialize_no_version():
    blob = native.serialize_predict_request(
        "m", -1, "", ["x"], [torch.zeros(1)], 0)
    req = pb.PredictRequest.FromString(blob)
    assert not req.model_spec.HasField("version")


def test_serialize_matches_python_protobuf_bytes():
    """For a single-input request the two serializers must emit identical
    bytes (deterministic field order; map with one entry)."""
    t = torch.tensor([1.0, 2.0], dtype=torch.float32)
    blob = native.serialize_predict_request("m", 1, "", ["x"], [t], 0)
    ref = pb.PredictRequest()
    ref.model_spec.name = "m"
    ref.model_spec.version.value = 1
    ref.inputs["x"].dtype = 1
    ref.inputs["x"].tensor_shape.dim.add().size = 2
    ref.inputs["x"].tensor_content = t.numpy().tobytes()
    assert blob == ref.SerializeToString()


def test_parse_python_built_request():
    req = pb.PredictRequest()
    req.model_spec.name = "m2"
    req.model_spec.version.value = 7
    req.inputs["a"].dtype = 1
    req.inputs["a"].tensor_shape.dim.add().size = 3
    req.inputs["a"].tensor_content = np.array(
        [1, 2, 3], dtype=np.float32).tobytes()
    req.output_filter.append("a")
    spec, outs, filt = native.parse_predict_request(
        req.SerializeToString(), "cpu", 0)
    assert spec["name"] == "m2" and spec["version"] == 7
    assert outs["a"].tolist() == [1.0, 2.0, 3.0]
    assert filt == ["a"]


@pytest.mark.parametrize("dtype,tfenum", [
    (torch.float32, 1), (torch.float64, 2), (torch.int32, 3),
    (torch.uint8, 4), (torch.int16, 5), (torch.int8, 6),
    (torch.int64, 9), (torch.bool, 10), (torch.bfloat16, 14),
    (torch.float16, 19),
])
def test_roundtrip_all_dtypes(dtype, tfenum):
    if dtype.is_floating_point:
        t = torch.rand(3, 5).to(dtype)
    elif dtype == torch.bool:
        t = torch.rand(3, 5) > 0.5
    else:
        t = torch.randint(0, 100, (3, 5)).to(dtype)
    blob = native.serialize_predict_request("m", -1, "", ["t"], [t], 0)
    req = pb.PredictRequest.FromString(blob)
    assert req.inputs["t"].dtype == tfenum
    _, outs, _ = native.parse_predict_request(blob, "cpu", 0)
    assert outs["t"].dtype == dtype
    assert torch.equal(outs["t"], t)


def test_parse_typed_field_fallback_with_fill():
    resp = pb.PredictResponse()
    resp.outputs["z"].dtype = 3
    resp.outputs["z"].tensor_shape.dim.add().size = 5
    resp.outputs["z"].int_val.extend([7, 9])
    _, outs, _ = native.parse_predict_response(
        resp.SerializeToString(), "cpu", 0)
    assert outs["z"].tolist() == [7, 9, 9, 9, 9]


def test_parse_typed_half_bit_pattern():
    resp = pb.PredictResponse()
    resp.outputs["h"].dtype = 19  # DT_HALF
    resp.outputs["h"].tensor_shape.dim.add().size = 2
    resp.outputs["h"].half_val.extend([0x3C00, 0xB800])  # 1.0, -0.5
    _, outs, _ = native.parse_predict_response(
        resp.SerializeToString(), "cpu", 0)
    assert outs["h"].dtype == torch.float16
    assert outs["h"].tolist() == [1.0, -0.5]


def test_parse_string_outputs():
    resp = pb.PredictResponse()
    resp.outputs["s"].dtype = 7
    resp.outputs["s"].tensor_shape.dim.add().size = 2
    resp.outputs["s"].string_val.extend([b"ab", b"cd"])
    _, outs, _ = native.parse_predict_response(
        resp.SerializeToString(), "cpu", 0)
    assert outs["s"] == [b"ab", b"cd"]


def test_echo_renames_input_suffix():
    t = torch.ones(2)
    blob = native.serialize_predict_request(
        "m", 1, "", ["float_input"], [t], 0)
    resp = pb.PredictResponse.FromString(native.echo_predict(blob))
    assert list(resp.outputs) == ["float_output"]


def test_echo_large_payload_integrity():
    t = torch.randn(8, 3, 64, 64)
    blob = native.serialize_predict_request("m", -1, "", ["x"], [t], 0)
    _, outs, _ = native.parse_predict_response(
        native.echo_predict(blob), "cpu", 0)
    assert torch.equal(outs["x"], t)


def test_tensor_content_bytes_cpu():
    t = torch.arange(10, dtype=torch.int64)
    assert native.tensor_content_bytes(t, 0) == t.numpy().tobytes()


def test_truncated_message_raises():
    t = torch.ones(4)
    blob = native.serialize_predict_request("m", -1, "", ["x"], [t], 0)
    with pytest.raises(Exception):
        native.parse_predict_request(blob[: len(blob) // 2], "cpu", 0)


def test_scalar_tensor():
    t = torch.tensor(3.5)
    blob = native.serialize_predict_request("m", -1, "", ["s"], [t], 0)
    req = pb.PredictRequest.FromString(blob)
    assert len(req.inputs["s"].tensor_shape.dim) == 0
    _, outs, _ = native.parse_predict_request(blob, "cpu", 0)
    assert outs["s"].shape == () and outs["s"].item() == 3.5
