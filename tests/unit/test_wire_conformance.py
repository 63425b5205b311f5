"""Byte-level wire conformance: golden serialized bytes for the messages the
protocol exercises, pinned against hand-derived protobuf encodings. This is
the *byte*-compatibility promise (SURVEY §4 blueprint item 1) — the
reference's tests pin only text-format semantics (tensors_test.py:14-22).

Wire facts cited from /root/reference/protobuf_srcs/... proto files.
"""
import numpy as np

from min_tfs_client_amd.tensors import ndarray_to_tensor_proto
from min_tfs_client_amd.wire import messages as pb
from min_tfs_client_amd.wire import schema


def test_tensorproto_field_numbers():
    """tensor.proto:14-94 field layout."""
    fields = {f.name: f.number
              for f in pb.TensorProto.DESCRIPTOR.fields}
    assert fields == {
        "dtype": 1, "tensor_shape": 2, "version_number": 3,
        "tensor_content": 4, "half_val": 13, "float_val": 5,
        "double_val": 6, "int_val": 7, "string_val": 8, "scomplex_val": 9,
        "int64_val": 10, "bool_val": 11, "dcomplex_val": 12,
        "resource_handle_val": 14, "variant_val": 15, "uint32_val": 16,
        "uint64_val": 17,
    }


def test_datatype_enum_values():
    """types.proto:12-68 incl. _REF offsets."""
    assert pb.DataType.DT_FLOAT == 1
    assert pb.DataType.DT_BFLOAT16 == 14
    assert pb.DataType.DT_HALF == 19
    assert pb.DataType.DT_UINT64 == 23
    assert pb.DataType.DT_FLOAT_REF == 101
    assert pb.DataType.DT_UINT64_REF == 123
    assert pb.DataType.Name(7) == "DT_STRING"


def test_golden_bytes_float_tensor_content():
    """field 4 (tensor_content): tag 0x22, then length-delimited LE bytes."""
    proto = ndarray_to_tensor_proto(np.array([1.0], dtype=np.float32))
    ser = proto.SerializeToString()
    # dtype=1: 08 01 | tensor_shape {dim {size:1}}: 12 04 12 02 08 01
    # tensor_content(4 bytes of 1.0f LE): 22 04 00 00 80 3f
    assert ser == bytes.fromhex("08011204120208012204" + "0000803f")


def test_golden_bytes_packed_half_val():
    """half_val is field 13, packed varints of raw fp16 bits
    (tensor.proto:51 [packed = true])."""
    proto = ndarray_to_tensor_proto(np.array([1.0], dtype=np.float16),
                                    use_tensor_content=False)
    ser = proto.SerializeToString()
    # dtype DT_HALF=19: 08 13 | shape: 12 04 12 02 08 01
    # half_val packed: tag 13<<3|2 = 0x6a, len 2, varint 0x3c00 -> 80 78
    assert ser == bytes.fromhex("08131204120208016a028078")


def test_golden_bytes_packed_bool_and_int64():
    proto = ndarray_to_tensor_proto(np.array([True, False]),
                                    use_tensor_content=False)
    ser = proto.SerializeToString()
    # dtype DT_BOOL=10: 08 0a | shape dim size 2 | bool_val field 11 packed:
    # tag 0x5a len 2: 01 00
    assert ser == bytes.fromhex("080a1204120208025a020100")

    proto = ndarray_to_tensor_proto(np.array([300], dtype=np.int64),
                                    use_tensor_content=False)
    # dtype DT_INT64=9 | int64_val field 10 packed: tag 0x52 len 2: ac 02
    assert proto.SerializeToString() == bytes.fromhex(
        "08091204120208015202ac02")


def test_golden_bytes_model_spec_int64value():
    """ModelSpec.version is google.protobuf.Int64Value at field 2
    (model.proto:9-33); the client sets .version.value
    (reference requests.py:45)."""
    spec = pb.ModelSpec()
    spec.name = "m"
    spec.version.value = 5
    # name: 0a 01 6d | version submessage: 12 02 08 05
    assert spec.SerializeToString() == bytes.fromhex("0a016d12020805")


def test_golden_bytes_predict_request():
    req = pb.PredictRequest()
    req.model_spec.name = "m"
    req.inputs["x"].CopyFrom(
        ndarray_to_tensor_proto(np.array([2.0], dtype=np.float32)))
    ser = req.SerializeToString()
    # model_spec: 0a 03 0a 01 6d
    # inputs map entry (field 2, 19 bytes): key "x" (0a 01 78) + value
    # (12 0e + 14-byte TensorProto: dtype/shape/tensor_content of 2.0f)
    expected = bytes.fromhex(
        "0a030a016d" "1213" "0a0178" "120e"
        "0801" "120412020801" "2204" "00000040")
    assert ser == expected


def test_golden_bytes_get_model_status():
    req = pb.GetModelStatusRequest()
    req.model_spec.name = "default"
    assert req.SerializeToString() == bytes.fromhex(
        "0a090a0764656661756c74")
    resp = pb.GetModelStatusResponse()
    s = resp.model_version_status.add()
    s.version = 1
    s.state = 30  # AVAILABLE (get_model_status.proto:27-45)
    assert resp.SerializeToString() == bytes.fromhex("0a040801101e")


def test_predict_response_parse_from_foreign_bytes():
    """Parse bytes as a TF-Serving server would emit them (map field, typed
    float_val) — cross-checks map entry numbering (predict.proto:35-40:
    outputs=1, model_spec=2)."""
    blob = bytes.fromhex(
        # outputs map entry (field 1, 19 bytes): key "y" + 14-byte
        # TensorProto with packed float_val [1.5]
        "0a13" "0a0179" "120e" "0801" "120412020801" "2a040000c03f"
        # model_spec (field 2): name "m"
        "12030a016d")
    resp = pb.PredictResponse.FromString(blob)
    assert resp.model_spec.name == "m"
    assert list(resp.outputs["y"].float_val) == [1.5]


def test_unknown_fields_preserved():
    """Fields we do not model (e.g. TensorInfo.composite_tensor) must
    round-trip as unknown fields, not be dropped."""
    # field 99, wiretype 0 (varint), value 7: tag = 99<<3 = 792 -> d8 06
    unknown = bytes([0xd8, 0x06, 0x07])
    info = pb.TensorInfo.FromString(bytes.fromhex("0a0178") + unknown)
    assert info.name == "x"
    assert unknown in info.SerializeToString()


def test_service_method_paths():
    """gRPC method paths must match prediction_service_pb2_grpc.py:51 and
    model_service_pb2_grpc.py."""
    svc = schema._pool.FindServiceByName(
        "tensorflow.serving.PredictionService")
    assert [m.name for m in svc.methods] == [
        "Classify", "Regress", "Predict", "MultiInference",
        "GetModelMetadata"]
    svc = schema._pool.FindServiceByName("tensorflow.serving.ModelService")
    assert [m.name for m in svc.methods] == [
        "GetModelStatus", "HandleReloadConfigRequest"]


def test_error_code_enum():
    assert pb.ErrorCode.OK == 0
    assert pb.ErrorCode.UNAUTHENTICATED == 16
    assert pb.ErrorCode.DATA_LOSS == 15


def test_example_feature_wire():
    ex = pb.Example()
    ex.features.feature["age"].int64_list.value.append(42)
    blob = ex.SerializeToString()
    back = pb.Example.FromString(blob)
    assert list(back.features.feature["age"].int64_list.value) == [42]


def test_versioned_model_server_config():
    cfg = pb.ModelServerConfig()
    mc = cfg.model_config_list.config.add()
    mc.name = "m"
    mc.base_path = "/models/m"
    mc.model_version_policy.specific.versions.extend([1, 3])
    blob = cfg.SerializeToString()
    back = pb.ModelServerConfig.FromString(blob)
    assert back.model_config_list.config[0].model_version_policy.WhichOneof(
        "policy_choice") == "specific"
    # Specific is field 102 inside ServableVersionPolicy
    assert list(back.model_config_list.config[0]
                .model_version_policy.specific.versions) == [1, 3]


def test_schema_builder_helpers():
    """wire/schema.py builder primitives produce valid descriptors."""
    from min_tfs_client_amd.wire.schema import (
        enum, field, map_field, message, proto_file, service)
    e = enum("E", [("A", 0), ("B", 5)])
    assert [v.number for v in e.value] == [0, 5]
    entry, f = map_field("m", 3, "string", "int64", ".pkg.Msg")
    assert entry.options.map_entry is True
    assert f.type_name == ".pkg.Msg.MEntry"
    m = message("Msg", fields=[field("x", 1, "int32")], nested=[entry],
                oneofs=["choice"])
    assert m.oneof_decl[0].name == "choice"
    svc = service("S", [("Do", ".pkg.Req", ".pkg.Resp")])
    assert svc.method[0].input_type == ".pkg.Req"
    fdp = proto_file("pkg/test_probe_unused.proto", "pkg", messages=[m],
                     enums=[e], services=[svc])
    assert fdp.syntax == "proto3"
    assert fdp.message_type[0].name == "Msg"


def test_all_wire_classes_instantiable():
    """Every exported message class constructs and serializes empty."""
    from min_tfs_client_amd.wire import messages as m
    count = 0
    for name in dir(m):
        cls = getattr(m, name)
        if isinstance(cls, type) and hasattr(cls, "SerializeToString") \
                and hasattr(cls, "DESCRIPTOR"):
            cls().SerializeToString()
            count += 1
    assert count >= 40
