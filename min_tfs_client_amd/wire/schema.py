"""Wire-format schema for the TF-Serving Predict protocol, built at import time.

This module declares — programmatically, via ``FileDescriptorProto`` — the
minimal transitive proto closure that the TF-Serving gRPC surface actually
exercises, with package names, message names, field names/numbers and enum
values byte-identical to upstream TensorFlow / TF-Serving:

* ``tensorflow/core/framework/{types,tensor_shape,resource_handle,tensor}.proto``
  (reference: /root/reference/protobuf_srcs/tensorflow/core/framework/tensor.proto:14-94,
  tensor_shape.proto:13-46, types.proto:12-68, resource_handle.proto:16-42)
* ``tensorflow/core/example/{feature,example}.proto``
* ``tensorflow/core/protobuf/error_codes.proto`` (tensorflow.error.Code)
* the TensorInfo / SignatureDef subset of ``tensorflow/core/protobuf/meta_graph.proto``
  (the ``composite_tensor`` encoding variant is intentionally omitted — its
  TypeSpecProto closure is enormous and unknown fields round-trip anyway)
* ``tensorflow_serving/apis/{model,predict,classification,regression,input,
  inference,get_model_metadata,get_model_status,model_service,
  prediction_service,model_management}.proto``
  (reference: /root/reference/protobuf_srcs/tensorflow_serving/apis/)
* ``tensorflow_serving/util/status.proto``,
  ``tensorflow_serving/config/{log_collector_config,logging_config,
  model_server_config}.proto``,
  ``tensorflow_serving/sources/storage_path/file_system_storage_path_source.proto``

Design note (MI355X-first): the reference build runs ``protoc`` over 149
vendored ``.proto`` files at wheel-build time (reference setup.py:15,41-49).
This container has no protoc and no grpcio-tools, and the schema here is a
stable, tiny closure — so the MI355X build declares the descriptors directly
in Python and registers them in the *default* descriptor pool. Wire bytes are
produced by the same protobuf runtime either way, so byte-compatibility holds;
golden-bytes tests in tests/unit/test_wire_conformance.py pin it.
"""
from __future__ import annotations

from google.protobuf import descriptor_pb2, descriptor_pool, message_factory

# Importing the well-known-type modules registers them in the default pool so
# our files can depend on them.
import google.protobuf.any_pb2  # noqa: F401
import google.protobuf.wrappers_pb2  # noqa: F401

_F = descriptor_pb2.FieldDescriptorProto

# ---------------------------------------------------------------------------
# Tiny declarative builder
# ---------------------------------------------------------------------------

_TYPE = {
    "double": _F.TYPE_DOUBLE,
    "float": _F.TYPE_FLOAT,
    "int64": _F.TYPE_INT64,
    "uint64": _F.TYPE_UINT64,
    "int32": _F.TYPE_INT32,
    "uint32": _F.TYPE_UINT32,
    "bool": _F.TYPE_BOOL,
    "string": _F.TYPE_STRING,
    "bytes": _F.TYPE_BYTES,
}


def field(name, number, ftype, repeated=False, oneof=None, json_name=None):
    """One proto field. ftype is a scalar name, '.pkg.Message' or '.pkg.Enum'."""
    f = _F()
    f.name = name
    f.number = number
    f.label = _F.LABEL_REPEATED if repeated else _F.LABEL_OPTIONAL
    if ftype in _TYPE:
        f.type = _TYPE[ftype]
    else:
        # Message vs enum is resolved by the pool from the type name; we must
        # still pick one of TYPE_MESSAGE/TYPE_ENUM. Callers mark enums with a
        # leading 'enum:'.
        if ftype.startswith("enum:"):
            f.type = _F.TYPE_ENUM
            f.type_name = ftype[len("enum:"):]
        else:
            f.type = _F.TYPE_MESSAGE
            f.type_name = ftype
    if oneof is not None:
        f.oneof_index = oneof
    if json_name is not None:
        f.json_name = json_name
    return f


def message(name, fields=(), nested=(), oneofs=(), enums=(), map_entry=False):
    m = descriptor_pb2.DescriptorProto()
    m.name = name
    for f in fields:
        m.field.add().CopyFrom(f)
    for n in nested:
        m.nested_type.add().CopyFrom(n)
    for o in oneofs:
        m.oneof_decl.add().name = o
    for e in enums:
        m.enum_type.add().CopyFrom(e)
    if map_entry:
        m.options.map_entry = True
    return m


def map_field(name, number, key_type, value_type, parent_fqn):
    """Returns (entry_message, field) implementing map<key_type, value_type>."""
    entry_name = "".join(p.capitalize() for p in name.split("_")) + "Entry"
    entry = message(
        entry_name,
        fields=[field("key", 1, key_type), field("value", 2, value_type)],
        map_entry=True,
    )
    f = field(name, number, f"{parent_fqn}.{entry_name}", repeated=True)
    return entry, f


def enum(name, values):
    e = descriptor_pb2.EnumDescriptorProto()
    e.name = name
    for vname, vnum in values:
        v = e.value.add()
        v.name = vname
        v.number = vnum
    return e


def proto_file(name, package, deps=(), messages=(), enums=(), services=()):
    fdp = descriptor_pb2.FileDescriptorProto()
    fdp.name = name
    fdp.package = package
    fdp.syntax = "proto3"
    for d in deps:
        fdp.dependency.append(d)
    for m in messages:
        fdp.message_type.add().CopyFrom(m)
    for e in enums:
        fdp.enum_type.add().CopyFrom(e)
    for s in services:
        fdp.service.add().CopyFrom(s)
    return fdp


def service(name, methods):
    s = descriptor_pb2.ServiceDescriptorProto()
    s.name = name
    for mname, req, resp in methods:
        m = s.method.add()
        m.name = mname
        m.input_type = req
        m.output_type = resp
    return s


# ---------------------------------------------------------------------------
# tensorflow/core — framework protos
# ---------------------------------------------------------------------------

_DT_BASE = [
    ("DT_INVALID", 0), ("DT_FLOAT", 1), ("DT_DOUBLE", 2), ("DT_INT32", 3),
    ("DT_UINT8", 4), ("DT_INT16", 5), ("DT_INT8", 6), ("DT_STRING", 7),
    ("DT_COMPLEX64", 8), ("DT_INT64", 9), ("DT_BOOL", 10), ("DT_QINT8", 11),
    ("DT_QUINT8", 12), ("DT_QINT32", 13), ("DT_BFLOAT16", 14),
    ("DT_QINT16", 15), ("DT_QUINT16", 16), ("DT_UINT16", 17),
    ("DT_COMPLEX128", 18), ("DT_HALF", 19), ("DT_RESOURCE", 20),
    ("DT_VARIANT", 21), ("DT_UINT32", 22), ("DT_UINT64", 23),
]
_DT_ALL = _DT_BASE + [(n + "_REF", v + 100) for n, v in _DT_BASE[1:]]

FILE_TYPES = proto_file(
    "tensorflow/core/framework/types.proto", "tensorflow",
    enums=[enum("DataType", _DT_ALL)],
)

FILE_TENSOR_SHAPE = proto_file(
    "tensorflow/core/framework/tensor_shape.proto", "tensorflow",
    messages=[
        message(
            "TensorShapeProto",
            fields=[
                field("dim", 2, ".tensorflow.TensorShapeProto.Dim", repeated=True),
                field("unknown_rank", 3, "bool"),
            ],
            nested=[
                message("Dim", fields=[field("size", 1, "int64"),
                                       field("name", 2, "string")]),
            ],
        ),
    ],
)

FILE_RESOURCE_HANDLE = proto_file(
    "tensorflow/core/framework/resource_handle.proto", "tensorflow",
    deps=["tensorflow/core/framework/tensor_shape.proto",
          "tensorflow/core/framework/types.proto"],
    messages=[
        message(
            "ResourceHandleProto",
            fields=[
                field("device", 1, "string"),
                field("container", 2, "string"),
                field("name", 3, "string"),
                field("hash_code", 4, "uint64"),
                field("maybe_type_name", 5, "string"),
                field("dtypes_and_shapes", 6,
                      ".tensorflow.ResourceHandleProto.DtypeAndShape",
                      repeated=True),
            ],
            nested=[
                message("DtypeAndShape", fields=[
                    field("dtype", 1, "enum:.tensorflow.DataType"),
                    field("shape", 2, ".tensorflow.TensorShapeProto"),
                ]),
            ],
        ),
    ],
)

FILE_TENSOR = proto_file(
    "tensorflow/core/framework/tensor.proto", "tensorflow",
    deps=["tensorflow/core/framework/resource_handle.proto",
          "tensorflow/core/framework/tensor_shape.proto",
          "tensorflow/core/framework/types.proto"],
    messages=[
        message(
            "TensorProto",
            fields=[
                field("dtype", 1, "enum:.tensorflow.DataType"),
                field("tensor_shape", 2, ".tensorflow.TensorShapeProto"),
                field("version_number", 3, "int32"),
                field("tensor_content", 4, "bytes"),
                field("half_val", 13, "int32", repeated=True),
                field("float_val", 5, "float", repeated=True),
                field("double_val", 6, "double", repeated=True),
                field("int_val", 7, "int32", repeated=True),
                field("string_val", 8, "bytes", repeated=True),
                field("scomplex_val", 9, "float", repeated=True),
                field("int64_val", 10, "int64", repeated=True),
                field("bool_val", 11, "bool", repeated=True),
                field("dcomplex_val", 12, "double", repeated=True),
                field("resource_handle_val", 14,
                      ".tensorflow.ResourceHandleProto", repeated=True),
                field("variant_val", 15, ".tensorflow.VariantTensorDataProto",
                      repeated=True),
                field("uint32_val", 16, "uint32", repeated=True),
                field("uint64_val", 17, "uint64", repeated=True),
            ],
        ),
        message(
            "VariantTensorDataProto",
            fields=[
                field("type_name", 1, "string"),
                field("metadata", 2, "bytes"),
                field("tensors", 3, ".tensorflow.TensorProto", repeated=True),
            ],
        ),
    ],
)

# --- example / feature -----------------------------------------------------

_features_entry, _features_field = map_field(
    "feature", 1, "string", ".tensorflow.Feature", ".tensorflow.Features")
_fl_entry, _fl_field = map_field(
    "feature_list", 1, "string", ".tensorflow.FeatureList",
    ".tensorflow.FeatureLists")

FILE_FEATURE = proto_file(
    "tensorflow/core/example/feature.proto", "tensorflow",
    messages=[
        message("BytesList", fields=[field("value", 1, "bytes", repeated=True)]),
        message("FloatList", fields=[field("value", 1, "float", repeated=True)]),
        message("Int64List", fields=[field("value", 1, "int64", repeated=True)]),
        message("Feature",
                fields=[field("bytes_list", 1, ".tensorflow.BytesList", oneof=0),
                        field("float_list", 2, ".tensorflow.FloatList", oneof=0),
                        field("int64_list", 3, ".tensorflow.Int64List", oneof=0)],
                oneofs=["kind"]),
        message("Features", fields=[_features_field], nested=[_features_entry]),
        message("FeatureList",
                fields=[field("feature", 1, ".tensorflow.Feature", repeated=True)]),
        message("FeatureLists", fields=[_fl_field], nested=[_fl_entry]),
    ],
)

FILE_EXAMPLE = proto_file(
    "tensorflow/core/example/example.proto", "tensorflow",
    deps=["tensorflow/core/example/feature.proto"],
    messages=[
        message("Example", fields=[field("features", 1, ".tensorflow.Features")]),
        message("SequenceExample",
                fields=[field("context", 1, ".tensorflow.Features"),
                        field("feature_lists", 2, ".tensorflow.FeatureLists")]),
    ],
)

FILE_ERROR_CODES = proto_file(
    "tensorflow/core/protobuf/error_codes.proto", "tensorflow.error",
    enums=[enum("Code", [
        ("OK", 0), ("CANCELLED", 1), ("UNKNOWN", 2), ("INVALID_ARGUMENT", 3),
        ("DEADLINE_EXCEEDED", 4), ("NOT_FOUND", 5), ("ALREADY_EXISTS", 6),
        ("PERMISSION_DENIED", 7), ("UNAUTHENTICATED", 16),
        ("RESOURCE_EXHAUSTED", 8), ("FAILED_PRECONDITION", 9), ("ABORTED", 10),
        ("OUT_OF_RANGE", 11), ("UNIMPLEMENTED", 12), ("INTERNAL", 13),
        ("UNAVAILABLE", 14), ("DATA_LOSS", 15),
        ("DO_NOT_USE_RESERVED_FOR_FUTURE_EXPANSION_USE_DEFAULT_IN_SWITCH_INSTEAD_",
         20),
    ])],
)

# --- meta_graph subset: TensorInfo + SignatureDef --------------------------

_sig_in_entry, _sig_in_field = map_field(
    "inputs", 1, "string", ".tensorflow.TensorInfo", ".tensorflow.SignatureDef")
_sig_out_entry, _sig_out_field = map_field(
    "outputs", 2, "string", ".tensorflow.TensorInfo", ".tensorflow.SignatureDef")

FILE_META_GRAPH = proto_file(
    "tensorflow/core/protobuf/meta_graph.proto", "tensorflow",
    deps=["tensorflow/core/framework/tensor_shape.proto",
          "tensorflow/core/framework/types.proto"],
    messages=[
        message(
            "TensorInfo",
            fields=[
                field("name", 1, "string", oneof=0),
                field("coo_sparse", 4, ".tensorflow.TensorInfo.CooSparse", oneof=0),
                field("dtype", 2, "enum:.tensorflow.DataType"),
                field("tensor_shape", 3, ".tensorflow.TensorShapeProto"),
            ],
            nested=[
                message("CooSparse", fields=[
                    field("values_tensor_name", 1, "string"),
                    field("indices_tensor_name", 2, "string"),
                    field("dense_shape_tensor_name", 3, "string"),
                ]),
            ],
            oneofs=["encoding"],
        ),
        message("SignatureDef",
                fields=[_sig_in_field, _sig_out_field,
                        field("method_name", 3, "string")],
                nested=[_sig_in_entry, _sig_out_entry]),
        message("AssetFileDef",
                fields=[field("tensor_info", 1, ".tensorflow.TensorInfo"),
                        field("filename", 2, "string")]),
    ],
)

# ---------------------------------------------------------------------------
# tensorflow_serving protos
# ---------------------------------------------------------------------------

FILE_MODEL = proto_file(
    "tensorflow_serving/apis/model.proto", "tensorflow.serving",
    deps=["google/protobuf/wrappers.proto"],
    messages=[
        message(
            "ModelSpec",
            fields=[
                field("name", 1, "string"),
                field("version", 2, ".google.protobuf.Int64Value", oneof=0),
                field("version_label", 4, "string", oneof=0),
                field("signature_name", 3, "string"),
            ],
            oneofs=["version_choice"],
        ),
    ],
)

_pred_in_entry, _pred_in_field = map_field(
    "inputs", 2, "string", ".tensorflow.TensorProto",
    ".tensorflow.serving.PredictRequest")
_pred_out_entry, _pred_out_field = map_field(
    "outputs", 1, "string", ".tensorflow.TensorProto",
    ".tensorflow.serving.PredictResponse")

FILE_PREDICT = proto_file(
    "tensorflow_serving/apis/predict.proto", "tensorflow.serving",
    deps=["tensorflow/core/framework/tensor.proto",
          "tensorflow_serving/apis/model.proto"],
    messages=[
        message("PredictRequest",
                fields=[field("model_spec", 1, ".tensorflow.serving.ModelSpec"),
                        _pred_in_field,
                        field("output_filter", 3, "string", repeated=True)],
                nested=[_pred_in_entry]),
        message("PredictResponse",
                fields=[field("model_spec", 2, ".tensorflow.serving.ModelSpec"),
                        _pred_out_field],
                nested=[_pred_out_entry]),
    ],
)

FILE_INPUT = proto_file(
    "tensorflow_serving/apis/input.proto", "tensorflow.serving",
    deps=["tensorflow/core/example/example.proto"],
    messages=[
        message("ExampleList",
                fields=[field("examples", 1, ".tensorflow.Example", repeated=True)]),
        message("ExampleListWithContext",
                fields=[field("examples", 1, ".tensorflow.Example", repeated=True),
                        field("context", 2, ".tensorflow.Example")]),
        message("Input",
                fields=[field("example_list", 1,
                              ".tensorflow.serving.ExampleList", oneof=0),
                        field("example_list_with_context", 2,
                              ".tensorflow.serving.ExampleListWithContext",
                              oneof=0)],
                oneofs=["kind"]),
    ],
)

FILE_CLASSIFICATION = proto_file(
    "tensorflow_serving/apis/classification.proto", "tensorflow.serving",
    deps=["tensorflow_serving/apis/input.proto",
          "tensorflow_serving/apis/model.proto"],
    messages=[
        message("Class", fields=[field("label", 1, "string"),
                                 field("score", 2, "float")]),
        message("Classifications",
                fields=[field("classes", 1, ".tensorflow.serving.Class",
                              repeated=True)]),
        message("ClassificationResult",
                fields=[field("classifications", 1,
                              ".tensorflow.serving.Classifications",
                              repeated=True)]),
        message("ClassificationRequest",
                fields=[field("model_spec", 1, ".tensorflow.serving.ModelSpec"),
                        field("input", 2, ".tensorflow.serving.Input")]),
        message("ClassificationResponse",
                fields=[field("model_spec", 2, ".tensorflow.serving.ModelSpec"),
                        field("result", 1,
                              ".tensorflow.serving.ClassificationResult")]),
    ],
)

FILE_REGRESSION = proto_file(
    "tensorflow_serving/apis/regression.proto", "tensorflow.serving",
    deps=["tensorflow_serving/apis/input.proto",
          "tensorflow_serving/apis/model.proto"],
    messages=[
        message("Regression", fields=[field("value", 1, "float")]),
        message("RegressionResult",
                fields=[field("regressions", 1, ".tensorflow.serving.Regression",
                              repeated=True)]),
        message("RegressionRequest",
                fields=[field("model_spec", 1, ".tensorflow.serving.ModelSpec"),
                        field("input", 2, ".tensorflow.serving.Input")]),
        message("RegressionResponse",
                fields=[field("model_spec", 2, ".tensorflow.serving.ModelSpec"),
                        field("result", 1,
                              ".tensorflow.serving.RegressionResult")]),
    ],
)

FILE_INFERENCE = proto_file(
    "tensorflow_serving/apis/inference.proto", "tensorflow.serving",
    deps=["tensorflow_serving/apis/classification.proto",
          "tensorflow_serving/apis/input.proto",
          "tensorflow_serving/apis/model.proto",
          "tensorflow_serving/apis/regression.proto"],
    messages=[
        message("InferenceTask",
                fields=[field("model_spec", 1, ".tensorflow.serving.ModelSpec"),
                        field("method_name", 2, "string")]),
        message("InferenceResult",
                fields=[field("model_spec", 1, ".tensorflow.serving.ModelSpec"),
                        field("classification_result", 2,
                              ".tensorflow.serving.ClassificationResult",
                              oneof=0),
                        field("regression_result", 3,
                              ".tensorflow.serving.RegressionResult", oneof=0)],
                oneofs=["result"]),
        message("MultiInferenceRequest",
                fields=[field("tasks", 1, ".tensorflow.serving.InferenceTask",
                              repeated=True),
                        field("input", 2, ".tensorflow.serving.Input")]),
        message("MultiInferenceResponse",
                fields=[field("results", 1, ".tensorflow.serving.InferenceResult",
                              repeated=True)]),
    ],
)

_sdm_entry, _sdm_field = map_field(
    "signature_def", 1, "string", ".tensorflow.SignatureDef",
    ".tensorflow.serving.SignatureDefMap")
_meta_entry, _meta_field = map_field(
    "metadata", 2, "string", ".google.protobuf.Any",
    ".tensorflow.serving.GetModelMetadataResponse")

FILE_GET_MODEL_METADATA = proto_file(
    "tensorflow_serving/apis/get_model_metadata.proto", "tensorflow.serving",
    deps=["google/protobuf/any.proto",
          "tensorflow/core/protobuf/meta_graph.proto",
          "tensorflow_serving/apis/model.proto"],
    messages=[
        message("SignatureDefMap", fields=[_sdm_field], nested=[_sdm_entry]),
        message("GetModelMetadataRequest",
                fields=[field("model_spec", 1, ".tensorflow.serving.ModelSpec"),
                        field("metadata_field", 2, "string", repeated=True)]),
        message("GetModelMetadataResponse",
                fields=[field("model_spec", 1, ".tensorflow.serving.ModelSpec"),
                        _meta_field],
                nested=[_meta_entry]),
    ],
)

FILE_STATUS = proto_file(
    "tensorflow_serving/util/status.proto", "tensorflow.serving",
    deps=["tensorflow/core/protobuf/error_codes.proto"],
    messages=[
        message("StatusProto",
                fields=[field("error_code", 1, "enum:.tensorflow.error.Code",
                              json_name="error_code"),
                        field("error_message", 2, "string",
                              json_name="error_message")]),
    ],
)

FILE_GET_MODEL_STATUS = proto_file(
    "tensorflow_serving/apis/get_model_status.proto", "tensorflow.serving",
    deps=["tensorflow_serving/apis/model.proto",
          "tensorflow_serving/util/status.proto"],
    messages=[
        message("GetModelStatusRequest",
                fields=[field("model_spec", 1, ".tensorflow.serving.ModelSpec")]),
        message(
            "ModelVersionStatus",
            fields=[
                field("version", 1, "int64"),
                field("state", 2,
                      "enum:.tensorflow.serving.ModelVersionStatus.State"),
                field("status", 3, ".tensorflow.serving.StatusProto"),
            ],
            enums=[enum("State", [("UNKNOWN", 0), ("START", 10), ("LOADING", 20),
                                  ("AVAILABLE", 30), ("UNLOADING", 40),
                                  ("END", 50)])],
        ),
        message("GetModelStatusResponse",
                fields=[field("model_version_status", 1,
                              ".tensorflow.serving.ModelVersionStatus",
                              repeated=True,
                              json_name="model_version_status")]),
    ],
)

FILE_LOG_COLLECTOR_CONFIG = proto_file(
    "tensorflow_serving/config/log_collector_config.proto", "tensorflow.serving",
    messages=[
        message("LogCollectorConfig",
                fields=[field("type", 1, "string"),
                        field("filename_prefix", 2, "string")]),
    ],
)

FILE_LOGGING_CONFIG = proto_file(
    "tensorflow_serving/config/logging_config.proto", "tensorflow.serving",
    deps=["tensorflow_serving/config/log_collector_config.proto"],
    messages=[
        message("SamplingConfig", fields=[field("sampling_rate", 1, "double")]),
        message("LoggingConfig",
                fields=[field("log_collector_config", 1,
                              ".tensorflow.serving.LogCollectorConfig"),
                        field("sampling_config", 2,
                              ".tensorflow.serving.SamplingConfig")]),
    ],
)

FILE_FS_STORAGE_PATH_SOURCE = proto_file(
    "tensorflow_serving/sources/storage_path/file_system_storage_path_source.proto",
    "tensorflow.serving",
    messages=[
        message(
            "FileSystemStoragePathSourceConfig",
            fields=[
                field("servables", 5,
                      ".tensorflow.serving.FileSystemStoragePathSourceConfig"
                      ".ServableToMonitor", repeated=True),
                field("servable_name", 1, "string"),
                field("base_path", 2, "string"),
                field("file_system_poll_wait_seconds", 3, "int64"),
                field("fail_if_zero_versions_at_startup", 4, "bool"),
                field("servable_versions_always_present", 6, "bool"),
            ],
            nested=[
                message(
                    "ServableVersionPolicy",
                    fields=[
                        field("latest", 100,
                              ".tensorflow.serving.FileSystemStoragePathSourceConfig"
                              ".ServableVersionPolicy.Latest", oneof=0),
                        field("all", 101,
                              ".tensorflow.serving.FileSystemStoragePathSourceConfig"
                              ".ServableVersionPolicy.All", oneof=0),
                        field("specific", 102,
                              ".tensorflow.serving.FileSystemStoragePathSourceConfig"
                              ".ServableVersionPolicy.Specific", oneof=0),
                    ],
                    nested=[
                        message("Latest",
                                fields=[field("num_versions", 1, "uint32")]),
                        message("All"),
                        message("Specific",
                                fields=[field("versions", 1, "int64",
                                              repeated=True)]),
                    ],
                    oneofs=["policy_choice"],
                ),
                message(
                    "ServableToMonitor",
                    fields=[
                        field("servable_name", 1, "string"),
                        field("base_path", 2, "string"),
                        field("servable_version_policy", 4,
                              ".tensorflow.serving.FileSystemStoragePathSourceConfig"
                              ".ServableVersionPolicy"),
                    ],
                ),
            ],
        ),
    ],
)

_vl_entry, _vl_field = map_field(
    "version_labels", 8, "string", "int64", ".tensorflow.serving.ModelConfig")

FILE_MODEL_SERVER_CONFIG = proto_file(
    "tensorflow_serving/config/model_server_config.proto", "tensorflow.serving",
    deps=["google/protobuf/any.proto",
          "tensorflow_serving/config/logging_config.proto",
          "tensorflow_serving/sources/storage_path/"
          "file_system_storage_path_source.proto"],
    enums=[enum("ModelType", [("MODEL_TYPE_UNSPECIFIED", 0), ("TENSORFLOW", 1),
                              ("OTHER", 2)])],
    messages=[
        message(
            "ModelConfig",
            fields=[
                field("name", 1, "string"),
                field("base_path", 2, "string"),
                field("model_type", 3, "enum:.tensorflow.serving.ModelType"),
                field("model_platform", 4, "string"),
                field("model_version_policy", 7,
                      ".tensorflow.serving.FileSystemStoragePathSourceConfig"
                      ".ServableVersionPolicy"),
                _vl_field,
                field("logging_config", 6, ".tensorflow.serving.LoggingConfig"),
            ],
            nested=[_vl_entry],
        ),
        message("ModelConfigList",
                fields=[field("config", 1, ".tensorflow.serving.ModelConfig",
                              repeated=True)]),
        message("ModelServerConfig",
                fields=[field("model_config_list", 1,
                              ".tensorflow.serving.ModelConfigList", oneof=0),
                        field("custom_model_config", 2,
                              ".google.protobuf.Any", oneof=0)],
                oneofs=["config"]),
    ],
)

FILE_MODEL_MANAGEMENT = proto_file(
    "tensorflow_serving/apis/model_management.proto", "tensorflow.serving",
    deps=["tensorflow_serving/config/model_server_config.proto",
          "tensorflow_serving/util/status.proto"],
    messages=[
        message("ReloadConfigRequest",
                fields=[field("config", 1,
                              ".tensorflow.serving.ModelServerConfig")]),
        message("ReloadConfigResponse",
                fields=[field("status", 1, ".tensorflow.serving.StatusProto")]),
    ],
)

FILE_MODEL_SERVICE = proto_file(
    "tensorflow_serving/apis/model_service.proto", "tensorflow.serving",
    deps=["tensorflow_serving/apis/get_model_status.proto",
          "tensorflow_serving/apis/model_management.proto"],
    services=[service("ModelService", [
        ("GetModelStatus", ".tensorflow.serving.GetModelStatusRequest",
         ".tensorflow.serving.GetModelStatusResponse"),
        ("HandleReloadConfigRequest", ".tensorflow.serving.ReloadConfigRequest",
         ".tensorflow.serving.ReloadConfigResponse"),
    ])],
)

FILE_PREDICTION_SERVICE = proto_file(
    "tensorflow_serving/apis/prediction_service.proto", "tensorflow.serving",
    deps=["tensorflow_serving/apis/classification.proto",
          "tensorflow_serving/apis/get_model_metadata.proto",
          "tensorflow_serving/apis/inference.proto",
          "tensorflow_serving/apis/predict.proto",
          "tensorflow_serving/apis/regression.proto"],
    services=[service("PredictionService", [
        ("Classify", ".tensorflow.serving.ClassificationRequest",
         ".tensorflow.serving.ClassificationResponse"),
        ("Regress", ".tensorflow.serving.RegressionRequest",
         ".tensorflow.serving.RegressionResponse"),
        ("Predict", ".tensorflow.serving.PredictRequest",
         ".tensorflow.serving.PredictResponse"),
        ("MultiInference", ".tensorflow.serving.MultiInferenceRequest",
         ".tensorflow.serving.MultiInferenceResponse"),
        ("GetModelMetadata", ".tensorflow.serving.GetModelMetadataRequest",
         ".tensorflow.serving.GetModelMetadataResponse"),
    ])],
)

FILE_SSL_CONFIG = proto_file(
    "tensorflow_serving/config/ssl_config.proto", "tensorflow.serving",
    messages=[
        message("SSLConfig",
                fields=[field("server_key", 1, "string"),
                        field("server_cert", 2, "string"),
                        field("custom_ca", 3, "string"),
                        field("client_verify", 4, "bool")]),
    ],
)

FILE_MONITORING_CONFIG = proto_file(
    "tensorflow_serving/config/monitoring_config.proto", "tensorflow.serving",
    messages=[
        message("PrometheusConfig", fields=[field("enable", 1, "bool"),
                                            field("path", 2, "string")]),
        message("MonitoringConfig",
                fields=[field("prometheus_config", 1,
                              ".tensorflow.serving.PrometheusConfig")]),
    ],
)

FILE_CORE_LOGGING = proto_file(
    "tensorflow_serving/core/logging.proto", "tensorflow.serving",
    deps=["tensorflow_serving/apis/model.proto",
          "tensorflow_serving/config/logging_config.proto"],
    messages=[
        message("LogMetadata",
                fields=[field("model_spec", 1,
                              ".tensorflow.serving.ModelSpec"),
                        field("sampling_config", 2,
                              ".tensorflow.serving.SamplingConfig"),
                        field("saved_model_tags", 3, "string",
                              repeated=True)]),
    ],
)

# prediction_log.proto (warmup/record-log format). The session_run_log
# oneof arm (field 5) is omitted — its SessionRunRequest closure pulls in
# the whole RunOptions/debugger graph; unknown fields round-trip anyway.
FILE_PREDICTION_LOG = proto_file(
    "tensorflow_serving/apis/prediction_log.proto", "tensorflow.serving",
    deps=["tensorflow_serving/apis/classification.proto",
          "tensorflow_serving/apis/inference.proto",
          "tensorflow_serving/apis/predict.proto",
          "tensorflow_serving/apis/regression.proto",
          "tensorflow_serving/core/logging.proto"],
    messages=[
        message("ClassifyLog",
                fields=[field("request", 1,
                              ".tensorflow.serving.ClassificationRequest"),
                        field("response", 2,
                              ".tensorflow.serving.ClassificationResponse")]),
        message("RegressLog",
                fields=[field("request", 1,
                              ".tensorflow.serving.RegressionRequest"),
                        field("response", 2,
                              ".tensorflow.serving.RegressionResponse")]),
        message("PredictLog",
                fields=[field("request", 1,
                              ".tensorflow.serving.PredictRequest"),
                        field("response", 2,
                              ".tensorflow.serving.PredictResponse")]),
        message("MultiInferenceLog",
                fields=[field("request", 1,
                              ".tensorflow.serving.MultiInferenceRequest"),
                        field("response", 2,
                              ".tensorflow.serving.MultiInferenceResponse")]),
        message("PredictionLog",
                fields=[field("log_metadata", 1,
                              ".tensorflow.serving.LogMetadata"),
                        field("classify_log", 2,
                              ".tensorflow.serving.ClassifyLog", oneof=0),
                        field("regress_log", 3,
                              ".tensorflow.serving.RegressLog", oneof=0),
                        field("predict_log", 6,
                              ".tensorflow.serving.PredictLog", oneof=0),
                        field("multi_inference_log", 4,
                              ".tensorflow.serving.MultiInferenceLog",
                              oneof=0)],
                oneofs=["log_type"]),
    ],
)

# --- profiler service (reference server.cc:324,339 registers it alongside
# Model/Prediction; proto tensorflow/core/profiler/profiler_service.proto).
# Subset closure: ProfileResponse fields 2/4/5 (GraphDef, op_profile,
# RunMetadata — deep TF-internal closures this server never emits) are
# omitted; proto3 peers treat absent fields as unset and python-protobuf
# preserves any unknown fields a TF peer might send. --------------------------

_tool_opts_entry, _tool_opts_field = map_field(
    "tool_options", 8, "string", ".tensorflow.ToolRequestOptions",
    ".tensorflow.ProfileRequest")

FILE_PROFILER_SERVICE = proto_file(
    "tensorflow/core/profiler/profiler_service.proto", "tensorflow",
    messages=[
        message("ProfileOptions",
                fields=[field("include_dataset_ops", 1, "bool")]),
        message("ToolRequestOptions",
                fields=[field("output_formats", 2, "string"),
                        field("save_to_repo", 3, "bool")]),
        message("ProfileRequest",
                fields=[field("duration_ms", 1, "uint64"),
                        field("max_events", 2, "uint64"),
                        field("tools", 3, "string", repeated=True),
                        _tool_opts_field,
                        field("opts", 4, ".tensorflow.ProfileOptions"),
                        field("repository_root", 5, "string"),
                        field("session_id", 6, "string"),
                        field("host_name", 7, "string")],
                nested=[_tool_opts_entry]),
        message("ProfileToolData",
                fields=[field("name", 1, "string"),
                        field("data", 2, "bytes")]),
        message("ProfileResponse",
                fields=[field("encoded_trace", 3, "bytes"),
                        field("tool_data", 6, ".tensorflow.ProfileToolData",
                              repeated=True),
                        field("empty_trace", 7, "bool")]),
        message("MonitorRequest",
                fields=[field("duration_ms", 1, "uint64"),
                        field("monitoring_level", 2, "int32"),
                        field("timestamp", 3, "bool")]),
        message("MonitorResponse",
                fields=[field("data", 1, "string")]),
    ],
    services=[service("ProfilerService", [
        ("Profile", ".tensorflow.ProfileRequest",
         ".tensorflow.ProfileResponse"),
        ("Monitor", ".tensorflow.MonitorRequest",
         ".tensorflow.MonitorResponse"),
    ])],
)

# Dependency-ordered registration list.
_ALL_FILES = [
    FILE_TYPES, FILE_TENSOR_SHAPE, FILE_RESOURCE_HANDLE, FILE_TENSOR,
    FILE_FEATURE, FILE_EXAMPLE, FILE_ERROR_CODES, FILE_META_GRAPH,
    FILE_MODEL, FILE_PREDICT, FILE_INPUT, FILE_CLASSIFICATION,
    FILE_REGRESSION, FILE_INFERENCE, FILE_GET_MODEL_METADATA, FILE_STATUS,
    FILE_GET_MODEL_STATUS, FILE_LOG_COLLECTOR_CONFIG, FILE_LOGGING_CONFIG,
    FILE_FS_STORAGE_PATH_SOURCE, FILE_MODEL_SERVER_CONFIG,
    FILE_MODEL_MANAGEMENT, FILE_MODEL_SERVICE, FILE_PREDICTION_SERVICE,
    FILE_MONITORING_CONFIG, FILE_CORE_LOGGING, FILE_PREDICTION_LOG,
    FILE_SSL_CONFIG, FILE_PROFILER_SERVICE,
]

_pool = descriptor_pool.Default()

for _f in _ALL_FILES:
    try:
        _pool.Add(_f)
    except Exception:
        # Already registered (re-import in the same process, or a conflicting
        # real TF install). FindFileByName below will surface real problems.
        pass


def get_message_class(full_name: str):
    return message_factory.GetMessageClass(_pool.FindMessageTypeByName(full_name))


def get_enum(full_name: str):
    return _pool.FindEnumTypeByName(full_name)
