"""Message classes for the TF-Serving wire surface.

Every class here serializes byte-identically to upstream TF/TF-Serving
generated code (same descriptors → same protobuf runtime encoder). The
``*_pb2`` shim modules under ``tensorflow/`` and ``tensorflow_serving/`` at
the repo root re-export these under the reference import paths
(reference: min_tfs_client imports e.g.
``tensorflow_serving.apis.predict_pb2`` — requests.py:7).
"""
from __future__ import annotations

from . import schema as _s

# --- tensorflow.core.framework ---------------------------------------------
TensorProto = _s.get_message_class("tensorflow.TensorProto")
VariantTensorDataProto = _s.get_message_class("tensorflow.VariantTensorDataProto")
TensorShapeProto = _s.get_message_class("tensorflow.TensorShapeProto")
ResourceHandleProto = _s.get_message_class("tensorflow.ResourceHandleProto")

# --- tensorflow.core.example ------------------------------------------------
BytesList = _s.get_message_class("tensorflow.BytesList")
FloatList = _s.get_message_class("tensorflow.FloatList")
Int64List = _s.get_message_class("tensorflow.Int64List")
Feature = _s.get_message_class("tensorflow.Feature")
Features = _s.get_message_class("tensorflow.Features")
FeatureList = _s.get_message_class("tensorflow.FeatureList")
FeatureLists = _s.get_message_class("tensorflow.FeatureLists")
Example = _s.get_message_class("tensorflow.Example")
SequenceExample = _s.get_message_class("tensorflow.SequenceExample")

# --- tensorflow.core.protobuf ----------------------------------------------
TensorInfo = _s.get_message_class("tensorflow.TensorInfo")
SignatureDef = _s.get_message_class("tensorflow.SignatureDef")
AssetFileDef = _s.get_message_class("tensorflow.AssetFileDef")

# --- tensorflow profiler service --------------------------------------------
ProfileOptions = _s.get_message_class("tensorflow.ProfileOptions")
ToolRequestOptions = _s.get_message_class("tensorflow.ToolRequestOptions")
ProfileRequest = _s.get_message_class("tensorflow.ProfileRequest")
ProfileToolData = _s.get_message_class("tensorflow.ProfileToolData")
ProfileResponse = _s.get_message_class("tensorflow.ProfileResponse")
MonitorRequest = _s.get_message_class("tensorflow.MonitorRequest")
MonitorResponse = _s.get_message_class("tensorflow.MonitorResponse")

# --- tensorflow_serving.apis ------------------------------------------------
ModelSpec = _s.get_message_class("tensorflow.serving.ModelSpec")
PredictRequest = _s.get_message_class("tensorflow.serving.PredictRequest")
PredictResponse = _s.get_message_class("tensorflow.serving.PredictResponse")
Input = _s.get_message_class("tensorflow.serving.Input")
ExampleList = _s.get_message_class("tensorflow.serving.ExampleList")
ExampleListWithContext = _s.get_message_class(
    "tensorflow.serving.ExampleListWithContext")
Class = _s.get_message_class("tensorflow.serving.Class")
Classifications = _s.get_message_class("tensorflow.serving.Classifications")
ClassificationResult = _s.get_message_class(
    "tensorflow.serving.ClassificationResult")
ClassificationRequest = _s.get_message_class(
    "tensorflow.serving.ClassificationRequest")
ClassificationResponse = _s.get_message_class(
    "tensorflow.serving.ClassificationResponse")
Regression = _s.get_message_class("tensorflow.serving.Regression")
RegressionResult = _s.get_message_class("tensorflow.serving.RegressionResult")
RegressionRequest = _s.get_message_class("tensorflow.serving.RegressionRequest")
RegressionResponse = _s.get_message_class("tensorflow.serving.RegressionResponse")
InferenceTask = _s.get_message_class("tensorflow.serving.InferenceTask")
InferenceResult = _s.get_message_class("tensorflow.serving.InferenceResult")
MultiInferenceRequest = _s.get_message_class(
    "tensorflow.serving.MultiInferenceRequest")
MultiInferenceResponse = _s.get_message_class(
    "tensorflow.serving.MultiInferenceResponse")
SignatureDefMap = _s.get_message_class("tensorflow.serving.SignatureDefMap")
GetModelMetadataRequest = _s.get_message_class(
    "tensorflow.serving.GetModelMetadataRequest")
GetModelMetadataResponse = _s.get_message_class(
    "tensorflow.serving.GetModelMetadataResponse")
StatusProto = _s.get_message_class("tensorflow.serving.StatusProto")
GetModelStatusRequest = _s.get_message_class(
    "tensorflow.serving.GetModelStatusRequest")
ModelVersionStatus = _s.get_message_class("tensorflow.serving.ModelVersionStatus")
GetModelStatusResponse = _s.get_message_class(
    "tensorflow.serving.GetModelStatusResponse")

# --- tensorflow_serving.config / sources ------------------------------------
LogCollectorConfig = _s.get_message_class("tensorflow.serving.LogCollectorConfig")
SamplingConfig = _s.get_message_class("tensorflow.serving.SamplingConfig")
LoggingConfig = _s.get_message_class("tensorflow.serving.LoggingConfig")
FileSystemStoragePathSourceConfig = _s.get_message_class(
    "tensorflow.serving.FileSystemStoragePathSourceConfig")
ModelConfig = _s.get_message_class("tensorflow.serving.ModelConfig")
ModelConfigList = _s.get_message_class("tensorflow.serving.ModelConfigList")
ModelServerConfig = _s.get_message_class("tensorflow.serving.ModelServerConfig")
ReloadConfigRequest = _s.get_message_class("tensorflow.serving.ReloadConfigRequest")
ReloadConfigResponse = _s.get_message_class(
    "tensorflow.serving.ReloadConfigResponse")
PrometheusConfig = _s.get_message_class("tensorflow.serving.PrometheusConfig")
SSLConfig = _s.get_message_class("tensorflow.serving.SSLConfig")
MonitoringConfig = _s.get_message_class("tensorflow.serving.MonitoringConfig")
LogMetadata = _s.get_message_class("tensorflow.serving.LogMetadata")
ClassifyLog = _s.get_message_class("tensorflow.serving.ClassifyLog")
RegressLog = _s.get_message_class("tensorflow.serving.RegressLog")
PredictLog = _s.get_message_class("tensorflow.serving.PredictLog")
MultiInferenceLog = _s.get_message_class(
    "tensorflow.serving.MultiInferenceLog")
PredictionLog = _s.get_message_class("tensorflow.serving.PredictionLog")


class _EnumShim:
    """Module-level enum access in generated-code style:
    ``types_pb2.DT_FLOAT`` and ``types_pb2.DataType.Name(1)``."""

    def __init__(self, enum_desc):
        self._desc = enum_desc
        for v in enum_desc.values:
            setattr(self, v.name, v.number)

    def Name(self, number: int) -> str:
        return self._desc.values_by_number[number].name

    def Value(self, name: str) -> int:
        return self._desc.values_by_name[name].number

    def keys(self):
        return [v.name for v in self._desc.values]

    def values(self):
        return [v.number for v in self._desc.values]

    def items(self):
        return [(v.name, v.number) for v in self._desc.values]


DataType = _EnumShim(_s.get_enum("tensorflow.DataType"))
ErrorCode = _EnumShim(_s.get_enum("tensorflow.error.Code"))
