"""gRPC service stubs for PredictionService and ModelService.

Hand-written (no grpcio-tools in this environment) with the exact method
paths and message types of the reference's checked-in generated stubs
(reference prediction_service_pb2_grpc.py:28-139,
model_service_pb2_grpc.py:26-69), so a server or client built on these is
wire-interoperable with any TF-Serving deployment.
"""
from __future__ import annotations

import grpc

from . import messages as pb

_PS = "/tensorflow.serving.PredictionService/"
_MS = "/tensorflow.serving.ModelService/"


class PredictionServiceStub:
    """Client stub; 5 rpcs (reference prediction_service.proto:15-31)."""

    def __init__(self, channel: grpc.Channel):
        self.Classify = channel.unary_unary(
            _PS + "Classify",
            request_serializer=pb.ClassificationRequest.SerializeToString,
            response_deserializer=pb.ClassificationResponse.FromString)
        self.Regress = channel.unary_unary(
            _PS + "Regress",
            request_serializer=pb.RegressionRequest.SerializeToString,
            response_deserializer=pb.RegressionResponse.FromString)
        self.Predict = channel.unary_unary(
            _PS + "Predict",
            request_serializer=pb.PredictRequest.SerializeToString,
            response_deserializer=pb.PredictResponse.FromString)
        self.MultiInference = channel.unary_unary(
            _PS + "MultiInference",
            request_serializer=pb.MultiInferenceRequest.SerializeToString,
            response_deserializer=pb.MultiInferenceResponse.FromString)
        self.GetModelMetadata = channel.unary_unary(
            _PS + "GetModelMetadata",
            request_serializer=pb.GetModelMetadataRequest.SerializeToString,
            response_deserializer=pb.GetModelMetadataResponse.FromString)


class PredictionServiceServicer:
    """Server-side service skeleton; override the methods you serve."""

    def Classify(self, request, context):
        context.set_code(grpc.StatusCode.UNIMPLEMENTED)
        context.set_details("Method not implemented!")
        raise NotImplementedError("Method not implemented!")

    def Regress(self, request, context):
        context.set_code(grpc.StatusCode.UNIMPLEMENTED)
        context.set_details("Method not implemented!")
        raise NotImplementedError("Method not implemented!")

    def Predict(self, request, context):
        context.set_code(grpc.StatusCode.UNIMPLEMENTED)
        context.set_details("Method not implemented!")
        raise NotImplementedError("Method not implemented!")

    def MultiInference(self, request, context):
        context.set_code(grpc.StatusCode.UNIMPLEMENTED)
        context.set_details("Method not implemented!")
        raise NotImplementedError("Method not implemented!")

    def GetModelMetadata(self, request, context):
        context.set_code(grpc.StatusCode.UNIMPLEMENTED)
        context.set_details("Method not implemented!")
        raise NotImplementedError("Method not implemented!")


def add_PredictionServiceServicer_to_server(servicer, server):
    rpc_method_handlers = {
        "Classify": grpc.unary_unary_rpc_method_handler(
            servicer.Classify,
            request_deserializer=pb.ClassificationRequest.FromString,
            response_serializer=pb.ClassificationResponse.SerializeToString),
        "Regress": grpc.unary_unary_rpc_method_handler(
            servicer.Regress,
            request_deserializer=pb.RegressionRequest.FromString,
            response_serializer=pb.RegressionResponse.SerializeToString),
        "Predict": grpc.unary_unary_rpc_method_handler(
            servicer.Predict,
            request_deserializer=pb.PredictRequest.FromString,
            response_serializer=pb.PredictResponse.SerializeToString),
        "MultiInference": grpc.unary_unary_rpc_method_handler(
            servicer.MultiInference,
            request_deserializer=pb.MultiInferenceRequest.FromString,
            response_serializer=pb.MultiInferenceResponse.SerializeToString),
        "GetModelMetadata": grpc.unary_unary_rpc_method_handler(
            servicer.GetModelMetadata,
            request_deserializer=pb.GetModelMetadataRequest.FromString,
            response_serializer=pb.GetModelMetadataResponse.SerializeToString),
    }
    generic_handler = grpc.method_handlers_generic_handler(
        "tensorflow.serving.PredictionService", rpc_method_handlers)
    server.add_generic_rpc_handlers((generic_handler,))


class ModelServiceStub:
    """Client stub; 2 rpcs (reference model_service.proto:12-24)."""

    def __init__(self, channel: grpc.Channel):
        self.GetModelStatus = channel.unary_unary(
            _MS + "GetModelStatus",
            request_serializer=pb.GetModelStatusRequest.SerializeToString,
            response_deserializer=pb.GetModelStatusResponse.FromString)
        self.HandleReloadConfigRequest = channel.unary_unary(
            _MS + "HandleReloadConfigRequest",
            request_serializer=pb.ReloadConfigRequest.SerializeToString,
            response_deserializer=pb.ReloadConfigResponse.FromString)


class ModelServiceServicer:
    def GetModelStatus(self, request, context):
        context.set_code(grpc.StatusCode.UNIMPLEMENTED)
        context.set_details("Method not implemented!")
        raise NotImplementedError("Method not implemented!")

    def HandleReloadConfigRequest(self, request, context):
        context.set_code(grpc.StatusCode.UNIMPLEMENTED)
        context.set_details("Method not implemented!")
        raise NotImplementedError("Method not implemented!")


def add_ModelServiceServicer_to_server(servicer, server):
    rpc_method_handlers = {
        "GetModelStatus": grpc.unary_unary_rpc_method_handler(
            servicer.GetModelStatus,
            request_deserializer=pb.GetModelStatusRequest.FromString,
            response_serializer=pb.GetModelStatusResponse.SerializeToString),
        "HandleReloadConfigRequest": grpc.unary_unary_rpc_method_handler(
            servicer.HandleReloadConfigRequest,
            request_deserializer=pb.ReloadConfigRequest.FromString,
            response_serializer=pb.ReloadConfigResponse.SerializeToString),
    }
    generic_handler = grpc.method_handlers_generic_handler(
        "tensorflow.serving.ModelService", rpc_method_handlers)
    server.add_generic_rpc_handlers((generic_handler,))


_PROF = "/tensorflow.ProfilerService/"


class ProfilerServiceStub:
    """Client stub; 2 rpcs (reference registers ProfilerService alongside
    Model/Prediction — server.cc:324,339; profiler_service.proto:12-17)."""

    def __init__(self, channel: grpc.Channel):
        self.Profile = channel.unary_unary(
            _PROF + "Profile",
            request_serializer=pb.ProfileRequest.SerializeToString,
            response_deserializer=pb.ProfileResponse.FromString)
        self.Monitor = channel.unary_unary(
            _PROF + "Monitor",
            request_serializer=pb.MonitorRequest.SerializeToString,
            response_deserializer=pb.MonitorResponse.FromString)


class ProfilerServiceServicer:
    def Profile(self, request, context):
        context.set_code(grpc.StatusCode.UNIMPLEMENTED)
        context.set_details("Method not implemented!")
        raise NotImplementedError("Method not implemented!")

    def Monitor(self, request, context):
        context.set_code(grpc.StatusCode.UNIMPLEMENTED)
        context.set_details("Method not implemented!")
        raise NotImplementedError("Method not implemented!")


def add_ProfilerServiceServicer_to_server(servicer, server):
    rpc_method_handlers = {
        "Profile": grpc.unary_unary_rpc_method_handler(
            servicer.Profile,
            request_deserializer=pb.ProfileRequest.FromString,
            response_serializer=pb.ProfileResponse.SerializeToString),
        "Monitor": grpc.unary_unary_rpc_method_handler(
            servicer.Monitor,
            request_deserializer=pb.MonitorRequest.FromString,
            response_serializer=pb.MonitorResponse.SerializeToString),
    }
    generic_handler = grpc.method_handlers_generic_handler(
        "tensorflow.ProfilerService", rpc_method_handlers)
    server.add_generic_rpc_handlers((generic_handler,))


# Raw-bytes stubs for the zero-(re)serialize hot path: the C++ codec emits
# finished request bytes; identity (de)serializers hand them to grpc's C core
# untouched, skipping python-protobuf entirely (design per the reference's
# EncodeTensorToByteBuffer zero-copy idea, grpc_tensor_coding.cc:140-248).
def _identity(x: bytes) -> bytes:
    return x


class RawPredictionServiceStub:
    def __init__(self, channel: grpc.Channel):
        self.Predict = channel.unary_unary(
            _PS + "Predict",
            request_serializer=_identity,
            response_deserializer=_identity)
