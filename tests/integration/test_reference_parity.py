"""Drop-in parity with zendesk/min-tfs-client: the reference's import
paths and call patterns must work unchanged against this framework
(reference requests.py:1-16, tensors.py:1-7, README usage)."""
import json

import numpy as np
import pytest
from google.protobuf.json_format import MessageToJson


def test_reference_import_paths():
    # the exact imports the reference package and its users make
    from tensorflow_serving.apis.predict_pb2 import (  # noqa: F401
        PredictRequest, PredictResponse)
    from tensorflow_serving.apis.classification_pb2 import (  # noqa: F401
        ClassificationRequest, ClassificationResponse)
    from tensorflow_serving.apis.regression_pb2 import (  # noqa: F401
        RegressionRequest, RegressionResponse)
    from tensorflow_serving.apis.get_model_status_pb2 import (  # noqa: F401
        GetModelStatusRequest, GetModelStatusResponse)
    from tensorflow_serving.apis.prediction_service_pb2_grpc import (  # noqa: F401,E501
        PredictionServiceStub)
    from tensorflow_serving.apis.model_service_pb2_grpc import (  # noqa: F401
        ModelServiceStub)
    from tensorflow.core.framework.tensor_pb2 import TensorProto  # noqa: F401
    from tensorflow.core.framework.tensor_shape_pb2 import (  # noqa: F401
        TensorShapeProto)
    from tensorflow.core.framework import types_pb2
    assert types_pb2.DataType.DT_FLOAT == 1

    from min_tfs_client.requests import TensorServingClient  # noqa: F401
    from min_tfs_client.tensors import (  # noqa: F401
        ndarray_to_tensor_proto, tensor_proto_to_ndarray)
    from min_tfs_client.types import DataType  # noqa: F401


def test_reference_readme_usage_pattern():
    """The usage block from the reference README, verbatim semantics,
    against our loopback server."""
    from min_tfs_client.requests import TensorServingClient
    from min_tfs_client.tensors import tensor_proto_to_ndarray
    from min_tfs_client_amd.server import ModelServer, identity_servable

    with ModelServer(port=0) as srv:
        srv.manager.load("default", identity_servable(), version=1)
        client = TensorServingClient(host="127.0.0.1", port=srv.port,
                                     credentials=None)
        response = client.predict_request(
            model_name="default",
            model_version=1,
            input_dict={
                "string_input": np.array(["hello world"]),
                "float_input": np.float32(np.random.rand(1, 100)),
                "int_input": np.array([1], dtype=np.int64),
            })
        float_output = tensor_proto_to_ndarray(
            response.outputs["float_output"])
        assert float_output.shape == (1, 100)
        client.close()


def test_reference_integration_test_shape():
    """Mirrors reference tests/integration/requests_test.py:39-50
    byte-for-byte on the JSON view of GetModelStatus."""
    from min_tfs_client.requests import TensorServingClient
    from min_tfs_client_amd.server import ModelServer, identity_servable

    with ModelServer(port=0) as srv:
        srv.manager.load("default", identity_servable(), version=1)
        client = TensorServingClient("127.0.0.1", srv.port)
        response = client.model_status_request(model_name="default")
        assert json.loads(MessageToJson(response)) == {
            "model_version_status": [
                {"version": "1", "state": "AVAILABLE", "status": {}}
            ]
        }
        client.close()


def test_reference_fixture_directory_layout(tmp_path):
    """The reference's checked-in model lives at
    fixtures/00000001/saved_model.pb; the zero-padded numeric version dir
    must resolve to version 1 (file_system_storage_path_source semantics)."""
    from min_tfs_client_amd.repository import FileSystemStoragePathSource
    from min_tfs_client_amd.server import ModelManager

    vdir = tmp_path / "default" / "00000001"
    vdir.mkdir(parents=True)
    (vdir / "saved_model.pb").write_bytes(b"")  # content unused: identity
    mgr = ModelManager()
    src = FileSystemStoragePathSource(mgr, poll_wait_seconds=0)
    src.set_models({"default": str(tmp_path / "default")})
    src.poll_once()
    assert mgr.version_statuses("default")[0][0] == 1
    assert mgr.version_statuses("default")[0][1] == 30


def test_ssl_channel(tmp_path):
    """Secure channel parity (reference requests.py:27-30 accepts
    ssl_channel_credentials)."""
    import grpc
    import os
    from min_tfs_client.requests import TensorServingClient
    from min_tfs_client_amd.server import ModelServer, identity_servable
    from min_tfs_client_amd.tensors import tensor_proto_to_ndarray

    certs = os.path.join(os.path.dirname(__file__), "fixtures", "certs")
    key = open(os.path.join(certs, "server.key"), "rb").read()
    crt = open(os.path.join(certs, "server.crt"), "rb").read()

    # TLS serving stays on the grpcio transport (the native
    # C++ transport is cleartext h2c)
    srv = ModelServer(port=0, transport="grpcio")
    srv.manager.load("default", identity_servable(), version=1)
    creds = grpc.ssl_server_credentials([(key, crt)])
    port = srv._server.add_secure_port("localhost:0", creds)
    srv.start()
    try:
        client = TensorServingClient(
            "localhost", port,
            credentials=grpc.ssl_channel_credentials(root_certificates=crt))
        x = np.array([1.0, 2.0], dtype=np.float32)
        resp = client.predict_request("default", {"x": x}, timeout=20)
        np.testing.assert_array_equal(
            tensor_proto_to_ndarray(resp.outputs["x"]), x)
        client.close()
    finally:
        srv.stop(0)
