"""mi355x-tfs-client — MI355X-native TensorFlow-Serving Predict client framework.

Reproduces the capabilities of zendesk/min-tfs-client (reference mounted at
/root/reference), redesigned MI355X-first:

* tensors live as PyTorch-ROCm device tensors (numpy accepted everywhere);
* the serialize hot path (dtype cast, NCHW<->NHWC transpose, TensorProto
  ``tensor_content`` pack/unpack) is hand-written CDNA4 HIP (gfx950);
* device<->pinned-host copies overlap protobuf encode and the gRPC send on a
  side HIP stream;
* data-parallel request sharding across the 8 GPUs of one node uses RCCL
  scatter / all-gather over xGMI via ``torch.distributed``.

Public API parity with the reference (requests.py:22-110):
``TensorServingClient`` with ``predict_request`` / ``classification_request``
/ ``regression_request`` / ``model_status_request``.
"""
from .version import __version__  # noqa: F401
from .client import TensorServingClient  # noqa: F401
from .tensors import (  # noqa: F401
    ndarray_to_tensor_proto,
    tensor_proto_to_ndarray,
    tensor_to_tensor_proto,
    tensor_proto_to_tensor,
    extract_shape,
)
from .types import DataType  # noqa: F401


def __getattr__(name):
    # heavier entry points, imported lazily to keep bare import cheap
    if name == "TurboPredictClient":
        from .turbo import TurboPredictClient
        return TurboPredictClient
    if name == "AsyncTurboPredictClient":
        from .aio import AsyncTurboPredictClient
        return AsyncTurboPredictClient
    if name == "ModelServer":
        from .server import ModelServer
        return ModelServer
    if name == "DataParallelPredictor":
        from .parallel import DataParallelPredictor
        return DataParallelPredictor
    if name == "ShmPredictClient":
        from .shm import ShmPredictClient
        return ShmPredictClient
    raise AttributeError(name)
