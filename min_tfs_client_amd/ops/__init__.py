"""Native ops: C++ wire codec + CDNA4 HIP pack kernels.

Loads the in-tree ``_native`` extension (built by ``setup.py build_ext
--inplace`` / ``__graft_entry__.build()``). On a GPU machine the extension
is REQUIRED — device-tensor paths raise rather than silently falling back
to eager PyTorch, so a passing GPU test means the HIP path actually ran.
"""
from __future__ import annotations

from typing import Optional

try:
    import torch
except ImportError:  # pragma: no cover
    torch = None

_native = None
_import_error: Optional[BaseException] = None
try:
    from min_tfs_client_amd import _native  # type: ignore
except Exception as e:  # pragma: no cover - exercised on unbuilt trees
    _import_error = e


def native_available() -> bool:
    return _native is not None


def require_native():
    """The GPU path must run the HIP extension; fail loudly otherwise."""
    if _native is None:
        raise RuntimeError(
            "min_tfs_client_amd._native extension is not built "
            "(run `python setup.py build_ext --inplace`); the HIP pack path "
            f"is mandatory on GPU machines. Import error: {_import_error}")
    return _native


def get_native():
    """Native module or None (CPU-only paths may fall back to the python
    codec)."""
    return _native


# ---------------------------------------------------------------------------
# public op wrappers
# ---------------------------------------------------------------------------

def cast(tensor, out_dtype):
    """Vectorized dtype cast on device (bf16/f16/f32 matrix)."""
    return require_native().cast(tensor, out_dtype)


def nchw_to_nhwc(tensor, out_dtype=None):
    """Fused NCHW->NHWC + cast in one CDNA4 kernel (BASELINE config 5)."""
    if out_dtype is None:
        out_dtype = tensor.dtype
    return require_native().nchw_to_nhwc(tensor, out_dtype)


def nhwc_to_nchw(tensor, out_dtype=None):
    """Inverse fused layout transform (unpack direction)."""
    if out_dtype is None:
        out_dtype = tensor.dtype
    return require_native().nhwc_to_nchw(tensor, out_dtype)


def quantize_q8(tensor, scale, zero_point=0.0):
    return require_native().quantize_q8(tensor, scale, zero_point)


def dequantize_q8(tensor, scale, zero_point=0.0):
    return require_native().dequantize_q8(tensor, scale, zero_point)


def pack_tensor_proto(tensor):
    """Device tensor -> python TensorProto message (for the non-turbo
    client path): native D2H pipeline for the payload, proto wrapper around
    it."""
    from ..tensors import tensor_to_tensor_proto
    from ..types import DataType
    from ..wire import messages as pb

    if not tensor.is_cuda:
        return tensor_to_tensor_proto(tensor)
    n = require_native()
    dtype = DataType(tensor.dtype)
    proto = pb.TensorProto()
    proto.dtype = dtype.enum
    for d in tensor.shape:
        proto.tensor_shape.dim.add().size = d
    proto.tensor_content = n.tensor_content_bytes(tensor, 1)
    return proto
