"""Alias of reference tensors.py — same import path, superset semantics."""
from min_tfs_client_amd.tensors import (  # noqa: F401
    coerce_to_bytes,
    extract_shape,
    ndarray_to_tensor_proto,
    tensor_proto_to_ndarray,
    write_values_to_tensor_proto,
)
