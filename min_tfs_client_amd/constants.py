"""Dtype mapping tables: numpy / torch <-> TF dtype string <-> DataType enum
<-> TensorProto field.

Parity with reference constants.py:13-50 (15 numpy dtypes), extended with:
* DT_BFLOAT16 (torch.bfloat16 is a first-class tensor dtype on ROCm; the
  reference has no bf16 entry at all),
* torch dtype mappings for every dtype torch can represent,
* per-dtype element size for the ``tensor_content`` fast path.

Wire facts from /root/reference/protobuf_srcs/tensorflow/core/framework/
tensor.proto:14-94 and types.proto:12-68.
"""
from __future__ import annotations

from typing import NamedTuple, Optional

import numpy as np

try:
    import torch
except ImportError:  # pragma: no cover - torch is present in every target env
    torch = None


class TFType(NamedTuple):
    TFDType: str
    TensorProtoField: str
    # bytes per element in tensor_content encoding (None => not fixed-width)
    itemsize: Optional[int]


# numpy type -> TF dtype metadata. Order mirrors reference constants.py:13-29.
NP_TO_TF_MAPPING = {
    np.float16: TFType("DT_HALF", "half_val", 2),
    np.float32: TFType("DT_FLOAT", "float_val", 4),
    np.float64: TFType("DT_DOUBLE", "double_val", 8),
    np.int8: TFType("DT_INT8", "int_val", 1),
    np.int16: TFType("DT_INT16", "int_val", 2),
    np.int32: TFType("DT_INT32", "int_val", 4),
    np.int64: TFType("DT_INT64", "int64_val", 8),
    np.uint8: TFType("DT_UINT8", "int_val", 1),
    np.uint16: TFType("DT_UINT16", "int_val", 2),
    np.uint32: TFType("DT_UINT32", "uint32_val", 4),
    np.uint64: TFType("DT_UINT64", "uint64_val", 8),
    np.complex64: TFType("DT_COMPLEX64", "scomplex_val", 8),
    np.complex128: TFType("DT_COMPLEX128", "dcomplex_val", 16),
    np.str_: TFType("DT_STRING", "string_val", None),
    np.bool_: TFType("DT_BOOL", "bool_val", 1),
}

# bytes is accepted as an alias for DT_STRING input (string_val holds bytes on
# the wire; tensor.proto:56-57).
_EXTRA_NP_ALIASES = {
    np.bytes_: np.str_,
}

TF_TO_NP_MAPPING = {v.TFDType: k for k, v in NP_TO_TF_MAPPING.items()}

# DT_BFLOAT16 has no numpy analogue; represent it as uint16 bit-patterns on
# the numpy side (TF itself stores bf16 as raw uint16 bits in half_val —
# reference tensor.cc:446-464).
TF_TO_NP_MAPPING["DT_BFLOAT16"] = np.uint16

# TF dtype string -> TFType, including bf16.
TF_TYPE_BY_NAME = {v.TFDType: v for v in NP_TO_TF_MAPPING.values()}
TF_TYPE_BY_NAME["DT_BFLOAT16"] = TFType("DT_BFLOAT16", "half_val", 2)

# TF dtype string -> enum int (types.proto:12-68).
TF_TO_ENUM_MAPPING = {
    "DT_FLOAT": 1, "DT_DOUBLE": 2, "DT_INT32": 3, "DT_UINT8": 4,
    "DT_INT16": 5, "DT_INT8": 6, "DT_STRING": 7, "DT_COMPLEX64": 8,
    "DT_INT64": 9, "DT_BOOL": 10, "DT_QINT8": 11, "DT_QUINT8": 12,
    "DT_QINT32": 13, "DT_BFLOAT16": 14, "DT_QINT16": 15, "DT_QUINT16": 16,
    "DT_UINT16": 17, "DT_COMPLEX128": 18, "DT_HALF": 19, "DT_RESOURCE": 20,
    "DT_VARIANT": 21, "DT_UINT32": 22, "DT_UINT64": 23,
}
ENUM_TO_TF_MAPPING = {v: k for k, v in TF_TO_ENUM_MAPPING.items()}

NP_TO_ENUM_MAPPING = {k: TF_TO_ENUM_MAPPING[v.TFDType]
                      for k, v in NP_TO_TF_MAPPING.items()}

NUMERICAL_TYPES = {
    np.float16, np.float32, np.float64,
    np.int8, np.int16, np.int32, np.int64,
    np.uint8, np.uint16, np.uint32, np.uint64,
    np.complex64, np.complex128, np.bool_,
}

# torch dtype <-> TF dtype string (torch has no uint16/32/64 tensors below
# 2.3; ROCm torch 2.10 has uint16/32/64 as barebones dtypes — map them).
if torch is not None:
    TORCH_TO_TF_MAPPING = {
        torch.float16: "DT_HALF",
        torch.bfloat16: "DT_BFLOAT16",
        torch.float32: "DT_FLOAT",
        torch.float64: "DT_DOUBLE",
        torch.int8: "DT_INT8",
        torch.int16: "DT_INT16",
        torch.int32: "DT_INT32",
        torch.int64: "DT_INT64",
        torch.uint8: "DT_UINT8",
        torch.complex64: "DT_COMPLEX64",
        torch.complex128: "DT_COMPLEX128",
        torch.bool: "DT_BOOL",
    }
    for _name, _tf in (("uint16", "DT_UINT16"), ("uint32", "DT_UINT32"),
                       ("uint64", "DT_UINT64")):
        _dt = getattr(torch, _name, None)
        if _dt is not None:
            TORCH_TO_TF_MAPPING[_dt] = _tf
    TF_TO_TORCH_MAPPING = {}
    for _k, _v in TORCH_TO_TF_MAPPING.items():
        TF_TO_TORCH_MAPPING.setdefault(_v, _k)
else:  # pragma: no cover
    TORCH_TO_TF_MAPPING = {}
    TF_TO_TORCH_MAPPING = {}
