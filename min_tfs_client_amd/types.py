"""Trivalent (quadrivalent, here) dtype resolution.

Parity with reference types.py:13-42: ``DataType`` accepts a numpy scalar
type, a TF dtype string (``"DT_FLOAT"``), or a proto enum int, and exposes
``.numpy_dtype``, ``.tf_dtype``, ``.enum``, ``.proto_field_name``,
``.is_numeric``. MI355X extension: also accepts a ``torch.dtype`` and
exposes ``.torch_dtype`` (None where torch cannot represent the dtype).
"""
from __future__ import annotations

from typing import Union

import numpy as np

try:
    import torch
    _TORCH_DTYPE = torch.dtype
except ImportError:  # pragma: no cover
    torch = None

    class _TORCH_DTYPE:  # type: ignore
        pass

from .constants import (
    ENUM_TO_TF_MAPPING,
    NUMERICAL_TYPES,
    NP_TO_TF_MAPPING,
    TF_TO_ENUM_MAPPING,
    TF_TO_NP_MAPPING,
    TF_TO_TORCH_MAPPING,
    TF_TYPE_BY_NAME,
    TORCH_TO_TF_MAPPING,
    _EXTRA_NP_ALIASES,
)

DTypeLike = Union[type, str, int, "_TORCH_DTYPE"]


class DataType:
    """Resolved dtype with every representation the wire format needs."""

    VALID_TYPES = NUMERICAL_TYPES.union({np.str_, np.bool_})

    def __init__(self, dtype: DTypeLike):
        self.tf_dtype = self._resolve_tf_name(dtype)
        meta = TF_TYPE_BY_NAME[self.tf_dtype]
        self.numpy_dtype = TF_TO_NP_MAPPING[self.tf_dtype]
        self.enum = TF_TO_ENUM_MAPPING[self.tf_dtype]
        self.proto_field_name = meta.TensorProtoField
        self.itemsize = meta.itemsize
        self.is_numeric = self.tf_dtype != "DT_STRING"
        self.torch_dtype = TF_TO_TORCH_MAPPING.get(self.tf_dtype)

    # -- resolution ---------------------------------------------------------

    def _resolve_tf_name(self, dtype: DTypeLike) -> str:
        if torch is not None and isinstance(dtype, torch.dtype):
            try:
                return TORCH_TO_TF_MAPPING[dtype]
            except KeyError:
                raise ValueError(
                    f"torch dtype {dtype} has no TensorProto mapping")
        if isinstance(dtype, type):
            np_type = _EXTRA_NP_ALIASES.get(dtype, dtype)
            self._validate_np(np_type)
            return NP_TO_TF_MAPPING[np_type].TFDType
        if isinstance(dtype, str):
            if dtype not in TF_TYPE_BY_NAME:
                raise ValueError(
                    f"Unknown TF dtype string {dtype!r}. Allowable: "
                    f"{', '.join(sorted(TF_TYPE_BY_NAME))}")
            return dtype
        if isinstance(dtype, int):
            if dtype not in ENUM_TO_TF_MAPPING:
                raise ValueError(f"Unknown DataType enum value {dtype}")
            name = ENUM_TO_TF_MAPPING[dtype]
            if name not in TF_TYPE_BY_NAME:
                raise ValueError(
                    f"DataType {name} is not supported by this client")
            return name
        raise ValueError(
            f"Expected dtype of types: type, str, int or torch.dtype, "
            f"got {type(dtype)}")

    def _validate_np(self, np_type: type) -> None:
        if np_type not in self.VALID_TYPES:
            raise ValueError(
                f"Dtype {np_type.__name__} is not valid. Allowable values: "
                f"{', '.join(sorted(t.__name__ for t in self.VALID_TYPES))}")

    def __repr__(self) -> str:  # pragma: no cover
        return f"DataType({self.tf_dtype})"

    def __eq__(self, other) -> bool:
        return isinstance(other, DataType) and other.tf_dtype == self.tf_dtype

    def __hash__(self) -> int:
        return hash(self.tf_dtype)
