"""Dtype-system conformance (mirror of reference types_test.py:7-43: the
15-dtype matrix across construction routes, asserting exact enum ints), plus
the torch-dtype route and error paths the reference leaves untested."""
import numpy as np
import pytest
import torch

from min_tfs_client_amd.types import DataType

# (numpy type, TF string, enum, proto field) — wire facts from
# /root/reference/protobuf_srcs/tensorflow/core/framework/types.proto:12-68
# and tensor.proto:14-94.
DTYPE_MATRIX = [
    (np.float16, "DT_HALF", 19, "half_val"),
    (np.float32, "DT_FLOAT", 1, "float_val"),
    (np.float64, "DT_DOUBLE", 2, "double_val"),
    (np.int8, "DT_INT8", 6, "int_val"),
    (np.int16, "DT_INT16", 5, "int_val"),
    (np.int32, "DT_INT32", 3, "int_val"),
    (np.int64, "DT_INT64", 9, "int64_val"),
    (np.uint8, "DT_UINT8", 4, "int_val"),
    (np.uint16, "DT_UINT16", 17, "int_val"),
    (np.uint32, "DT_UINT32", 22, "uint32_val"),
    (np.uint64, "DT_UINT64", 23, "uint64_val"),
    (np.complex64, "DT_COMPLEX64", 8, "scomplex_val"),
    (np.complex128, "DT_COMPLEX128", 18, "dcomplex_val"),
    (np.str_, "DT_STRING", 7, "string_val"),
    (np.bool_, "DT_BOOL", 10, "bool_val"),
]


@pytest.mark.parametrize("np_type,tf_name,enum,field", DTYPE_MATRIX)
def test_construct_from_numpy_type(np_type, tf_name, enum, field):
    dt = DataType(np_type)
    assert dt.tf_dtype == tf_name
    assert dt.enum == enum
    assert dt.proto_field_name == field
    assert dt.numpy_dtype is np_type


@pytest.mark.parametrize("np_type,tf_name,enum,field", DTYPE_MATRIX)
def test_construct_from_tf_string(np_type, tf_name, enum, field):
    dt = DataType(tf_name)
    assert dt.enum == enum
    assert dt.numpy_dtype is np_type
    assert dt.proto_field_name == field


@pytest.mark.parametrize("np_type,tf_name,enum,field", DTYPE_MATRIX)
def test_construct_from_enum(np_type, tf_name, enum, field):
    dt = DataType(enum)
    assert dt.tf_dtype == tf_name
    assert dt.numpy_dtype is np_type


TORCH_MATRIX = [
    (torch.float16, "DT_HALF", 19),
    (torch.bfloat16, "DT_BFLOAT16", 14),
    (torch.float32, "DT_FLOAT", 1),
    (torch.float64, "DT_DOUBLE", 2),
    (torch.int8, "DT_INT8", 6),
    (torch.int16, "DT_INT16", 5),
    (torch.int32, "DT_INT32", 3),
    (torch.int64, "DT_INT64", 9),
    (torch.uint8, "DT_UINT8", 4),
    (torch.complex64, "DT_COMPLEX64", 8),
    (torch.complex128, "DT_COMPLEX128", 18),
    (torch.bool, "DT_BOOL", 10),
]


@pytest.mark.parametrize("torch_dtype,tf_name,enum", TORCH_MATRIX)
def test_construct_from_torch_dtype(torch_dtype, tf_name, enum):
    dt = DataType(torch_dtype)
    assert dt.tf_dtype == tf_name
    assert dt.enum == enum
    assert dt.torch_dtype == torch_dtype


def test_bfloat16_mapping():
    dt = DataType("DT_BFLOAT16")
    assert dt.enum == 14
    assert dt.proto_field_name == "half_val"  # bf16 shares half_val bits
    assert dt.numpy_dtype is np.uint16  # numpy carrier is the bit-pattern
    assert dt.torch_dtype == torch.bfloat16


def test_invalid_numpy_type_raises():
    with pytest.raises(ValueError, match="not valid"):
        DataType(np.void)


def test_invalid_string_raises():
    with pytest.raises(ValueError, match="Unknown TF dtype"):
        DataType("DT_BOGUS")


def test_invalid_enum_raises():
    with pytest.raises(ValueError, match="Unknown DataType enum"):
        DataType(999)


def test_unsupported_enum_raises():
    with pytest.raises(ValueError, match="not supported"):
        DataType(20)  # DT_RESOURCE


def test_invalid_kind_raises():
    with pytest.raises(ValueError, match="Expected dtype"):
        DataType(1.5)
