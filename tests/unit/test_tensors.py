"""Tensor-codec tests: reference tensors_test.py:25-117 coverage (golden
text-format, round-trips, bytes coercion, shape extraction) plus the gaps
the reference never covered: fp16/bf16 bit-pattern semantics,
complex interleave, tensor_content encode/decode, repeat-last-fill decode."""
import textwrap

import numpy as np
import pytest
import torch
from google.protobuf import text_format

from min_tfs_client_amd.tensors import (
    coerce_to_bytes,
    extract_shape,
    ndarray_to_tensor_proto,
    tensor_proto_to_ndarray,
    tensor_proto_to_tensor,
    tensor_to_tensor_proto,
)
from min_tfs_client_amd.wire import messages as pb


def assert_proto_text(proto, expected: str):
    assert text_format.MessageToString(proto) == textwrap.dedent(expected)


# -- bytes coercion (reference tensors_test.py:25-39) -----------------------

def test_coerce_str_to_bytes():
    assert coerce_to_bytes("héllo") == "héllo".encode("utf-8")


def test_coerce_bytes_passthrough():
    assert coerce_to_bytes(b"\x00\x01") == b"\x00\x01"


# -- golden text-format (reference tensors_test.py:42-83) --------------------

def test_float_typed_golden():
    proto = ndarray_to_tensor_proto(
        np.array([[1.0, 2.0], [3.0, 4.0]], dtype=np.float32),
        use_tensor_content=False)
    assert_proto_text(proto, """\
        dtype: DT_FLOAT
        tensor_shape {
          dim {
            size: 2
          }
          dim {
            size: 2
          }
        }
        float_val: 1.0
        float_val: 2.0
        float_val: 3.0
        float_val: 4.0
        """)


def test_string_typed_golden():
    proto = ndarray_to_tensor_proto(np.array(["ab", "cd"]),
                                    use_tensor_content=False)
    assert_proto_text(proto, """\
        dtype: DT_STRING
        tensor_shape {
          dim {
            size: 2
          }
        }
        string_val: "ab"
        string_val: "cd"
        """)


def test_float_content_golden():
    proto = ndarray_to_tensor_proto(np.array([1.0], dtype=np.float32))
    assert_proto_text(proto, """\
        dtype: DT_FLOAT
        tensor_shape {
          dim {
            size: 1
          }
        }
        tensor_content: "\\000\\000\\200?"
        """)


# -- round trips -------------------------------------------------------------

NUMERIC_DTYPES = [np.float16, np.float32, np.float64, np.int8, np.int16,
                  np.int32, np.int64, np.uint8, np.uint16, np.uint32,
                  np.uint64, np.complex64, np.complex128, np.bool_]


@pytest.mark.parametrize("np_dtype", NUMERIC_DTYPES)
@pytest.mark.parametrize("use_content", [True, False])
def test_numeric_roundtrip(np_dtype, use_content):
    rng = np.random.default_rng(0)
    if np_dtype is np.bool_:
        arr = rng.random((3, 4)) > 0.5
    elif np.issubdtype(np_dtype, np.complexfloating):
        arr = (rng.random((3, 4)) + 1j * rng.random((3, 4))).astype(np_dtype)
    elif np.issubdtype(np_dtype, np.floating):
        arr = rng.random((3, 4)).astype(np_dtype)
    else:
        arr = rng.integers(0, 100, (3, 4)).astype(np_dtype)
    proto = ndarray_to_tensor_proto(arr, use_tensor_content=use_content)
    out = tensor_proto_to_ndarray(proto)
    assert out.dtype == arr.dtype
    np.testing.assert_array_equal(out, arr)


def test_string_roundtrip():
    arr = np.array([["a", "bb"], ["ccc", "dddd"]])
    proto = ndarray_to_tensor_proto(arr)
    out = tensor_proto_to_ndarray(proto)
    assert out.shape == (2, 2)
    assert out[1][1] == b"dddd"


def test_half_bit_pattern_semantics():
    """DT_HALF half_val holds raw uint16 bits (reference tensor.cc:446-464),
    NOT numeric values — the reference client gets this wrong (SURVEY §2.2
    fact 3); verify our encoding matches TF semantics exactly."""
    arr = np.array([1.0, -0.5, 65504.0], dtype=np.float16)
    proto = ndarray_to_tensor_proto(arr, use_tensor_content=False)
    assert list(proto.half_val) == [0x3C00, 0xB800, 0x7BFF]


def test_complex64_interleave():
    arr = np.array([1 + 2j, 3 - 4j], dtype=np.complex64)
    proto = ndarray_to_tensor_proto(arr, use_tensor_content=False)
    assert list(proto.scomplex_val) == [1.0, 2.0, 3.0, -4.0]


# -- torch round trips -------------------------------------------------------

TORCH_DTYPES = [torch.float16, torch.bfloat16, torch.float32, torch.float64,
                torch.int8, torch.int16, torch.int32, torch.int64,
                torch.uint8, torch.complex64, torch.bool]


@pytest.mark.parametrize("torch_dtype", TORCH_DTYPES)
@pytest.mark.parametrize("use_content", [True, False])
def test_torch_roundtrip(torch_dtype, use_content):
    if torch_dtype is torch.bool:
        t = torch.rand(2, 3) > 0.5
    elif torch_dtype.is_complex:
        t = torch.complex(torch.rand(2, 3), torch.rand(2, 3)).to(torch_dtype)
    elif torch_dtype.is_floating_point:
        t = torch.rand(2, 3).to(torch_dtype)
    else:
        t = torch.randint(0, 100, (2, 3), dtype=torch_dtype)
    proto = tensor_to_tensor_proto(t, use_tensor_content=use_content)
    out = tensor_proto_to_tensor(proto)
    assert out.dtype == torch_dtype
    assert torch.equal(out, t)


def test_bfloat16_enum_and_bits():
    t = torch.tensor([1.0, -2.0], dtype=torch.bfloat16)
    proto = tensor_to_tensor_proto(t, use_tensor_content=False)
    assert proto.dtype == 14  # DT_BFLOAT16
    assert list(proto.half_val) == [0x3F80, 0xC000]


def test_noncontiguous_torch_tensor():
    t = torch.arange(12, dtype=torch.float32).reshape(3, 4).t()
    proto = tensor_to_tensor_proto(t)
    out = tensor_proto_to_tensor(proto)
    assert torch.equal(out, t.contiguous())


# -- decode: superset behaviors ---------------------------------------------

def test_decode_hand_built_typed_proto():
    proto = pb.TensorProto()
    proto.dtype = 1
    proto.tensor_shape.dim.add().size = 3
    proto.float_val.extend([1.0, 2.0, 3.0])
    np.testing.assert_array_equal(
        tensor_proto_to_ndarray(proto),
        np.array([1.0, 2.0, 3.0], dtype=np.float32))


def test_decode_repeat_last_fill():
    """Short typed field fills trailing elements with the last value
    (reference tensor.cc:487-527)."""
    proto = pb.TensorProto()
    proto.dtype = 3  # DT_INT32
    proto.tensor_shape.dim.add().size = 5
    proto.int_val.extend([7, 9])
    np.testing.assert_array_equal(
        tensor_proto_to_ndarray(proto), np.array([7, 9, 9, 9, 9],
                                                 dtype=np.int32))


def test_decode_empty_typed_field_is_zeros():
    proto = pb.TensorProto()
    proto.dtype = 1
    proto.tensor_shape.dim.add().size = 4
    np.testing.assert_array_equal(
        tensor_proto_to_ndarray(proto), np.zeros(4, dtype=np.float32))


def test_decode_tensor_content_preferred():
    """A tensor_content-bearing response decodes (the reference client would
    return an empty array — SURVEY §2.2 fact 1)."""
    arr = np.arange(6, dtype=np.int64).reshape(2, 3)
    proto = pb.TensorProto()
    proto.dtype = 9
    for d in arr.shape:
        proto.tensor_shape.dim.add().size = d
    proto.tensor_content = arr.tobytes()
    np.testing.assert_array_equal(tensor_proto_to_ndarray(proto), arr)


def test_extract_shape():
    proto = ndarray_to_tensor_proto(np.zeros((2, 5, 7), dtype=np.float32))
    assert extract_shape(proto) == (2, 5, 7)


def test_scalar_roundtrip():
    proto = ndarray_to_tensor_proto(np.float32(3.5))
    out = tensor_proto_to_ndarray(proto)
    assert out.shape == ()
    assert out == np.float32(3.5)
