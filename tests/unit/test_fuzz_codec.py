"""Property-based cross-validation of the C++ wire codec against python
protobuf, plus malformed-input robustness (the python-level complement of
the ASAN wire_test)."""
import numpy as np
import pytest
import torch
from hypothesis import given, settings, strategies as st

from min_tfs_client_amd.wire import messages as pb

native = pytest.importorskip(
    "min_tfs_client_amd._native", reason="_native extension not built")

_DTYPES = [torch.float32, torch.float64, torch.int32, torch.int64,
           torch.int16, torch.int8, torch.uint8, torch.bool,
           torch.float16, torch.bfloat16]


def _rand_tensor(draw):
    dtype = draw(st.sampled_from(_DTYPES))
    rank = draw(st.integers(0, 4))
    shape = tuple(draw(st.integers(1, 5)) for _ in range(rank))
    n = int(np.prod(shape)) if shape else 1
    g = torch.Generator().manual_seed(draw(st.integers(0, 2**31)))
    if dtype is torch.bool:
        t = torch.rand(n, generator=g) > 0.5
    elif dtype.is_floating_point:
        t = (torch.rand(n, generator=g) * 100 - 50).to(dtype)
    else:
        t = torch.randint(-100 if not dtype.is_signed else -100, 100,
                          (n,), generator=g).to(dtype)
    return t.reshape(shape)


names_st = st.text(
    alphabet=st.characters(min_codepoint=33, max_codepoint=0x2FF),
    min_size=1, max_size=12)


@settings(max_examples=60, deadline=None)
@given(st.data())
def test_serialize_parse_random_requests(data):
    n_inputs = data.draw(st.integers(1, 4))
    names = sorted({data.draw(names_st) for _ in range(n_inputs)})
    tensors = [_rand_tensor(data.draw) for _ in names]
    model = data.draw(names_st)
    version = data.draw(st.integers(-1, 1 << 40))
    sig = data.draw(st.sampled_from(["", "serving_default", "sig"]))
    blob = native.serialize_predict_request(model, version, sig, names,
                                            tensors, 0)
    # python protobuf must parse it identically
    req = pb.PredictRequest.FromString(blob)
    assert req.model_spec.name == model
    if version >= 0:
        assert req.model_spec.version.value == version
    assert sorted(req.inputs) == names
    # and the C++ parser round-trips every tensor bit-exactly
    _spec, outs, _ = native.parse_predict_request(blob, "cpu", 0)
    for name, t in zip(names, tensors):
        assert outs[name].dtype == t.dtype
        assert outs[name].shape == t.shape
        assert torch.equal(outs[name], t)


@settings(max_examples=60, deadline=None)
@given(st.data())
def test_python_built_parses_in_cpp(data):
    """Requests built by python protobuf (either encoding) parse in C++."""
    req = pb.PredictRequest()
    req.model_spec.name = data.draw(names_st)
    t = _rand_tensor(data.draw)
    from min_tfs_client_amd.tensors import tensor_to_tensor_proto
    use_content = data.draw(st.booleans())
    # skip typed-field arm for dtypes whose typed decode is C++-supported
    req.inputs["x"].CopyFrom(tensor_to_tensor_proto(t, use_content))
    _spec, outs, _ = native.parse_predict_request(
        req.SerializeToString(), "cpu", 0)
    assert outs["x"].dtype == t.dtype
    assert torch.equal(outs["x"], t)


@settings(max_examples=120, deadline=None)
@given(st.binary(min_size=0, max_size=300))
def test_garbage_bytes_never_crash(blob):
    """Arbitrary bytes either parse (unknown fields are legal) or raise a
    python exception — never crash the process."""
    for is_request in (True, False):
        try:
            native.parse_predict_request(blob, "cpu", 0) if is_request \
                else native.parse_predict_response(blob, "cpu", 0)
        except Exception:
            pass


@settings(max_examples=60, deadline=None)
@given(st.data())
def test_truncations_never_crash(data):
    t = torch.randn(16)
    blob = native.serialize_predict_request("m", 1, "", ["x"], [t], 0)
    cut = data.draw(st.integers(0, len(blob)))
    flip = data.draw(st.integers(-1, len(blob) - 1))
    mutated = bytearray(blob[:cut])
    if 0 <= flip < len(mutated):
        mutated[flip] ^= data.draw(st.integers(1, 255))
    try:
        native.parse_predict_request(bytes(mutated), "cpu", 0)
    except Exception:
        pass


def test_uint_barebones_dtypes_roundtrip():
    """torch's barebones uint16/32/64 dtypes map to DT_UINT16/32/64."""
    for dt, enum in ((torch.uint16, 17), (torch.uint32, 22),
                     (torch.uint64, 23)):
        t = torch.arange(6).to(dt)
        blob = native.serialize_predict_request("m", -1, "", ["x"], [t], 0)
        assert pb.PredictRequest.FromString(blob).inputs["x"].dtype == enum
        _s, outs, _ = native.parse_predict_request(blob, "cpu", 0)
        assert outs["x"].dtype == dt and torch.equal(outs["x"], t)
