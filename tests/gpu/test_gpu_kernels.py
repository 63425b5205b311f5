"""HIP kernel numerics on MI355X (gfx950): every kernel is compared against
a plain PyTorch fp32 reference of the same op. These tests REQUIRE the
native extension — no eager fallback is permitted on a GPU box."""
import pytest
import torch

pytestmark = pytest.mark.gpu

from min_tfs_client_amd import ops  # noqa: E402


@pytest.fixture(scope="module")
def native():
    n = ops.require_native()  # raises if missing: HIP path is mandatory
    assert n.hip_available()
    return n


DEV = "cuda:0"


# ---------------------------------------------------------------------------
# cast kernels
# ---------------------------------------------------------------------------

@pytest.mark.parametrize("src,dst", [
    (torch.bfloat16, torch.float32),
    (torch.float16, torch.float32),
    (torch.float32, torch.bfloat16),
    (torch.float32, torch.float16),
    (torch.bfloat16, torch.float16),
    (torch.float16, torch.bfloat16),
])
def test_cast_matches_torch(native, src, dst):
    x = torch.randn(1 << 20, device=DEV).to(src)
    out = ops.cast(x, dst)
    ref = x.to(dst)  # torch eager reference
    assert out.dtype == dst
    assert torch.equal(out, ref)


def test_cast_odd_tail(native):
    x = torch.randn(1_000_003, device=DEV, dtype=torch.bfloat16)
    assert torch.equal(ops.cast(x, torch.float32), x.to(torch.float32))


def test_cast_special_values(native):
    x = torch.tensor([float("inf"), float("-inf"), float("nan"), 0.0, -0.0,
                      65504.0, 1e-8], device=DEV, dtype=torch.float32)
    out = ops.cast(x, torch.bfloat16)
    ref = x.to(torch.bfloat16)
    assert torch.equal(out.isnan(), ref.isnan())
    assert torch.equal(out[~out.isnan()], ref[~ref.isnan()])


# ---------------------------------------------------------------------------
# fused NCHW->NHWC + cast (BASELINE config 5)
# ---------------------------------------------------------------------------

@pytest.mark.parametrize("shape", [
    (32, 3, 224, 224),      # the headline image config (small-C path)
    (4, 8, 17, 31),         # small-C boundary, ragged HW
    (2, 256, 56, 56),       # generic tiled path
    (1, 64, 64, 64),
    (3, 65, 7, 9),          # ragged C, tiled path
])
def test_nchw_nhwc_bf16_to_f32(native, shape):
    x = torch.randn(*shape, device=DEV, dtype=torch.bfloat16)
    out = ops.nchw_to_nhwc(x, torch.float32)
    ref = x.permute(0, 2, 3, 1).contiguous().to(torch.float32)
    assert out.shape == ref.shape
    assert torch.equal(out, ref)


def test_nchw_nhwc_f32_identity_dtype(native):
    x = torch.randn(2, 96, 28, 28, device=DEV)
    out = ops.nchw_to_nhwc(x, torch.float32)
    assert torch.equal(out, x.permute(0, 2, 3, 1).contiguous())


def test_nchw_nhwc_f32_to_bf16(native):
    x = torch.randn(2, 16, 32, 32, device=DEV)
    out = ops.nchw_to_nhwc(x, torch.bfloat16)
    ref = x.permute(0, 2, 3, 1).contiguous().to(torch.bfloat16)
    assert torch.equal(out, ref)


# ---------------------------------------------------------------------------
# quantize / dequantize
# ---------------------------------------------------------------------------

def test_quantize_q8_matches_torch(native):
    x = torch.randn(1 << 18, device=DEV) * 10
    scale, zp = 0.1, 3.0
    q = ops.quantize_q8(x, scale, zp)
    # pinned semantics: multiply by float(1/scale), separate add, RNE —
    # expressed identically on the torch side (torch's own x/scale also
    # multiplies by the reciprocal but may differ at half-way ties)
    inv = float(1.0 / scale)
    ref = torch.clamp(torch.round(x * inv + zp), -128, 127).to(torch.int8)
    assert torch.equal(q, ref)


def test_dequantize_roundtrip(native):
    x = torch.randn(1 << 16, device=DEV)
    scale = 0.05
    q = ops.quantize_q8(x, scale, 0.0)
    d = ops.dequantize_q8(q, scale, 0.0)
    assert (d - x).abs().max().item() <= scale / 2 + 1e-6


# ---------------------------------------------------------------------------
# staging copies
# ---------------------------------------------------------------------------

@pytest.mark.parametrize("copy_mode", [0, 1])
@pytest.mark.parametrize("nelem", [17, 1 << 16, (8 << 20) + 13])
def test_tensor_content_bytes_gpu(native, copy_mode, nelem):
    x = torch.randn(nelem, device=DEV)
    blob = native.tensor_content_bytes(x, copy_mode)
    assert blob == x.cpu().numpy().tobytes()


@pytest.mark.parametrize("copy_mode", [0, 1])
def test_serialize_parse_gpu_roundtrip(native, copy_mode):
    t = torch.randn(32, 3, 64, 64, device=DEV)
    blob = native.serialize_predict_request(
        "m", 1, "", ["x"], [t], copy_mode)
    _, outs, _ = native.parse_predict_request(blob, DEV, copy_mode)
    assert outs["x"].is_cuda
    assert torch.equal(outs["x"], t)


def test_serialize_mixed_cpu_gpu_inputs(native):
    a = torch.randn(4, 4, device=DEV)
    b = torch.arange(6, dtype=torch.int64)  # cpu
    blob = native.serialize_predict_request(
        "m", -1, "", ["a", "b"], [a, b], 0)
    _, outs, _ = native.parse_predict_request(blob, "cpu", 0)
    assert torch.equal(outs["a"], a.cpu())
    assert torch.equal(outs["b"], b)


# ---------------------------------------------------------------------------
# inverse layout transform (unpack direction)
# ---------------------------------------------------------------------------

@pytest.mark.parametrize("shape", [
    (32, 224, 224, 3),      # NHWC image (small-C path)
    (4, 17, 31, 3),
    (2, 56, 56, 256),       # generic tiled path
    (3, 7, 9, 65),
])
def test_nhwc_nchw_f32_to_bf16(native, shape):
    x = torch.randn(*shape, device=DEV, dtype=torch.float32)
    out = ops.nhwc_to_nchw(x, torch.bfloat16)
    ref = x.permute(0, 3, 1, 2).contiguous().to(torch.bfloat16)
    assert out.shape == ref.shape
    assert torch.equal(out, ref)


def test_nhwc_nchw_roundtrip_with_forward(native):
    x = torch.randn(8, 3, 64, 64, device=DEV, dtype=torch.bfloat16)
    nhwc = ops.nchw_to_nhwc(x, torch.float32)
    back = ops.nhwc_to_nchw(nhwc, torch.bfloat16)
    assert torch.equal(back, x)
