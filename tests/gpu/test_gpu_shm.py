"""Shared-memory transport with device tensors: serialize DMAs from HBM
directly into the shm segment; responses unpack back to HBM."""
import pytest
import torch

pytestmark = pytest.mark.gpu

from min_tfs_client_amd.server import ModelManager, Servable, identity_servable  # noqa: E402
from min_tfs_client_amd.shm import ShmListener, ShmPredictClient  # noqa: E402

DEV = "cuda:0"


def test_shm_gpu_roundtrip(tmp_path):
    mgr = ModelManager()
    mgr.load("m", identity_servable(), version=1)

    def gpu_double(inputs):
        return {k: v * 2 for k, v in inputs.items()}

    mgr.load("double", Servable(gpu_double), version=1)
    with ShmListener(mgr, str(tmp_path / "hs"), device=DEV):
        with ShmPredictClient(str(tmp_path / "hs"),
                              slot_bytes=64 << 20) as c:
            x = torch.randn(32, 3, 224, 224, device=DEV)
            out = c.predict("m", {"images": x}, output_device=DEV)
            assert out["images"].is_cuda
            assert torch.equal(out["images"], x)
            # GPU servable through shm: inputs land on device, math runs
            y = torch.randn(64, 128, device=DEV)
            out = c.predict("double", {"y": y}, output_device=DEV)
            assert torch.equal(out["y"], y * 2)
