"""Repository -> GPU serving end-to-end: model.json family loading on
cuda:0 through the filesystem source, served via the raw path."""
import json

import pytest
import torch

pytestmark = pytest.mark.gpu

from min_tfs_client_amd.repository import FileSystemStoragePathSource  # noqa: E402
from min_tfs_client_amd.server import ModelManager, ModelServer  # noqa: E402
from min_tfs_client_amd.turbo import TurboPredictClient  # noqa: E402


def test_model_json_resnet_on_gpu(tmp_path):
    vdir = tmp_path / "resnet50" / "00000001"
    vdir.mkdir(parents=True)
    (vdir / "model.json").write_text(json.dumps(
        {"family": "resnet50", "device": "cuda:0"}))
    srv = ModelServer(port=0, raw_predict=True, device="cuda:0")
    src = FileSystemStoragePathSource(srv.manager, poll_wait_seconds=0)
    src.set_models({"resnet50": str(tmp_path / "resnet50")})
    src.poll_once()
    srv.start()
    try:
        with TurboPredictClient(srv.address) as c:
            x = torch.randn(2, 3, 224, 224, device="cuda:0")
            out = c.predict("resnet50", {"images": x},
                            output_device="cuda:0", timeout=120)
            assert out["logits"].shape == (2, 1000)
            assert out["logits"].is_cuda
    finally:
        srv.stop(0)


def test_torchscript_on_gpu(tmp_path):
    from min_tfs_client_amd.repository import default_loader

    class Scale(torch.nn.Module):
        def forward(self, x):
            return x * 3

    vdir = tmp_path / "m" / "1"
    vdir.mkdir(parents=True)
    torch.jit.script(Scale()).save(str(vdir / "model.pt"))
    s = default_loader("m", str(vdir), device="cuda:0")
    out = s({"x": torch.ones(4, device="cuda:0")})
    assert out["output"].is_cuda
    assert torch.equal(out["output"], torch.full((4,), 3.0,
                                                 device="cuda:0"))
