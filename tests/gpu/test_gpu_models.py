"""Model-family serving on the GPU: real torch models behind the raw
server, full turbo round trips with device tensors."""
import pytest
import torch

pytestmark = pytest.mark.gpu

from min_tfs_client_amd.batching import BatchingServable  # noqa: E402
from min_tfs_client_amd.models import bert_servable, resnet50_servable  # noqa: E402
from min_tfs_client_amd.server import ModelServer  # noqa: E402
from min_tfs_client_amd.turbo import TurboPredictClient  # noqa: E402

DEV = "cuda:0"


@pytest.fixture(scope="module")
def server(tmp_path_factory):
    sock = f"unix://{tmp_path_factory.mktemp('m')}/models.sock"
    with ModelServer(address=sock, raw_predict=True, device=DEV) as srv:
        srv.manager.load("resnet50", resnet50_servable(DEV), version=1)
        srv.manager.load("bert", bert_servable(DEV), version=1)
        srv.manager.load(
            "resnet50_batched",
            BatchingServable(resnet50_servable(DEV), max_batch_size=64,
                             batch_timeout_s=0.005),
            version=1)
        yield srv


def test_resnet50_gpu_serving(server):
    with TurboPredictClient(server.address) as client:
        x = torch.randn(8, 3, 224, 224, device=DEV)
        out = client.predict("resnet50", {"images": x}, output_device=DEV)
        assert out["logits"].shape == (8, 1000)
        assert out["logits"].is_cuda
        assert torch.isfinite(out["logits"]).all()


def test_bert_gpu_serving(server):
    with TurboPredictClient(server.address) as client:
        ids = torch.randint(0, 30522, (4, 128), dtype=torch.int32,
                            device=DEV)
        mask = torch.ones(4, 128, dtype=torch.int32, device=DEV)
        out = client.predict("bert",
                             {"input_ids": ids, "attention_mask": mask},
                             output_device=DEV)
        assert out["last_hidden_state"].shape == (4, 128, 768)
        assert out["pooled_output"].shape == (4, 768)


def test_batched_resnet_concurrent(server):
    import threading
    results = {}

    def call(i):
        with TurboPredictClient(server.address) as client:
            x = torch.randn(4, 3, 224, 224, device=DEV)
            out = client.predict("resnet50_batched", {"images": x},
                                 timeout=120)
            results[i] = out["logits"].shape

    threads = [threading.Thread(target=call, args=(i,)) for i in range(4)]
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    assert all(results[i] == (4, 1000) for i in range(4))
    inner = server.manager.get("resnet50_batched")
    assert inner.batches_run >= 1
