"""Model-family sanity on CPU (tiny shapes; full shapes are GPU tests)."""
import numpy as np
import pytest
import torch

from min_tfs_client_amd.models import (
    bert_base,
    bert_servable,
    resnet50,
    resnet50_servable,
)


def test_resnet50_shapes_and_param_count():
    m = resnet50()
    n_params = sum(p.numel() for p in m.parameters())
    # canonical ResNet-50: 25.557M params
    assert abs(n_params - 25_557_032) < 10_000, n_params
    with torch.no_grad():
        y = m(torch.randn(2, 3, 64, 64))
    assert y.shape == (2, 1000)


def test_resnet50_servable():
    s = resnet50_servable()
    out = s({"images": np.random.rand(1, 3, 64, 64).astype(np.float32)})
    assert out["logits"].shape == (1, 1000)


def test_bert_base_param_count():
    m = bert_base()
    n_params = sum(p.numel() for p in m.parameters())
    # BERT-base ~110M params (109.48M encoder+embeddings+pooler)
    assert 105e6 < n_params < 115e6, n_params


def test_bert_servable_multi_input():
    s = bert_servable()
    ids = np.random.randint(0, 30522, (2, 16), dtype=np.int32)
    mask = np.ones((2, 16), dtype=np.int32)
    out = s({"input_ids": ids, "attention_mask": mask})
    assert out["last_hidden_state"].shape == (2, 16, 768)
    assert out["pooled_output"].shape == (2, 768)


def test_bert_attention_mask_effect():
    s = bert_servable()
    ids = np.random.randint(0, 30522, (1, 8), dtype=np.int32)
    full = s({"input_ids": ids,
              "attention_mask": np.ones((1, 8), np.int32)})
    half_mask = np.ones((1, 8), np.int32)
    half_mask[:, 4:] = 0
    half = s({"input_ids": ids, "attention_mask": half_mask})
    # masking must change the pooled output
    assert not np.allclose(full["pooled_output"].numpy(),
                           half["pooled_output"].numpy())


def test_bert_seq_too_long_raises():
    s = bert_servable()
    ids = np.zeros((1, 513), dtype=np.int32)
    with pytest.raises(ValueError, match="sequence length"):
        s({"input_ids": ids, "attention_mask": np.ones((1, 513), np.int32)})


def test_bert_without_mask():
    s = bert_servable()
    ids = np.random.randint(0, 30522, (1, 8), dtype=np.int32)
    out = s({"input_ids": ids})
    assert out["pooled_output"].shape == (1, 768)


def test_resnet_wrong_channels_raises():
    s = resnet50_servable()
    with pytest.raises(Exception):
        s({"images": np.zeros((1, 4, 32, 32), np.float32)})
