"""BERT-base encoder in plain PyTorch — the BASELINE config-3 model family
(multi-input: input_ids + attention_mask int32, batch 128 x seq 512).

Self-contained implementation (no checkpoint download; random init).
"""
from __future__ import annotations

import math

import torch
import torch.nn as nn


class BertConfig:
    vocab_size = 30522
    hidden = 768
    layers = 12
    heads = 12
    intermediate = 3072
    max_pos = 512
    type_vocab = 2
    eps = 1e-12


class BertLayer(nn.Module):
    def __init__(self, cfg: BertConfig):
        super().__init__()
        h = cfg.hidden
        self.qkv = nn.Linear(h, 3 * h)
        self.attn_out = nn.Linear(h, h)
        self.ln1 = nn.LayerNorm(h, eps=cfg.eps)
        self.ff1 = nn.Linear(h, cfg.intermediate)
        self.ff2 = nn.Linear(cfg.intermediate, h)
        self.ln2 = nn.LayerNorm(h, eps=cfg.eps)
        self.heads = cfg.heads
        self.head_dim = h // cfg.heads

    def forward(self, x, attn_mask):
        B, S, H = x.shape
        qkv = self.qkv(x).view(B, S, 3, self.heads, self.head_dim)
        q, k, v = qkv.unbind(2)  # each B,S,heads,hd
        q = q.transpose(1, 2)
        k = k.transpose(1, 2)
        v = v.transpose(1, 2)
        # scaled_dot_product_attention uses the fused path on ROCm
        attn = torch.nn.functional.scaled_dot_product_attention(
            q, k, v, attn_mask=attn_mask)
        attn = attn.transpose(1, 2).reshape(B, S, H)
        x = self.ln1(x + self.attn_out(attn))
        x = self.ln2(x + self.ff2(torch.nn.functional.gelu(self.ff1(x))))
        return x


class BertEncoder(nn.Module):
    def __init__(self, cfg: BertConfig = BertConfig()):
        super().__init__()
        self.cfg = cfg
        self.tok = nn.Embedding(cfg.vocab_size, cfg.hidden)
        self.pos = nn.Embedding(cfg.max_pos, cfg.hidden)
        self.typ = nn.Embedding(cfg.type_vocab, cfg.hidden)
        self.ln = nn.LayerNorm(cfg.hidden, eps=cfg.eps)
        self.blocks = nn.ModuleList(
            [BertLayer(cfg) for _ in range(cfg.layers)])
        self.pooler = nn.Linear(cfg.hidden, cfg.hidden)

    def forward(self, input_ids, attention_mask=None):
        B, S = input_ids.shape
        if S > self.cfg.max_pos:
            raise ValueError(
                f"sequence length {S} exceeds max_position_embeddings "
                f"{self.cfg.max_pos}")
        pos_ids = torch.arange(S, device=input_ids.device)
        x = self.tok(input_ids.long()) + self.pos(pos_ids) \
            + self.typ(torch.zeros_like(input_ids.long()))
        x = self.ln(x)
        mask = None
        if attention_mask is not None:
            # additive mask: (B,1,1,S), 0 where attend, -inf where padded
            m = attention_mask[:, None, None, :].to(x.dtype)
            mask = (1.0 - m) * torch.finfo(x.dtype).min
        for blk in self.blocks:
            x = blk(x, mask)
        pooled = torch.tanh(self.pooler(x[:, 0]))
        return x, pooled


def bert_base() -> BertEncoder:
    return BertEncoder()


def bert_servable(device: str = "cpu", dtype=torch.float32, seed: int = 0):
    """BERT-base Servable: inputs "input_ids"/"attention_mask" int32,
    outputs "last_hidden_state" + "pooled_output"."""
    from ..server import Servable

    torch.manual_seed(seed)
    model = bert_base().to(device=device, dtype=dtype).eval()

    @torch.no_grad()
    def fn(inputs):
        ids = inputs["input_ids"]
        mask = inputs.get("attention_mask")
        if not isinstance(ids, torch.Tensor):
            ids = torch.as_tensor(ids)
        ids = ids.to(device=device)
        if mask is not None:
            if not isinstance(mask, torch.Tensor):
                mask = torch.as_tensor(mask)
            mask = mask.to(device=device)
        hidden, pooled = model(ids, mask)
        return {"last_hidden_state": hidden.float(),
                "pooled_output": pooled.float()}

    s = Servable(
        fn,
        signature={
            "method_name": "tensorflow/serving/predict",
            "inputs": {"input_ids": (3, [-1, -1]),
                       "attention_mask": (3, [-1, -1])},  # DT_INT32
            "outputs": {"last_hidden_state": (1, [-1, -1, 768]),
                        "pooled_output": (1, [-1, 768])},
        })
    s.module = model
    return s
