#!/usr/bin/env python3
"""Extended randomized GPU validation: many shapes/dtype pairs for every
kernel, each checked bit-exact against the torch eager reference."""
import os
import random
import sys

_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, _ROOT)

import torch  # noqa: E402

from min_tfs_client_amd import ops  # noqa: E402


def main(iters=200, seed=0):
    rng = random.Random(seed)
    assert torch.cuda.is_available()
    dev = "cuda:0"
    checked = {"cast": 0, "nchw": 0, "nhwc": 0, "quant": 0, "bytes": 0}
    native = ops.require_native()
    for i in range(iters):
        kind = rng.choice(list(checked))
        if kind == "cast":
            src, dst = rng.choice([
                (torch.bfloat16, torch.float32),
                (torch.float16, torch.float32),
                (torch.float32, torch.bfloat16),
                (torch.float32, torch.float16),
                (torch.bfloat16, torch.float16),
                (torch.float16, torch.bfloat16)])
            n = rng.randint(1, 1 << 20)
            x = torch.randn(n, device=dev).to(src)
            assert torch.equal(ops.cast(x, dst), x.to(dst)), (src, dst, n)
        elif kind in ("nchw", "nhwc"):
            N = rng.randint(1, 8)
            C = rng.choice([1, 2, 3, 4, 5, 7, 8, 16, 31, 64, 129, 256])
            H = rng.randint(1, 64)
            W = rng.randint(1, 64)
            src, dst = rng.choice([(torch.bfloat16, torch.float32),
                                   (torch.float32, torch.float32),
                                   (torch.float32, torch.bfloat16)])
            if kind == "nchw":
                if src == torch.float32 and dst == torch.bfloat16:
                    src = torch.float32
                x = torch.randn(N, C, H, W, device=dev).to(src)
                out = ops.nchw_to_nhwc(x, dst)
                ref = x.permute(0, 2, 3, 1).contiguous().to(dst)
            else:
                if src == torch.bfloat16 and dst == torch.float32:
                    pass
                x = torch.randn(N, H, W, C, device=dev).to(src)
                try:
                    out = ops.nhwc_to_nchw(x, dst)
                except RuntimeError:
                    continue  # unsupported dtype pair for inverse
                ref = x.permute(0, 3, 1, 2).contiguous().to(dst)
            assert out.shape == ref.shape, (kind, N, C, H, W)
            assert torch.equal(out, ref), (kind, N, C, H, W, src, dst)
        elif kind == "quant":
            n = rng.randint(1, 1 << 18)
            scale = rng.choice([0.1, 0.05, 1.0, 2.5])
            zp = rng.choice([0.0, 3.0, -5.0])
            x = torch.randn(n, device=dev) * 10
            q = ops.quantize_q8(x, scale, zp)
            inv = float(1.0 / scale)
            ref = torch.clamp(torch.round(x * inv + zp), -128,
                              127).to(torch.int8)
            assert torch.equal(q, ref), (n, scale, zp)
        else:  # bytes roundtrip
            n = rng.randint(1, 1 << 20)
            x = torch.randn(n, device=dev)
            mode = rng.choice([0, 1])
            assert native.tensor_content_bytes(x, mode) == \
                x.cpu().numpy().tobytes()
        checked[kind] += 1
    print("kernel sweep OK:", checked)


if __name__ == "__main__":
    main(int(sys.argv[1]) if len(sys.argv) > 1 else 200)
