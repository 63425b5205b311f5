#!/usr/bin/env python3
"""Kernel micro-benchmarks with effective-bandwidth reporting.

Run under rocprofv3 for per-kernel hardware stats:
  cd /tmp && rocprofv3 --kernel-trace --stats -d gpurun_out/prof -- \
      python tools/profile_kernels.py
"""
import json
import os
import sys

_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, _ROOT)

import torch  # noqa: E402

from min_tfs_client_amd import ops  # noqa: E402


def time_kernel(fn, reps=50, warmup=10):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    start = torch.cuda.Event(enable_timing=True)
    end = torch.cuda.Event(enable_timing=True)
    start.record()
    for _ in range(reps):
        fn()
    end.record()
    torch.cuda.synchronize()
    return start.elapsed_time(end) / reps  # ms


def report(name, ms, bytes_moved):
    gbps = bytes_moved / 1e9 / (ms / 1e3)
    print(json.dumps({"kernel": name, "ms": round(ms, 4),
                      "GB_s": round(gbps, 1)}))


def main():
    assert torch.cuda.is_available()
    dev = "cuda:0"
    results = []

    # --- elementwise cast, large (HBM-bound) -------------------------------
    n = 256 << 20  # 256M elements: 512MB bf16 in, 1GB f32 out
    x_bf16 = torch.randn(n, device=dev, dtype=torch.bfloat16)
    ms = time_kernel(lambda: ops.cast(x_bf16, torch.float32))
    report("cast_bf16_to_f32_256M", ms, n * 2 + n * 4)
    del x_bf16

    x_f32 = torch.randn(n, device=dev, dtype=torch.float32)
    ms = time_kernel(lambda: ops.cast(x_f32, torch.bfloat16))
    report("cast_f32_to_bf16_256M", ms, n * 4 + n * 2)
    del x_f32

    # --- fused NCHW->NHWC + cast -------------------------------------------
    # headline image shape (small-C path)
    x = torch.randn(32, 3, 224, 224, device=dev, dtype=torch.bfloat16)
    nbytes = x.numel() * (2 + 4)
    ms = time_kernel(lambda: ops.nchw_to_nhwc(x, torch.float32))
    report("nchw_nhwc_smallC_32x3x224x224_bf16f32", ms, nbytes)
    # torch eager comparison (permute+contiguous+to = 2 kernels)
    ms = time_kernel(
        lambda: x.permute(0, 2, 3, 1).contiguous().to(torch.float32))
    report("torch_eager_permute_cast_same_shape", ms, nbytes)

    # generic tiled path, big C
    x = torch.randn(32, 256, 56, 56, device=dev, dtype=torch.bfloat16)
    nbytes = x.numel() * (2 + 4)
    ms = time_kernel(lambda: ops.nchw_to_nhwc(x, torch.float32))
    report("nchw_nhwc_tiled_32x256x56x56_bf16f32", ms, nbytes)
    ms = time_kernel(
        lambda: x.permute(0, 2, 3, 1).contiguous().to(torch.float32))
    report("torch_eager_permute_cast_big_C", ms, nbytes)

    # large batch variant (config-5 at scale)
    x = torch.randn(256, 3, 224, 224, device=dev, dtype=torch.bfloat16)
    nbytes = x.numel() * (2 + 4)
    ms = time_kernel(lambda: ops.nchw_to_nhwc(x, torch.float32))
    report("nchw_nhwc_smallC_256x3x224x224_bf16f32", ms, nbytes)

    # --- quantize ----------------------------------------------------------
    x = torch.randn(128 << 20, device=dev)
    nbytes = x.numel() * (4 + 1)
    ms = time_kernel(lambda: ops.quantize_q8(x, 0.1, 0.0))
    report("quantize_q8_128M", ms, nbytes)


if __name__ == "__main__":
    main()
