#!/usr/bin/env python3
"""Raw host<->device copy ceilings on this box (context for the latency
decomposition): pinned vs pageable, D2H and H2D, 19.27MB payload."""
import json
import os
import sys
import time
import statistics

_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, _ROOT)
import torch  # noqa: E402


def med_bw(fn, nbytes, n=30):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    ts = []
    for _ in range(n):
        t0 = time.perf_counter()
        fn()
        torch.cuda.synchronize()
        ts.append(time.perf_counter() - t0)
    t = statistics.median(ts)
    return round(nbytes / t / 1e9, 2), round(t * 1e3, 3)


def main():
    assert torch.cuda.is_available()
    n = 32 * 3 * 224 * 224
    nbytes = n * 4
    dev = torch.randn(n, device="cuda:0")
    pinned = torch.empty(n, pin_memory=True)
    pageable = torch.empty(n)
    out = {"payload_mb": round(nbytes / 1e6, 2)}
    out["d2h_pinned_GBps"], out["d2h_pinned_ms"] = med_bw(
        lambda: pinned.copy_(dev, non_blocking=True), nbytes)
    out["d2h_pageable_GBps"], out["d2h_pageable_ms"] = med_bw(
        lambda: pageable.copy_(dev), nbytes)
    out["h2d_pinned_GBps"], out["h2d_pinned_ms"] = med_bw(
        lambda: dev.copy_(pinned, non_blocking=True), nbytes)
    out["h2d_pageable_GBps"], out["h2d_pageable_ms"] = med_bw(
        lambda: dev.copy_(pageable), nbytes)
    print(json.dumps(out))


if __name__ == "__main__":
    main()
