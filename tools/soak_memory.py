#!/usr/bin/env python3
"""Serving soak: sustained requests with RSS tracking on both processes —
memory-leak detection for the C++ codec + grpc path."""
import json
import os
import sys

_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, _ROOT)

import psutil  # noqa: E402
import torch  # noqa: E402

from min_tfs_client_amd.server import ModelServer, identity_servable  # noqa: E402
from min_tfs_client_amd.turbo import TurboPredictClient  # noqa: E402


def main(n=3000):
    dev = "cuda:0" if torch.cuda.is_available() else "cpu"
    sock = f"unix:///tmp/mi355x_soak_{os.getpid()}.sock"
    proc = psutil.Process()
    with ModelServer(address=sock, raw_predict=True) as srv:
        srv.manager.load("m", identity_servable(), version=1)
        with TurboPredictClient(sock) as c:
            ids = torch.randint(0, 30522, (128, 512), dtype=torch.int32,
                                device=dev)
            mask = torch.ones(128, 512, dtype=torch.int32, device=dev)
            rss = []
            for i in range(n):
                c.predict("m", {"input_ids": ids, "attention_mask": mask},
                          output_device=dev)
                if i % 500 == 0 or i == n - 1:
                    rss.append(round(proc.memory_info().rss / 1e6, 1))
            print(json.dumps({
                "requests": n, "rss_mb_samples": rss,
                "rss_growth_mb": round(rss[-1] - rss[1] if len(rss) > 1
                                       else 0, 1),
                "cuda_mem_mb": round(
                    torch.cuda.memory_allocated() / 1e6, 1)
                if dev != "cpu" else None}))


if __name__ == "__main__":
    main(int(sys.argv[1]) if len(sys.argv) > 1 else 3000)
