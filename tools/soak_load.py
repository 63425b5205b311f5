#!/usr/bin/env python3
"""Sustained-load stability run: N seconds of concurrent 19MB-class
traffic; reports req/s, error count, RSS trajectory."""
import json
import os
import sys
import threading
import time

_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, _ROOT)

import psutil  # noqa: E402
import torch  # noqa: E402

from min_tfs_client_amd.server import ModelServer, identity_servable  # noqa: E402
from min_tfs_client_amd.turbo import TurboPredictClient  # noqa: E402


def main(seconds=60, threads=4):
    dev = "cuda:0" if torch.cuda.is_available() else "cpu"
    sock = f"unix:///tmp/mi355x_load_{os.getpid()}.sock"
    stats = {"done": 0, "errors": 0}
    lock = threading.Lock()
    stop = time.monotonic() + seconds
    proc = psutil.Process()

    with ModelServer(address=sock, raw_predict=True, max_workers=16) as srv:
        srv.manager.load("m", identity_servable(), version=1)

        def worker():
            with TurboPredictClient(sock) as c:
                x = torch.randn(32, 3, 224, 224, device=dev)
                while time.monotonic() < stop:
                    try:
                        out = c.predict("m", {"images": x},
                                        output_device=dev, timeout=60)
                        assert out["images"].shape == x.shape
                        with lock:
                            stats["done"] += 1
                    except Exception:
                        with lock:
                            stats["errors"] += 1

        rss = [round(proc.memory_info().rss / 1e6, 1)]
        ts = [threading.Thread(target=worker) for _ in range(threads)]
        t0 = time.monotonic()
        for t in ts:
            t.start()
        while any(t.is_alive() for t in ts):
            time.sleep(5)
            rss.append(round(proc.memory_info().rss / 1e6, 1))
        for t in ts:
            t.join()
        elapsed = time.monotonic() - t0
    print(json.dumps({
        "seconds": round(elapsed, 1), "threads": threads,
        "requests": stats["done"], "errors": stats["errors"],
        "req_per_s": round(stats["done"] / elapsed, 1),
        "rss_mb": rss,
    }))


if __name__ == "__main__":
    main(int(sys.argv[1]) if len(sys.argv) > 1 else 60,
         int(sys.argv[2]) if len(sys.argv) > 2 else 4)
