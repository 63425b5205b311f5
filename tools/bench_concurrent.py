#!/usr/bin/env python3
"""Concurrent-clients serving benchmark: aggregate throughput of K client
threads against a real model servable, batched vs unbatched — quantifies
what the BatchingServable buys on MI355X (merging batch-4 requests into
batch-32+ launches fills the 256-CU chip)."""
import json
import os
import statistics
import sys
import threading
import time

_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, _ROOT)

import torch  # noqa: E402

from min_tfs_client_amd.batching import BatchingServable  # noqa: E402
from min_tfs_client_amd.models import resnet50_servable  # noqa: E402
from min_tfs_client_amd.server import ModelServer  # noqa: E402
from min_tfs_client_amd.turbo import TurboPredictClient  # noqa: E402


def run_load(address, model, n_threads, per_thread, batch, device):
    lat = []
    lock = threading.Lock()

    def worker():
        with TurboPredictClient(address) as c:
            x = torch.randn(batch, 3, 224, 224, device=device)
            local = []
            for _ in range(per_thread):
                t0 = time.perf_counter()
                c.predict(model, {"images": x}, timeout=120)
                local.append(time.perf_counter() - t0)
            with lock:
                lat.extend(local)

    threads = [threading.Thread(target=worker) for _ in range(n_threads)]
    t0 = time.perf_counter()
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    elapsed = time.perf_counter() - t0
    total_images = n_threads * per_thread * batch
    return {
        "req_per_s": round(n_threads * per_thread / elapsed, 1),
        "images_per_s": round(total_images / elapsed, 1),
        "p50_ms": round(statistics.median(lat) * 1e3, 2),
        "p99_ms": round(sorted(lat)[int(0.99 * len(lat)) - 1] * 1e3, 2),
    }


def main():
    device = "cuda:0" if torch.cuda.is_available() else "cpu"
    sock = f"unix:///tmp/mi355x_conc_{os.getpid()}.sock"
    with ModelServer(address=sock, raw_predict=True, device=device,
                     max_workers=32) as srv:
        srv.manager.load("plain", resnet50_servable(device), version=1)
        # allowed_batch_sizes pins the merged shapes so MIOpen's per-shape
        # kernel search happens once per size, not per request mix (the
        # reason TF-Serving has the same knob, batching_session.h:92-97)
        srv.manager.load(
            "batched",
            BatchingServable(resnet50_servable(device), max_batch_size=64,
                             batch_timeout_s=0.003,
                             allowed_batch_sizes=[8, 16, 32, 64]),
            version=1)
        n_threads, per_thread, batch = 8, 12, 4
        if device == "cpu":
            n_threads, per_thread = 2, 3
        # shape warmup: one request per allowed batch size
        with TurboPredictClient(sock) as wc:
            for b in (4, 8, 16, 32, 64):
                x = torch.randn(b, 3, 224, 224, device=device)
                wc.predict("plain", {"images": x}, timeout=300)
                wc.predict("batched", {"images": x}, timeout=300)
        for model in ("plain", "batched"):
            # concurrency warmup
            run_load(sock, model, 2, 2, batch, device)
            r = run_load(sock, model, n_threads, per_thread, batch, device)
            r.update({"model": model, "threads": n_threads,
                      "batch_per_request": batch, "device": device})
            if model == "batched":
                r["batches_run"] = srv.manager.get("batched").batches_run
            print(json.dumps(r))


if __name__ == "__main__":
    main()
