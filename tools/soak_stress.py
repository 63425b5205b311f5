#!/usr/bin/env python3
"""Combined-stress soak: concurrent sequential/pipelined/sharded clients,
echo + GPU model + batched servables, every response verified."""
import json
import os
import sys
import threading
import time
import warnings

_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, _ROOT)

import torch  # noqa: E402

from min_tfs_client_amd.batching import BatchingServable  # noqa: E402
from min_tfs_client_amd.models import resnet50_servable  # noqa: E402
from min_tfs_client_amd.server import ModelServer, Servable, identity_servable  # noqa: E402
from min_tfs_client_amd.shm import ShmPredictClient  # noqa: E402
from min_tfs_client_amd.turbo import TurboPredictClient  # noqa: E402


def main(seconds=300):
    dev = "cuda:0" if torch.cuda.is_available() else "cpu"
    sock = f"unix:///tmp/mi355x_stress_{os.getpid()}.sock"
    stop_at = time.monotonic() + seconds
    stats = {}
    lock = threading.Lock()
    errors = []

    def record(name, ok, err=None):
        with lock:
            s = stats.setdefault(name, {"ok": 0, "fail": 0})
            s["ok" if ok else "fail"] += 1
            if err and len(errors) < 5:
                errors.append(f"{name}: {err}")

    def scaled(inputs):
        return {k: v * 2 for k, v in inputs.items()}

    shm_dir = f"/tmp/mi355x_stress_hs_{os.getpid()}"
    with ModelServer(address=sock, raw_predict=True, device=dev,
                     max_workers=32, shm_handshake_dir=shm_dir) as srv:
        srv.manager.load("echo", identity_servable(), version=1)
        srv.manager.load("scale", Servable(scaled), version=1)
        if dev != "cpu":
            srv.manager.load(
                "resnet", BatchingServable(
                    resnet50_servable(dev), max_batch_size=32,
                    batch_timeout_s=0.002,
                    allowed_batch_sizes=[8, 16, 32]), version=1)

        def seq_worker(wid):
            with TurboPredictClient(sock) as c:
                g = torch.Generator().manual_seed(wid)
                while time.monotonic() < stop_at:
                    x = torch.randn(8, 3, 64, 64, generator=g).to(dev)
                    try:
                        out = c.predict("scale", {"x": x},
                                        output_device=dev, timeout=60)
                        record("seq", torch.allclose(out["x"], x * 2))
                    except Exception as e:  # noqa: BLE001
                        record("seq", False, e)

        def pipe_worker():
            with TurboPredictClient(sock, num_channels=2) as c:
                x = torch.randn(16, 3, 64, 64, device=dev)
                while time.monotonic() < stop_at:
                    futs = [c.predict_future("echo", {"x": x})
                            for _ in range(4)]
                    try:
                        for fut, dec in futs:
                            out = dec(fut.result(), output_device=dev)
                            record("pipe", torch.equal(out["x"], x))
                    except Exception as e:  # noqa: BLE001
                        record("pipe", False, e)

        def shard_worker():
            with TurboPredictClient(sock, num_channels=4) as c:
                x = torch.randn(33, 3, 32, 32, device=dev)  # uneven
                while time.monotonic() < stop_at:
                    try:
                        out = c.predict_sharded("echo", {"x": x}, shards=4,
                                                output_device=dev,
                                                timeout=60)
                        record("shard", torch.equal(out["x"], x))
                    except Exception as e:  # noqa: BLE001
                        record("shard", False, e)

        def model_worker():
            if dev == "cpu":
                return
            with TurboPredictClient(sock) as c:
                x = torch.randn(8, 3, 224, 224, device=dev)
                while time.monotonic() < stop_at:
                    try:
                        out = c.predict("resnet", {"images": x},
                                        output_device=dev, timeout=120)
                        ok = (out["logits"].shape == (8, 1000)
                              and bool(torch.isfinite(out["logits"]).all()))
                        record("model", ok)
                    except Exception as e:  # noqa: BLE001
                        record("model", False, e)

        def shm_worker(wid):
            with ShmPredictClient(shm_dir, slot_bytes=64 << 20) as c:
                g = torch.Generator().manual_seed(1000 + wid)
                while time.monotonic() < stop_at:
                    x = torch.randn(16, 3, 96, 96, generator=g).to(dev)
                    try:
                        out = c.predict("scale", {"x": x},
                                        output_device=dev, timeout=60)
                        record("shm", torch.allclose(out["x"], x * 2))
                    except Exception as e:  # noqa: BLE001
                        record("shm", False, e)

        def zc_worker():
            with TurboPredictClient(sock) as c:
                x = torch.randn(4, 256)
                with warnings.catch_warnings():
                    warnings.simplefilter("ignore")
                    while time.monotonic() < stop_at:
                        try:
                            out = c.predict("echo", {"x": x},
                                            zero_copy=True, timeout=60)
                            record("zerocopy", torch.equal(out["x"], x))
                        except Exception as e:  # noqa: BLE001
                            record("zerocopy", False, e)

        workers = ([threading.Thread(target=seq_worker, args=(i,))
                    for i in range(2)] +
                   [threading.Thread(target=shm_worker, args=(i,))
                    for i in range(2)] +
                   [threading.Thread(target=pipe_worker),
                    threading.Thread(target=shard_worker),
                    threading.Thread(target=model_worker),
                    threading.Thread(target=zc_worker)])
        t0 = time.monotonic()
        for t in workers:
            t.start()
        for t in workers:
            t.join()
        elapsed = time.monotonic() - t0
    total_ok = sum(s["ok"] for s in stats.values())
    total_fail = sum(s["fail"] for s in stats.values())
    print(json.dumps({"seconds": round(elapsed, 1), "patterns": stats,
                      "total_ok": total_ok, "total_fail": total_fail,
                      "sample_errors": errors}))
    return 1 if total_fail else 0


if __name__ == "__main__":
    sys.exit(main(int(sys.argv[1]) if len(sys.argv) > 1 else 300))
