#!/usr/bin/env python3
"""Peak shm-transport aggregate: N connections (threads) x sequential
19MB echo requests, server in a separate process."""
import json
import multiprocessing
import os
import sys
import threading
import time

_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, _ROOT)


def _server(shm_dir, ready, stop):
    sys.path.insert(0, _ROOT)
    from min_tfs_client_amd.server import ModelServer, identity_servable
    with ModelServer(address=f"unix:///tmp/shm_scale_{os.getpid()}.sock",
                     raw_predict=True, shm_handshake_dir=shm_dir) as srv:
        srv.manager.load("m", identity_servable(), version=1)
        ready.set()
        stop.wait()


def main(conns=4, per_conn=150):
    import torch
    from min_tfs_client_amd.shm import ShmPredictClient
    dev = "cuda:0" if torch.cuda.is_available() else "cpu"
    shm_dir = f"/tmp/shm_scale_hs_{os.getpid()}"
    ctx = multiprocessing.get_context("spawn")
    ready, stop = ctx.Event(), ctx.Event()
    p = ctx.Process(target=_server, args=(shm_dir, ready, stop), daemon=True)
    p.start()
    assert ready.wait(300)
    lat = []
    lock = threading.Lock()

    def worker():
        with ShmPredictClient(shm_dir, slot_bytes=64 << 20) as c:
            x = torch.randn(32, 3, 224, 224, device=dev)
            for _ in range(5):
                c.predict("m", {"images": x}, output_device=dev)
            local = []
            for _ in range(per_conn):
                t0 = time.perf_counter()
                c.predict("m", {"images": x}, output_device=dev)
                local.append(time.perf_counter() - t0)
            with lock:
                lat.extend(local)

    threads = [threading.Thread(target=worker) for _ in range(conns)]
    t0 = time.perf_counter()
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    wall = time.perf_counter() - t0
    stop.set()
    p.join(timeout=10)
    import statistics
    print(json.dumps({
        "connections": conns, "total_requests": conns * per_conn,
        "agg_req_per_s": round(conns * per_conn / wall, 1),
        "agg_GBps": round(conns * per_conn * 2 * 19.27e-3 / wall, 2),
        "p50_ms": round(statistics.median(lat) * 1e3, 3),
        "p99_ms": round(sorted(lat)[int(len(lat) * 0.99) - 1] * 1e3, 3),
    }))


if __name__ == "__main__":
    main(*(int(a) for a in sys.argv[1:3]))
