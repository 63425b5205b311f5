#!/usr/bin/env python3
"""Peak 1-GPU throughput hunt: K client PROCESSES (no shared GIL), each
pipelining requests against one shared server."""
import json
import multiprocessing
import os
import sys
import time

_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, _ROOT)


def _client_proc(address, n_requests, pipeline, q):
    sys.path.insert(0, _ROOT)
    import torch
    from min_tfs_client_amd.turbo import TurboPredictClient
    dev = "cuda:0" if torch.cuda.is_available() else "cpu"
    with TurboPredictClient(address, num_channels=2) as c:
        x = torch.randn(32, 3, 224, 224, device=dev)
        blob_futs = []
        t0 = time.perf_counter()
        done = 0
        inflight = [c.predict_future("m", {"images": x})
                    for _ in range(pipeline)]
        submitted = pipeline
        while done < n_requests:
            fut, dec = inflight.pop(0)
            fut.result()
            done += 1
            if submitted < n_requests:
                inflight.append(c.predict_future("m", {"images": x}))
                submitted += 1
        q.put((done, time.perf_counter() - t0))


def main(n_procs=3, per_proc=150, pipeline=8, pair_servers=0):
    from min_tfs_client_amd.server import ModelServer, identity_servable
    ctx = multiprocessing.get_context("spawn")
    servers = []
    socks = []
    n_servers = n_procs if pair_servers else 1
    for i in range(n_servers):
        sock = f"unix:///tmp/mi355x_mp_{os.getpid()}_{i}.sock"
        srv = ModelServer(address=sock, raw_predict=True, max_workers=32)
        srv.manager.load("m", identity_servable(), version=1)
        srv.start()
        servers.append(srv)
        socks.append(sock)
    try:
        q = ctx.Queue()
        procs = [ctx.Process(target=_client_proc,
                             args=(socks[i % len(socks)], per_proc,
                                   pipeline, q))
                 for i in range(n_procs)]
        for p in procs:
            p.start()
        results = [q.get(timeout=600) for _ in procs]
        for p in procs:
            p.join(timeout=30)
    finally:
        for srv in servers:
            srv.stop(0)
    total = sum(r[0] for r in results)
    # clients start within ~a second of each other (spawn+import outside
    # their timed regions); the slowest child's request-loop time is the
    # honest aggregate window
    window = max(r[1] for r in results)
    print(json.dumps({
        "client_procs": n_procs, "pipeline": pipeline,
        "paired_servers": bool(pair_servers),
        "total_requests": total,
        "window_s": round(window, 2),
        "agg_req_per_s": round(total / window, 1),
        "agg_GBps": round(total * 2 * 19.27e-3 / window, 2),
    }))


if __name__ == "__main__":
    main(*(int(a) for a in sys.argv[1:5]))
