#!/usr/bin/env python3
"""A/B matrix over gRPC channel options, malloc tuning and pipeline depth
for the 19 MB Predict round trip. Drives the transport-tuning decisions."""
import json
import os
import statistics
import sys
import time

_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, _ROOT)

import grpc  # noqa: E402
import torch  # noqa: E402

from min_tfs_client_amd import _native as native  # noqa: E402
from min_tfs_client_amd.server import ModelServer, identity_servable  # noqa: E402
from min_tfs_client_amd.utils.allocator import tune_malloc  # noqa: E402

PATH = "/tensorflow.serving.PredictionService/Predict"
BASE = [("grpc.max_send_message_length", 1 << 30),
        ("grpc.max_receive_message_length", 1 << 30)]

VARIANTS = {
    "base": [],
    "frame16m": [("grpc.http2.max_frame_size", 16 * 1024 * 1024 - 1)],
    "frame16m+nobdp": [
        ("grpc.http2.max_frame_size", 16 * 1024 * 1024 - 1),
        ("grpc.http2.bdp_probe", 0),
        ("grpc.http2.lookahead_bytes", 64 << 20),
    ],
    "frame16m+wbuf": [
        ("grpc.http2.max_frame_size", 16 * 1024 * 1024 - 1),
        ("grpc.http2.write_buffer_size", 64 << 20),
    ],
}


def rtt(stub, blob, reps=25, warmup=5, depth=1):
    for _ in range(warmup):
        stub(blob, 30)
    ts = []
    if depth == 1:
        for _ in range(reps):
            t0 = time.perf_counter()
            stub(blob, 30)
            ts.append(time.perf_counter() - t0)
        return statistics.median(ts) * 1e3, None
    # pipelined: keep `depth` in flight, measure aggregate throughput
    t0 = time.perf_counter()
    inflight = [stub.future(blob, 30) for _ in range(depth)]
    done = 0
    total = reps * depth
    while done < total:
        f = inflight.pop(0)
        f.result()
        done += 1
        if done + len(inflight) < total:
            inflight.append(stub.future(blob, 30))
    el = time.perf_counter() - t0
    return el / total * 1e3, total / el  # ms-per-req, req/s


def main():
    if "--tuned-malloc" in sys.argv:
        print("malloc tuned:", tune_malloc())
    dev = "cuda:0" if torch.cuda.is_available() else "cpu"
    x = torch.randn(32, 3, 224, 224, device=dev)
    blob = native.serialize_predict_request("m", -1, "", ["x"], [x], 0)
    print(json.dumps({"payload_mb": round(len(blob) / 1e6, 2)}))

    sock = f"unix:///tmp/mi355x_ab_{os.getpid()}.sock"
    with ModelServer(address=sock, raw_predict=True) as srv:
        srv.manager.load("m", identity_servable(), version=1)
        for name, extra in VARIANTS.items():
            ch = grpc.insecure_channel(sock, options=BASE + extra)
            stub = ch.unary_unary(PATH, request_serializer=lambda b: b,
                                  response_deserializer=lambda b: b)
            ms, _ = rtt(stub, blob)
            out = {"variant": name, "rtt_ms": round(ms, 3)}
            for depth in (2, 4):
                mspr, rps = rtt(stub, blob, reps=15, depth=depth)
                out[f"pipe{depth}_ms_per_req"] = round(mspr, 3)
            ch.close()
            print(json.dumps(out))

    # serialize cost with tuned malloc (fresh allocations each call)
    t = []
    for _ in range(20):
        t0 = time.perf_counter()
        native.serialize_predict_request("m", -1, "", ["x"], [x], 0)
        t.append(time.perf_counter() - t0)
    print(json.dumps({"serialize_staged_ms_after_tune":
                      round(statistics.median(t) * 1e3, 3)}))


if __name__ == "__main__":
    main()
