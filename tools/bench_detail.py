#!/usr/bin/env python3
"""Latency decomposition of the turbo Predict round trip.

Times each stage separately so the RTT is attributable (SURVEY §7 'honest
benchmarking' requirement): HIP pack/serialize, wire parse, server echo,
raw gRPC transport, full client predict.
"""
import json
import os
import statistics
import sys
import time

_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, _ROOT)

import torch  # noqa: E402

from min_tfs_client_amd import _native as native  # noqa: E402
from min_tfs_client_amd.server import ModelServer, identity_servable  # noqa: E402
from min_tfs_client_amd.turbo import TurboPredictClient  # noqa: E402


def timeit(fn, reps=20, warmup=5):
    for _ in range(warmup):
        fn()
    ts = []
    for _ in range(reps):
        t0 = time.perf_counter()
        fn()
        ts.append(time.perf_counter() - t0)
    return statistics.median(ts) * 1e3  # ms


def main():
    has_gpu = torch.cuda.is_available()
    dev = "cuda:0" if has_gpu else "cpu"
    x = torch.randn(32, 3, 224, 224, device=dev)
    nbytes = x.numel() * 4
    results = {"payload_mb": round(nbytes / 1e6, 2), "gpu": has_gpu}

    # 1. serialize (pack + skeleton + D2H into wire bytes)
    results["serialize_staged_ms"] = timeit(
        lambda: native.serialize_predict_request("m", -1, "", ["x"], [x], 0))
    results["serialize_pageable_ms"] = timeit(
        lambda: native.serialize_predict_request("m", -1, "", ["x"], [x], 1))
    blob = native.serialize_predict_request("m", -1, "", ["x"], [x], 0)
    results["wire_bytes_mb"] = round(len(blob) / 1e6, 2)

    # 2. server-side echo (parse + host memcpy + reserialize)
    results["echo_ms"] = timeit(lambda: native.echo_predict(blob))
    resp = native.echo_predict(blob)

    # 3. parse response
    results["parse_to_cpu_ms"] = timeit(
        lambda: native.parse_predict_response(resp, "cpu", 0))
    if has_gpu:
        results["parse_to_gpu_staged_ms"] = timeit(
            lambda: native.parse_predict_response(resp, dev, 0))
        results["parse_to_gpu_pageable_ms"] = timeit(
            lambda: native.parse_predict_response(resp, dev, 1))

    # 4. host memcpy bandwidth reference
    a = bytearray(nbytes)
    b = bytes(nbytes)
    def host_copy():
        a[:] = b
    results["host_memcpy_ms"] = timeit(host_copy)

    # 4b. streaming serialize: skeleton only, payloads stay in place
    def ser_stream():
        parts = native.serialize_predict_streaming(True, "m", -1, "",
                                                   ["x"], [x])
        if has_gpu:
            torch.cuda.current_stream().synchronize()
        return parts
    results["serialize_streaming_skeleton_ms"] = timeit(ser_stream)

    # 5. raw gRPC round trip (pre-built blob, decode skipped)
    sock = f"unix:///tmp/mi355x_detail_{os.getpid()}.sock"
    with ModelServer(address=sock, raw_predict=True) as srv:
        srv.manager.load("m", identity_servable(), version=1)
        client = TurboPredictClient(sock)
        rpc = client._predict
        results["grpc_rtt_prebuilt_ms"] = timeit(lambda: rpc(blob, 30))
        if client.backend == "native":
            sb, sregions, skeep = client._serialize_streaming(
                "m", {"x": x}, None, "")
            results["grpc_rtt_streaming_prebuilt_ms"] = timeit(
                lambda: rpc.call_streaming(sb, list(sregions), 30))
        # tiny request for the protocol floor
        tiny = native.serialize_predict_request(
            "m", -1, "", ["x"], [torch.zeros(1)], 0)
        results["grpc_rtt_tiny_ms"] = timeit(lambda: rpc(tiny, 30))
        # full client predict, streaming vs buffered send
        results["predict_full_ms"] = timeit(
            lambda: client.predict("m", {"x": x}, output_device=dev), reps=20)
        if client.backend == "native":
            results["predict_full_buffered_ms"] = timeit(
                lambda: client.predict("m", {"x": x}, output_device=dev,
                                       streaming=False), reps=20)
        client.close()

    print(json.dumps(results, indent=1))


if __name__ == "__main__":
    main()
