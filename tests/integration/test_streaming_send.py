"""Differential tests for the streaming (skeleton + regions) send path.

The streaming path must put byte-identical gRPC messages on the wire as
the buffered path, in both directions, including through python-grpcio
peers whose 64KB flow-control windows force the DataMessageWriter's
window-wait loop. (North-star overlap: BASELINE.json "hipMemcpyAsync
overlaps protobuf encode and the gRPC send" — the CPU-reachable half is
host-region zero-copy send; the DMA half is pinned by tests/gpu.)
"""
import threading
from concurrent import futures

import grpc
import numpy as np
import pytest
import torch

pytest.importorskip("min_tfs_client_amd._transport")
from min_tfs_client_amd import _transport as T  # noqa: E402
from min_tfs_client_amd.native_transport import StreamingReply  # noqa: E402
from min_tfs_client_amd.ops import require_native  # noqa: E402

native = require_native()

OPTS = [
    ("grpc.max_send_message_length", 1 << 30),
    ("grpc.max_receive_message_length", 1 << 30),
]


def _mixed_inputs():
    """small (<64KB, inlined into the skeleton) + large (host region)
    + medium tensors of several dtypes."""
    g = torch.Generator().manual_seed(7)
    return {
        "small": torch.randn(4, 3, generator=g),
        "large": torch.randn(64, 3, 224, 224, generator=g),  # ~38 MB
        "ids": torch.arange(128 * 512, dtype=torch.int32).reshape(128, 512),
        "flags": torch.zeros(2, dtype=torch.bool),
    }


def _streaming_parts(inputs, is_request=True, name="m", version=3,
                     signature="serving_default"):
    names = list(inputs.keys())
    tensors = [inputs[k] for k in names]
    return native.serialize_predict_streaming(
        is_request, name, version, signature, names, tensors)


def _buffered(inputs, is_request=True, name="m", version=3,
              signature="serving_default"):
    names = list(inputs.keys())
    tensors = [inputs[k] for k in names]
    fn = (native.serialize_predict_request if is_request
          else native.serialize_predict_response)
    return fn(name, version, signature, names, tensors, 1)


def test_streaming_regions_shape():
    inputs = _mixed_inputs()
    blob, regions, keepalive = _streaming_parts(inputs)
    # large + ids become host regions; small + flags are inlined
    assert len(regions) == 2
    offsets = [r[0] for r in regions]
    assert offsets == sorted(offsets)
    for off, nbytes, ptr, is_dev in regions:
        assert not is_dev
        assert ptr != 0
        assert off + nbytes <= len(blob)
    assert len(keepalive) == 2


def test_client_streaming_bytes_identical_via_grpcio_server():
    """native client streaming request -> grpcio server: the server must
    receive exactly the bytes the buffered serializer produces."""
    received = []

    class Handler(grpc.GenericRpcHandler):
        def service(self, hcd):
            def echo(req, ctx):
                received.append(req)
                return b"ok"
            return grpc.unary_unary_rpc_method_handler(
                echo, request_deserializer=lambda x: x,
                response_serializer=lambda x: x)

    srv = grpc.server(futures.ThreadPoolExecutor(max_workers=4),
                      options=OPTS)
    srv.add_generic_rpc_handlers((Handler(),))
    port = srv.add_insecure_port("127.0.0.1:0")
    srv.start()
    try:
        inputs = _mixed_inputs()
        blob, regions, keepalive = _streaming_parts(inputs)
        ch = T.GrpcChannel(f"127.0.0.1:{port}")
        try:
            resp = ch.call_streaming("/t.S/Echo", blob, list(regions), 30.0)
            assert bytes(resp) == b"ok"
        finally:
            ch.close()
        expect = _buffered(inputs)
        assert len(received) == 1
        assert received[0] == expect
    finally:
        srv.stop(0)


def test_server_streaming_reply_bytes_identical_via_grpcio_client():
    """native server StreamingReply -> grpcio client: response bytes must
    equal the buffered serializer's output."""
    outputs = _mixed_inputs()
    expect = _buffered(outputs, is_request=False)

    def handler(view):
        blob, regions, keepalive = _streaming_parts(
            outputs, is_request=False)
        return StreamingReply(blob, list(regions), keepalive)

    srv = T.GrpcServer("127.0.0.1:0", 2)
    srv.register_handler("/t.S/Pred", handler)
    addr = srv.start()
    try:
        ch = grpc.insecure_channel(addr, options=OPTS)
        stub = ch.unary_unary("/t.S/Pred",
                              request_serializer=lambda x: x,
                              response_deserializer=lambda x: x)
        got = stub(b"x", timeout=30)
        assert got == expect
        ch.close()
    finally:
        srv.stop()


def test_streaming_native_to_native_roundtrip():
    """turbo client (streaming default) -> native server with a
    NON-identity servable, so the server's python handler produces a
    StreamingReply: both streaming paths exercised in one round trip."""
    from min_tfs_client_amd.server import ModelServer, Servable

    inputs = {"images": torch.randn(32, 3, 224, 224)}
    with ModelServer(address="127.0.0.1:0", raw_predict=True) as server:
        server.manager.load(
            "double", Servable(lambda d: {"images": d["images"] * 2}),
            version=1)
        from min_tfs_client_amd.turbo import TurboPredictClient
        with TurboPredictClient(server.address, backend="native") as c:
            out = c.predict("double", inputs, timeout=30)
            assert torch.equal(out["images"], inputs["images"] * 2)
            # explicit non-streaming must agree
            out2 = c.predict("double", inputs, timeout=30, streaming=False)
            assert torch.equal(out2["images"], inputs["images"] * 2)


def test_streaming_concurrent_calls():
    """8 threads x streaming calls through one channel: frames from
    different streams interleave on one connection without corruption."""
    srv = T.GrpcServer("127.0.0.1:0", 8)
    srv.register_handler("/t.S/Echo", lambda v: bytes(v))
    addr = srv.start()
    try:
        inputs = {"x": torch.randn(8, 3, 224, 224)}
        blob, regions, keepalive = _streaming_parts(inputs)
        expect = _buffered(inputs)
        ch = T.GrpcChannel(addr)
        errs = []

        def worker():
            try:
                for _ in range(4):
                    r = ch.call_streaming("/t.S/Echo", blob,
                                          list(regions), 30.0)
                    if bytes(r) != expect:
                        errs.append("mismatch")
            except Exception as e:  # noqa: BLE001
                errs.append(repr(e))

        ts = [threading.Thread(target=worker) for _ in range(8)]
        for t in ts:
            t.start()
        for t in ts:
            t.join()
        ch.close()
        assert errs == []
    finally:
        srv.stop()


def test_call_streaming_parsed_no_gpu_fallback():
    """On a machine without a GPU, parse_device requests must degrade to
    (None, raw_buf) — never error — so callers can always pass a device
    and fall back to the ordinary parse."""
    import torch
    if torch.cuda.is_available():
        pytest.skip("covered by tests/gpu on GPU machines")
    srv = T.GrpcServer("127.0.0.1:0", 2)
    srv.register_handler("/t.S/Echo", lambda v: bytes(v))
    addr = srv.start()
    try:
        inputs = {"x": torch.randn(512, 512)}  # 1MB+, host region
        blob, regions, keep = _streaming_parts(inputs)
        ch = T.GrpcChannel(addr)
        try:
            outs, raw = ch.call_streaming_parsed(
                "/t.S/Echo", blob, list(regions), 0, 30.0)
            assert outs is None
            assert bytes(raw) == _buffered(inputs)
        finally:
            ch.close()
    finally:
        srv.stop()


def test_register_handler_parsed_no_gpu_fallback():
    """Server-side: a parsed-registered handler on a GPU-less machine
    still gets called (with None, None) and serves correctly."""
    import torch
    if torch.cuda.is_available():
        pytest.skip("covered by tests/gpu on GPU machines")
    calls = []

    def handler(view, spec, outs):
        calls.append((spec, outs))
        return bytes(view)

    srv = T.GrpcServer("127.0.0.1:0", 2)
    srv.register_handler_parsed("/t.S/Echo", handler, 0)
    addr = srv.start()
    try:
        inputs = {"x": torch.randn(512, 512)}
        blob, regions, keep = _streaming_parts(inputs)
        ch = T.GrpcChannel(addr)
        try:
            got = bytes(ch.call_streaming("/t.S/Echo", blob,
                                          list(regions), 30.0))
            assert got == _buffered(inputs)
        finally:
            ch.close()
        assert calls and calls[0] == (None, None)
    finally:
        srv.stop()


def test_call_streaming_parsed_deadline(tmp_path):
    """Deadline on a parse-ahead call: DEADLINE_EXCEEDED surfaces, the
    channel and server both stay usable."""
    import time as _time

    def slow(view):
        _time.sleep(1.0)
        return bytes(view)

    srv = T.GrpcServer("127.0.0.1:0", 2)
    srv.register_handler("/t.S/Slow", slow)
    srv.register_handler("/t.S/Echo", lambda v: bytes(v))
    addr = srv.start()
    try:
        inputs = {"x": torch.randn(512, 512)}
        blob, regions, keep = _streaming_parts(inputs)
        ch = T.GrpcChannel(addr)
        try:
            with pytest.raises(T.NativeRpcError) as ei:
                ch.call_streaming_parsed("/t.S/Slow", blob, list(regions),
                                         0, 0.05)
            assert ei.value.code_int == 4  # DEADLINE_EXCEEDED
            # channel still healthy afterwards
            outs, raw = ch.call_streaming_parsed(
                "/t.S/Echo", blob, list(regions), 0, 30.0)
            assert bytes(raw) == _buffered(inputs)
        finally:
            ch.close()
    finally:
        srv.stop()
