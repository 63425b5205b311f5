"""Interop + protocol tests for the native C++ gRPC transport.

The transport (ops/csrc/grpc_transport.cpp) is a from-scratch HTTP/2 +
HPACK implementation; these tests pin it against python-grpcio — an
independent gRPC implementation — in BOTH directions, which is the
protocol-conformance oracle (reference behavior: the client interops with
a real tensorflow_model_server, reference actions.yml:48).
"""
import threading
from concurrent import futures

import grpc
import numpy as np
import pytest
import torch

pytest.importorskip("min_tfs_client_amd._transport")
from min_tfs_client_amd import _transport as T  # noqa: E402

OPTS = [
    ("grpc.max_send_message_length", 1 << 30),
    ("grpc.max_receive_message_length", 1 << 30),
]
BIG = b"\xa5\x5a\xff\x00" * (19 * 1024 * 1024 // 4)  # ~19MB


@pytest.fixture
def native_echo_server():
    srv = T.GrpcServer("127.0.0.1:0", 4)
    srv.register_handler("/t.S/Echo", lambda v: bytes(v))

    def fail(view):
        e = Exception("boom")
        e.grpc_code = 3
        e.grpc_details = "bad arg: payload rejected"
        raise e

    srv.register_handler("/t.S/Fail", fail)
    addr = srv.start()
    yield addr
    srv.stop()


@pytest.fixture
def grpcio_echo_server():
    class Handler(grpc.GenericRpcHandler):
        def service(self, hcd):
            if hcd.method == "/t.S/Echo":
                return grpc.unary_unary_rpc_method_handler(
                    lambda req, ctx: req,
                    request_deserializer=lambda x: x,
                    response_serializer=lambda x: x)
            if hcd.method == "/t.S/Slow":
                import time

                def slow(req, ctx):
                    time.sleep(5)
                    return req

                return grpc.unary_unary_rpc_method_handler(
                    slow, request_deserializer=lambda x: x,
                    response_serializer=lambda x: x)
            return None

    srv = grpc.server(futures.ThreadPoolExecutor(max_workers=8),
                      options=OPTS)
    srv.add_generic_rpc_handlers((Handler(),))
    port = srv.add_insecure_port("127.0.0.1:0")
    srv.start()
    yield f"127.0.0.1:{port}"
    srv.stop(0)


# ---------------------------------------------------------------------------
# direction 1: standard grpcio client -> native server
# ---------------------------------------------------------------------------

def test_grpcio_client_to_native_server(native_echo_server):
    ch = grpc.insecure_channel(native_echo_server, options=OPTS)
    stub = ch.unary_unary("/t.S/Echo", request_serializer=lambda x: x,
                          response_deserializer=lambda x: x)
    assert stub(b"x" * 100, timeout=10) == b"x" * 100
    # 19MB exercises flow control against grpcio's 4MB windows + frames
    assert stub(BIG, timeout=60) == BIG
    ch.close()


def test_grpcio_client_error_status(native_echo_server):
    ch = grpc.insecure_channel(native_echo_server, options=OPTS)
    stub = ch.unary_unary("/t.S/Fail", request_serializer=lambda x: x,
                          response_deserializer=lambda x: x)
    with pytest.raises(grpc.RpcError) as err:
        stub(b"x", timeout=10)
    assert err.value.code() == grpc.StatusCode.INVALID_ARGUMENT
    assert "payload rejected" in err.value.details()
    # unknown method -> UNIMPLEMENTED
    stub2 = ch.unary_unary("/t.S/Nope", request_serializer=lambda x: x,
                           response_deserializer=lambda x: x)
    with pytest.raises(grpc.RpcError) as err:
        stub2(b"x", timeout=10)
    assert err.value.code() == grpc.StatusCode.UNIMPLEMENTED
    ch.close()


def test_grpcio_concurrent_streams(native_echo_server):
    ch = grpc.insecure_channel(native_echo_server, options=OPTS)
    stub = ch.unary_unary("/t.S/Echo", request_serializer=lambda x: x,
                          response_deserializer=lambda x: x)
    payloads = [bytes([i]) * (50000 + i) for i in range(24)]
    futs = [stub.future(p, timeout=30) for p in payloads]
    for p, f in zip(payloads, futs):
        assert f.result() == p
    ch.close()


# ---------------------------------------------------------------------------
# direction 2: native client -> standard grpcio server
# ---------------------------------------------------------------------------

def test_native_client_to_grpcio_server(grpcio_echo_server):
    ch = T.GrpcChannel(grpcio_echo_server)
    out = ch.call("/t.S/Echo", b"ping", 10.0)
    assert bytes(memoryview(out)) == b"ping"
    # big payload: our sender must respect grpcio's 4MB frames + windows
    out = ch.call("/t.S/Echo", BIG, 60.0)
    assert bytes(memoryview(out)) == BIG
    ch.close()


def test_native_client_grpcio_error(grpcio_echo_server):
    ch = T.GrpcChannel(grpcio_echo_server)
    with pytest.raises(T.NativeRpcError) as err:
        ch.call("/t.S/Missing", b"x", 10.0)
    assert err.value.code_int == 12  # UNIMPLEMENTED
    ch.close()


def test_native_client_deadline(grpcio_echo_server):
    ch = T.GrpcChannel(grpcio_echo_server)
    with pytest.raises(T.NativeRpcError) as err:
        ch.call("/t.S/Slow", b"x", 0.5)
    assert err.value.code_int == 4  # DEADLINE_EXCEEDED
    # channel must stay usable after a timed-out call
    out = ch.call("/t.S/Echo", b"after-timeout", 10.0)
    assert bytes(memoryview(out)) == b"after-timeout"
    ch.close()


# ---------------------------------------------------------------------------
# native <-> native
# ---------------------------------------------------------------------------

def test_native_pair_unix_socket(tmp_path):
    sock = f"unix://{tmp_path}/t.sock"
    srv = T.GrpcServer(sock, 4)
    srv.register_handler("/t.S/Echo", lambda v: bytes(v))
    addr = srv.start()
    ch = T.GrpcChannel(addr)
    assert bytes(memoryview(ch.call("/t.S/Echo", BIG, 60.0))) == BIG
    ch.close()
    srv.stop()


def test_native_pair_pipelined(native_echo_server):
    ch = T.GrpcChannel(native_echo_server)
    ids = [ch.start("/t.S/Echo", bytes([i]) * 10000, 30.0)
           for i in range(16)]
    for i, call_id in enumerate(ids):
        assert bytes(memoryview(ch.wait(call_id, 30.0))) == bytes([i]) * 10000
    ch.close()


def test_native_pair_parallel_callers(native_echo_server):
    ch = T.GrpcChannel(native_echo_server)
    errors = []

    def worker(seed):
        try:
            for i in range(20):
                payload = bytes([seed, i]) * 5000
                out = ch.call("/t.S/Echo", payload, 30.0)
                assert bytes(memoryview(out)) == payload
        except Exception as e:  # noqa: BLE001
            errors.append(e)

    threads = [threading.Thread(target=worker, args=(s,)) for s in range(8)]
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    assert not errors, errors
    ch.close()


def test_empty_message_round_trip(native_echo_server):
    ch = T.GrpcChannel(native_echo_server)
    assert bytes(memoryview(ch.call("/t.S/Echo", b"", 10.0))) == b""
    ch.close()


# ---------------------------------------------------------------------------
# hpack unit vectors (RFC 7541 Appendix C)
# ---------------------------------------------------------------------------

def test_hpack_huffman_via_grpcio_headers(native_echo_server):
    """grpcio huffman-codes its header literals; a successful RPC proves
    the native decoder handles huffman strings + dynamic table entries."""
    ch = grpc.insecure_channel(native_echo_server, options=OPTS)
    # long, compressible method path exercises huffman + CONTINUATION-free
    # multi-header blocks; metadata adds literal headers
    stub = ch.unary_unary("/t.S/Echo", request_serializer=lambda x: x,
                          response_deserializer=lambda x: x)
    out, _call = stub.with_call(
        b"hdrs", timeout=10,
        metadata=(("x-custom-header", "value-with-dashes-and-12345"),
                  ("another-header-name", "AAAA" * 40)))
    assert out == b"hdrs"
    ch.close()


# ---------------------------------------------------------------------------
# full stack: ModelServer(native) under load from mixed clients
# ---------------------------------------------------------------------------

def test_mixed_clients_one_native_server():
    from min_tfs_client_amd.client import TensorServingClient
    from min_tfs_client_amd.server import ModelServer, identity_servable
    from min_tfs_client_amd.tensors import tensor_proto_to_ndarray
    from min_tfs_client_amd.turbo import TurboPredictClient

    with ModelServer(port=0) as srv:
        srv.manager.load("m", identity_servable(), version=1)
        x = torch.arange(24, dtype=torch.float32).reshape(2, 3, 4)
        with TurboPredictClient(srv.address) as turbo:
            out = turbo.predict("m", {"x": x})
            assert torch.equal(out["x"], x)
        host, port = srv.address.split(":")
        proto_client = TensorServingClient(host, int(port))
        resp = proto_client.predict_request(
            "m", {"x": x.numpy()}, timeout=20)
        np.testing.assert_array_equal(
            tensor_proto_to_ndarray(resp.outputs["x"]), x.numpy())
        proto_client.close()
        # echo fast path was exercised and accounted
        q = srv.metrics.latency_quantiles("predict")
        assert q.get("count", 0) >= 1
