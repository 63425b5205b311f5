"""Adversarial robustness of the from-scratch C++ HTTP/2 server.

The transport (ops/csrc/grpc_transport.cpp, h2core.h, hpack.h) parses
untrusted bytes off a socket; every malformed input must at worst kill
THAT connection — the server must keep serving well-formed peers. After
each abuse pattern a control request through a real grpcio client must
still succeed.
"""
import os
import random
import socket
import struct
import time

import grpc
import pytest

pytest.importorskip("min_tfs_client_amd._transport")
from min_tfs_client_amd import _transport as T  # noqa: E402

PREFACE = b"PRI * HTTP/2.0\r\n\r\nSM\r\n\r\n"


@pytest.fixture(scope="module")
def server():
    srv = T.GrpcServer("127.0.0.1:0", 4)
    srv.register_handler("/t.S/Echo", lambda v: bytes(v))
    addr = srv.start()
    host, port = addr.rsplit(":", 1)
    yield (host, int(port), addr)
    srv.stop()


def _control_request_ok(addr):
    """A well-formed request through grpcio still round-trips."""
    ch = grpc.insecure_channel(addr)
    try:
        stub = ch.unary_unary("/t.S/Echo", request_serializer=lambda x: x,
                              response_deserializer=lambda x: x)
        assert stub(b"healthy", timeout=10) == b"healthy"
    finally:
        ch.close()


def _raw(host, port):
    s = socket.create_connection((host, port), timeout=5)
    s.settimeout(5)
    return s


def _frame(ftype, flags, stream, payload=b""):
    return (struct.pack("!I", len(payload))[1:]
            + bytes([ftype, flags]) + struct.pack("!I", stream) + payload)


def test_bad_preface(server):
    host, port, addr = server
    s = _raw(host, port)
    s.sendall(b"GET / HTTP/1.1\r\nHost: x\r\n\r\n")
    # server must drop the connection, not hang or die
    s.settimeout(5)
    assert s.recv(65536) in (b"",) or True  # any response then close
    s.close()
    _control_request_ok(addr)


def test_garbage_after_preface(server):
    host, port, addr = server
    s = _raw(host, port)
    s.sendall(PREFACE + os.urandom(512))
    time.sleep(0.1)
    s.close()
    _control_request_ok(addr)


def test_truncated_frame_then_close(server):
    host, port, addr = server
    s = _raw(host, port)
    # declare a 1000-byte SETTINGS frame, send 3 bytes, vanish
    s.sendall(PREFACE + _frame(4, 0, 0, b"")[:9][:3])
    s.close()
    _control_request_ok(addr)


def test_oversized_frame_length(server):
    host, port, addr = server
    s = _raw(host, port)
    # 16MB-1 declared DATA on an unopened stream
    hdr = struct.pack("!I", 0xFFFFFF)[1:] + bytes([0, 0]) \
        + struct.pack("!I", 7)
    s.sendall(PREFACE + _frame(4, 0, 0, b"") + hdr)
    time.sleep(0.1)
    s.close()
    _control_request_ok(addr)


def test_garbage_hpack_headers(server):
    host, port, addr = server
    s = _raw(host, port)
    block = os.urandom(64)
    s.sendall(PREFACE + _frame(4, 0, 0, b"")
              + _frame(1, 0x4 | 0x1, 1, block))  # END_HEADERS|END_STREAM
    time.sleep(0.1)
    s.close()
    _control_request_ok(addr)


def test_data_overruns_grpc_length(server):
    host, port, addr = server
    # valid-ish HEADERS via hpack never-indexed literals is complex to
    # hand-roll; instead overrun the gRPC message length on a stream the
    # server accepted from a real client preamble. Simplest: declare a
    # 1-byte message then send 100 bytes of DATA.
    host_, port_, _ = server
    s = _raw(host_, port_)
    payload = b"\x00" + struct.pack("!I", 1) + b"x" * 100
    s.sendall(PREFACE + _frame(4, 0, 0, b"")
              + _frame(1, 0x4, 1, b"")         # empty header block
              + _frame(0, 0x1, 1, payload))    # DATA + END_STREAM
    time.sleep(0.1)
    s.close()
    _control_request_ok(addr)


def test_abrupt_disconnect_mid_message(server):
    host, port, addr = server
    s = _raw(host, port)
    payload = b"\x00" + struct.pack("!I", 1 << 20) + b"y" * 1000
    s.sendall(PREFACE + _frame(4, 0, 0, b"")
              + _frame(1, 0x4, 1, b"")
              + _frame(0, 0, 1, payload))
    s.close()  # vanish with 1MB-declared message 1KB-sent
    _control_request_ok(addr)


def test_random_frame_fuzz(server):
    """200 random frame sequences; the server must survive them all."""
    host, port, addr = server
    rng = random.Random(1234)
    for _ in range(200):
        s = _raw(host, port)
        try:
            s.sendall(PREFACE)
            for _ in range(rng.randint(1, 5)):
                ftype = rng.randint(0, 12)
                flags = rng.randint(0, 255)
                stream = rng.randint(0, 3)
                payload = os.urandom(rng.randint(0, 200))
                s.sendall(_frame(ftype, flags, stream, payload))
        except OSError:
            pass  # server already dropped us: fine
        finally:
            s.close()
    _control_request_ok(addr)


def test_many_connections_churn(server):
    """open/close 100 connections rapidly (thread-per-conn server must
    reap, not leak into unbounded threads)."""
    host, port, addr = server
    for i in range(100):
        s = _raw(host, port)
        if i % 3 == 0:
            s.sendall(PREFACE)
        s.close()
    _control_request_ok(addr)


def test_huge_declared_message_rejected(server):
    """A 5-byte gRPC prefix declaring a 2GB message must be refused
    (RESOURCE_EXHAUSTED semantics) without allocating the 2GB."""
    host, port, addr = server
    s = _raw(host, port)
    # handshake enough for the server to accept a stream: preface,
    # SETTINGS, empty HEADERS (END_HEADERS), then DATA with huge prefix
    prefix = b"\x00" + struct.pack("!I", (2 << 30) - 1)
    s.sendall(PREFACE + _frame(4, 0, 0, b"")
              + _frame(1, 0x4, 1, b"")
              + _frame(0, 0, 1, prefix + b"xx"))
    time.sleep(0.2)
    s.close()
    _control_request_ok(addr)


def test_connection_churn_releases_fds(server):
    """Dead connections must be reaped while the server runs: 300
    connect/close cycles may not grow this process's fd count
    (regression for the EMFILE hang found by tools/soak_churn.py)."""
    host, port, addr = server

    def nfds():
        return len(os.listdir("/proc/self/fd"))

    # settle, then measure
    for _ in range(10):
        _raw(host, port).close()
    time.sleep(0.3)
    before = nfds()
    for i in range(300):
        s = _raw(host, port)
        if i % 2 == 0:
            s.sendall(PREFACE)
        s.close()
    time.sleep(1.0)
    after = nfds()
    assert after - before < 40, (before, after)
    _control_request_ok(addr)
