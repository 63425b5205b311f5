"""Pins for round-2 fixes: native complex decode, packed-field bounds,
reload-through-source consistency, native multi-listener."""
import os

import numpy as np
import pytest
import torch

from min_tfs_client_amd.ops import require_native
from min_tfs_client_amd.wire import messages as pb

native = pytest.importorskip("min_tfs_client_amd._native") and require_native()


# ---------------------------------------------------------------------------
# complex typed-field decode (VERDICT r01 missing #5; tensor.proto:59-61)
# ---------------------------------------------------------------------------

def _resp_with(tensor_proto_fill):
    resp = pb.PredictResponse()
    t = resp.outputs["z"]
    tensor_proto_fill(t)
    return resp.SerializeToString()


def test_native_scomplex_typed_decode():
    vals = np.array([1 + 2j, -3.5 + 0.25j, 7 - 1j], dtype=np.complex64)

    def fill(t):
        t.dtype = 8  # DT_COMPLEX64
        t.tensor_shape.dim.add().size = 3
        for z in vals:
            t.scomplex_val.extend([float(z.real), float(z.imag)])

    _s, outs, _ = native.parse_predict_response(_resp_with(fill), "cpu", 1)
    assert outs["z"].dtype == torch.complex64
    assert torch.equal(outs["z"], torch.from_numpy(vals.copy()))


def test_native_dcomplex_typed_decode_with_fill():
    # repeat-last-fill repeats the last complex PAIR (tensor.cc:487-527)
    def fill(t):
        t.dtype = 18  # DT_COMPLEX128
        t.tensor_shape.dim.add().size = 4
        t.dcomplex_val.extend([1.0, -1.0, 2.5, 0.5])  # two values, 4 elems

    _s, outs, _ = native.parse_predict_response(_resp_with(fill), "cpu", 1)
    expect = torch.tensor([1 - 1j, 2.5 + 0.5j, 2.5 + 0.5j, 2.5 + 0.5j],
                          dtype=torch.complex128)
    assert torch.equal(outs["z"], expect)


def test_native_complex_roundtrip_via_turbo_codec():
    vals = torch.view_as_complex(torch.randn(5, 2)).to(torch.complex64)
    blob = native.serialize_predict_request("m", -1, "", ["z"], [vals], 1)
    _s, outs, _ = native.parse_predict_request(blob, "cpu", 1)
    assert torch.equal(outs["z"], vals)


# ---------------------------------------------------------------------------
# packed-field bounds: malformed lengths must error,
# not read out of bounds
# ---------------------------------------------------------------------------

def _tensor_entry_bytes(field: int, payload: bytes) -> bytes:
    """hand-rolls a PredictResponse{outputs{key:'x', value:TensorProto{
    <field>: packed payload}}} with a deliberately bad payload length."""
    tp = bytes([field << 3 | 2, len(payload)]) + payload
    entry = bytes([0x0A, 1]) + b"x" + bytes([0x12, len(tp)]) + tp
    return bytes([0x0A, len(entry)]) + entry


def test_truncated_packed_float_rejected():
    bad = _tensor_entry_bytes(5, b"\x00\x00\x80?\xff")  # 5 bytes: 4+1
    with pytest.raises(Exception, match="truncated packed float"):
        native.parse_predict_response(bad, "cpu", 1)


def test_truncated_packed_double_rejected():
    bad = _tensor_entry_bytes(6, b"\x01" * 7)
    with pytest.raises(Exception, match="truncated packed double"):
        native.parse_predict_response(bad, "cpu", 1)


# ---------------------------------------------------------------------------
# reload-through-source: standard TF layout
# (base_path/<version>/...) loads via ReloadConfig, and a polled model
# unloaded by reload can come back
# ---------------------------------------------------------------------------

def _reload_request(configs):
    req = pb.ReloadConfigRequest()
    for name, base in configs.items():
        c = req.config.model_config_list.config.add()
        c.name = name
        c.base_path = base
        c.model_platform = "tensorflow"
    return req


def test_reload_with_version_dirs_via_source(tmp_path):
    from min_tfs_client_amd.repository import FileSystemStoragePathSource
    from min_tfs_client_amd.server import (
        ModelManager,
        ModelServiceImpl,
        STATE_AVAILABLE,
    )
    for name in ("a", "b"):
        vdir = tmp_path / name / "3"
        vdir.mkdir(parents=True)
        (vdir / "identity").touch()
    manager = ModelManager()
    source = FileSystemStoragePathSource(manager, poll_wait_seconds=0)
    ms = ModelServiceImpl(manager, storage_source=source)
    resp = ms.HandleReloadConfigRequest(
        _reload_request({"a": str(tmp_path / "a"),
                         "b": str(tmp_path / "b")}), None)
    assert resp.status.error_code == pb.ErrorCode.OK
    assert manager.get("a") is not None  # version 3 resolved from dir
    sts = manager.version_statuses("a")
    assert sts == [(3, STATE_AVAILABLE, None)]
    # remove 'b' via reload; source bookkeeping must follow
    resp = ms.HandleReloadConfigRequest(
        _reload_request({"a": str(tmp_path / "a")}), None)
    assert resp.status.error_code == pb.ErrorCode.OK
    with pytest.raises(KeyError):
        manager.get("b")
    # re-adding 'b' reloads it (round-1 bug: stale _loaded kept it END)
    resp = ms.HandleReloadConfigRequest(
        _reload_request({"a": str(tmp_path / "a"),
                         "b": str(tmp_path / "b")}), None)
    assert resp.status.error_code == pb.ErrorCode.OK
    assert manager.get("b") is not None


def test_reload_factory_fallback_scans_versions(tmp_path):
    from min_tfs_client_amd.repository import default_loader
    from min_tfs_client_amd.server import ModelManager, ModelServiceImpl
    vdir = tmp_path / "m" / "7"
    vdir.mkdir(parents=True)
    (vdir / "identity").touch()
    manager = ModelManager()
    ms = ModelServiceImpl(manager, servable_factory=default_loader)
    resp = ms.HandleReloadConfigRequest(
        _reload_request({"m": str(tmp_path / "m")}), None)
    assert resp.status.error_code == pb.ErrorCode.OK
    assert manager.version_statuses("m")[0][0] == 7


# ---------------------------------------------------------------------------
# native server multi-listener (--grpc_socket_path analogue)
# ---------------------------------------------------------------------------

def test_native_server_extra_unix_listener(tmp_path):
    pytest.importorskip("min_tfs_client_amd._transport")
    from min_tfs_client_amd import _transport as T
    srv = T.GrpcServer("127.0.0.1:0", 2)
    sock = f"unix://{tmp_path}/extra.sock"
    srv.add_address(sock)
    srv.register_handler("/t.S/Echo", lambda v: bytes(v))
    tcp_addr = srv.start()
    try:
        for target in (tcp_addr, sock):
            ch = T.GrpcChannel(target)
            out = ch.call("/t.S/Echo", b"dual", 10.0)
            assert bytes(memoryview(out)) == b"dual"
            ch.close()
    finally:
        srv.stop()


def test_shm_handshake_rejects_non_shm_paths(tmp_path):
    """handshake naming a non-/dev/shm path must be
    rejected before the server maps (and would overwrite) it."""
    import json
    import time
    from min_tfs_client_amd.server import ModelManager
    from min_tfs_client_amd.shm import ShmListener
    victim = tmp_path / "victim.dat"
    victim.write_bytes(b"\x00" * 4096)
    hs_dir = tmp_path / "hs"
    hs_dir.mkdir()
    listener = ShmListener(ModelManager(), str(hs_dir))
    listener.start()
    try:
        (hs_dir / "evil.json").write_text(json.dumps(
            {"req": str(victim), "resp": str(victim)}))
        deadline = time.time() + 5
        while (hs_dir / "evil.json").exists() and time.time() < deadline:
            time.sleep(0.05)
        time.sleep(0.3)  # give a (wrong) attach a chance to write the ack
        assert victim.read_bytes() == b"\x00" * 4096  # untouched
    finally:
        listener.stop()
