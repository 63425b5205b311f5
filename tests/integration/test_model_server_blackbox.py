"""Black-box test of the standalone model server binary: launch
``python -m min_tfs_client_amd.model_server`` as a subprocess and speak
gRPC + REST to it (the reference's own server test pattern,
model_servers/tensorflow_model_server_test.py:86-392)."""
import json
import os
import signal
import socket
import subprocess
import sys
import time
import urllib.request

import numpy as np
import pytest

_ROOT = os.path.dirname(os.path.dirname(
    os.path.dirname(os.path.abspath(__file__))))


def _free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


@pytest.fixture(scope="module")
def server_proc(tmp_path_factory):
    base = tmp_path_factory.mktemp("models")
    vdir = base / "default" / "00000001"
    vdir.mkdir(parents=True)
    (vdir / "identity").touch()
    grpc_port, rest_port = _free_port(), _free_port()
    env = dict(os.environ)
    env["PYTHONPATH"] = _ROOT
    proc = subprocess.Popen(
        [sys.executable, "-m", "min_tfs_client_amd.model_server",
         "--port", str(grpc_port), "--rest_api_port", str(rest_port),
         "--model_name", "default",
         "--model_base_path", str(base / "default"),
         "--file_system_poll_wait_seconds", "0.5"],
        cwd=_ROOT, env=env,
        stdout=subprocess.PIPE, stderr=subprocess.STDOUT)
    # wait for readiness via REST status
    deadline = time.time() + 60
    ready = False
    while time.time() < deadline:
        try:
            body = urllib.request.urlopen(
                f"http://127.0.0.1:{rest_port}/v1/models/default",
                timeout=2).read()
            if b"AVAILABLE" in body:
                ready = True
                break
        except Exception:
            time.sleep(0.3)
    if not ready:
        proc.send_signal(signal.SIGTERM)
        out = proc.communicate(timeout=10)[0]
        pytest.fail(f"server did not become ready:\n{out.decode()[-2000:]}")
    yield {"grpc": grpc_port, "rest": rest_port, "proc": proc,
           "base": base}
    proc.send_signal(signal.SIGTERM)
    try:
        proc.wait(timeout=15)
    except subprocess.TimeoutExpired:
        proc.kill()


def test_grpc_predict(server_proc):
    from min_tfs_client.requests import TensorServingClient
    from min_tfs_client.tensors import tensor_proto_to_ndarray
    c = TensorServingClient("127.0.0.1", server_proc["grpc"])
    try:
        x = np.random.rand(2, 4).astype(np.float32)
        resp = c.predict_request("default", {"x": x}, timeout=20)
        np.testing.assert_array_equal(
            tensor_proto_to_ndarray(resp.outputs["x"]), x)
    finally:
        c.close()


def test_grpc_turbo_predict(server_proc):
    pytest.importorskip("min_tfs_client_amd._native")
    import torch
    from min_tfs_client_amd.turbo import TurboPredictClient
    with TurboPredictClient(f"127.0.0.1:{server_proc['grpc']}") as c:
        x = torch.randn(3, 5)
        out = c.predict("default", {"x": x}, timeout=20)
        assert torch.equal(out["x"], x)


def test_grpc_model_status(server_proc):
    from min_tfs_client.requests import TensorServingClient
    c = TensorServingClient("127.0.0.1", server_proc["grpc"])
    try:
        st = c.model_status_request("default")
        assert st.model_version_status[0].state == 30
    finally:
        c.close()


def test_rest_predict(server_proc):
    req = urllib.request.Request(
        f"http://127.0.0.1:{server_proc['rest']}/v1/models/default:predict",
        data=json.dumps({"instances": [[1.0, 2.0]]}).encode(),
        headers={"Content-Type": "application/json"})
    body = json.loads(urllib.request.urlopen(req, timeout=10).read())
    assert body == {"predictions": [[1.0, 2.0]]}


def test_hot_version_swap(server_proc):
    """A new version dir appearing on disk is picked up by the polling
    source; latest policy swaps v1 -> v2 live."""
    from min_tfs_client.requests import TensorServingClient
    v2 = server_proc["base"] / "default" / "00000002"
    v2.mkdir(parents=True)
    (v2 / "identity").touch()
    c = TensorServingClient("127.0.0.1", server_proc["grpc"])
    try:
        deadline = time.time() + 30
        states = {}
        while time.time() < deadline:
            st = c.model_status_request("default")
            states = {s.version: s.state for s in st.model_version_status}
            if states.get(2) == 30:
                break
            time.sleep(0.5)
        assert states.get(2) == 30, states
        assert states.get(1) == 50  # unloaded by latest-1 policy
    finally:
        c.close()
