"""Shared-memory local transport: in-process and cross-process, wire-byte
fidelity, error paths, concurrent connections."""
import multiprocessing
import os

import numpy as np
import pytest
import torch

pytest.importorskip("min_tfs_client_amd._native",
                    reason="_native extension not built")

from min_tfs_client_amd.server import (  # noqa: E402
    ModelManager,
    ModelServer,
    Servable,
    identity_servable,
)
from min_tfs_client_amd.shm import ShmListener, ShmPredictClient  # noqa: E402


@pytest.fixture()
def listener(tmp_path):
    mgr = ModelManager()
    mgr.load("m", identity_servable(), version=1)

    def double(inputs):
        return {k: torch.as_tensor(np.asarray(v)) * 2
                for k, v in inputs.items()}

    mgr.load("double", Servable(double), version=1)
    with ShmListener(mgr, str(tmp_path / "hs")) as lst:
        yield lst


def test_shm_roundtrip(listener):
    with ShmPredictClient(listener.dir, slot_bytes=8 << 20) as c:
        x = torch.randn(4, 3, 16, 16)
        out = c.predict("m", {"x": x})
        assert torch.equal(out["x"], x)


def test_shm_multiple_dtypes(listener):
    with ShmPredictClient(listener.dir, slot_bytes=8 << 20) as c:
        ins = {"a": torch.randn(3, 5),
               "b": torch.arange(6, dtype=torch.int64),
               "c": torch.tensor([1.5, -2.0], dtype=torch.bfloat16)}
        out = c.predict("m", ins)
        for k, v in ins.items():
            assert torch.equal(out[k], v)


def test_shm_servable_and_version_errors(listener):
    with ShmPredictClient(listener.dir, slot_bytes=4 << 20) as c:
        out = c.predict("double", {"x": torch.ones(3)})
        assert torch.equal(out["x"], torch.full((3,), 2.0))
        with pytest.raises(RuntimeError, match="Servable not found"):
            c.predict("missing", {"x": torch.ones(1)})
        # connection survives errors
        out = c.predict("double", {"x": torch.ones(2)})
        assert torch.equal(out["x"], torch.full((2,), 2.0))


def test_shm_slot_overflow_raises(listener):
    with ShmPredictClient(listener.dir, slot_bytes=1 << 20) as c:
        with pytest.raises(RuntimeError, match="slot too small"):
            c.predict("m", {"x": torch.randn(1 << 20)})  # 4MB > 1MB slot


def test_shm_concurrent_connections(listener):
    import threading
    results = {}

    def worker(i):
        with ShmPredictClient(listener.dir, slot_bytes=4 << 20) as c:
            x = torch.full((16,), float(i))
            for _ in range(20):
                out = c.predict("m", {"x": x})
                assert torch.equal(out["x"], x)
            results[i] = True

    threads = [threading.Thread(target=worker, args=(i,)) for i in range(4)]
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    assert all(results.get(i) for i in range(4))


def _client_proc(hs_dir, q):
    import sys
    root = os.path.dirname(os.path.dirname(
        os.path.dirname(os.path.abspath(__file__))))
    sys.path.insert(0, root)
    import torch as _t
    from min_tfs_client_amd.shm import ShmPredictClient as C
    try:
        with C(hs_dir, slot_bytes=8 << 20) as c:
            x = _t.arange(128, dtype=_t.float32).reshape(8, 16)
            out = c.predict("m", {"x": x})
            q.put(bool(_t.equal(out["x"], x)))
    except Exception as e:  # noqa: BLE001
        q.put(f"FAIL: {e}")


@pytest.mark.timeout(180)
def test_shm_cross_process(tmp_path):
    """The real deployment shape: server (with gRPC + shm listener) in this
    process, client in a spawned process."""
    hs = str(tmp_path / "hs")
    with ModelServer(port=0, raw_predict=True, shm_handshake_dir=hs) as srv:
        srv.manager.load("m", identity_servable(), version=1)
        ctx = multiprocessing.get_context("spawn")
        q = ctx.Queue()
        p = ctx.Process(target=_client_proc, args=(hs, q))
        p.start()
        result = q.get(timeout=150)
        p.join(timeout=30)
        assert result is True, result


def test_shm_segments_cleaned_up(listener):
    c = ShmPredictClient(listener.dir, slot_bytes=1 << 20)
    req_path, resp_path = c._req.path, c._resp.path
    assert os.path.exists(req_path)
    c.close()
    assert not os.path.exists(req_path)
    assert not os.path.exists(resp_path)


def test_shm_connect_timeout_no_server(tmp_path, monkeypatch):
    import time
    created = []
    from min_tfs_client_amd import shm as shm_mod
    orig = shm_mod._Segment

    class Tracking(orig):
        def __init__(self, path, size, create):
            super().__init__(path, size, create)
            if create:
                created.append(path)

    monkeypatch.setattr(shm_mod, "_Segment", Tracking)
    t0 = time.monotonic()
    with pytest.raises(TimeoutError, match="did not attach"):
        ShmPredictClient(str(tmp_path / "nobody"), slot_bytes=1 << 20,
                         connect_timeout=0.5)
    assert time.monotonic() - t0 < 5
    # exactly this client's segments were unlinked on the failed connect
    assert len(created) == 2
    for path in created:
        assert not os.path.exists(path)


def test_predict_after_close_raises(listener):
    c = ShmPredictClient(listener.dir, slot_bytes=1 << 20)
    c.close()
    with pytest.raises(RuntimeError, match="closed"):
        c.predict("m", {"x": torch.ones(1)})
