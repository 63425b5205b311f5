"""BatchingServable semantics (BatchingSession analogue)."""
import threading
import time

import numpy as np
import pytest

from min_tfs_client_amd.batching import BatchingServable
from min_tfs_client_amd.server import Servable


class CountingInner(Servable):
    def __init__(self):
        super().__init__(self._fn)
        self.calls = []
        self.lock = threading.Lock()

    def _fn(self, inputs):
        with self.lock:
            self.calls.append(
                {k: (np.asarray(v).shape[0] if np.asarray(v).ndim else 0)
                 for k, v in inputs.items()})
        return {k: np.asarray(v) * 2 for k, v in inputs.items()}


def test_merges_concurrent_requests():
    inner = CountingInner()
    b = BatchingServable(inner, max_batch_size=8, batch_timeout_s=0.05)
    results = {}

    def call(i):
        x = np.full((2, 3), i, dtype=np.float32)
        results[i] = b({"x": x})

    threads = [threading.Thread(target=call, args=(i,)) for i in range(4)]
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    # each caller got its own doubled slice
    for i in range(4):
        np.testing.assert_array_equal(
            results[i]["x"], np.full((2, 3), i * 2, dtype=np.float32))
    # and fewer inner calls than requests (merging happened)
    assert len(inner.calls) < 4
    assert sum(c["x"] for c in inner.calls) == 8


def test_full_batch_fires_immediately():
    inner = CountingInner()
    b = BatchingServable(inner, max_batch_size=4, batch_timeout_s=10.0)
    results = {}

    def call(i):
        results[i] = b({"x": np.full((2,), i, dtype=np.float32)})

    t0 = time.monotonic()
    threads = [threading.Thread(target=call, args=(i,)) for i in range(2)]
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    # 2 requests x batch 2 = max_batch_size 4 -> no timeout wait
    assert time.monotonic() - t0 < 5.0
    assert len(inner.calls) >= 1


def test_allowed_batch_sizes_padding():
    inner = CountingInner()
    b = BatchingServable(inner, max_batch_size=8,
                         batch_timeout_s=0.01,
                         allowed_batch_sizes=[4, 8])
    out = b({"x": np.ones((3, 2), dtype=np.float32)})
    assert out["x"].shape == (3, 2)  # caller sees its own rows
    assert inner.calls[-1]["x"] == 4  # padded to allowed size


def test_allowed_batch_sizes_validation():
    inner = CountingInner()
    with pytest.raises(ValueError, match="final allowed_batch_sizes"):
        BatchingServable(inner, max_batch_size=8,
                         allowed_batch_sizes=[2, 4])


def test_oversize_request_rejected():
    inner = CountingInner()
    b = BatchingServable(inner, max_batch_size=2, batch_timeout_s=0.01)
    with pytest.raises(ValueError, match="max_batch_size"):
        b({"x": np.ones((5, 1), dtype=np.float32)})


def test_scalar_request_unbatched():
    inner = CountingInner()
    b = BatchingServable(inner, max_batch_size=4, batch_timeout_s=0.01)
    out = b({"x": np.float32(3.0)})
    assert out["x"] == 6.0


def test_inner_error_propagates():
    def bad(inputs):
        raise RuntimeError("boom")

    b = BatchingServable(Servable(bad), max_batch_size=4,
                         batch_timeout_s=0.01)
    with pytest.raises(RuntimeError, match="boom"):
        b({"x": np.ones((1, 1), dtype=np.float32)})


def test_through_server():
    """Batching behind the gRPC server: concurrent clients are merged."""
    from min_tfs_client_amd.server import ModelServer
    from min_tfs_client_amd.client import TensorServingClient
    from min_tfs_client_amd.tensors import tensor_proto_to_ndarray

    inner = CountingInner()
    batched = BatchingServable(inner, max_batch_size=16,
                               batch_timeout_s=0.05)
    with ModelServer(port=0, max_workers=8) as srv:
        srv.manager.load("m", batched, version=1)
        results = {}

        def call(i):
            c = TensorServingClient("127.0.0.1", srv.port)
            try:
                x = np.full((2, 2), i, dtype=np.float32)
                r = c.predict_request("m", {"x": x})
                results[i] = tensor_proto_to_ndarray(r.outputs["x"])
            finally:
                c.close()

        threads = [threading.Thread(target=call, args=(i,))
                   for i in range(6)]
        for t in threads:
            t.start()
        for t in threads:
            t.join()
        for i in range(6):
            np.testing.assert_array_equal(
                results[i], np.full((2, 2), i * 2, dtype=np.float32))
        assert len(inner.calls) < 6


def test_unload_closes_batcher_thread():
    import time
    from min_tfs_client_amd.server import ModelManager
    inner = CountingInner()
    b = BatchingServable(inner, max_batch_size=4, batch_timeout_s=0.01)
    mgr = ModelManager()
    mgr.load("m", b, version=1)
    assert b._thread.is_alive()
    mgr.unload("m", 1)
    b._thread.join(timeout=5)
    assert not b._thread.is_alive()


def test_closed_batcher_rejects_and_drains():
    inner = CountingInner()
    b = BatchingServable(inner, max_batch_size=4, batch_timeout_s=0.01)
    b.close()
    b._thread.join(timeout=5)
    with pytest.raises(RuntimeError, match="shut down"):
        b({"x": np.ones((1, 1), dtype=np.float32)})
