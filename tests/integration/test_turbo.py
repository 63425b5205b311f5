"""Turbo (raw-bytes) serving path integration, CPU: raw server + turbo
client, and cross-interop with the python-protobuf client — the wire bytes
are the same protocol, so each client must work against each server mode."""
import numpy as np
import pytest
import torch

from min_tfs_client_amd.client import TensorServingClient
from min_tfs_client_amd.server import ModelServer, Servable, identity_servable
from min_tfs_client_amd.tensors import tensor_proto_to_ndarray

pytest.importorskip(
    "min_tfs_client_amd._native",
    reason="_native extension not built")

from min_tfs_client_amd.turbo import TurboPredictClient  # noqa: E402


@pytest.fixture(scope="module")
def raw_server():
    with ModelServer(port=0, raw_predict=True) as srv:
        srv.manager.load("default", identity_servable(), version=1)

        def double_fn(inputs):
            return {k: v * 2 for k, v in inputs.items()}

        srv.manager.load("double", Servable(double_fn), version=1)
        yield srv


def test_turbo_client_echo(raw_server):
    with TurboPredictClient(raw_server.address) as client:
        x = torch.randn(4, 3, 8, 8)
        out = client.predict("default", {"images": x})
        assert torch.equal(out["images"], x)


def test_turbo_client_suffix_rename(raw_server):
    with TurboPredictClient(raw_server.address) as client:
        x = torch.ones(3)
        out = client.predict("default", {"float_input": x})
        assert torch.equal(out["float_output"], x)


def test_turbo_client_real_servable(raw_server):
    """Non-identity servable: parse -> torch fn -> serialize."""
    with TurboPredictClient(raw_server.address) as client:
        x = torch.arange(6, dtype=torch.float32)
        out = client.predict("double", {"x": x})
        assert torch.equal(out["x"], x * 2)


def test_turbo_client_not_found(raw_server):
    import grpc
    with TurboPredictClient(raw_server.address) as client:
        with pytest.raises(grpc.RpcError) as err:
            client.predict("missing", {"x": torch.zeros(1)})
        assert err.value.code() == grpc.StatusCode.NOT_FOUND


def test_proto_client_against_raw_server(raw_server):
    """A standard python-protobuf client must interoperate with the raw
    C++-codec server byte-for-byte."""
    host, port = raw_server.address.rsplit(":", 1)
    c = TensorServingClient(host, int(port))
    try:
        x = np.random.rand(2, 3).astype(np.float32)
        resp = c.predict_request("default", {"x": x}, model_version=1)
        np.testing.assert_array_equal(
            tensor_proto_to_ndarray(resp.outputs["x"]), x)
    finally:
        c.close()


def test_turbo_client_against_proto_server():
    """And the turbo client must interoperate with the python-protobuf
    server."""
    with ModelServer(port=0) as srv:
        srv.manager.load("default", identity_servable(), version=1)
        with TurboPredictClient(srv.address) as client:
            x = torch.randn(5, 2)
            out = client.predict("default", {"x": x})
            assert torch.equal(out["x"], x)


def test_turbo_unix_socket(tmp_path):
    sock = f"unix://{tmp_path}/turbo.sock"
    with ModelServer(address=sock, raw_predict=True) as srv:
        srv.manager.load("default", identity_servable(), version=1)
        with TurboPredictClient(sock) as client:
            x = torch.randn(16, 3, 32, 32)
            out = client.predict("default", {"images": x})
            assert torch.equal(out["images"], x)


def test_turbo_pipelined_futures(raw_server):
    with TurboPredictClient(raw_server.address) as client:
        futs = []
        xs = [torch.randn(4, 4) for _ in range(8)]
        for x in xs:
            futs.append((client.predict_future("default", {"x": x}), x))
        for (fut, decode), x in futs:
            out = decode(fut.result())
            assert torch.equal(out["x"], x)


def test_turbo_multi_input_bert_shapes(raw_server):
    with TurboPredictClient(raw_server.address) as client:
        ids = torch.randint(0, 30522, (4, 64), dtype=torch.int32)
        mask = torch.ones(4, 64, dtype=torch.int32)
        out = client.predict("default",
                             {"input_ids": ids, "attention_mask": mask})
        assert torch.equal(out["input_ids"], ids)
        assert torch.equal(out["attention_mask"], mask)


def test_predict_sharded(raw_server):
    with TurboPredictClient(raw_server.address, num_channels=2) as client:
        x = torch.randn(7, 3, 8, 8)  # uneven over 2 shards
        y = torch.arange(7, dtype=torch.int64)
        out = client.predict_sharded("default", {"x": x, "y": y}, shards=2)
        assert torch.equal(out["x"], x)
        assert torch.equal(out["y"], y)


def test_predict_sharded_fallbacks(raw_server):
    with TurboPredictClient(raw_server.address, num_channels=2) as client:
        # scalar input: falls back to plain predict
        out = client.predict_sharded("default", {"s": torch.tensor(2.5)},
                                     shards=4)
        assert out["s"].item() == 2.5
        # batch smaller than shard count
        out = client.predict_sharded("default", {"x": torch.ones(1, 3)},
                                     shards=4)
        assert torch.equal(out["x"], torch.ones(1, 3))


def test_turbo_version_label_via_raw_server(raw_server):
    """version_label travels through the C++ parse path too (ModelSpec
    field 4)."""
    raw_server.manager.load("lbl", identity_servable(), version=3)
    raw_server.manager.set_version_label("lbl", "prod", 3)
    from min_tfs_client_amd.wire import messages as pb
    from min_tfs_client_amd import _native as native
    req = pb.PredictRequest()
    req.model_spec.name = "lbl"
    req.model_spec.version_label = "prod"
    import numpy as np
    from min_tfs_client_amd.tensors import ndarray_to_tensor_proto
    req.inputs["x"].CopyFrom(ndarray_to_tensor_proto(
        np.ones(2, np.float32)))
    with TurboPredictClient(raw_server.address) as c:
        resp = c._predict(req.SerializeToString(), 30)
        _s, outs, _ = native.parse_predict_response(resp, "cpu", 1)
        assert torch.equal(outs["x"], torch.ones(2))


def test_async_client(raw_server):
    import asyncio
    from min_tfs_client_amd.aio import AsyncTurboPredictClient

    async def run():
        async with AsyncTurboPredictClient(raw_server.address) as c:
            xs = [torch.randn(4, 4) for _ in range(6)]
            outs = await asyncio.gather(
                *[c.predict("default", {"x": x}) for x in xs])
            for out, x in zip(outs, xs):
                assert torch.equal(out["x"], x)

    asyncio.run(run())


def test_async_client_both_backends(raw_server):
    import asyncio
    from min_tfs_client_amd.aio import AsyncTurboPredictClient

    async def run(backend):
        async with AsyncTurboPredictClient(raw_server.address,
                                           backend=backend) as c:
            assert c.backend == backend
            x = torch.randn(3, 5)
            out = await c.predict("default", {"x": x})
            assert torch.equal(out["x"], x)

    asyncio.run(run("native"))
    asyncio.run(run("grpcio"))


def test_deadline_exceeded():
    """gRPC deadline semantics: timeout is seconds, slow servables abort
    with DEADLINE_EXCEEDED (reference requests.py:49 passes timeout
    positionally the same way)."""
    import grpc
    import time as _time

    def slow(inputs):
        _time.sleep(2.0)
        return inputs

    with ModelServer(port=0, raw_predict=True) as srv:
        srv.manager.load("slow", Servable(slow), version=1)
        with TurboPredictClient(srv.address) as c:
            with pytest.raises(grpc.RpcError) as err:
                c.predict("slow", {"x": torch.ones(1)}, timeout=0.3)
            assert err.value.code() == grpc.StatusCode.DEADLINE_EXCEEDED


def test_client_bytes_metrics(raw_server):
    with TurboPredictClient(raw_server.address) as c:
        c.predict("default", {"x": torch.ones(100)})
        counters = c.metrics.counters()
        assert counters["bytes_total{direction='tx'}"] > 400
        assert counters["bytes_total{direction='rx'}"] > 400


def test_output_filter_raw_path(raw_server):
    """output_filter must be honored on the raw C++ path too (for the
    identity servable this disables the echo fast path)."""
    from min_tfs_client_amd.wire import messages as pb
    from min_tfs_client_amd import _native as native
    import numpy as np
    from min_tfs_client_amd.tensors import ndarray_to_tensor_proto
    req = pb.PredictRequest()
    req.model_spec.name = "default"
    req.inputs["a"].CopyFrom(ndarray_to_tensor_proto(np.ones(1, np.float32)))
    req.inputs["b"].CopyFrom(ndarray_to_tensor_proto(np.ones(1, np.float32)))
    req.output_filter.append("a")
    with TurboPredictClient(raw_server.address) as c:
        resp_bytes = c._predict(req.SerializeToString(), 30)
    resp = pb.PredictResponse.FromString(
        bytes(memoryview(resp_bytes)))  # native returns a buffer view
    assert sorted(resp.outputs) == ["a"]


def test_zero_copy_parse(raw_server):
    import warnings
    with TurboPredictClient(raw_server.address) as c:
        x = torch.randn(8, 16)
        with warnings.catch_warnings():
            warnings.simplefilter("ignore")  # frombuffer non-writable note
            out = c.predict("default", {"x": x}, zero_copy=True)
        assert torch.equal(out["x"], x)
        # it's a view over the response buffer, not a copy
        assert out["x"].data_ptr() != 0


def test_multi_target_client():
    """A client over a fleet of server instances: round-robin +
    shard-parallel across targets."""
    with ModelServer(port=0, raw_predict=True) as s1, \
            ModelServer(port=0, raw_predict=True) as s2:
        s1.manager.load("m", identity_servable(), version=1)
        s2.manager.load("m", identity_servable(), version=1)
        with TurboPredictClient([s1.address, s2.address]) as c:
            assert len(c._stubs) == 2
            x = torch.randn(8, 4)
            for _ in range(3):
                out = c.predict_sharded("m", {"x": x}, shards=2)
                assert torch.equal(out["x"], x)
            # both servers actually served requests
            n1 = s1.metrics.latency_quantiles("predict").get("count", 0)
            n2 = s2.metrics.latency_quantiles("predict").get("count", 0)
            assert n1 > 0 and n2 > 0


def test_predict_sharded_error_cancels(raw_server):
    """A failing shard rpc propagates and cancels the siblings."""
    import grpc
    with TurboPredictClient(raw_server.address, num_channels=2) as client:
        with pytest.raises(grpc.RpcError):
            client.predict_sharded("missing_model",
                                   {"x": torch.randn(8, 2)}, shards=2)
        # client still usable
        out = client.predict_sharded("default",
                                     {"x": torch.ones(4, 2)}, shards=2)
        assert torch.equal(out["x"], torch.ones(4, 2))


def test_predict_transform_cpu_fallback(raw_server):
    """transform= applies layout/dtype conversion before packing (CPU
    fallback path; the GPU path runs the fused CDNA4 kernel)."""
    with TurboPredictClient(raw_server.address) as c:
        x = torch.randn(2, 3, 4, 5, dtype=torch.bfloat16)
        out = c.predict("default", {"images": x},
                        transform={"images": ("nhwc", torch.float32)})
        ref = x.permute(0, 2, 3, 1).contiguous().to(torch.float32)
        assert out["images"].shape == (2, 4, 5, 3)
        assert torch.equal(out["images"], ref)
