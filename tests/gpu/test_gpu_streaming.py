"""Device-region streaming send on real hardware: the pinned-staging DMA
chunks must reach the wire byte-identical to the buffered serializer, in
both directions (client request regions, server StreamingReply regions).
This is the DMA half of the north-star overlap ("hipMemcpyAsync overlaps
the gRPC send"); tests/integration/test_streaming_send.py pins the
CPU-reachable half."""
import pytest
import torch

pytestmark = pytest.mark.gpu

from min_tfs_client_amd import _transport as T  # noqa: E402
from min_tfs_client_amd.native_transport import StreamingReply  # noqa: E402
from min_tfs_client_amd.ops import require_native  # noqa: E402
from min_tfs_client_amd.server import ModelServer, Servable  # noqa: E402
from min_tfs_client_amd.turbo import TurboPredictClient  # noqa: E402

native = require_native()
DEV = "cuda:0"


def _inputs():
    g = torch.Generator(device=DEV).manual_seed(11)
    return {
        # > 2 staging chunks (4 MiB each) to exercise the DMA pipeline
        "images": torch.randn(64, 3, 224, 224, device=DEV, generator=g),
        "small": torch.randn(4, 4, device=DEV, generator=g),
        "ids": torch.randint(0, 30000, (128, 512), dtype=torch.int32,
                             device=DEV, generator=g),
    }


def _streaming(inputs, is_request=True):
    names = list(inputs.keys())
    tensors = [inputs[k] for k in names]
    parts = native.serialize_predict_streaming(
        is_request, "m", 1, "serving_default", names, tensors)
    torch.cuda.current_stream().synchronize()
    return parts


def _buffered(inputs, is_request=True):
    names = list(inputs.keys())
    tensors = [inputs[k] for k in names]
    fn = (native.serialize_predict_request if is_request
          else native.serialize_predict_response)
    return fn("m", 1, "serving_default", names, tensors, 1)


def test_device_regions_detected():
    blob, regions, keepalive = _streaming(_inputs())
    assert len(regions) == 3  # every cuda tensor becomes a device region
    assert all(r[3] for r in regions)


def test_client_device_streaming_bytes_identical():
    srv = T.GrpcServer("127.0.0.1:0", 2)
    received = []
    srv.register_handler("/t.S/Echo",
                         lambda v: received.append(bytes(v)) or b"ok")
    addr = srv.start()
    try:
        inputs = _inputs()
        blob, regions, keepalive = _streaming(inputs)
        ch = T.GrpcChannel(addr)
        try:
            assert bytes(ch.call_streaming("/t.S/Echo", blob,
                                           list(regions), 60.0)) == b"ok"
        finally:
            ch.close()
        assert received[0] == _buffered(inputs)
    finally:
        srv.stop()


def test_server_device_streaming_reply_bytes_identical():
    outputs = _inputs()
    expect = _buffered(outputs, is_request=False)

    def handler(view):
        blob, regions, keepalive = _streaming(outputs, is_request=False)
        return StreamingReply(blob, list(regions), keepalive)

    srv = T.GrpcServer("127.0.0.1:0", 2)
    srv.register_handler("/t.S/Pred", handler)
    addr = srv.start()
    try:
        ch = T.GrpcChannel(addr)
        try:
            got = bytes(ch.call("/t.S/Pred", b"x", 60.0))
        finally:
            ch.close()
        assert got == expect
    finally:
        srv.stop()


def test_end_to_end_gpu_streaming_predict(tmp_path):
    """Full turbo round trip with device tensors through both streaming
    paths (client request + server non-identity StreamingReply)."""
    sock = f"unix://{tmp_path}/stream.sock"
    with ModelServer(address=sock, raw_predict=True, device=DEV) as srv:
        srv.manager.load(
            "double", Servable(lambda d: {k: v * 2 for k, v in d.items()}),
            version=1)
        with TurboPredictClient(srv.address, backend="native") as c:
            x = torch.randn(64, 3, 224, 224, device=DEV)
            out = c.predict("double", {"images": x}, output_device=DEV,
                            timeout=60)
            assert out["images"].is_cuda
            assert torch.equal(out["images"], x * 2)


def test_receive_side_parsed_unpack():
    """call_streaming_parsed: the reader H2Ds tensor_content spans while
    the response streams in; results must equal the ordinary parse."""
    outputs = _inputs()
    payload = _buffered(outputs, is_request=False)  # response layout
    srv = T.GrpcServer("127.0.0.1:0", 2)
    srv.register_handler("/t.S/Fixed", lambda v: payload)
    addr = srv.start()
    try:
        req_blob, req_regions, keep = _streaming(
            {"q": torch.zeros(4, device=DEV)})
        ch = T.GrpcChannel(addr)
        try:
            outs, raw = ch.call_streaming_parsed(
                "/t.S/Fixed", req_blob, list(req_regions), 0, 60.0)
            assert outs is not None, "canonical response must parse-ahead"
            _spec, ref, _f = native.parse_predict_response(
                bytes(raw), "cuda:0", 1)
            assert set(outs) == set(ref)
            for k in ref:
                assert outs[k].is_cuda
                assert outs[k].dtype == ref[k].dtype
                assert torch.equal(outs[k], ref[k]), k
        finally:
            ch.close()
    finally:
        srv.stop()


def test_receive_side_parse_fallback_non_canonical():
    """A response with typed *_val fields (no tensor_content) must return
    None from the prospector, and the raw buffer must parse normally."""
    from min_tfs_client_amd.wire import messages as pb

    resp = pb.PredictResponse()
    resp.model_spec.name = "m"
    tp = resp.outputs["x"]
    tp.dtype = 1
    tp.tensor_shape.dim.add().size = 3
    tp.float_val.extend([1.0, 2.0, 3.0])
    # pad so the message crosses the 1MB pinned threshold and the
    # prospector actually engages before bailing on float_val
    big = resp.outputs["pad"]
    big.dtype = 1
    big.tensor_shape.dim.add().size = 1 << 19
    big.tensor_content = b"\x00" * (4 << 19)
    payload = resp.SerializeToString()

    srv = T.GrpcServer("127.0.0.1:0", 2)
    srv.register_handler("/t.S/Fixed", lambda v: payload)
    addr = srv.start()
    try:
        inputs = {"x": torch.randn(4, device=DEV)}
        blob, regions, keepalive = _streaming(inputs)
        ch = T.GrpcChannel(addr)
        try:
            outs, raw = ch.call_streaming_parsed(
                "/t.S/Fixed", blob, list(regions), 0, 60.0)
            # typed-field entry comes first -> prospector bails -> None
            assert outs is None
            assert bytes(raw) == payload
        finally:
            ch.close()
    finally:
        srv.stop()


def test_predict_uses_parsed_receive(tmp_path):
    """turbo.predict(output_device=cuda) end-to-end through the parse-
    ahead path with a multi-tensor response."""
    sock = f"unix://{tmp_path}/pr.sock"
    with ModelServer(address=sock, raw_predict=True, device=DEV) as srv:
        srv.manager.load(
            "mix", Servable(lambda d: {
                "a": d["a"] * 2,
                "b": d["b"].to(torch.float64),
                "c": torch.empty(0, 4, device=DEV),
            }), version=1)
        with TurboPredictClient(sock, backend="native") as c:
            a = torch.randn(64, 3, 224, 224, device=DEV)
            b = torch.randint(0, 100, (128, 512), dtype=torch.int32,
                              device=DEV)
            out = c.predict("mix", {"a": a, "b": b}, output_device=DEV,
                            timeout=60)
            assert torch.equal(out["a"], a * 2)
            assert torch.equal(out["b"], b.to(torch.float64))
            assert out["c"].shape == (0, 4)
            assert all(v.is_cuda for v in out.values())


def test_concurrent_device_streaming():
    """8 concurrent streaming sends: each leases its own staging context
    (pooled pipeline) — results must not cross-corrupt."""
    import threading

    srv = T.GrpcServer("127.0.0.1:0", 8)
    srv.register_handler("/t.S/Echo", lambda v: bytes(v))
    addr = srv.start()
    try:
        payloads = []
        for i in range(4):
            t = torch.full((24, 3, 224, 224), float(i + 1), device=DEV)
            blob, regions, keep = _streaming({"x": t})
            expect = _buffered({"x": t})
            payloads.append((blob, list(regions), keep, expect))
        ch = T.GrpcChannel(addr)
        errs = []

        def worker(idx):
            blob, regions, keep, expect = payloads[idx % len(payloads)]
            try:
                for _ in range(3):
                    r = ch.call_streaming("/t.S/Echo", blob, regions, 60.0)
                    if bytes(r) != expect:
                        errs.append(f"mismatch idx={idx}")
            except Exception as e:  # noqa: BLE001
                errs.append(repr(e))

        ts = [threading.Thread(target=worker, args=(i,)) for i in range(8)]
        for t in ts:
            t.start()
        for t in ts:
            t.join()
        ch.close()
        assert errs == []
    finally:
        srv.stop()


def test_sharded_predict_parse_ahead(tmp_path):
    """predict_sharded with cuda outputs goes through per-shard
    parse-ahead futures; results must match the single-shot path."""
    sock = f"unix://{tmp_path}/shard_pa.sock"
    with ModelServer(address=sock, raw_predict=True, device=DEV) as srv:
        srv.manager.load(
            "double", Servable(lambda d: {k: v * 2 for k, v in d.items()}),
            version=1)
        with TurboPredictClient(sock, backend="native",
                                num_channels=4) as c:
            x = torch.randn(33, 3, 224, 224, device=DEV)  # uneven shards
            out = c.predict_sharded("double", {"images": x}, shards=4,
                                    output_device=DEV, timeout=60)
            assert out["images"].is_cuda
            assert torch.equal(out["images"], x * 2)


def test_server_request_prospecting():
    """register_handler_parsed: the server H2Ds request tensor_content
    while it streams in and hands the handler device tensors + spec."""
    seen = {}

    def handler(view, spec, outs):
        seen["spec"] = spec
        seen["outs"] = outs
        return b"ok"

    srv = T.GrpcServer("127.0.0.1:0", 2)
    srv.register_handler_parsed(
        "/tensorflow.serving.PredictionService/Predict", handler, 0)
    addr = srv.start()
    try:
        inputs = _inputs()
        names = list(inputs.keys())
        blob, regions, keep = native.serialize_predict_streaming(
            True, "promodel", 7, "sig", names, [inputs[k] for k in names])
        torch.cuda.current_stream().synchronize()
        ch = T.GrpcChannel(addr)
        try:
            assert bytes(ch.call_streaming(
                "/tensorflow.serving.PredictionService/Predict",
                blob, list(regions), 60.0)) == b"ok"
        finally:
            ch.close()
        assert seen["outs"] is not None, "canonical request must prospect"
        assert seen["spec"]["name"] == "promodel"
        assert seen["spec"]["version"] == 7
        assert seen["spec"]["signature_name"] == "sig"
        for k in inputs:
            assert seen["outs"][k].is_cuda
            assert torch.equal(seen["outs"][k], inputs[k]), k
    finally:
        srv.stop()


def test_server_prospecting_end_to_end_gpu_servable(tmp_path):
    """ModelServer(device=cuda): full round trip where BOTH directions
    are prospected (request spans H2D'd during receive, response spans
    H2D'd client-side during receive)."""
    sock = f"unix://{tmp_path}/prospect.sock"
    with ModelServer(address=sock, raw_predict=True, device=DEV) as srv:
        srv.manager.load(
            "triple", Servable(lambda d: {k: v * 3 for k, v in d.items()}),
            version=1)
        with TurboPredictClient(sock, backend="native") as c:
            x = torch.randn(64, 3, 224, 224, device=DEV)
            out = c.predict("triple", {"images": x}, output_device=DEV,
                            timeout=60)
            assert torch.equal(out["images"], x * 3)


def test_echo_model_skips_prospecting(tmp_path):
    """Identity models are served by the all-C++ echo path; the skip_model
    hook must keep their requests off the GPU while still echoing
    correctly on a cuda-device server."""
    from min_tfs_client_amd.server import identity_servable

    sock = f"unix://{tmp_path}/echo_skip.sock"
    with ModelServer(address=sock, raw_predict=True, device=DEV) as srv:
        srv.manager.load("default", identity_servable(), version=1)
        with TurboPredictClient(sock, backend="native") as c:
            x = torch.randn(32, 3, 224, 224, device=DEV)
            before = torch.cuda.memory_allocated()
            for _ in range(3):
                out = c.predict("default", {"images": x},
                                output_device=DEV, timeout=60)
                assert torch.equal(out["images"], x)
            # server-side prospecting for the echo model would have
            # allocated ~19MB x 3 on the device inside the SERVER (same
            # process here): allow client-side parse-ahead allocs only
            after = torch.cuda.memory_allocated()
            assert after - before < 200 << 20


def test_parse_ahead_fuzz_random_responses():
    """50 randomized canonical responses (mixed dtypes, scalars, empty
    tensors, long names, multi-chunk payloads): parse-ahead must either
    match the ordinary parse exactly or fall back to None."""
    import random

    rng = random.Random(20260914)
    dtypes = [torch.float32, torch.float64, torch.int32, torch.int64,
              torch.int16, torch.uint8, torch.bfloat16, torch.float16,
              torch.bool]

    srv = T.GrpcServer("127.0.0.1:0", 2)
    payload_holder = {}
    srv.register_handler("/t.S/Fuzz", lambda v: payload_holder["p"])
    addr = srv.start()
    try:
        ch = T.GrpcChannel(addr)
        req_blob, req_regions, keep = _streaming(
            {"q": torch.zeros(4, device=DEV)})
        try:
            for it in range(50):
                n_tensors = rng.randint(1, 5)
                outputs = {}
                for i in range(n_tensors):
                    dt = rng.choice(dtypes)
                    kind = rng.random()
                    if kind < 0.15:
                        shape = ()  # scalar
                    elif kind < 0.3:
                        shape = (0, rng.randint(1, 8))  # empty
                    elif kind < 0.5:
                        shape = (rng.randint(1, 2048),)
                    else:
                        shape = (rng.randint(1, 48), 3,
                                 rng.randint(8, 224), rng.randint(8, 224))
                    name = ("t" * rng.randint(1, 40)) + str(i)
                    if dt == torch.bool:
                        t = torch.randint(0, 2, shape, device=DEV
                                          ).to(torch.bool)
                    elif dt.is_floating_point:
                        t = torch.randn(shape, device=DEV).to(dt)
                    else:
                        t = torch.randint(0, 100, shape, dtype=dt,
                                          device=DEV)
                    outputs[name] = t
                payload_holder["p"] = _buffered(outputs, is_request=False)
                outs, raw = ch.call_streaming_parsed(
                    "/t.S/Fuzz", req_blob, list(req_regions), 0, 60.0)
                _s, ref, _f = native.parse_predict_response(
                    bytes(raw), "cuda:0", 1)
                if outs is None:
                    # acceptable only for sub-1MB messages (machinery
                    # skips them); big canonical ones must prospect
                    assert len(bytes(raw)) < (1 << 20), it
                    outs = ref
                assert set(outs) == set(ref), it
                for k in ref:
                    assert outs[k].dtype == ref[k].dtype, (it, k)
                    assert outs[k].shape == ref[k].shape, (it, k)
                    assert torch.equal(outs[k], ref[k]), (it, k)
        finally:
            ch.close()
    finally:
        srv.stop()
