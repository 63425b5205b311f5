"""ProfilerService (reference server.cc:324,339) on both server paths:
Profile returns the chrome-trace spans captured during the window as
tool_data; Monitor returns the formatted metrics snapshot."""
import json
import threading

import grpc
import pytest
import torch

from min_tfs_client_amd.server import ModelServer, identity_servable
from min_tfs_client_amd.turbo import TurboPredictClient
from min_tfs_client_amd.wire import messages as pb
from min_tfs_client_amd.wire.grpc_stubs import ProfilerServiceStub


@pytest.mark.parametrize("raw", [True, False],
                         ids=["native-transport", "grpcio"])
def test_profile_and_monitor(raw, tmp_path):
    addr = f"unix://{tmp_path}/prof.sock" if raw else "127.0.0.1:0"
    with ModelServer(address=addr, raw_predict=raw,
                     transport="native" if raw else "grpcio") as srv:
        srv.manager.load("m", identity_servable(), version=1)

        if raw:
            # native transport speaks standard gRPC: use the C++ channel
            from min_tfs_client_amd import _transport as T
            ch_native = T.GrpcChannel(srv.address)

            def call(path, req, resp_cls):
                out = ch_native.call(path, req.SerializeToString(), 30.0)
                return resp_cls.FromString(bytes(out))
        else:
            ch = grpc.insecure_channel(f"127.0.0.1:{srv.port}")
            stub = ProfilerServiceStub(ch)

            def call(path, req, resp_cls):
                fn = stub.Profile if path.endswith("Profile") \
                    else stub.Monitor
                return fn(req, timeout=30)

        # traffic inside the profiling window -> spans in the trace
        target = srv.address if raw else f"127.0.0.1:{srv.port}"

        def traffic():
            with TurboPredictClient(target) as c:
                for _ in range(5):
                    c.predict("m", {"x": torch.randn(4, 4)}, timeout=20)

        t = threading.Thread(target=traffic)
        preq = pb.ProfileRequest()
        preq.duration_ms = 700
        preq.tools.append("trace_viewer")
        t.start()
        presp = call("/tensorflow.ProfilerService/Profile", preq,
                     pb.ProfileResponse)
        t.join()
        assert len(presp.tool_data) == 1
        assert presp.tool_data[0].name == "trace_viewer.json"
        trace = json.loads(presp.tool_data[0].data)
        events = trace["traceEvents"] if isinstance(trace, dict) else trace
        names = {e.get("name") for e in events}
        assert not presp.empty_trace
        assert any(n and n.startswith("turbo.") for n in names), names

        mreq = pb.MonitorRequest()
        mreq.monitoring_level = 2
        mreq.timestamp = True
        mresp = call("/tensorflow.ProfilerService/Monitor", mreq,
                     pb.MonitorResponse)
        assert "timestamp: " in mresp.data
        assert "request_count" in mresp.data or "predict" in mresp.data

        if raw:
            ch_native.close()
        else:
            ch.close()


def test_profiler_pb2_shim_paths():
    from tensorflow.core.profiler import profiler_service_pb2 as p
    r = p.ProfileRequest()
    r.session_id = "s1"
    assert p.ProfileRequest.FromString(r.SerializeToString()).session_id \
        == "s1"
