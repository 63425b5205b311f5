import sys, os
sys.path.insert(0, "/root/repo")
import torch, warnings
from min_tfs_client_amd.server import ModelServer, identity_servable
from min_tfs_client_amd.turbo import TurboPredictClient
from min_tfs_client_amd.utils.tracing import Tracer
sock = f"unix:///tmp/tr_{os.getpid()}.sock"
with ModelServer(address=sock, raw_predict=True) as srv:
    srv.manager.load("m", identity_servable(), version=1)
    with TurboPredictClient(sock) as c:
        x = torch.randn(32, 3, 224, 224, device="cuda:0")
        for _ in range(5):
            c.predict("m", {"images": x}, output_device="cuda:0")
        t = Tracer.get(); t.clear(); t.start()
        for _ in range(20):
            c.predict("m", {"images": x}, output_device="cuda:0")
        t.stop()
        n = t.export("gpurun_out/trace_sample.json")
        print("exported", n, "spans")
