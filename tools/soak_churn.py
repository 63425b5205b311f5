#!/usr/bin/env python3
"""Teardown-path stress: connection churn + deadline RSTs against a
cuda-device server with request prospecting and parse-ahead clients.

The hardened paths this targets (DeviceParse RAII quiesce): streams
erased by RST with H2D copies in flight, connections dying mid-request,
failed prospects, pinned receive buffers returned to the pool under
churn. Every completed response is content-verified."""
import json
import os
import sys
import threading
import time

_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, _ROOT)

import torch  # noqa: E402

from min_tfs_client_amd.server import ModelServer, Servable  # noqa: E402
from min_tfs_client_amd.turbo import TurboPredictClient  # noqa: E402


def main(seconds=240):
    dev = "cuda:0" if torch.cuda.is_available() else "cpu"
    sock = f"unix:///tmp/mi355x_churn_{os.getpid()}.sock"
    stop_at = time.monotonic() + seconds
    stats = {"ok": 0, "deadline": 0, "conn_churns": 0, "fail": 0}
    lock = threading.Lock()
    errors = []

    def scaled(inputs):
        return {k: v * 2 for k, v in inputs.items()}

    def slow(inputs):
        time.sleep(0.05)
        return inputs

    with ModelServer(address=sock, raw_predict=True, device=dev,
                     max_workers=24) as srv:
        srv.manager.load("scale", Servable(scaled), version=1)
        srv.manager.load("slow", Servable(slow), version=1)

        def churn_worker(wid):
            """Reconnect every 5 requests: connection teardown with
            possible prospect state on the server side."""
            g = torch.Generator().manual_seed(wid)
            while time.monotonic() < stop_at:
                with TurboPredictClient(sock, backend="native") as c:
                    with lock:
                        stats["conn_churns"] += 1
                    for _ in range(5):
                        if time.monotonic() >= stop_at:
                            break
                        x = torch.randn(16, 3, 128, 128,
                                        generator=g).to(dev)
                        try:
                            out = c.predict("scale", {"x": x},
                                            output_device=dev, timeout=30)
                            ok = torch.allclose(out["x"], x * 2)
                            with lock:
                                stats["ok" if ok else "fail"] += 1
                                if not ok and len(errors) < 5:
                                    errors.append("content mismatch")
                        except Exception as e:  # noqa: BLE001
                            with lock:
                                stats["fail"] += 1
                                if len(errors) < 5:
                                    errors.append(repr(e))

        def deadline_worker(wid):
            """Tiny deadlines against the slow model: constant
            DEADLINE_EXCEEDED -> client RST while the server-side
            request prospect may have copies in flight."""
            import grpc
            g = torch.Generator().manual_seed(100 + wid)
            with TurboPredictClient(sock, backend="native") as c:
                while time.monotonic() < stop_at:
                    x = torch.randn(8, 3, 224, 224, generator=g).to(dev)
                    try:
                        c.predict("slow", {"x": x}, output_device=dev,
                                  timeout=0.02)
                        with lock:
                            stats["ok"] += 1
                    except grpc.RpcError as e:
                        code = e.code()
                        with lock:
                            if code == grpc.StatusCode.DEADLINE_EXCEEDED:
                                stats["deadline"] += 1
                            else:
                                stats["fail"] += 1
                                if len(errors) < 5:
                                    errors.append(repr(e))
                    except Exception as e:  # noqa: BLE001
                        with lock:
                            stats["fail"] += 1
                            if len(errors) < 5:
                                errors.append(repr(e))

        def abrupt_worker(wid):
            """Close the channel mid-flight (no clean wait): exercises
            fail_all_pending + Pending RAII destruction."""
            g = torch.Generator().manual_seed(200 + wid)
            while time.monotonic() < stop_at:
                c = TurboPredictClient(sock, backend="native")
                try:
                    x = torch.randn(16, 3, 224, 224, generator=g).to(dev)
                    futs = [c._stubs[0].future_streaming_parsed(
                        *c._serialize_streaming("scale", {"x": x},
                                                None, "")[:2],
                        0 if dev != "cpu" else -1, 30)
                        for _ in range(2)]
                    # collect one, abandon the other, close
                    outs, raw = futs[0].result_parsed()
                    with lock:
                        stats["ok"] += 1
                finally:
                    c.close()
                with lock:
                    stats["conn_churns"] += 1

        workers = ([threading.Thread(target=churn_worker, args=(i,))
                    for i in range(3)] +
                   [threading.Thread(target=deadline_worker, args=(i,))
                    for i in range(2)] +
                   [threading.Thread(target=abrupt_worker, args=(i,))
                    for i in range(2)])
        t0 = time.monotonic()
        for t in workers:
            t.start()
        for t in workers:
            t.join()
        elapsed = time.monotonic() - t0
    print(json.dumps({"seconds": round(elapsed, 1), **stats,
                      "sample_errors": errors}))
    return 1 if stats["fail"] else 0


if __name__ == "__main__":
    sys.exit(main(int(sys.argv[1]) if len(sys.argv) > 1 else 240))
