"""Data-parallel sharding tests on CPU: 2-process gloo group, scatter ->
local predict -> all-gather (the config-4 path without GPUs; on MI355X the
same code runs over RCCL/xGMI)."""
import multiprocessing
import os

import numpy as np
import pytest
import torch

from min_tfs_client_amd.parallel import shard_sizes


def test_shard_sizes_even():
    assert shard_sizes(256, 8) == [32] * 8


def test_shard_sizes_uneven():
    assert shard_sizes(10, 4) == [3, 3, 2, 2]
    assert shard_sizes(3, 4) == [1, 1, 1, 0]


def _dp_worker(rank, world, port, q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        import torch.distributed as dist
        dist.init_process_group("gloo", rank=rank, world_size=world)

        import sys
        root = os.path.dirname(os.path.dirname(
            os.path.dirname(os.path.abspath(__file__))))
        if root not in sys.path:
            sys.path.insert(0, root)
        from min_tfs_client_amd.parallel import DataParallelPredictor
        from min_tfs_client_amd.server import ModelServer, identity_servable
        from min_tfs_client_amd.turbo import TurboPredictClient

        sock = f"unix:///tmp/dp_test_{os.getpid()}_{rank}.sock"
        with ModelServer(address=sock, raw_predict=True) as srv:
            srv.manager.load("m", identity_servable(), version=1)
            with TurboPredictClient(sock) as client:
                dp = DataParallelPredictor(client, device="cpu")
                # uneven batch: 7 rows over 2 ranks
                if rank == 0:
                    torch.manual_seed(0)
                    full = {"x": torch.randn(7, 3),
                            "y": torch.arange(14,
                                              dtype=torch.int64).view(7, 2)}
                else:
                    full = None
                out = dp.predict("m", full)
                # every rank must hold the full gathered batch
                assert out["x"].shape == (7, 3)
                assert out["y"].shape == (7, 2)
                if rank == 0:
                    assert torch.equal(out["x"], full["x"])
                    assert torch.equal(out["y"], full["y"])
                # shard-only mode
                out_shard = dp.predict("m", full, gather_outputs=False)
                assert out_shard["x"].shape[0] == shard_sizes(7, world)[rank]
        dist.destroy_process_group()
        q.put((rank, "ok"))
    except Exception as e:  # noqa: BLE001
        import traceback
        q.put((rank, f"FAIL: {e}\n{traceback.format_exc()}"))


@pytest.mark.timeout(120)
def test_dp_predict_two_ranks():
    import socket
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    ctx = multiprocessing.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_dp_worker, args=(r, 2, port, q))
             for r in range(2)]
    for p in procs:
        p.start()
    results = [q.get(timeout=100) for _ in range(2)]
    for p in procs:
        p.join(timeout=30)
    for rank, status in results:
        assert status == "ok", f"rank {rank}: {status}"


def _dp_rowchange_worker(rank, world, port, q):
    """Servable whose output dim-0 differs from its input shard's (one
    summary row per shard): the gather must use ACTUAL output row counts
    (round-1 weakness #5 — row-aligned assumption mis-gathered this)."""
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        import torch.distributed as dist
        dist.init_process_group("gloo", rank=rank, world_size=world)
        import sys
        root = os.path.dirname(os.path.dirname(
            os.path.dirname(os.path.abspath(__file__))))
        if root not in sys.path:
            sys.path.insert(0, root)
        from min_tfs_client_amd.parallel import DataParallelPredictor
        from min_tfs_client_amd.server import (
            ModelServer,
            Servable,
        )
        from min_tfs_client_amd.turbo import TurboPredictClient

        def reduce_fn(inputs):
            x = inputs["x"]
            x = torch.as_tensor(np.asarray(x))
            # one row out per shard regardless of shard row count
            return {"sum": x.sum(dim=0, keepdim=True)}

        sock = f"unix:///tmp/dp_rc_{os.getpid()}_{rank}.sock"
        with ModelServer(address=sock) as srv:
            srv.manager.load("m", Servable(reduce_fn), version=1)
            with TurboPredictClient(sock) as client:
                dp = DataParallelPredictor(client, device="cpu")
                full = None
                if rank == 0:
                    torch.manual_seed(3)
                    full = {"x": torch.randn(7, 5)}
                out = dp.predict("m", full)
                # world rows: one summary row per rank
                assert out["sum"].shape == (world, 5)
                if rank == 0:
                    sizes = shard_sizes(7, world)
                    expect = torch.cat([
                        full["x"][sum(sizes[:r]):sum(sizes[:r + 1])].sum(
                            dim=0, keepdim=True)
                        for r in range(world)], dim=0)
                    assert torch.allclose(out["sum"], expect, atol=1e-5)
        dist.destroy_process_group()
        q.put((rank, "ok"))
    except Exception as e:  # noqa: BLE001
        import traceback
        q.put((rank, f"FAIL: {e}\n{traceback.format_exc()}"))


@pytest.mark.timeout(120)
def test_dp_gather_with_row_count_change():
    import socket
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    ctx = multiprocessing.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_dp_rowchange_worker, args=(r, 2, port, q))
             for r in range(2)]
    for p in procs:
        p.start()
    results = [q.get(timeout=100) for _ in range(2)]
    for p in procs:
        p.join(timeout=30)
    for rank, status in results:
        assert status == "ok", f"rank {rank}: {status}"


def _dp_shm_worker(rank, world, port, q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        import torch.distributed as dist
        dist.init_process_group("gloo", rank=rank, world_size=world)
        import sys
        root = os.path.dirname(os.path.dirname(
            os.path.dirname(os.path.abspath(__file__))))
        if root not in sys.path:
            sys.path.insert(0, root)
        from min_tfs_client_amd.parallel import DataParallelPredictor
        from min_tfs_client_amd.server import ModelManager, identity_servable
        from min_tfs_client_amd.shm import ShmListener, ShmPredictClient

        mgr = ModelManager()
        mgr.load("m", identity_servable(), version=1)
        hs = f"/tmp/dp_shm_{os.getpid()}_{rank}"
        with ShmListener(mgr, hs):
            with ShmPredictClient(hs, slot_bytes=8 << 20) as client:
                dp = DataParallelPredictor(client, device="cpu")
                full = None
                if rank == 0:
                    torch.manual_seed(0)
                    full = {"x": torch.randn(9, 4)}
                out = dp.predict("m", full)
                assert out["x"].shape == (9, 4)
                if rank == 0:
                    assert torch.equal(out["x"], full["x"])
        dist.destroy_process_group()
        q.put((rank, "ok"))
    except Exception as e:  # noqa: BLE001
        import traceback
        q.put((rank, f"FAIL: {e}\n{traceback.format_exc()}"))


@pytest.mark.timeout(180)
def test_dp_predict_over_shm_transport():
    """Config-4 composition with the shm local transport — the optimal
    multi-GPU deployment shape (RCCL scatter/all-gather between ranks,
    shared-memory hop to each rank's local server)."""
    import socket
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    ctx = multiprocessing.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_dp_shm_worker, args=(r, 2, port, q))
             for r in range(2)]
    for p in procs:
        p.start()
    results = [q.get(timeout=150) for _ in range(2)]
    for p in procs:
        p.join(timeout=30)
    for rank, status in results:
        assert status == "ok", f"rank {rank}: {status}"
