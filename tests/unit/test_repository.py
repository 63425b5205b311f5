"""Model repository: TFRecord framing, version policies, filesystem
polling, load retries, warmup replay."""
import os

import numpy as np
import pytest
import torch

from min_tfs_client_amd.repository import (
    FileSystemStoragePathSource,
    VersionPolicy,
    crc32c,
    default_loader,
    read_tfrecord,
    run_warmup,
    write_tfrecord,
    write_warmup_file,
)
from min_tfs_client_amd.server import ModelManager, Servable
from min_tfs_client_amd.tensors import ndarray_to_tensor_proto
from min_tfs_client_amd.wire import messages as pb


# -- TFRecord framing --------------------------------------------------------

def test_crc32c_known_vectors():
    # RFC 3720 test vectors
    assert crc32c(b"") == 0
    assert crc32c(b"123456789") == 0xE3069283
    assert crc32c(bytes(32)) == 0x8A9136AA


def test_tfrecord_roundtrip(tmp_path):
    path = str(tmp_path / "records")
    recs = [b"hello", b"", b"x" * 1000]
    write_tfrecord(path, recs)
    assert read_tfrecord(path) == recs


def test_tfrecord_corruption_detected(tmp_path):
    path = str(tmp_path / "records")
    write_tfrecord(path, [b"payload"])
    blob = bytearray(open(path, "rb").read())
    blob[14] ^= 0xFF  # flip a payload byte
    open(path, "wb").write(bytes(blob))
    with pytest.raises(ValueError, match="corrupt"):
        read_tfrecord(path)


# -- version policy ----------------------------------------------------------

def test_policy_latest_default():
    assert VersionPolicy().aspired([3, 1, 7]) == [7]


def test_policy_latest_n():
    assert VersionPolicy("latest", 2).aspired([3, 1, 7]) == [3, 7]


def test_policy_all():
    assert VersionPolicy("all").aspired([3, 1]) == [1, 3]


def test_policy_specific():
    assert VersionPolicy("specific", specific=[1, 9]).aspired(
        [3, 1, 7]) == [1]


def test_policy_from_proto():
    cfg = pb.FileSystemStoragePathSourceConfig.ServableVersionPolicy()
    cfg.latest.num_versions = 3
    p = VersionPolicy.from_proto(cfg)
    assert p.kind == "latest" and p.num_versions == 3
    cfg2 = pb.FileSystemStoragePathSourceConfig.ServableVersionPolicy()
    cfg2.specific.versions.extend([2, 5])
    assert VersionPolicy.from_proto(cfg2).specific == [2, 5]


# -- loaders -----------------------------------------------------------------

def _make_version(tmp_path, model, version, kind="identity"):
    vdir = tmp_path / model / str(version)
    vdir.mkdir(parents=True)
    if kind == "identity":
        (vdir / "identity").touch()
    elif kind == "json-identity":
        (vdir / "model.json").write_text('{"family": "identity"}')
    elif kind == "torchscript":
        class Doubler(torch.nn.Module):
            def forward(self, x):
                return x * 2
        torch.jit.script(Doubler()).save(str(vdir / "model.pt"))
    return str(vdir)


def test_default_loader_identity(tmp_path):
    vdir = _make_version(tmp_path, "m", 1)
    s = default_loader("m", vdir)
    out = s({"x": np.ones(3, np.float32)})
    assert np.array_equal(out["x"], np.ones(3, np.float32))


def test_default_loader_torchscript(tmp_path):
    vdir = _make_version(tmp_path, "m", 1, "torchscript")
    s = default_loader("m", vdir)
    out = s({"x": torch.ones(3)})
    assert torch.equal(out["output"], torch.full((3,), 2.0))


def test_default_loader_missing_raises(tmp_path):
    vdir = tmp_path / "m" / "1"
    vdir.mkdir(parents=True)
    with pytest.raises(FileNotFoundError):
        default_loader("m", str(vdir))


# -- polling source ----------------------------------------------------------

def test_poll_loads_latest_and_hot_swaps(tmp_path):
    mgr = ModelManager()
    src = FileSystemStoragePathSource(mgr, poll_wait_seconds=0)
    _make_version(tmp_path, "m", 1)
    src.set_models({"m": str(tmp_path / "m")})
    src.poll_once()
    assert [v for v, s, _ in mgr.version_statuses("m") if s == 30] == [1]
    # new version appears -> latest policy loads 2, unloads 1
    _make_version(tmp_path, "m", 2)
    src.poll_once()
    states = {v: s for v, s, _ in mgr.version_statuses("m")}
    assert states[2] == 30 and states[1] == 50


def test_poll_policy_all_keeps_both(tmp_path):
    mgr = ModelManager()
    src = FileSystemStoragePathSource(mgr, poll_wait_seconds=0)
    _make_version(tmp_path, "m", 1)
    _make_version(tmp_path, "m", 2)
    src.set_models({"m": str(tmp_path / "m")},
                   {"m": VersionPolicy("all")})
    src.poll_once()
    states = {v: s for v, s, _ in mgr.version_statuses("m")}
    assert states == {1: 30, 2: 30}


def test_load_failure_records_error_state(tmp_path):
    mgr = ModelManager()
    src = FileSystemStoragePathSource(mgr, poll_wait_seconds=0,
                                      max_num_load_retries=1,
                                      load_retry_interval_s=0)
    vdir = tmp_path / "m" / "1"
    vdir.mkdir(parents=True)  # empty: loader raises
    src.set_models({"m": str(tmp_path / "m")})
    src.poll_once()
    version, state, err = mgr.version_statuses("m")[0]
    assert state == 50 and err is not None
    assert err[0] == pb.ErrorCode.UNKNOWN


def test_load_retry_succeeds_on_flaky_loader(tmp_path):
    mgr = ModelManager()
    attempts = []

    def flaky_loader(name, vdir):
        attempts.append(1)
        if len(attempts) < 3:
            raise RuntimeError("transient")
        return Servable(lambda i: i)

    src = FileSystemStoragePathSource(mgr, loader=flaky_loader,
                                      poll_wait_seconds=0,
                                      max_num_load_retries=5,
                                      load_retry_interval_s=0)
    _make_version(tmp_path, "m", 1)
    src.set_models({"m": str(tmp_path / "m")})
    src.poll_once()
    assert len(attempts) == 3
    assert mgr.version_statuses("m")[0][1] == 30


def test_removed_model_unloaded(tmp_path):
    mgr = ModelManager()
    src = FileSystemStoragePathSource(mgr, poll_wait_seconds=0)
    _make_version(tmp_path, "m", 1)
    src.set_models({"m": str(tmp_path / "m")})
    src.poll_once()
    src.set_models({})
    states = {v: s for v, s, _ in mgr.version_statuses("m")}
    assert states[1] == 50


# -- warmup ------------------------------------------------------------------

def test_warmup_replay(tmp_path):
    vdir = _make_version(tmp_path, "m", 1)
    req = pb.PredictRequest()
    req.model_spec.name = "m"
    req.inputs["x"].CopyFrom(
        ndarray_to_tensor_proto(np.ones((2, 2), np.float32)))
    write_warmup_file(vdir, [req, req])

    seen = []
    servable = Servable(lambda inputs: seen.append(inputs) or inputs)
    n = run_warmup(servable, vdir)
    assert n == 2
    assert seen[0]["x"].shape == (2, 2)


def test_warmup_in_poll_cycle(tmp_path):
    mgr = ModelManager()
    src = FileSystemStoragePathSource(mgr, poll_wait_seconds=0)
    vdir = _make_version(tmp_path, "m", 1)
    req = pb.PredictRequest()
    req.inputs["x"].CopyFrom(
        ndarray_to_tensor_proto(np.zeros(1, np.float32)))
    write_warmup_file(vdir, [req])
    src.set_models({"m": str(tmp_path / "m")})
    src.poll_once()  # must not raise; warmup runs through identity
    assert mgr.version_statuses("m")[0][1] == 30


def test_warmup_disabled(tmp_path):
    mgr = ModelManager()
    src = FileSystemStoragePathSource(mgr, poll_wait_seconds=0,
                                      enable_warmup=False)
    vdir = _make_version(tmp_path, "m", 1)
    (tmp_path / "m" / "1" / "assets.extra").mkdir()
    # corrupt warmup file would raise if read
    (tmp_path / "m" / "1" / "assets.extra" /
     "tf_serving_warmup_requests").write_bytes(b"garbage")
    src.set_models({"m": str(tmp_path / "m")})
    src.poll_once()
    assert mgr.version_statuses("m")[0][1] == 30


def test_loader_state_dict(tmp_path):
    """model.json with a state_dict: weights are loaded into the family
    module."""
    import json as _json
    vdir = tmp_path / "m" / "1"
    vdir.mkdir(parents=True)
    from min_tfs_client_amd.models import resnet50_servable
    donor = resnet50_servable()
    torch.save(donor.module.state_dict(), str(vdir / "weights.pt"))
    (vdir / "model.json").write_text(_json.dumps(
        {"family": "resnet50", "state_dict": "weights.pt"}))
    s = default_loader("m", str(vdir))
    # weights equal the donor's, not a fresh random init
    for (n1, p1), (n2, p2) in zip(s.module.state_dict().items(),
                                  donor.module.state_dict().items()):
        assert n1 == n2
        assert torch.equal(p1, p2)


def test_failed_version_not_retried_until_dir_removed(tmp_path):
    mgr = ModelManager()
    attempts = []

    def failing_loader(name, vdir):
        attempts.append(1)
        raise RuntimeError("permanent")

    src = FileSystemStoragePathSource(mgr, loader=failing_loader,
                                      poll_wait_seconds=0,
                                      max_num_load_retries=1,
                                      load_retry_interval_s=0)
    _make_version(tmp_path, "m", 1)
    src.set_models({"m": str(tmp_path / "m")})
    src.poll_once()
    n_after_first = len(attempts)
    src.poll_once()
    src.poll_once()
    assert len(attempts) == n_after_first  # no re-attempts
    # removing and recreating the version dir clears the failure memory
    import shutil
    shutil.rmtree(str(tmp_path / "m" / "1"))
    src.poll_once()
    _make_version(tmp_path, "m", 1)
    src.poll_once()
    assert len(attempts) > n_after_first
