"""Model repository: filesystem version polling, version policies, loaders
and SavedModel-style warmup.

Native analogue of the reference's servable-lifecycle stack:
``FileSystemStoragePathSource`` (file_system_storage_path_source.cc) polls
each model's base path for numeric version directories every
``file_system_poll_wait_seconds``; an aspired-versions policy (latest-N /
all / specific — file_system_storage_path_source.proto ServableVersionPolicy)
decides which versions are loaded; loads retry up to ``max_num_load_retries``
(util/retrier.cc, main.cc:107-116); and warmup replays recorded
``PredictionLog`` records from ``assets.extra/tf_serving_warmup_requests``
before a version goes AVAILABLE (saved_model_warmup.cc:55-83).

On-disk version-directory formats understood by the default loader:
  * ``identity``            — empty marker file -> identity echo servable
  * ``model.json``          — {"family": "resnet50"|"bert_base",
                               "device": ..., "state_dict": "weights.pt"?}
  * ``model.pt``            — TorchScript module (torch.jit.load)
plus TF-style ``assets.extra/tf_serving_warmup_requests`` (a TFRecord of
serialized PredictionLog protos — framing per TF's RecordWriter: length,
masked-crc32c(length), payload, masked-crc32c(payload)).
"""
from __future__ import annotations

import json
import logging
import os
import struct
import threading
import time
from typing import Callable, Dict, List, Optional

from .server import ModelManager, Servable, identity_servable
from .wire import messages as pb

logger = logging.getLogger("mi355x_tfs.repository")

# ---------------------------------------------------------------------------
# TFRecord framing (crc32c, masked) — for warmup files
# ---------------------------------------------------------------------------

_CRC_TABLE = []


def _crc32c_table():
    global _CRC_TABLE
    if _CRC_TABLE:
        return _CRC_TABLE
    poly = 0x82F63B78
    table = []
    for i in range(256):
        crc = i
        for _ in range(8):
            crc = (crc >> 1) ^ poly if crc & 1 else crc >> 1
        table.append(crc)
    _CRC_TABLE = table
    return table


def crc32c(data: bytes) -> int:
    table = _crc32c_table()
    crc = 0xFFFFFFFF
    for b in data:
        crc = table[(crc ^ b) & 0xFF] ^ (crc >> 8)
    return crc ^ 0xFFFFFFFF


def _masked_crc(data: bytes) -> int:
    crc = crc32c(data)
    return ((crc >> 15) | (crc << 17)) + 0xA282EAD8 & 0xFFFFFFFF


def write_tfrecord(path: str, records: List[bytes],
                   append: bool = False) -> None:
    with open(path, "ab" if append else "wb") as f:
        for rec in records:
            length = struct.pack("<Q", len(rec))
            f.write(length)
            f.write(struct.pack("<I", _masked_crc(length)))
            f.write(rec)
            f.write(struct.pack("<I", _masked_crc(rec)))


def read_tfrecord(path: str, verify_crc: bool = True) -> List[bytes]:
    records = []
    with open(path, "rb") as f:
        while True:
            header = f.read(12)
            if len(header) < 12:
                break
            (length,) = struct.unpack("<Q", header[:8])
            (lcrc,) = struct.unpack("<I", header[8:])
            if verify_crc and _masked_crc(header[:8]) != lcrc:
                raise ValueError(f"{path}: corrupt length crc")
            rec = f.read(length)
            (dcrc,) = struct.unpack("<I", f.read(4))
            if verify_crc and _masked_crc(rec) != dcrc:
                raise ValueError(f"{path}: corrupt record crc")
            records.append(rec)
    return records


WARMUP_FILE = os.path.join("assets.extra", "tf_serving_warmup_requests")


def write_warmup_file(version_dir: str, requests: List) -> None:
    """requests: list of PredictRequest messages (or serialized bytes)."""
    os.makedirs(os.path.join(version_dir, "assets.extra"), exist_ok=True)
    records = []
    for r in requests:
        log = pb.PredictionLog()
        if isinstance(r, bytes):
            log.predict_log.request.MergeFromString(r)
        else:
            log.predict_log.request.CopyFrom(r)
        records.append(log.SerializeToString())
    write_tfrecord(os.path.join(version_dir, WARMUP_FILE), records)


def run_warmup(servable: Servable, version_dir: str,
               max_records: int = 1000) -> int:
    """Replays PredictionLog records through the servable (PredictLog arms
    only, like saved_model_warmup.cc RunWarmupRequest). Returns #replayed."""
    from .tensors import tensor_proto_to_ndarray

    path = os.path.join(version_dir, WARMUP_FILE)
    if not os.path.exists(path):
        return 0
    n = 0
    for rec in read_tfrecord(path)[:max_records]:
        log = pb.PredictionLog.FromString(rec)
        if log.WhichOneof("log_type") != "predict_log":
            continue
        req = log.predict_log.request
        inputs = {k: tensor_proto_to_ndarray(v)
                  for k, v in req.inputs.items()}
        servable(inputs)
        n += 1
    return n


# ---------------------------------------------------------------------------
# Loaders
# ---------------------------------------------------------------------------

def default_loader(name: str, version_dir: str,
                   device: str = "cpu") -> Servable:
    """Builds a Servable from a version directory (formats in module
    docstring). ``device`` is the default placement for formats that do
    not name one themselves (model.json may override it)."""
    if os.path.exists(os.path.join(version_dir, "identity")) or \
            os.path.exists(os.path.join(version_dir, "saved_model.pb")):
        # saved_model.pb acceptance keeps reference-style fixture layouts
        # loadable (we serve it as identity, like the reference test model)
        return identity_servable()
    cfg_path = os.path.join(version_dir, "model.json")
    if os.path.exists(cfg_path):
        with open(cfg_path) as f:
            cfg = json.load(f)
        family = cfg.get("family")
        device = cfg.get("device", device)
        if family == "resnet50":
            from .models import resnet50_servable
            servable = resnet50_servable(device)
        elif family == "bert_base":
            from .models import bert_servable
            servable = bert_servable(device)
        elif family == "identity":
            servable = identity_servable()
        else:
            raise ValueError(f"unknown model family {family!r} for {name}")
        state = cfg.get("state_dict")
        if state:
            import torch
            module = getattr(servable, "module", None)
            if module is None:
                raise ValueError(
                    f"model family {family!r} does not expose a module for "
                    f"state_dict loading")
            sd = torch.load(os.path.join(version_dir, state),
                            map_location=device, weights_only=True)
            module.load_state_dict(sd)
        return servable
    pt_path = os.path.join(version_dir, "model.pt")
    if os.path.exists(pt_path):
        import torch
        module = torch.jit.load(pt_path, map_location=device)
        module.eval()

        @torch.no_grad()
        def fn(inputs):
            keys = sorted(inputs.keys())
            tensors = [(inputs[k] if isinstance(inputs[k], torch.Tensor)
                        else torch.as_tensor(inputs[k])).to(device)
                       for k in keys]
            out = module(*tensors)
            if isinstance(out, dict):
                return out
            if isinstance(out, (tuple, list)):
                return {f"output_{i}": o for i, o in enumerate(out)}
            return {"output": out}

        return Servable(fn)
    raise FileNotFoundError(
        f"no loadable model found in {version_dir} (expected identity, "
        f"model.json, model.pt or saved_model.pb)")


# ---------------------------------------------------------------------------
# Version policy + polling source
# ---------------------------------------------------------------------------

class VersionPolicy:
    """latest-N (default N=1, TF's default), all, or specific versions —
    FileSystemStoragePathSourceConfig.ServableVersionPolicy semantics."""

    def __init__(self, kind: str = "latest", num_versions: int = 1,
                 specific: Optional[List[int]] = None):
        assert kind in ("latest", "all", "specific")
        self.kind = kind
        self.num_versions = num_versions
        self.specific = specific or []

    def aspired(self, found: List[int]) -> List[int]:
        if self.kind == "all":
            return sorted(found)
        if self.kind == "specific":
            return sorted(v for v in found if v in self.specific)
        return sorted(found)[-self.num_versions:]

    @classmethod
    def from_proto(cls, policy) -> "VersionPolicy":
        which = policy.WhichOneof("policy_choice") if policy else None
        if which == "all":
            return cls("all")
        if which == "specific":
            return cls("specific", specific=list(policy.specific.versions))
        if which == "latest":
            return cls("latest", num_versions=policy.latest.num_versions or 1)
        return cls("latest", 1)


class FileSystemStoragePathSource:
    """Polls {model_name: base_path} for numeric version dirs and drives
    the manager's load/unload to match the aspired set."""

    def __init__(self, manager: ModelManager,
                 loader: Callable[[str, str], Servable] = default_loader,
                 poll_wait_seconds: float = 1.0,
                 max_num_load_retries: int = 5,
                 load_retry_interval_s: float = 0.1,
                 enable_warmup: bool = True,
                 fail_if_zero_versions_at_startup: bool = False):
        self.manager = manager
        self.loader = loader
        self.poll_wait_seconds = poll_wait_seconds
        self.max_num_load_retries = max_num_load_retries
        self.load_retry_interval_s = load_retry_interval_s
        self.enable_warmup = enable_warmup
        self.fail_if_zero = fail_if_zero_versions_at_startup
        self._configs: Dict[str, tuple] = {}  # name -> (base_path, policy)
        self._loaded: Dict[str, Dict[int, str]] = {}
        # versions that exhausted their retries: not re-attempted until
        # their directory disappears (TF gives up after max retries too)
        self._failed: Dict[str, set] = {}
        self._lock = threading.Lock()
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None

    # -- config ---------------------------------------------------------
    def set_models(self, configs: Dict[str, str],
                   policies: Optional[Dict[str, VersionPolicy]] = None):
        with self._lock:
            old = set(self._configs)
            self._configs = {
                name: (path, (policies or {}).get(name, VersionPolicy()))
                for name, path in configs.items()}
            for gone in old - set(self._configs):
                self.manager.unload(gone)
                self._loaded.pop(gone, None)

    # -- one poll cycle (public for tests / manual control) ------------
    def poll_once(self):
        with self._lock:
            configs = dict(self._configs)
        for name, (base, policy) in configs.items():
            try:
                found = self._scan_versions(base)
            except FileNotFoundError:
                logger.warning("model %s: base path %s missing", name, base)
                continue
            if not found and self.fail_if_zero:
                raise RuntimeError(
                    f"model {name}: no versions at startup under {base}")
            aspired = set(policy.aspired(sorted(found)))
            loaded = self._loaded.setdefault(name, {})
            failed = self._failed.setdefault(name, set())
            failed.intersection_update(found)   # dir removed -> forget
            for ver in sorted(aspired - set(loaded) - failed):
                vdir = os.path.join(base, found[ver])
                if self._load_with_retries(name, ver, vdir):
                    loaded[ver] = vdir
                else:
                    failed.add(ver)
            for ver in sorted(set(loaded) - aspired):
                self.manager.unload(name, ver)
                del loaded[ver]

    def _scan_versions(self, base: str) -> Dict[int, str]:
        """{version int: dir entry name} — zero-padded names like the
        reference fixture's 00000001 resolve to version 1."""
        if not os.path.isdir(base):
            raise FileNotFoundError(base)
        out: Dict[int, str] = {}
        for entry in os.listdir(base):
            if entry.isdigit() and os.path.isdir(os.path.join(base, entry)):
                out[int(entry)] = entry
        return out

    def _load_with_retries(self, name: str, version: int,
                           vdir: str) -> bool:
        # retries per --max_num_load_retries semantics (main.cc:107-116)
        for attempt in range(self.max_num_load_retries + 1):
            try:
                servable = self.loader(name, vdir)
                if self.enable_warmup:
                    n = run_warmup(servable, vdir)
                    if n:
                        logger.info("model %s v%d: replayed %d warmup "
                                    "records", name, version, n)
                self.manager.load(name, servable, version)
                return True
            except Exception as e:  # noqa: BLE001
                logger.warning("load %s v%d attempt %d failed: %s", name,
                               version, attempt, e)
                if attempt == self.max_num_load_retries:
                    self.manager.fail_load(name, version,
                                           pb.ErrorCode.UNKNOWN, str(e))
                    return False
                time.sleep(self.load_retry_interval_s)
        return False

    # -- polling thread -------------------------------------------------
    def start(self):
        self.poll_once()
        if self.poll_wait_seconds > 0:
            self._thread = threading.Thread(target=self._loop, daemon=True)
            self._thread.start()
        return self

    def stop(self):
        self._stop.set()
        if self._thread:
            self._thread.join(timeout=5)

    def _loop(self):
        while not self._stop.wait(self.poll_wait_seconds):
            try:
                self.poll_once()
            except Exception:  # noqa: BLE001
                logger.exception("poll cycle failed")
