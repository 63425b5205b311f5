"""Pluggable request logging with sampling.

Analogue of the reference's request-logger stack (SURVEY §5:
core/request_logger.h:33, core/server_request_logger.h:40,
config/logging_config.proto): each logged request/response pair becomes a
``PredictionLog`` record (prediction_log.proto) written through a log
collector — here a TFRecord file collector compatible with our warmup
reader, so logged traffic can be replayed as warmup directly.
"""
from __future__ import annotations

import random
import threading
from typing import Dict, Optional

from .repository import write_tfrecord
from .wire import messages as pb


class FileLogCollector:
    """LogCollectorConfig{type="file", filename_prefix=...} analogue:
    appends TFRecord-framed PredictionLog records to one file per model."""

    def __init__(self, filename_prefix: str):
        self.filename_prefix = filename_prefix
        self._lock = threading.Lock()
        self._buffers: Dict[str, list] = {}

    def collect(self, model_name: str, record: bytes,
                flush_every: int = 256):
        with self._lock:
            buf = self._buffers.setdefault(model_name, [])
            buf.append(record)
            if len(buf) >= flush_every:
                self._flush_locked(model_name)

    def _flush_locked(self, model: str):
        records = self._buffers.get(model) or []
        if records:
            # append: a long-running server must not re-buffer or rewrite
            # its whole log history
            write_tfrecord(f"{self.filename_prefix}.{model}.log", records,
                           append=True)
            self._buffers[model] = []

    def flush(self):
        with self._lock:
            for model in list(self._buffers):
                self._flush_locked(model)


class RequestLogger:
    """Per-model logging with SamplingConfig.sampling_rate semantics
    (logging_config.proto:7-9): each request is logged with probability
    sampling_rate."""

    def __init__(self, collector: FileLogCollector,
                 sampling_rate: float = 1.0,
                 seed: Optional[int] = None):
        self.collector = collector
        self.sampling_rate = sampling_rate
        self._rng = random.Random(seed)
        self.logged = 0
        self.seen = 0

    def log_predict(self, request_bytes_or_msg, response_bytes_or_msg,
                    model_name: str = ""):
        self.seen += 1
        if self._rng.random() >= self.sampling_rate:
            return False
        log = pb.PredictionLog()
        if isinstance(request_bytes_or_msg, bytes):
            log.predict_log.request.MergeFromString(request_bytes_or_msg)
        else:
            log.predict_log.request.CopyFrom(request_bytes_or_msg)
        if response_bytes_or_msg is not None:
            if isinstance(response_bytes_or_msg, bytes):
                log.predict_log.response.MergeFromString(
                    response_bytes_or_msg)
            else:
                log.predict_log.response.CopyFrom(response_bytes_or_msg)
        name = model_name or log.predict_log.request.model_spec.name
        log.log_metadata.model_spec.CopyFrom(
            log.predict_log.request.model_spec)
        log.log_metadata.sampling_config.sampling_rate = self.sampling_rate
        self.collector.collect(name, log.SerializeToString())
        self.logged += 1
        return True


class ServerRequestLogger:
    """Maps model name -> RequestLogger (server_request_logger.h:40
    analogue); built from {model_name: LoggingConfig}."""

    def __init__(self):
        self._loggers: Dict[str, RequestLogger] = {}
        # observers called on every (re)configure — the native transport
        # drops logged models from its C++ echo fast path so every logged
        # request passes through the python handler
        self._subscribers = []

    def subscribe(self, callback) -> None:
        self._subscribers.append(callback)

    def logged_models(self):
        return set(self._loggers)

    def configure(self, model_name: str, logging_config,
                  collector: Optional[FileLogCollector] = None):
        """logging_config: pb.LoggingConfig (or None to remove)."""
        if logging_config is None:
            self._loggers.pop(model_name, None)
        else:
            prefix = (logging_config.log_collector_config.filename_prefix
                      or f"/tmp/prediction_log_{model_name}")
            rate = logging_config.sampling_config.sampling_rate or 1.0
            self._loggers[model_name] = RequestLogger(
                collector or FileLogCollector(prefix), sampling_rate=rate)
        for cb in list(self._subscribers):
            try:
                cb()
            except Exception:  # noqa: BLE001
                pass

    def get(self, model_name: str) -> Optional[RequestLogger]:
        return self._loggers.get(model_name)

    def log_predict(self, model_name, request, response):
        logger = self._loggers.get(model_name)
        if logger is not None:
            logger.log_predict(request, response, model_name)

    def flush_all(self):
        for logger in self._loggers.values():
            logger.collector.flush()
