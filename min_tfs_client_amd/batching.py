"""Server-side request batching along dim 0.

Analogue of TF-Serving's BatchingSession (reference
batching/batching_session.h:80-101 + basic_batch_scheduler.h):
concurrent Predict requests for the same servable are merged along the
batch dimension, run once, and split back — with ``allowed_batch_sizes``
padding semantics (pad to the next allowed size, strip after) and a
``batch_timeout`` bound on queue latency.

MI355X note: merging requests is what fills the 256-CU chip — a batch-1
model launch leaves most CUs idle; the batcher turns k concurrent batch-b
requests into one batch-k*b launch.
"""
from __future__ import annotations

import threading
import time
from typing import Dict, List, Optional, Sequence

import numpy as np

try:
    import torch
except ImportError:  # pragma: no cover
    torch = None

from .server import Servable


class _Pending:
    __slots__ = ("inputs", "batch_size", "event", "result", "error")

    def __init__(self, inputs, batch_size):
        self.inputs = inputs
        self.batch_size = batch_size
        self.event = threading.Event()
        self.result = None
        self.error = None


def _dim0(v) -> int:
    if torch is not None and isinstance(v, torch.Tensor):
        return int(v.shape[0]) if v.dim() > 0 else -1
    arr = np.asarray(v)
    return int(arr.shape[0]) if arr.ndim > 0 else -1


def _cat(tensors: Sequence):
    if torch is not None and isinstance(tensors[0], torch.Tensor):
        return torch.cat(list(tensors), dim=0)
    return np.concatenate([np.asarray(t) for t in tensors], axis=0)


def _zeros_like_rows(v, rows: int):
    if torch is not None and isinstance(v, torch.Tensor):
        return torch.zeros((rows,) + tuple(v.shape[1:]), dtype=v.dtype,
                           device=v.device)
    arr = np.asarray(v)
    return np.zeros((rows,) + arr.shape[1:], dtype=arr.dtype)


class BatchingServable(Servable):
    """Wraps a Servable; callers block until their slice of the merged
    batch's result is ready (BasicBatchScheduler semantics: a full batch
    fires immediately, otherwise the oldest request waits at most
    ``batch_timeout_s``)."""

    def __init__(self, inner: Servable, max_batch_size: int = 256,
                 batch_timeout_s: float = 0.002,
                 allowed_batch_sizes: Optional[List[int]] = None,
                 max_enqueued_batches: int = 64):
        super().__init__(self._call_inner, inner.signature,
                         inner.signature_name)
        self.inner = inner
        self.max_batch_size = max_batch_size
        self.batch_timeout_s = batch_timeout_s
        if allowed_batch_sizes is not None:
            if sorted(allowed_batch_sizes) != list(allowed_batch_sizes):
                raise ValueError("allowed_batch_sizes must be sorted")
            if allowed_batch_sizes[-1] != max_batch_size:
                # TF requires the final entry to equal max_batch_size
                # (batching_session.h:92-97)
                raise ValueError("final allowed_batch_sizes entry must "
                                 "equal max_batch_size")
        self.allowed_batch_sizes = allowed_batch_sizes
        self._lock = threading.Condition()
        self._queue: List[_Pending] = []
        self._max_queue = max_enqueued_batches
        self._closed = False
        self.batches_run = 0  # introspection for tests/metrics
        self._thread = threading.Thread(target=self._loop, daemon=True)
        self._thread.start()

    # -- client-facing entry (runs on gRPC worker threads) ---------------
    def _call_inner(self, inputs: Dict):
        sizes = {k: _dim0(v) for k, v in inputs.items()}
        batch = next(iter(sizes.values()))
        if batch < 0 or any(s != batch for s in sizes.values()):
            # unbatchable request: run it alone (TF rejects; we degrade)
            return self.inner(inputs)
        if batch > self.max_batch_size:
            raise ValueError(
                f"request batch size {batch} > max_batch_size "
                f"{self.max_batch_size}")
        p = _Pending(inputs, batch)
        with self._lock:
            if self._closed:
                raise RuntimeError("batching servable is shut down")
            if len(self._queue) >= self._max_queue:
                raise RuntimeError("batching queue full")
            self._queue.append(p)
            self._lock.notify()
        p.event.wait()
        if p.error is not None:
            raise p.error
        return p.result

    def close(self):
        with self._lock:
            self._closed = True
            self._lock.notify()

    # -- batcher thread ---------------------------------------------------
    def _take_batch(self) -> List[_Pending]:
        """Blocks until a batch is ready: full, or timeout after first."""
        with self._lock:
            while not self._queue and not self._closed:
                self._lock.wait()
            if self._closed and not self._queue:
                return []
            deadline = time.monotonic() + self.batch_timeout_s
            while True:
                total = sum(p.batch_size for p in self._queue)
                if total >= self.max_batch_size or self._closed:
                    break
                remaining = deadline - time.monotonic()
                if remaining <= 0:
                    break
                self._lock.wait(timeout=remaining)
            taken, acc = [], 0
            while self._queue and \
                    acc + self._queue[0].batch_size <= self.max_batch_size:
                p = self._queue.pop(0)
                taken.append(p)
                acc += p.batch_size
            return taken

    def _loop(self):
        while True:
            taken = self._take_batch()
            if not taken:
                # drain: fail anything that raced past the closed check
                with self._lock:
                    leftover = self._queue
                    self._queue = []
                for p in leftover:
                    p.error = RuntimeError("batching servable is shut down")
                    p.event.set()
                return
            try:
                self._run_batch(taken)
            except Exception as e:  # noqa: BLE001
                for p in taken:
                    p.error = e
                    p.event.set()

    def _run_batch(self, taken: List[_Pending]):
        keys = list(taken[0].inputs.keys())
        # singleton with mismatched keys: run alone
        groups = [p for p in taken if list(p.inputs.keys()) == keys]
        rest = [p for p in taken if list(p.inputs.keys()) != keys]
        for p in rest:
            try:
                p.result = self.inner(p.inputs)
            except Exception as e:  # noqa: BLE001
                p.error = e
            p.event.set()
        if not groups:
            return
        total = sum(p.batch_size for p in groups)
        padded = self._padded_size(total)
        merged = {}
        for k in keys:
            parts = [p.inputs[k] for p in groups]
            if padded > total:
                parts.append(_zeros_like_rows(parts[0], padded - total))
            merged[k] = _cat(parts)
        outputs = self.inner(merged)
        self.batches_run += 1
        offset = 0
        for p in groups:
            sliced = {}
            for k, v in outputs.items():
                if _dim0(v) == padded:
                    sliced[k] = v[offset:offset + p.batch_size]
                else:  # non-batched output: replicate reference behavior
                    sliced[k] = v
            p.result = sliced
            p.event.set()
            offset += p.batch_size

    def _padded_size(self, n: int) -> int:
        if not self.allowed_batch_sizes:
            return n
        for s in self.allowed_batch_sizes:
            if s >= n:
                return s
        return n
