"""Data-parallel request sharding over RCCL/xGMI (BASELINE config 4).

One process per GPU (``torch.distributed``, backend "nccl" == RCCL on
ROCm; "gloo" for CPU tests). Rank 0 holds the global batch; ``predict``
scatters it along dim 0 across the group, every rank round-trips its shard
against its local PredictionService, and the responses are all-gathered so
every rank holds the full output batch.

xGMI design note: each MI355X has 7 point-to-point xGMI links
(~153 GB/s each). ``dist.scatter`` from rank 0 is a 1->N fan-out that uses
all of the root's links simultaneously — exactly the right collective for
this topology (a ring would serialize onto one link). The all-gather is
RCCL's standard algorithm over the same links. The reference system has no
multi-GPU path at all (client is single-process synchronous);
the vendored NcclManager (nccl_manager.cc:688-760) is the pattern source
for stream/event discipline, which torch.distributed supplies natively.
"""
from __future__ import annotations

from typing import Dict, List, Optional

import torch
import torch.distributed as dist


def shard_sizes(total: int, world: int) -> List[int]:
    """Uneven-tail split of `total` rows over `world` ranks (generality
    requirement from SURVEY §7 hard-part 4)."""
    base, rem = divmod(total, world)
    return [base + (1 if r < rem else 0) for r in range(world)]


class DataParallelPredictor:
    """Scatter -> local predict -> all-gather, over an initialized process
    group."""

    def __init__(self, client, group=None, device: Optional[str] = None):
        self.client = client
        self.group = group
        self.rank = dist.get_rank(group)
        self.world = dist.get_world_size(group)
        if device is None:
            device = (f"cuda:{torch.cuda.current_device()}"
                      if torch.cuda.is_available() else "cpu")
        self.device = torch.device(device)

    # -- collectives ----------------------------------------------------
    def _scatter(self, full: Optional[torch.Tensor],
                 meta: Dict) -> torch.Tensor:
        """Scatter dim-0 shards of `full` (defined on rank 0) to all
        ranks. Returns this rank's shard. Pads to equal shard sizes for
        the collective, trims after."""
        sizes = meta["sizes"]
        pad_rows = max(sizes)
        shape_tail = meta["shape_tail"]
        dtype = meta["dtype"]
        shard = torch.empty((pad_rows, *shape_tail), dtype=dtype,
                            device=self.device)
        if self.rank == 0:
            chunks = []
            off = 0
            for s in sizes:
                c = full[off:off + s]
                if s < pad_rows:
                    pad = torch.zeros((pad_rows - s, *shape_tail),
                                      dtype=dtype, device=full.device)
                    c = torch.cat([c, pad], dim=0)
                chunks.append(c.contiguous().to(self.device))
                off += s
            dist.scatter(shard, chunks, src=0, group=self.group)
        else:
            dist.scatter(shard, None, src=0, group=self.group)
        return shard[:sizes[self.rank]]

    def _gathered_row_counts(self, shard: torch.Tensor) -> List[int]:
        """All-gathers every rank's ACTUAL output dim-0 size (a model's
        output rows need not match its input shard rows — e.g. detection
        models emit variable counts). Small fixed-size tensor collective,
        so it works identically on gloo and RCCL."""
        local = torch.tensor([shard.shape[0]], dtype=torch.int64,
                             device=shard.device)
        out = [torch.empty_like(local) for _ in range(self.world)]
        dist.all_gather(out, local, group=self.group)
        return [int(t.item()) for t in out]

    def _all_gather(self, shard: torch.Tensor,
                    sizes: List[int]) -> torch.Tensor:
        pad_rows = max(sizes)
        if pad_rows == 0:
            return shard[:0]
        padded = shard
        if shard.shape[0] < pad_rows:
            pad = torch.zeros((pad_rows - shard.shape[0],
                               *shard.shape[1:]), dtype=shard.dtype,
                              device=shard.device)
            padded = torch.cat([shard, pad], dim=0)
        out = [torch.empty_like(padded) for _ in range(self.world)]
        dist.all_gather(out, padded.contiguous(), group=self.group)
        return torch.cat([t[:s] for t, s in zip(out, sizes)], dim=0)

    # -- public ---------------------------------------------------------
    def predict(self, model_name: str,
                global_inputs: Optional[Dict[str, torch.Tensor]],
                timeout: float = 60.0,
                model_version: Optional[int] = None,
                gather_outputs: bool = True
                ) -> Dict[str, torch.Tensor]:
        """`global_inputs` is significant on rank 0 only. Returns the full
        gathered outputs on every rank (or this rank's shard outputs when
        gather_outputs=False)."""
        # metadata broadcast: key order, dtypes, shapes, shard sizes
        if self.rank == 0:
            keys = sorted(global_inputs.keys())
            metas = {}
            for k in keys:
                t = global_inputs[k]
                metas[k] = {
                    "sizes": shard_sizes(int(t.shape[0]), self.world),
                    "shape_tail": tuple(t.shape[1:]),
                    "dtype": t.dtype,
                }
            obj = [keys, metas]
        else:
            obj = [None, None]
        dist.broadcast_object_list(obj, src=0, group=self.group)
        keys, metas = obj

        shard_inputs = {}
        for k in keys:
            full = global_inputs[k] if self.rank == 0 else None
            shard_inputs[k] = self._scatter(full, metas[k])

        shard_out = self.client.predict(
            model_name, shard_inputs, timeout=timeout,
            model_version=model_version,
            output_device=str(self.device))
        if not gather_outputs:
            return shard_out

        gathered = {}
        for k in sorted(shard_out.keys()):
            v = shard_out[k]
            if not isinstance(v, torch.Tensor) or v.dim() == 0:
                gathered[k] = v
                continue
            # gather each rank's true output row count first (round-1 bug:
            # assuming row alignment with input shard sizes mis-gathers
            # models whose output dim-0 differs per shard)
            v = v.to(self.device)
            out_sizes = self._gathered_row_counts(v)
            gathered[k] = self._all_gather(v, out_sizes)
        return gathered
