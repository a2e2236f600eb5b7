"""K6: distributed gradient accumulation.

The reference's APRIL-ANN task reduces per-worker gradients through the
MapReduce shuffle (emit(weight_name, grad) -> reducefn axpy sum,
common.lua:85-137).  On an MI355X node the gradients of all ranks are
co-resident in HBM, so the reduce maps to a bucketed RCCL allreduce over
xGMI; the local many-shard sum is the grad_colsum kernel
(VALU float4 path; an MFMA variant exists for the record — A/B in
profiles/, the op is bandwidth-bound so VALU wins)."""

from __future__ import annotations

from typing import Dict

import torch
import torch.distributed as dist

from .. import ops
from . import dist as dx


def local_shard_sum(grads: torch.Tensor, use_mfma: bool = False
                    ) -> torch.Tensor:
    """Sum G local gradient shards [G, D] -> [D]."""
    if grads.is_cuda:
        return ops.ext().grad_colsum(grads, use_mfma)
    return grads.sum(0)


def allreduce_gradients(named_grads: Dict[str, torch.Tensor],
                        group=None) -> Dict[str, torch.Tensor]:
    """Bucketed cross-rank gradient sum: flatten into one buffer, ONE
    RCCL allreduce (xGMI ring is per-link bound, so fewer+larger
    collectives — guide), then unflatten."""
    rank, world = dx.world_info(group)
    names = sorted(named_grads)
    flats = [named_grads[n].reshape(-1) for n in names]
    buf = torch.cat(flats)
    if world > 1:
        dist.all_reduce(buf, op=dist.ReduceOp.SUM, group=group)
    out = {}
    off = 0
    for n in names:
        g = named_grads[n]
        out[n] = buf[off:off + g.numel()].view_as(g)
        off += g.numel()
    return out
