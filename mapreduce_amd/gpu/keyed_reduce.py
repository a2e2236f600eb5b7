"""Distributed reduce-by-key over raw (key, value) tensor columns.

The K5/K6 hot path as a public primitive: users who
already hold their emitted pairs as device tensors — i64 keys (u64 bit
patterns) and i64/f64 values — get the reference's declared-property
reducer semantics (job.lua:104-106: associative+commutative sum, or the
idempotent min/max) without writing a task script:

    job = KeyedReduceJob(device, op="sum")
    ukeys, reduced = job.run(keys, vals)

Two-level reduction, exactly the reference's combiner-then-reduce split
(job.lua:198-201 map-side combine, :264-284 reduce): sort + segmented
reduce locally first (the combiner — shrinks the shuffle to unique
keys), mulhi-partition the unique keys, one RCCL all-to-all per column
over xGMI, then sort + reduce the received runs.  Exact for sum/min/max
by associativity+commutativity (min/max also idempotent, so re-applying
at both levels is safe).  Multi-rank key ownership matches the engine:
rank r owns keys with mulhi(key, world) == r.
"""

from __future__ import annotations

from typing import Tuple

import torch

from .. import ops
from . import dist as dx


class KeyedReduceJob:
    def __init__(self, device, group=None, op: str = "sum"):
        if op not in ("sum", "min", "max"):
            raise ValueError(f"unsupported op {op!r}")
        self.device = torch.device(device)
        self.group = group
        self.rank, self.world = dx.world_info(group)
        self.op = op

    def run(self, keys: torch.Tensor,
            vals: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor]:
        """keys: i64 (u64 bit order); vals: i64 or f64, one per key.
        Returns (unique_keys_sorted, reduced) — this rank's partition of
        the global reduction."""
        if keys.numel() != vals.numel():
            raise ValueError("one value per key required")
        # local combine: sort + segmented reduce (K1 + K5)
        sk, sv = ops.sort_by_key(keys, vals)
        uk, uv, _, _ = ops.reduce_by_key_sorted(sk, sv, op=self.op)
        if self.world == 1:
            return uk, uv
        # shuffle unique keys by mulhi partition (C5/C6): sorted keys are
        # partition-contiguous, so the exchange is two sliced all-to-alls
        counts = ops.partition_counts(uk, self.world)
        recv = dx.exchange_counts(counts, self.group)
        sc = counts.cpu().tolist()
        rc = recv.cpu().tolist()
        rk = dx.exchange(uk, sc, rc, self.group)
        rv = dx.exchange(uv, sc, rc, self.group)
        # final reduce of the received per-rank runs
        k2, v2 = ops.sort_by_key(rk, rv)
        fk, fv, _, _ = ops.reduce_by_key_sorted(k2, v2, op=self.op)
        return fk, fv
