"""Parameter-id -> shard partitioners.

Reference: the custom partitioners built inside
``FlinkParameterServer.transform`` (``hash(paramId) % psParallelism``; Scala
``Int.hashCode`` is the identity) and the contiguous range partitioning
option of the passive-aggressive path
(``PassiveAggressiveParameterServer.transformBinary(rangePartitioning=...)``).

Device execution note: these functions are pure torch tensor ops, so on
CUDA inputs the SAME code runs as device kernels — host/device
bit-match is by construction (pinned by tests/test_partitioner.py and
the GPU engine tests), with no separate HIP twin to keep in sync.
"""
from __future__ import annotations

from typing import Tuple

import torch


class Partitioner:
    """Maps int64 parameter ids to (shard, local_row)."""

    n_shards: int

    def shard_of(self, ids: torch.Tensor) -> torch.Tensor:
        raise NotImplementedError

    def local_row(self, ids: torch.Tensor) -> torch.Tensor:
        """Row of each id inside its shard's dense slab."""
        raise NotImplementedError

    def global_id(self, shard: int, rows: torch.Tensor) -> torch.Tensor:
        """Inverse of (shard_of, local_row) — used by checkpoint dump."""
        raise NotImplementedError

    def shard_rows(self, shard: int) -> int:
        """Dense slab height for `shard`."""
        raise NotImplementedError


class ModPartitioner(Partitioner):
    """shard = id % n_shards, local_row = id // n_shards.

    Bijective for ids in [0, num_ids); matches the reference's hash
    partitioner (identity hashCode mod psParallelism).
    """

    def __init__(self, num_ids: int, n_shards: int):
        self.num_ids = int(num_ids)
        self.n_shards = int(n_shards)

    def shard_of(self, ids: torch.Tensor) -> torch.Tensor:
        if self.n_shards == 1:
            return torch.zeros_like(ids)
        return ids % self.n_shards

    def local_row(self, ids: torch.Tensor) -> torch.Tensor:
        if self.n_shards == 1:  # identity — skip the int64 div kernel
            return ids
        return ids // self.n_shards

    def global_id(self, shard: int, rows: torch.Tensor) -> torch.Tensor:
        return rows * self.n_shards + shard

    def shard_rows(self, shard: int) -> int:
        if shard >= self.n_shards:  # rank hosts no shard (ps_parallelism
            return 0                # < world — see algorithms.py)
        return (self.num_ids - shard + self.n_shards - 1) // self.n_shards


class RangePartitioner(Partitioner):
    """Contiguous blocks: shard s owns [s*block, (s+1)*block) (last shard
    takes the remainder).  The PA/LR path option."""

    def __init__(self, num_ids: int, n_shards: int):
        self.num_ids = int(num_ids)
        self.n_shards = int(n_shards)
        self.block = (self.num_ids + n_shards - 1) // n_shards

    def shard_of(self, ids: torch.Tensor) -> torch.Tensor:
        return torch.clamp(ids // self.block, max=self.n_shards - 1)

    def local_row(self, ids: torch.Tensor) -> torch.Tensor:
        shard = self.shard_of(ids)
        return ids - shard * self.block

    def global_id(self, shard: int, rows: torch.Tensor) -> torch.Tensor:
        return rows + shard * self.block

    def shard_rows(self, shard: int) -> int:
        if shard >= self.n_shards:
            return 0
        lo = shard * self.block
        hi = min(self.num_ids, (shard + 1) * self.block)
        return max(0, hi - lo)


def make_partitioner(kind: str, num_ids: int, n_shards: int) -> Partitioner:
    if kind == "mod":
        return ModPartitioner(num_ids, n_shards)
    if kind == "range":
        return RangePartitioner(num_ids, n_shards)
    raise ValueError(f"unknown partitioner {kind!r}")


def bucket_by_shard(ids: torch.Tensor, part: Partitioner
                    ) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    """Stable-partition ids by destination shard (K10 bucket_pack host/CPU
    reference).

    Returns (order, counts, shards):
      order  : (n,) permutation so that ids[order] is grouped by shard asc.
      counts : (n_shards,) int64 number of ids per shard
      shards : (n,) shard of each (un-permuted) id
    """
    shards = part.shard_of(ids)
    order = torch.argsort(shards, stable=True)
    counts = torch.bincount(shards, minlength=part.n_shards)
    return order, counts, shards
