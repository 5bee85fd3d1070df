"""Sketch algorithms on the parameter server.

Reference: ``sketch/**`` ([LOW] — bloom-filter /
tug-of-war sketch frequency estimation on the PS; the reference mount
was empty in rounds 1 and 2, so existence on the main branch could not
be verified either way).  Built defensively to close the inventory row:
the three classic PS-sketches, expressed MI355X-first as *batched*
pushes of pre-aggregated deltas into the '+' sharded table — a sketch is
exactly a parameter table whose ids are hash cells, so the whole
existing engine (bucketing, alltoallv, scatter-apply kernels, dump/load)
carries them with no new kernel:

* :class:`CountMinWorkerLogic` — count-min sketch: d hash rows x w
  cells; per occurrence push +1 into cell ``(r, h_r(id))``; the
  frequency estimate is ``min_r cell[r][h_r(id)]`` (one-sided: never an
  under-estimate).
* :class:`BloomFilterWorkerLogic` — counting Bloom filter (membership,
  no false negatives): same cells with d rows; present iff every cell
  is non-zero.
* :class:`AMSSketchWorkerLogic` — tug-of-war (AMS) F2 sketch: push the
  +/-1 sign ``s_r(id)``; the second-moment estimate is the median over
  rows of ``sum_w cell^2``.

All hashing is the framework's deterministic splitmix64 (bit-identical
host/device — ops/reference.py), so any rank/shard can compute any
cell id with no coordination, and the sketch state re-shards through
the ordinary checkpoint path.
"""
from __future__ import annotations

from typing import List

import torch

from ..api import ParameterServerClient, WorkerLogic
from ..ops.reference import splitmix64


def _cell_ids(ids: torch.Tensor, row: int, width: int, seed: int
              ) -> torch.Tensor:
    """Global parameter id of sketch cell (row, h_row(id)) — rows are
    laid out as contiguous [row*width, (row+1)*width) blocks so a range
    partitioner keeps each row contiguous per shard if desired."""
    h = splitmix64(ids.to(torch.int64) + (seed + row) * 0x9E3779B9)
    return (h % width).abs() + row * width


def _signs(ids: torch.Tensor, row: int, seed: int) -> torch.Tensor:
    h = splitmix64(ids.to(torch.int64) + (seed + row) * 0x51ED2705)
    return ((h & 1) * 2 - 1).to(torch.float32)


class _SketchBase(WorkerLogic):
    """Shared machinery: per batch of ids, aggregate occurrence counts
    (sorted-unique push — the table's non-atomic fast path), map to d
    rows of cells, push one coalesced delta batch."""

    def __init__(self, depth: int, width: int, seed: int = 17,
                 device: torch.device = torch.device("cpu")):
        if depth < 1 or width < 1:
            raise ValueError("depth and width must be >= 1")
        self.depth = int(depth)
        self.width = int(width)
        self.seed = int(seed)
        self.device = torch.device(device)
        self.pushed_ids = 0

    @property
    def num_ids(self) -> int:
        """Parameter-table size this sketch needs (PSConfig.num_ids)."""
        return self.depth * self.width

    def _delta_for(self, ids: torch.Tensor, counts: torch.Tensor,
                   row: int) -> torch.Tensor:
        raise NotImplementedError

    def on_recv(self, data: torch.Tensor,
                client: ParameterServerClient) -> None:
        ids = torch.as_tensor(data, dtype=torch.int64,
                              device=self.device).reshape(-1)
        uids, counts = torch.unique(ids, return_counts=True)
        cells: List[torch.Tensor] = []
        deltas: List[torch.Tensor] = []
        for r in range(self.depth):
            cells.append(_cell_ids(uids, r, self.width, self.seed))
            deltas.append(self._delta_for(uids, counts, r))
        cell = torch.cat(cells)
        delta = torch.cat(deltas).reshape(-1, 1)
        # distinct rows can hash-collide onto one cell within the batch:
        # aggregate so the push is sorted-unique (table fast path)
        ucell, inv = torch.unique(cell, return_inverse=True)
        agg = torch.zeros(ucell.numel(), dtype=torch.float32,
                          device=delta.device)
        agg.index_add_(0, inv, delta.reshape(-1))
        client.push(ucell, agg.reshape(-1, 1))
        self.pushed_ids += int(ucell.numel())

    def on_pull_recv(self, ids: torch.Tensor, values: torch.Tensor,
                     client: ParameterServerClient) -> None:
        # queries are answered synchronously from the merged table in
        # estimate(); streaming pull-based queries would land here
        pass

    # -- offline estimation over a dumped/merged table -----------------

    def _table_cells(self, table: torch.Tensor, ids: torch.Tensor
                     ) -> torch.Tensor:
        """(len(ids), depth) cell values for each query id."""
        ids = torch.as_tensor(ids, dtype=torch.int64).reshape(-1)
        cols = []
        for r in range(self.depth):
            cols.append(table.reshape(-1)[_cell_ids(ids, r, self.width,
                                                    self.seed)])
        return torch.stack(cols, dim=1)


class CountMinWorkerLogic(_SketchBase):
    """Count-min sketch: estimate(id) = min over rows; one-sided error
    <= eps*N with prob 1-delta at width=ceil(e/eps), depth=ceil(ln 1/δ)."""

    def _delta_for(self, ids, counts, row):
        return counts.to(torch.float32)

    def estimate(self, table: torch.Tensor, ids: torch.Tensor
                 ) -> torch.Tensor:
        return self._table_cells(table, ids).min(dim=1).values


class BloomFilterWorkerLogic(_SketchBase):
    """Counting Bloom filter: contains(id) = all cells non-zero (no
    false negatives; false-positive rate set by depth/width)."""

    def _delta_for(self, ids, counts, row):
        return torch.ones_like(counts, dtype=torch.float32)

    def contains(self, table: torch.Tensor, ids: torch.Tensor
                 ) -> torch.Tensor:
        return (self._table_cells(table, ids) > 0).all(dim=1)


class AMSSketchWorkerLogic(_SketchBase):
    """Tug-of-war (AMS) sketch for the second frequency moment F2:
    each row r holds sum_id s_r(id)*count(id) per cell; the estimator is
    the median over rows of sum_cells cell^2."""

    def _delta_for(self, ids, counts, row):
        return _signs(ids, row, self.seed) * counts.to(torch.float32)

    def f2_estimate(self, table: torch.Tensor) -> float:
        t = table.reshape(self.depth, self.width)
        per_row = (t * t).sum(dim=1)
        return float(per_row.median())


def merge_shard_dumps(dumps, depth: int, width: int) -> torch.Tensor:
    """Reassemble the full (depth*width, 1) sketch table from per-shard
    ``table.dump()`` outputs (any shard count — the checkpoint layout
    contract)."""
    full = torch.zeros(depth * width, 1)
    for gids, vals in dumps:
        full[gids.cpu()] = vals.cpu().reshape(-1, 1)
    return full
