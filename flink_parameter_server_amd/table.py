"""HBM-resident sharded parameter table.

This is the MI355X-native replacement for the reference's
``server/SimplePSLogic.scala`` ``mutable.HashMap[Int, P]`` (SURVEY.md
§2.1/§2.3 K1–K3): each shard keeps its parameter partition as one dense
f32 slab in HBM (288 GB per GPU fits every BASELINE config dense — e.g.
1e9 features x 2 f32 = 8 GB over 8 shards), initialized eagerly with the
deterministic per-id hash init (semantically identical to the reference's
lazy ``getOrElseUpdate(id, paramInit(id))`` because init is a pure function
of the id), updated by HIP scatter-apply kernels.
"""
from __future__ import annotations

from typing import Tuple

import torch

from . import ops
from .parallel.partitioner import Partitioner


class ShardedTable:
    """One shard (this rank's partition) of the global parameter table.

    optimizer:
      "add"     — W[id] += delta           (reference paramUpdate = vectorSum /
                                            '+'; the worker scales by lr)
      "adagrad" — G += g^2; W -= lr*g/sqrt(G+eps)   (new capability)
    """

    def __init__(self, part: Partitioner, shard: int, dim: int,
                 device: torch.device,
                 init: Tuple[float, float] = (0.0, 0.0),
                 optimizer: str = "add", lr: float = 0.0,
                 eps: float = 1e-8, seed: int = 0):
        self.part = part
        self.shard = shard
        self._full_ids: "torch.Tensor | None" = None
        self.dim = int(dim)
        self.device = torch.device(device)
        self.optimizer = optimizer
        self.lr = float(lr)
        self.eps = float(eps)
        rows = part.shard_rows(shard)
        self.rows = rows
        self.data = torch.empty(rows, dim, dtype=torch.float32,
                                device=self.device)
        lo, hi = init
        if lo == 0.0 and hi == 0.0:
            self.data.zero_()
        else:
            # Deterministic per-global-id init: row r holds global id
            # part.global_id(shard, r).  For ModPartitioner that is
            # first_id=shard, stride=n_shards; for RangePartitioner
            # first_id=shard*block, stride=1.
            first = int(self.part.global_id(
                self.shard, torch.zeros(1, dtype=torch.int64)).item()) if rows else 0
            if rows >= 2:
                second = int(self.part.global_id(
                    self.shard, torch.ones(1, dtype=torch.int64)).item())
                stride = second - first
            else:
                stride = 1
            if rows:
                ops.hash_init_rows(self.data, first, stride, lo, hi, seed)
        if optimizer == "adagrad":
            self.accum = torch.zeros(rows, dim, dtype=torch.float32,
                                     device=self.device)
        elif optimizer == "add":
            self.accum = None
        else:
            raise ValueError(f"unknown optimizer {optimizer!r}")

    # -- server-side ops (K3 gather / K1 sgd apply / K2 adagrad apply) ----

    def gather(self, ids: torch.Tensor) -> torch.Tensor:
        if self._is_full_range(ids):
            # dense pull of the whole shard: the "gather" is the slab
            # itself.  Answer values alias the live table; the engine's
            # round order (worker consumes answers BEFORE this round's
            # pushes apply) makes that observationally identical to a
            # copy taken at answer time under fresh/merged schedules.
            # Under "pipelined" a self-addressed alias is consumed one
            # round later, after one more local push-apply — the worker
            # then sees FRESHER values than gather time, which the
            # async pull contract allows (answers carry no freshness
            # guarantee; remote segments are real copies either way).
            return self.data
        rows = self.part.local_row(ids)
        return ops.gather(self.data, rows)

    def _is_full_range(self, ids: torch.Tensor) -> bool:
        """ids == every row of this shard in local-row order?  Verified
        with one tensor compare on first sight, then recognized by
        buffer IDENTITY (dense-pull workers re-send one cached tensor).
        The verified tensor itself is retained in the cache entry — as
        runtime._shard_counts and the HIP position cache do — so its
        storage stays alive and a recycled allocation of the same
        pointer (e.g. a fresh multi-GPU recv buffer each round) can
        never be misclassified as the cached full-range pull."""
        if ids.numel() != self.rows or self.rows == 0:
            return False
        cached = self._full_ids
        if cached is not None and ids is cached:
            return True
        rows = self.part.local_row(ids)
        expect = torch.arange(self.rows, dtype=rows.dtype,
                              device=rows.device)
        if bool(torch.equal(rows, expect)):
            self._full_ids = ids
            return True
        return False

    def apply(self, ids: torch.Tensor, deltas: torch.Tensor,
              unique: bool = False) -> None:
        rows = self.part.local_row(ids)
        if self.optimizer == "add":
            ops.apply_add(self.data, rows, deltas, unique)
        else:
            # `unique` batches arrive as sorted unique pulled ids, and
            # both partitioners are monotone per shard, so the local
            # rows are already sorted — the adagrad pre-sort is skipped.
            ops.apply_adagrad(self.data, self.accum, rows, deltas,
                              self.lr, self.eps, sorted_unique=unique)

    def dump(self) -> Tuple[torch.Tensor, torch.Tensor]:
        """(global_ids, values) of every row of the shard — the reference's
        ``ParameterServerLogic.close`` model dump (unordered contract;
        ours happens to be local-row order)."""
        rows = torch.arange(self.rows, dtype=torch.int64)
        gids = self.part.global_id(self.shard, rows)
        return gids, self.data

    def load(self, ids: torch.Tensor, values: torch.Tensor) -> None:
        """Warm start (``transformWithModelLoad``): overwrite rows by id."""
        rows = self.part.local_row(ids).to(self.device)
        self.data.index_copy_(0, rows, values.to(self.device,
                                                 self.data.dtype))


class TableServerLogic:
    """ServerLogic over a ShardedTable — the production default server.

    Semantics parity: ``SimplePSLogic(paramInit, paramUpdate)`` with
    paramInit = deterministic ranged-random init and paramUpdate = '+'
    (or the Adagrad rule).
    """

    #: answers every pull batch in full, in received order, same round —
    #: lets the engine elide answer ids from the feedback edge.
    answers_in_order = True

    def __init__(self, table: ShardedTable, dump_on_close: bool = True,
                 pushes_unique: bool = False):
        """``pushes_unique=True``: the application guarantees every push
        batch carries SORTED distinct ids (all bundled model logics do —
        they push pre-aggregated deltas per sorted unique pulled id, one
        push per round), enabling the non-atomic add fast path and the
        sort-free adagrad path.  Leave False when pushes may repeat or
        interleave ids (e.g. under a count/time flush strategy that
        coalesces several logical pushes into one round)."""
        self.table = table
        self.dump_on_close = dump_on_close
        self.pushes_unique = pushes_unique

    def on_pull_recv(self, ids: torch.Tensor, worker: int, ps) -> None:
        ps.answer_pull(ids, self.table.gather(ids), worker)

    def on_push_recv(self, ids: torch.Tensor, deltas: torch.Tensor,
                     ps) -> None:
        self.table.apply(ids, deltas, unique=self.pushes_unique)

    def close(self, ps) -> None:
        if self.dump_on_close:
            ps.output(self.table.dump())
