"""Shared worker-side machinery for the model logics.

``PullBuffer`` implements the "wait until every pulled parameter of a batch
has been answered" reassembly the reference does per instance in the
passive-aggressive worker (``PassiveAggressiveParameterServer`` inner
logic) and per rating in the MF worker
(``PSOnlineMatrixFactorization`` pending-pull queueing) —
here batched: one buffer per training batch, filled as answer segments
arrive (answers may arrive split across rounds under a pull limit, and the
same id may be answered more than once when overlapping batches pull it;
a duplicate answer must not double-count).
"""
from __future__ import annotations

from typing import Optional

import torch


class PullBuffer:
    """Collects pulled rows for a fixed set of unique ids.

    ``resolver``: optional callable mapping an id tensor to buffer rows
    (foreign ids -> -1).  Used by the hash-coalesced mode, whose uids
    are unique but UNSORTED (hash-position order) — searchsorted does
    not apply there.  Without a resolver uids must be (or are made)
    sorted.
    """

    def __init__(self, uids: torch.Tensor, dim: int,
                 assume_sorted: bool = False, resolver=None):
        if not assume_sorted and resolver is None:
            uids, _ = torch.sort(uids)
        self.uids = uids
        self.resolver = resolver
        self.values = torch.empty(uids.numel(), dim, dtype=torch.float32,
                                  device=uids.device)
        # lazily allocated: the whole-answer fast path never needs the
        # per-row flags (two kernel launches per round saved)
        self._have_t: Optional[torch.Tensor] = None
        self._n_have = 0

    @property
    def _have(self) -> torch.Tensor:
        if self._have_t is None:
            self._have_t = torch.zeros(self.uids.numel(),
                                       dtype=torch.bool,
                                       device=self.uids.device)
        return self._have_t

    def slots_of(self, ids: torch.Tensor) -> torch.Tensor:
        """Buffer row of each id (caller guarantees membership)."""
        if self.resolver is not None:
            return self.resolver(ids)
        return torch.searchsorted(self.uids, ids)

    def offer_exact(self, ids: torch.Tensor, values: torch.Tensor) -> None:
        """Fast path for a whole-pull answer.  ONLY the pointer-identity
        case may skip verification (in-order answers return the very
        tensor we pulled — dense pull / single shard); any other
        delivery goes through the membership-checked :meth:`offer`.
        (A count-match heuristic here once let a WRONG-BATCH answer
        with near-total id overlap fill the buffer unverified — with
        merged/pipelined schedules keeping two overlapping pulls in
        flight, the misalignment cascaded batch-by-batch into NaN
        training.  Never fill rows from unverified ids.)"""
        if (ids.data_ptr() == self.uids.data_ptr()
                and ids.numel() == self.uids.numel()
                and values.dtype == self.values.dtype):
            # adopt the payload, skip the permute
            self.values = values
            if self._have_t is not None:
                self._have_t.fill_(True)
            self._n_have = int(self.uids.numel())
            return
        self.offer(ids, values)

    def offer(self, ids: torch.Tensor, values: torch.Tensor) -> None:
        """Fill rows answered by (ids, values); ignores foreign ids and
        rows already filled."""
        if self.complete or not self.uids.numel():
            return
        if self.resolver is not None:
            pos = self.resolver(ids).long()
            hit = pos >= 0
            rows = pos[hit]
        else:
            pos = torch.searchsorted(self.uids, ids)
            pos_c = torch.clamp(pos, max=self.uids.numel() - 1)
            hit = self.uids[pos_c] == ids
            rows = pos_c[hit]
        if not bool(hit.any()):
            return
        fresh = ~self._have[rows]
        rows = rows[fresh]
        if not rows.numel():
            return
        self.values[rows] = values[hit][fresh].to(self.values.dtype)
        self._have[rows] = True
        self._n_have += int(rows.numel())

    @property
    def complete(self) -> bool:
        return self._n_have >= self.uids.numel()


class DenseCoalescer:
    """Sort-free id coalescing for bounded id spaces: a dense flag
    scatter + int32 cumsum replaces sort-based ``torch.unique`` and gives
    the id->slot map for free (measured: rocPRIM sort-unique was the
    second-largest kernel cost of the MF/SGNS/LR rounds).

    Usable when the id space fits a dense flag array (space <= 2^27 by
    default); returns (sorted unique ids, slot of each input id).
    """

    MAX_SPACE = 1 << 27

    def __init__(self, space: int, device):
        self.space = int(space)
        self.present = torch.zeros(self.space, dtype=torch.bool,
                                   device=device)

    def __call__(self, ids: torch.Tensor):
        """Returns (sorted unique int64 ids, int32 slot of each input id)
        — int32 slots feed the HIP kernels directly (half the metadata
        bytes of the sort/pack passes)."""
        present = self.present
        present.zero_()
        idx = ids if ids.dtype == torch.int64 else ids.long()
        present[idx] = True
        lookup = torch.cumsum(present, 0, dtype=torch.int32)
        uids = present.nonzero().reshape(-1)
        slots = lookup[idx] - 1
        return uids, slots


class HashCoalescer:
    """Device-side id coalescing for UNBOUNDED id spaces (the 1e9-
    feature LR config — anything too big for :class:`DenseCoalescer`'s
    dense flag array).

    One open-addressing hash pass replaces the torch.unique radix sort
    + the per-element torch.searchsorted that dominated the LR round
    (measured 54% + ~20%: profiles/PERF_NOTES.md round 2).  Returns
    (unsorted-unique uids, int32 slots) and keeps the table alive so
    later answer segments resolve rows via :meth:`resolver` (pull
    answers arrive shard-bucketed, i.e. permuted, at world > 1).
    Unsorted uids are sufficient everywhere: unique=True pulls skip the
    engine's coalescing, bucketing handles any order, and the adagrad
    apply only needs duplicates adjacent — unique ids have none.
    """

    def __init__(self):
        self._keys = None
        self._slotmap = None

    def __call__(self, ids: torch.Tensor):
        from .. import ops
        uids, slots, self._keys, self._slotmap = \
            ops.hip.ext().hash_coalesce(ids)
        return uids, slots

    @property
    def resolver(self):
        keys, slotmap = self._keys, self._slotmap
        from .. import ops
        ext = ops.hip.ext()

        def _resolve(ids: torch.Tensor) -> torch.Tensor:
            return ext.hash_lookup(keys, slotmap, ids)

        return _resolve
