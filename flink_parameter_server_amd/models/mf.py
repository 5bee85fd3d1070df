"""Matrix-factorization SGD on the parameter server.

Reference: ``matrix/factorization/PSOnlineMatrixFactorization.scala``
(``psOnlineMF(ratings, numFactors, rangeMin, rangeMax, learningRate,
negativeSampleRate, userMemory, pullLimit, workerParallelism,
psParallelism, iterationWaitTime)``) and
``PSOfflineMatrixFactorization.scala`` (epoch sweeps) — SURVEY.md §3.3.

Model split (reference behavior, kept): user factors are WORKER-LOCAL
state (the reference bounds them with a ``userMemory`` LRU; on MI355X with
288 GB HBM we keep the worker's user partition dense and resident); item
factors live on the parameter-server shards.  Per batch of ratings the
worker pulls the item vectors, runs the fused SGD step (HIP kernel K4 on
GPU: e = r - u.i; u += lr*(e*i - reg*u); push lr*(e*u - reg*i)), and pushes
the item deltas; the server applies them with paramUpdate='+'
(``SimplePSLogic(RangedRandomFactorInitializer, vectorSum)``).

Negative sampling (reference: ``negativeSampleRate`` random items per
positive, rating 0) is generated on the fly with the deterministic hash
RNG so no host RNG state is carried.
"""
from __future__ import annotations

from typing import Iterator, List, NamedTuple, Optional, Tuple

import torch

from .. import ops
from ..api import ParameterServerClient, WorkerLogic
from .common import DenseCoalescer, HashCoalescer, PullBuffer


class Rating(NamedTuple):
    """Reference: ``matrix/factorization/utils/Rating.scala`` —
    ``Rating(user: Int, item: Int, rating: Double)`` (ids widened to
    int)."""
    user: int
    item: int
    rating: float


class SGDUpdater:
    """Record-mode parity twin of the fused HIP step (K4) — reference:
    ``matrix/factorization/utils/SGDUpdater.scala``:
    ``delta(rating, userVec, itemVec)`` returns ``(userDelta, itemDelta)``
    with ``e = r - u.i``, ``du = lr*(e*i - reg*u)``, ``di = lr*(e*u -
    reg*i)`` (reg = 0 matches the reference's plain rule)."""

    def __init__(self, learning_rate: float, reg: float = 0.0):
        self.lr = float(learning_rate)
        self.reg = float(reg)

    def delta(self, rating: float, user_vec: torch.Tensor,
              item_vec: torch.Tensor
              ) -> Tuple[torch.Tensor, torch.Tensor]:
        e = float(rating) - float(user_vec @ item_vec)
        du = self.lr * (e * item_vec - self.reg * user_vec)
        di = self.lr * (e * user_vec - self.reg * item_vec)
        return du, di


class MFBatch:
    """One batch of ratings local to a worker: user rows are indices into
    the worker's local user-factor table."""

    __slots__ = ("user_rows", "item_ids", "ratings")

    def __init__(self, user_rows: torch.Tensor, item_ids: torch.Tensor,
                 ratings: torch.Tensor):
        self.user_rows = user_rows
        self.item_ids = item_ids
        self.ratings = ratings

    def __len__(self) -> int:
        return int(self.user_rows.numel())


class OnlineMFWorkerLogic(WorkerLogic):
    def __init__(self, n_local_users: int, n_factors: int,
                 learning_rate: float, reg: float = 0.0,
                 range_min: float = -0.01, range_max: float = 0.01,
                 negative_sample_rate: int = 0, n_items: int = 0,
                 device: torch.device = torch.device("cpu"),
                 pull_limit: Optional[int] = None,
                 output_user_factors: bool = False,
                 output_user_factors_on_close: bool = False,
                 assume_static_batches: bool = False,
                 aggregate: str = "sum",
                 dense_pull: Optional[bool] = None,
                 seed: int = 1):
        """``assume_static_batches=True``: the data source re-yields the
        SAME batch objects with unchanged index tensors (epoch sweeps
        over a resident dataset — the offline-MF pattern); the fused
        kernel then caches its sort positions per batch.  Never set it
        for sources that rewrite batch buffers in place.

        ``aggregate``: "sum" (reference parity — a batch's updates for a
        row are the SUM over its occurrences, so a hot key's effective
        step scales with its batch multiplicity) or "mean" (new
        capability — item deltas AND user updates averaged per row,
        keeping the effective per-row step at lr regardless of batch
        composition; the stable choice for large batches over skewed
        keys)."""
        self.n_factors = int(n_factors)
        self.lr = float(learning_rate)
        self.reg = float(reg)
        self.neg_rate = int(negative_sample_rate)
        self.n_items = int(n_items)
        self.device = torch.device(device)
        self.pull_limit = pull_limit
        self.output_user_factors = output_user_factors
        self.output_on_close = (output_user_factors or
                                output_user_factors_on_close)
        self.assume_static = bool(assume_static_batches)
        if aggregate not in ("sum", "mean"):
            raise ValueError(f"unknown aggregate {aggregate!r}")
        self.aggregate = aggregate
        self._cnt_cache: dict = {}
        self._uid_cache: dict = {}
        self._neg_counter = 0
        self._seed = seed
        # Worker-local user table (reference: per-subtask user map with
        # deterministic ranged-random init).
        self.user_table = torch.empty(n_local_users, n_factors,
                                      dtype=torch.float32, device=self.device)
        if n_local_users:
            ops.hash_init_rows(self.user_table, 0, 1, range_min, range_max,
                               seed=seed + 7777)
        self._pending: List[Tuple[MFBatch, PullBuffer, Optional[torch.Tensor]]] = []
        # Pull policy for a known item space, fastest first:
        # * dense pull (n_items <= DENSE_PULL_MAX): pull the WHOLE item
        #   range every round — at MovieLens-like item counts a 1M-rating
        #   batch touches essentially every item anyway, the answer is a
        #   sub-MB payload, slots are the raw item ids, and the whole
        #   coalescing pass (flag scatter + cumsum + nonzero + lookup,
        #   ~15% of a round's kernel time) disappears.
        # * dense-flag coalescing: O(n) scatter + cumsum instead of a
        #   sort-based torch.unique, rating->slot lookup for free.
        # * torch.unique fallback for unbounded id spaces.
        self._coalescer: Optional[DenseCoalescer] = None
        self._all_items: Optional[torch.Tensor] = None
        # dense_pull=False forces per-batch sparse pulls (the
        # coalescer/unique path) even for small item spaces — the honest
        # pull/push-ops regime the HBM co-headline benchmarks measure;
        # None = auto by item-space size.
        self._dense_pull = (0 < self.n_items <= self.DENSE_PULL_MAX
                            if dense_pull is None else
                            bool(dense_pull and self.n_items > 0))
        if (not self._dense_pull and
                0 < self.n_items <= DenseCoalescer.MAX_SPACE):
            self._coalescer = DenseCoalescer(self.n_items, self.device)

    def batch_size(self, batch: MFBatch) -> int:
        return len(batch)

    #: item spaces up to this size are pulled whole each round
    DENSE_PULL_MAX = 65536

    def on_recv(self, data: MFBatch, client: ParameterServerClient) -> None:
        if self.neg_rate > 0:
            data = self._with_negatives(data)
        if self._dense_pull:
            if self._all_items is None:
                self._all_items = torch.arange(
                    self.n_items, dtype=torch.int64,
                    device=data.item_ids.device)
            uids, slots = self._all_items, data.item_ids
            buf = PullBuffer(uids, self.n_factors, assume_sorted=True)
            self._pending.append((data, buf, slots))
            client.pull(uids, unique=True)
            return
        # sparse pull: per static batch the coalescing (uids, slots) is
        # computed once and reused on every epoch revisit — the stable
        # slot tensors then also unlock the fused kernel's record cache
        cached = None
        if self.assume_static and self.neg_rate == 0:
            hit = self._uid_cache.get(id(data))
            if hit is not None and hit[0] is data.item_ids:
                cached = hit
        if cached is not None:
            _, uids, slots, resolver = cached
        elif self._coalescer is not None:
            uids, slots = self._coalescer(data.item_ids)
            resolver = None
        elif data.item_ids.is_cuda:
            # unbounded item space: hash coalescing (no sort/searchsorted)
            hc = HashCoalescer()
            uids, slots = hc(data.item_ids.long())
            resolver = hc.resolver
        else:
            uids = torch.unique(data.item_ids)
            slots = resolver = None
        if cached is None and self.assume_static and self.neg_rate == 0:
            if len(self._uid_cache) > 64:
                self._uid_cache.clear()
            self._uid_cache[id(data)] = (data.item_ids, uids, slots,
                                         resolver)
        buf = (PullBuffer(uids, self.n_factors, resolver=resolver)
               if resolver is not None else
               PullBuffer(uids, self.n_factors, assume_sorted=True))
        self._pending.append((data, buf, slots))
        client.pull(uids, unique=True)

    def on_pull_recv(self, ids: torch.Tensor, values: torch.Tensor,
                     client: ParameterServerClient) -> None:
        if (len(self._pending) == 1 and
                self._pending[0][1]._n_have == 0 and
                ids.numel() == self._pending[0][1].uids.numel()):
            # likely a whole-pull answer: offer_exact adopts it sync-
            # free on pointer identity, else verifies membership.  Only
            # a COMPLETE buffer trains — a same-sized answer for a
            # different overlapping pull (possible when merged/
            # pipelined schedules keep two pulls in flight) fills just
            # the overlap and keeps waiting for its own answer.
            batch, buf, slots = self._pending[0]
            buf.offer_exact(ids, values)
            if buf.complete:
                self._pending = []
                self._train(batch, buf, client, slots)
            return
        still: List[Tuple[MFBatch, PullBuffer, Optional[torch.Tensor]]] = []
        for batch, buf, slots in self._pending:
            buf.offer(ids, values)
            if buf.complete:
                self._train(batch, buf, client, slots)
            else:
                still.append((batch, buf, slots))
        self._pending = still

    def _train(self, batch: MFBatch, buf: PullBuffer,
               client: ParameterServerClient,
               slots: Optional[torch.Tensor] = None) -> None:
        if slots is None:
            slots = buf.slots_of(batch.item_ids)
        # cacheable when the slot tensors are stable across revisits:
        # dense pull (slots are the batch's own item ids) or the cached
        # sparse coalescing above
        stable = self._dense_pull
        if not stable and self.assume_static and self.neg_rate == 0:
            hit = self._uid_cache.get(id(batch))
            stable = hit is not None and hit[2] is slots
        tag = id(batch) if (self.assume_static and stable and
                            self.neg_rate == 0) else 0
        deltas = ops.mf_sgd_step(self.user_table, batch.user_rows,
                                 buf.values, slots, batch.ratings,
                                 self.lr, self.reg, reuse_tag=tag,
                                 mean_users=self.aggregate == "mean")
        if self.aggregate == "mean":
            deltas = deltas * self._inv_counts(batch, slots,
                                               buf.uids.numel())
        client.push(buf.uids, deltas)
        if self.output_user_factors:
            client.output((batch.user_rows,
                           self.user_table[batch.user_rows].clone()))

    def _inv_counts(self, batch: MFBatch, slots: torch.Tensor,
                    m: int) -> torch.Tensor:
        """(m, 1) reciprocal occurrence counts of each pulled row in the
        batch (cached per static batch object)."""
        if self.assume_static:
            hit = self._cnt_cache.get(id(batch))
            if hit is not None and hit[0] is slots:
                return hit[1]
        cnt = torch.bincount(slots.long(), minlength=m).clamp_(min=1)
        inv = cnt.reciprocal().to(torch.float32).unsqueeze(1)
        if self.assume_static:
            if len(self._cnt_cache) > 64:
                self._cnt_cache.clear()
            self._cnt_cache[id(batch)] = (slots, inv)
        return inv

    def _with_negatives(self, data: MFBatch) -> MFBatch:
        """rate-N hash-RNG negative items per positive, rating 0
        (reference: negativeSampleRate)."""
        n = len(data)
        k = self.neg_rate
        keys = (torch.arange(n * k, dtype=torch.int64,
                             device=self.device)
                + self._neg_counter * (n * k + 1))
        self._neg_counter += 1
        neg_items = ops.reference.splitmix64(keys + self._seed).abs() \
            % max(1, self.n_items)
        neg_users = data.user_rows.repeat_interleave(k)
        return MFBatch(
            torch.cat([data.user_rows, neg_users]),
            torch.cat([data.item_ids, neg_items]),
            torch.cat([data.ratings,
                       torch.zeros(n * k, dtype=torch.float32,
                                   device=self.device)]))

    def close(self, client: ParameterServerClient) -> None:
        if self.output_on_close:
            rows = torch.arange(self.user_table.shape[0],
                                dtype=torch.int64, device=self.device)
            client.output((rows, self.user_table.clone()))


def epoch_iterator(batches: List[MFBatch], epochs: int
                   ) -> Iterator[MFBatch]:
    """Offline MF (``PSOfflineMatrixFactorization``): the worker sweeps its
    buffered rating partition in epochs; user factors persist across
    epochs in the worker-local table, item factors accumulate on the PS.
    """
    for _ in range(epochs):
        for b in batches:
            yield b
