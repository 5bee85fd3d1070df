"""Word2vec skip-gram with negative sampling (SGNS) on the PS.

New capability (BASELINE.json configs[2]: 1M vocab x dim 300, embedding
table sharded across 8x MI355X) — SURVEY.md K8.

Model: ONE sharded table of 2*V rows — rows [0, V) are center ("input")
embeddings, rows [V, 2V) are context ("output") embeddings; both sharded
by id like any other parameter, so the pull/push machinery is unchanged.
Per (center, context) positive pair the worker adds `neg` hash-RNG
negative context ids (unigram sampling left to the data generator),
pulls all touched rows, runs the fused SGNS step (dots + sigmoid + axpys,
HIP kernel K8) and pushes deltas for center and context rows.
"""
from __future__ import annotations

from typing import List, Optional

import torch

from .. import ops
from ..api import ParameterServerClient, WorkerLogic
from .common import DenseCoalescer, HashCoalescer, PullBuffer


class SGNSBatch:
    """Positive skip-gram pairs (center, context) of vocab ids [0, V)."""

    __slots__ = ("centers", "contexts")

    def __init__(self, centers: torch.Tensor, contexts: torch.Tensor):
        self.centers = centers
        self.contexts = contexts

    def __len__(self) -> int:
        return int(self.centers.numel())


class SGNSWorkerLogic(WorkerLogic):
    def __init__(self, vocab_size: int, dim: int, learning_rate: float,
                 negatives: int = 5,
                 pull_limit: Optional[int] = None,
                 dense_pull: bool = False, seed: int = 11):
        """``dense_pull=True``: pull the WHOLE 2V-row table every round
        (slots are then the raw table ids — the per-batch coalescing
        pass and the answer gather disappear; with one shard the answer
        aliases the table slab, so the pull is free).  Same policy as
        the MF worker's dense-pull mode; right for world=1 or small
        vocabularies, wrong for a big table sharded across ranks (the
        answer payload would ship whole shards over xGMI every round)."""
        self.V = int(vocab_size)
        self.dim = int(dim)
        self.lr = float(learning_rate)
        self.neg = int(negatives)
        self.pull_limit = pull_limit
        self.dense_pull = bool(dense_pull)
        self._seed = seed
        self._counter = 0
        # pending: (center_ids, ctx_table_ids per pair row, labels, buffer)
        self._pending: List[tuple] = []
        self._coalescer: Optional[DenseCoalescer] = None
        self._all_rows: Optional[torch.Tensor] = None
        self._device: Optional[torch.device] = None

    def batch_size(self, batch: SGNSBatch) -> int:
        return len(batch)

    def on_recv(self, data: SGNSBatch,
                client: ParameterServerClient) -> None:
        n = len(data)
        dev = data.centers.device
        k = self.neg
        # expand each positive into (1 positive + k negatives) pair rows;
        # on device one fused kernel builds rows + hash-RNG negatives +
        # labels (bit-identical negative sequence to this torch path)
        base_key = self._counter * (n * k + 1) + self._seed
        self._counter += 1
        if dev.type == "cuda":
            from ..ops import hip
            centers, ctx_table_ids, labels = hip.ext().sgns_expand(
                data.centers, data.contexts, k, self.V, base_key)
        else:
            centers = data.centers.repeat_interleave(k + 1)
            keys = (torch.arange(n * k, dtype=torch.int64, device=dev)
                    + base_key)
            neg_ctx = ops.reference.splitmix64(keys).abs() % self.V
            ctx = torch.empty(n * (k + 1), dtype=torch.int64, device=dev)
            ctx.view(n, k + 1)[:, 0] = data.contexts
            ctx.view(n, k + 1)[:, 1:] = neg_ctx.view(n, k)
            labels = torch.zeros(n * (k + 1), dtype=torch.float32,
                                 device=dev)
            labels.view(n, k + 1)[:, 0] = 1.0
            ctx_table_ids = ctx + self.V  # context rows live at [V, 2V)
        if self.dense_pull:
            # whole-table pull: slots ARE the table ids, no coalescing
            if self._all_rows is None or self._all_rows.device != dev:
                self._all_rows = torch.arange(2 * self.V,
                                              dtype=torch.int64,
                                              device=dev)
            uids = self._all_rows
            c_slots, o_slots = centers, ctx_table_ids
            buf = PullBuffer(uids, self.dim, assume_sorted=True)
            self._pending.append((centers, ctx_table_ids, labels, buf,
                                  c_slots, o_slots))
            client.pull(uids, unique=True)
            return
        allids = torch.cat([centers, ctx_table_ids])
        if self._coalescer is None and 2 * self.V <= DenseCoalescer.MAX_SPACE:
            self._coalescer = DenseCoalescer(2 * self.V, dev)
        if self._coalescer is not None:
            uids, slots = self._coalescer(allids)
            c_slots, o_slots = slots[:n * (k + 1)], slots[n * (k + 1):]
            buf = PullBuffer(uids, self.dim, assume_sorted=True)
        elif dev.type == "cuda":
            # unbounded vocab: hash coalescing (no sort/searchsorted)
            hc = HashCoalescer()
            uids, slots = hc(allids)
            c_slots, o_slots = slots[:n * (k + 1)], slots[n * (k + 1):]
            buf = PullBuffer(uids, self.dim, resolver=hc.resolver)
        else:
            uids = torch.unique(allids)
            c_slots = o_slots = None
            buf = PullBuffer(uids, self.dim, assume_sorted=True)
        self._pending.append((centers, ctx_table_ids, labels, buf,
                              c_slots, o_slots))
        client.pull(uids, unique=True)

    def on_pull_recv(self, ids: torch.Tensor, values: torch.Tensor,
                     client: ParameterServerClient) -> None:
        if (len(self._pending) == 1 and
                self._pending[0][3]._n_have == 0 and
                ids.numel() == self._pending[0][3].uids.numel()):
            centers, ctx_ids, labels, buf, cs, os_ = self._pending[0]
            buf.offer_exact(ids, values)
            if buf.complete:  # see mf.py: same-size wrong-pull guard
                self._pending = []
                self._train(centers, ctx_ids, labels, buf, client,
                            cs, os_)
            return
        still = []
        for centers, ctx_ids, labels, buf, cs, os_ in self._pending:
            buf.offer(ids, values)
            if buf.complete:
                self._train(centers, ctx_ids, labels, buf, client, cs, os_)
            else:
                still.append((centers, ctx_ids, labels, buf, cs, os_))
        self._pending = still

    def _train(self, centers: torch.Tensor, ctx_ids: torch.Tensor,
               labels: torch.Tensor, buf: PullBuffer,
               client: ParameterServerClient,
               c_slots: Optional[torch.Tensor] = None,
               o_slots: Optional[torch.Tensor] = None) -> None:
        if c_slots is None:
            c_slots = buf.slots_of(centers)
            o_slots = buf.slots_of(ctx_ids)
        # One buffer holds both center and context rows; slot spaces are
        # disjoint ids so the deltas share one accumulator.
        deltas = ops.sgns_step(buf.values, c_slots, o_slots, labels,
                               self.lr)
        client.push(buf.uids, deltas)
