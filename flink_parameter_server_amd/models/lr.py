"""Sparse logistic regression with async Adagrad on the PS.

New capability (BASELINE.json configs[3]: 1e9 hashed features, async
Adagrad, sparse pull/push on 8x MI355X) — not in the reference, but built
on the same PA skeleton: worker pulls the weights of a
sparse batch's features, computes sigmoid margins, pushes raw gradients;
the SERVER applies Adagrad (G += g^2; W -= lr*g/sqrt(G+eps)) so the
optimizer state lives with the shard (HIP kernel K2).
"""
from __future__ import annotations

from typing import List, Optional, Tuple

import torch

from .. import ops
from ..api import ParameterServerClient, WorkerLogic
from .common import DenseCoalescer, HashCoalescer, PullBuffer
from .pa import SparseBatch


class LRWorkerLogic(WorkerLogic):
    def __init__(self, pull_limit: Optional[int] = None,
                 emit_predictions: bool = False,
                 n_features: int = 0,
                 device: torch.device = torch.device("cpu")):
        self.pull_limit = pull_limit
        self.emit_predictions = emit_predictions
        self._pending: List[Tuple[SparseBatch, PullBuffer,
                                  Optional[torch.Tensor]]] = []
        self._coalescer: Optional[DenseCoalescer] = None
        if 0 < n_features <= DenseCoalescer.MAX_SPACE:
            self._coalescer = DenseCoalescer(n_features, device)

    def batch_size(self, batch: SparseBatch) -> int:
        return len(batch)

    def on_recv(self, data: SparseBatch,
                client: ParameterServerClient) -> None:
        if self._coalescer is not None:
            uids, slots = self._coalescer(data.feat_ids)
            buf = PullBuffer(uids, 1, assume_sorted=True)
        elif data.feat_ids.is_cuda:
            # unbounded id space (1e9 hashed features): hash coalescing
            # — no radix sort, no searchsorted; uids unsorted-unique
            hc = HashCoalescer()
            uids, slots = hc(data.feat_ids)
            buf = PullBuffer(uids, 1, resolver=hc.resolver)
        else:
            uids, slots = torch.unique(data.feat_ids), None
            buf = PullBuffer(uids, 1, assume_sorted=True)
        self._pending.append((data, buf, slots))
        client.pull(uids, unique=True)

    def on_pull_recv(self, ids: torch.Tensor, values: torch.Tensor,
                     client: ParameterServerClient) -> None:
        if (len(self._pending) == 1 and
                self._pending[0][1]._n_have == 0 and
                ids.numel() == self._pending[0][1].uids.numel()):
            batch, buf, slots = self._pending[0]
            buf.offer_exact(ids, values)
            self._pending = []
            self._train(batch, buf, client, slots)
            return
        still = []
        for batch, buf, slots in self._pending:
            buf.offer(ids, values)
            if buf.complete:
                self._train(batch, buf, client, slots)
            else:
                still.append((batch, buf, slots))
        self._pending = still

    def _train(self, batch: SparseBatch, buf: PullBuffer,
               client: ParameterServerClient,
               slots: Optional[torch.Tensor] = None) -> None:
        if slots is None:
            slots = buf.slots_of(batch.feat_ids)
        probs, grads = ops.lr_step(batch.indptr, slots, batch.x_vals,
                                   batch.labels, buf.values.reshape(-1))
        if self.emit_predictions:
            unlabeled = torch.isnan(batch.labels)
            if bool(unlabeled.any()):
                rows = torch.nonzero(unlabeled).reshape(-1)
                out_ids = batch.ids[rows] if batch.ids is not None \
                    else rows
                client.output((out_ids, probs[unlabeled]))
        client.push(buf.uids, grads.reshape(-1, 1))
