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
from .common import PullBuffer
from .pa import SparseBatch


class LRWorkerLogic(WorkerLogic):
    def __init__(self, pull_limit: Optional[int] = None,
                 emit_predictions: bool = False):
        self.pull_limit = pull_limit
        self.emit_predictions = emit_predictions
        self._pending: List[Tuple[SparseBatch, PullBuffer]] = []

    def batch_size(self, batch: SparseBatch) -> int:
        return len(batch)

    def on_recv(self, data: SparseBatch,
                client: ParameterServerClient) -> None:
        uids = torch.unique(data.feat_ids)
        buf = PullBuffer(uids, 1)
        self._pending.append((data, buf))
        client.pull(uids)

    def on_pull_recv(self, ids: torch.Tensor, values: torch.Tensor,
                     client: ParameterServerClient) -> None:
        still = []
        for batch, buf in self._pending:
            buf.offer(ids, values)
            if buf.complete:
                self._train(batch, buf, client)
            else:
                still.append((batch, buf))
        self._pending = still

    def _train(self, batch: SparseBatch, buf: PullBuffer,
               client: ParameterServerClient) -> None:
        slots = buf.slots_of(batch.feat_ids)
        probs, grads = ops.lr_step(batch.indptr, slots, batch.x_vals,
                                   batch.labels, buf.values.reshape(-1))
        if self.emit_predictions:
            unlabeled = torch.isnan(batch.labels)
            if bool(unlabeled.any()):
                client.output((torch.nonzero(unlabeled).reshape(-1),
                               probs[unlabeled]))
        client.push(buf.uids, grads.reshape(-1, 1))
