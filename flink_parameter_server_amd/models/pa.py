"""Passive-aggressive classification (binary + multiclass) on the PS.

Reference: ``passive/aggressive/PassiveAggressiveParameterServer.scala``
(``transformBinary(model?)(trainingData, workerParallelism, psParallelism,
paMethod, pullLimit, featureCount, rangePartitioning, iterationWaitTime)``
and ``transformMulticlass(...)``) plus
``passive/aggressive/algorithm/*.scala`` (PA / PA-I / PA-II tau rules) —
SURVEY.md §3.4.

Model: per-feature weights sharded by feature id (paramId = feature
index; P = scalar for binary, per-class row for multiclass).  Data =
sparse instances with optional labels — labeled instances train,
unlabeled ones emit a prediction (the reference's streaming-predict path).
The worker pulls the weights of the batch's non-zero features, waits for
all of them (the reference's per-instance reassembly barrier), computes
margins and tau, and pushes tau*y*x deltas; server update is '+' with
zero init.
"""
from __future__ import annotations

from typing import List, Optional, Tuple

import torch

from .. import ops
from ..api import ParameterServerClient, WorkerLogic
from .common import DenseCoalescer, HashCoalescer, PullBuffer


class SparseBatch:
    """CSR batch of sparse instances.

    labels: binary — f32 in {-1,+1}, NaN = unlabeled (predict);
            multiclass — int64 class index, -1 = unlabeled.
    ids: optional per-instance global ids carried through to prediction
         outputs (None = outputs use batch-local row indices).
    """

    __slots__ = ("indptr", "feat_ids", "x_vals", "labels", "ids")

    def __init__(self, indptr: torch.Tensor, feat_ids: torch.Tensor,
                 x_vals: torch.Tensor, labels: torch.Tensor,
                 ids: Optional[torch.Tensor] = None):
        self.indptr = indptr
        self.feat_ids = feat_ids
        self.x_vals = x_vals
        self.labels = labels
        self.ids = ids

    def __len__(self) -> int:
        return int(self.indptr.numel()) - 1


class PAWorkerLogic(WorkerLogic):
    """Binary (n_classes=0) or multiclass (n_classes>=2) PA worker."""

    def __init__(self, variant: str = "PA-I", C: float = 1.0,
                 n_classes: int = 0,
                 pull_limit: Optional[int] = None,
                 emit_predictions: bool = True,
                 n_features: int = 0,
                 device: torch.device = torch.device("cpu")):
        if variant not in ("PA", "PA-I", "PA-II"):
            raise ValueError(f"unknown PA variant {variant!r}")
        self.variant = variant
        self.C = float(C)
        self.n_classes = int(n_classes)
        self.dim = max(1, self.n_classes)
        self.pull_limit = pull_limit
        self.emit_predictions = emit_predictions
        self._pending: List[Tuple[SparseBatch, PullBuffer]] = []
        # feature-space-aware coalescing like the LR worker: dense-flag
        # scatter for bounded spaces, sort-free hash for unbounded ones
        # on device; torch.unique fallback otherwise
        self._coalescer = None
        if 0 < n_features <= DenseCoalescer.MAX_SPACE:
            self._coalescer = DenseCoalescer(n_features, device)

    def batch_size(self, batch: SparseBatch) -> int:
        return len(batch)

    def on_recv(self, data: SparseBatch,
                client: ParameterServerClient) -> None:
        if self._coalescer is not None:
            uids, _ = self._coalescer(data.feat_ids)
            buf = PullBuffer(uids, self.dim, assume_sorted=True)
        elif data.feat_ids.is_cuda:
            hc = HashCoalescer()
            uids, _ = hc(data.feat_ids.long())
            buf = PullBuffer(uids, self.dim, resolver=hc.resolver)
        else:
            uids = torch.unique(data.feat_ids)
            buf = PullBuffer(uids, self.dim, assume_sorted=True)
        self._pending.append((data, buf))
        client.pull(uids, unique=True)

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
        if self.n_classes == 0:
            margins, deltas = ops.pa_step(
                batch.indptr, slots, batch.x_vals, batch.labels,
                buf.values.reshape(-1), self.variant, self.C)
            if self.emit_predictions:
                unlabeled = torch.isnan(batch.labels)
                if bool(unlabeled.any()):
                    pred = torch.sign(margins[unlabeled])
                    rows = torch.nonzero(unlabeled).reshape(-1)
                    out_ids = batch.ids[rows] if batch.ids is not None \
                        else rows
                    client.output((out_ids, pred))
            client.push(buf.uids, deltas.reshape(-1, 1))
        else:
            scores, deltas = ops.pa_multiclass_step(
                batch.indptr, slots, batch.x_vals, batch.labels,
                buf.values, self.variant, self.C)
            if self.emit_predictions:
                unlabeled = batch.labels < 0
                if bool(unlabeled.any()):
                    pred = scores[unlabeled].argmax(dim=1)
                    rows = torch.nonzero(unlabeled).reshape(-1)
                    out_ids = batch.ids[rows] if batch.ids is not None \
                        else rows
                    client.output((out_ids, pred))
            client.push(buf.uids, deltas)
