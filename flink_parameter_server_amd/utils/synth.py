"""Synthetic data generators for tests and benchmarks.

There is no network in this environment, so every benchmark
runs on synthetic data of the named shapes: MovieLens-1M-shaped ratings
for MF, zipfian sparse instances for PA/LR, zipf-unigram skip-gram pairs
for SGNS.  Shapes follow BASELINE.json's configs.
"""
from __future__ import annotations

import math
from typing import Iterator, List, Optional, Tuple

import torch

from ..models.mf import MFBatch
from ..models.pa import SparseBatch
from ..models.sgns import SGNSBatch


def mf_ratings(n_users: int, n_items: int, n_ratings: int, batch: int,
               device: torch.device = torch.device("cpu"),
               seed: int = 0, zipf_items: bool = True
               ) -> Iterator[MFBatch]:
    """Rating stream shaped like MovieLens (zipfian item popularity).
    user_rows are already worker-local row indices in [0, n_users)."""
    g = torch.Generator().manual_seed(seed)
    remaining = n_ratings
    while remaining > 0:
        b = min(batch, remaining)
        remaining -= b
        users = torch.randint(0, n_users, (b,), generator=g)
        if zipf_items:
            u = torch.rand(b, generator=g)
            # approximate zipf via inverse-power transform
            items = (n_items * u.pow(3.0)).long().clamp_(0, n_items - 1)
        else:
            items = torch.randint(0, n_items, (b,), generator=g)
        ratings = (torch.rand(b, generator=g) * 4.0 + 1.0)
        yield MFBatch(users.to(device), items.to(device),
                      ratings.to(device))


def planted_mf_ratings(n_users: int, n_items: int, rank: int,
                       n_ratings: int, batch: int, seed: int = 0,
                       device: torch.device = torch.device("cpu")
                       ) -> Tuple[List[MFBatch], torch.Tensor, torch.Tensor]:
    """Ratings from a planted low-rank model (for convergence tests).
    Returns (batches, U_true, V_true)."""
    g = torch.Generator().manual_seed(seed)
    U = torch.randn(n_users, rank, generator=g) / math.sqrt(rank)
    V = torch.randn(n_items, rank, generator=g) / math.sqrt(rank)
    batches = []
    for start in range(0, n_ratings, batch):
        b = min(batch, n_ratings - start)
        users = torch.randint(0, n_users, (b,), generator=g)
        items = torch.randint(0, n_items, (b,), generator=g)
        r = (U[users] * V[items]).sum(dim=1)
        batches.append(MFBatch(users.to(device), items.to(device),
                               r.to(device)))
    return batches, U, V


def sparse_instances(n_features: int, nnz_per_instance: int,
                     n_instances: int, batch: int,
                     task: str = "binary",
                     n_classes: int = 3,
                     unlabeled_frac: float = 0.0,
                     device: torch.device = torch.device("cpu"),
                     seed: int = 0,
                     teacher_seed: Optional[int] = None
                     ) -> Iterator[SparseBatch]:
    """Zipfian sparse classification instances with a planted linear
    teacher (so training can be checked to converge).

    task: "binary" -> labels in {-1,+1} (NaN = unlabeled), teacher sign;
          "lr"     -> labels in {0,1}   (NaN = unlabeled);
          "multiclass" -> int64 labels (-1 = unlabeled).
    """
    tg = torch.Generator().manual_seed(
        seed if teacher_seed is None else teacher_seed)
    if task == "multiclass":
        teacher = torch.randn(n_features, n_classes, generator=tg)
    else:
        teacher = torch.randn(n_features, generator=tg)
    g = torch.Generator().manual_seed(seed)
    remaining = n_instances
    while remaining > 0:
        b = min(batch, remaining)
        remaining -= b
        nnz = b * nnz_per_instance
        u = torch.rand(nnz, generator=g)
        feats = (n_features * u.pow(2.0)).long().clamp_(0, n_features - 1)
        vals = torch.randn(nnz, generator=g)
        indptr = torch.arange(0, nnz + 1, nnz_per_instance,
                              dtype=torch.int64)
        seg = torch.repeat_interleave(torch.arange(b), nnz_per_instance)
        if task == "multiclass":
            scores = torch.zeros(b, n_classes)
            scores.index_add_(0, seg, teacher[feats] * vals.unsqueeze(1))
            labels = scores.argmax(dim=1)
            if unlabeled_frac > 0:
                mask = torch.rand(b, generator=g) < unlabeled_frac
                labels = labels.masked_fill(mask, -1)
        else:
            margins = torch.zeros(b)
            margins.index_add_(0, seg, teacher[feats] * vals)
            if task == "binary":
                labels = torch.where(margins >= 0,
                                     torch.ones(b), -torch.ones(b))
            else:  # lr
                labels = (margins >= 0).float()
            if unlabeled_frac > 0:
                mask = torch.rand(b, generator=g) < unlabeled_frac
                labels = labels.masked_fill(mask, float("nan"))
        yield SparseBatch(indptr.to(device), feats.to(device),
                          vals.to(device), labels.to(device))


def sgns_pairs(vocab: int, n_pairs: int, batch: int,
               device: torch.device = torch.device("cpu"),
               seed: int = 0) -> Iterator[SGNSBatch]:
    """Skip-gram positive pairs with zipfian unigram frequencies."""
    g = torch.Generator().manual_seed(seed)
    remaining = n_pairs
    while remaining > 0:
        b = min(batch, remaining)
        remaining -= b
        u = torch.rand(b, 2, generator=g)
        ids = (vocab * u.pow(3.0)).long().clamp_(0, vocab - 1)
        yield SGNSBatch(ids[:, 0].to(device), ids[:, 1].to(device))
