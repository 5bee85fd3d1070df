"""Device-op dispatch.

CPU tensors -> plain-PyTorch fp32 reference (``ops.reference``), which is
also the semantics oracle for the kernels.
CUDA (= ROCm/MI355X) tensors -> hand-written HIP/CDNA4 kernels from the
in-tree extension; a missing extension raises (never a silent fallback).
"""
from __future__ import annotations

from typing import Tuple

import torch

from . import reference
from . import hip

__all__ = [
    "gather", "apply_add", "apply_adagrad", "hash_init",
    "mf_sgd_step", "pa_step", "pa_multiclass_step", "lr_step", "sgns_step",
    "reference", "hip",
]


def _cuda(t: torch.Tensor) -> bool:
    return t.is_cuda


def gather(table: torch.Tensor, rows: torch.Tensor) -> torch.Tensor:
    if _cuda(table):
        return hip.ext().gather(table, rows)
    return reference.gather(table, rows)


def apply_add(table: torch.Tensor, rows: torch.Tensor,
              deltas: torch.Tensor, unique: bool = False) -> None:
    if _cuda(table):
        hip.ext().apply_add(table, rows, deltas, unique)
        return
    reference.apply_add(table, rows, deltas, unique)


def apply_adagrad(table: torch.Tensor, accum: torch.Tensor,
                  rows: torch.Tensor, grads: torch.Tensor,
                  lr: float, eps: float = 1e-8,
                  sorted_unique: bool = False) -> None:
    """``sorted_unique``: the caller guarantees ``rows`` ascending with no
    duplicates (our model logics push per sorted unique pulled id), so
    the pre-sort is skipped."""
    if _cuda(table):
        # K2 contract: duplicates pre-aggregated, one RMW per distinct
        # row.  Sort the batch (device radix sort) and let segment
        # leaders aggregate in-kernel.
        if sorted_unique:
            hip.ext().apply_adagrad_sorted(table, accum,
                                           rows.contiguous(),
                                           grads.contiguous(), lr, eps)
            return
        srows, order = torch.sort(rows)
        hip.ext().apply_adagrad_sorted(table, accum, srows.contiguous(),
                                       grads.index_select(0, order)
                                       .contiguous(), lr, eps)
        return
    reference.apply_adagrad(table, accum, rows, grads, lr, eps)


def hash_init(ids: torch.Tensor, dim: int, lo: float, hi: float,
              seed: int = 0) -> torch.Tensor:
    if _cuda(ids):
        return hip.ext().hash_init(ids, dim, lo, hi, seed)
    return reference.hash_init(ids, dim, lo, hi, seed)


def hash_init_rows(out: torch.Tensor, first_id: int, stride: int,
                   lo: float, hi: float, seed: int = 0) -> None:
    """Initialize a whole dense shard slab in place: row r of ``out`` is the
    deterministic init vector of global id ``first_id + r * stride``.
    Device path runs one kernel over the slab; CPU path materializes ids.
    """
    if _cuda(out):
        hip.ext().hash_init_rows(out, first_id, stride, lo, hi, seed)
        return
    ids = torch.arange(out.shape[0], dtype=torch.int64) * stride + first_id
    out.copy_(reference.hash_init(ids, out.shape[1], lo, hi, seed))


def mf_sgd_step(user_table: torch.Tensor, user_rows: torch.Tensor,
                item_vecs: torch.Tensor, item_slots: torch.Tensor,
                ratings: torch.Tensor, lr: float, reg: float = 0.0,
                reuse_tag: int = 0,
                mean_users: bool = False) -> torch.Tensor:
    """``reuse_tag != 0`` promises that whenever the same (tag,
    user_rows buffer, item_slots buffer) recur, their CONTENT is
    unchanged (epoch resweeps over resident batches) — the HIP path then
    reuses the cached sort positions instead of re-sorting, and caches
    both sorted record arrays outright (zero per-visit scatter rebuilds)
    when additionally the SAME ratings buffer recurs with unchanged
    content; a different ratings buffer safely re-scatters only the
    first-pass records."""
    if _cuda(user_table):
        return hip.ext().mf_sgd_step(user_table, user_rows, item_vecs,
                                     item_slots, ratings, lr, reg,
                                     reuse_tag, mean_users)
    # torch index ops need int64; the HIP path takes int32 directly
    return reference.mf_sgd_step(user_table, user_rows.long(), item_vecs,
                                 item_slots.long(), ratings, lr, reg,
                                 mean_users=mean_users)


def pa_step(indptr: torch.Tensor, feat_slots: torch.Tensor,
            x_vals: torch.Tensor, labels: torch.Tensor,
            weights: torch.Tensor, variant: str = "PA-I", C: float = 1.0
            ) -> Tuple[torch.Tensor, torch.Tensor]:
    if _cuda(weights):
        return hip.ext().pa_step(indptr, feat_slots, x_vals, labels,
                                 weights, variant, C)
    return reference.pa_step(indptr, feat_slots, x_vals, labels, weights,
                             variant, C)


def pa_multiclass_step(indptr: torch.Tensor, feat_slots: torch.Tensor,
                       x_vals: torch.Tensor, labels: torch.Tensor,
                       weights: torch.Tensor, variant: str = "PA-I",
                       C: float = 1.0) -> Tuple[torch.Tensor, torch.Tensor]:
    if _cuda(weights):
        return hip.ext().pa_multiclass_step(indptr, feat_slots, x_vals,
                                            labels, weights, variant, C)
    return reference.pa_multiclass_step(indptr, feat_slots, x_vals, labels,
                                        weights, variant, C)


def lr_step(indptr: torch.Tensor, feat_slots: torch.Tensor,
            x_vals: torch.Tensor, labels: torch.Tensor,
            weights: torch.Tensor, reuse_tag: int = 0
            ) -> Tuple[torch.Tensor, torch.Tensor]:
    """``reuse_tag != 0`` promises the same (tag, feat_slots buffer,
    x_vals buffer) recur with unchanged content (epoch resweeps); the
    HIP path then runs the slot-sorted record walk with the records and
    per-instance norms cached per batch — the per-element gather/
    scatter form (random 4-B line traffic) leaves the steady round."""
    if _cuda(weights):
        return hip.ext().lr_step(indptr, feat_slots, x_vals, labels,
                                 weights, reuse_tag)
    return reference.lr_step(indptr, feat_slots.long(), x_vals, labels,
                             weights)


def sgns_step(vecs: torch.Tensor, center_slots: torch.Tensor,
              ctx_slots: torch.Tensor, labels: torch.Tensor, lr: float
              ) -> torch.Tensor:
    """Fused SGNS step over one shared row buffer (center and context
    slots index disjoint rows); returns the combined deltas buffer."""
    if _cuda(vecs):
        return hip.ext().sgns_step(vecs, center_slots, ctx_slots, labels,
                                   lr)
    return reference.sgns_step(vecs, center_slots.long(),
                               ctx_slots.long(), labels, lr)
