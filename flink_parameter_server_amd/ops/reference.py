"""Plain-PyTorch fp32 reference implementations of every device op.

These are the *semantics oracle*: the HIP/CDNA4 kernels in ``csrc/`` are
tested element-wise against these (tests/test_ops_gpu.py), and they are the
execution path on CPU (tests, record-mode parity, gloo multi-process runs).
On a GPU box the HIP extension is mandatory — see ``ops/__init__.py``.

Reference hot spots these correspond to (mount empty, so
cites are file/symbol level):

* ``gather``        <- ``SimplePSLogic.onPullRecv`` answer assembly
* ``apply_add``     <- ``SimplePSLogic.onPushRecv`` with ``paramUpdate = vectorSum``
* ``apply_adagrad`` <- (new capability, BASELINE configs[3])
* ``hash_init``     <- ``matrix/factorization/utils/RangedRandomFactorInitializer``
                       (deterministic per-id init so any shard can init a
                       factor without coordination)
* ``mf_sgd_step``   <- ``matrix/factorization/utils/SGDUpdater.delta`` per rating
* ``pa_step``       <- ``passive/aggressive/algorithm/*`` margin + tau + update
* ``lr_step``       <- (new capability: sparse logistic regression)
* ``sgns_step``     <- (new capability: word2vec skip-gram negative sampling)
"""
from __future__ import annotations

from typing import Tuple

import torch

# ---------------------------------------------------------------------------
# Deterministic per-element hashing (splitmix64) — must bit-match the HIP
# kernel in csrc/ps_kernels.hip.  torch has no uint64, so we emulate logical
# shifts on int64 (arithmetic shift + mask); multiplication wraps mod 2^64
# identically for both signednesses.
# ---------------------------------------------------------------------------

_M1 = -4658895280553007687   # 0xBF58476D1CE4E5B9 as signed int64
_M2 = -7723592293110705685   # 0x94D049BB133111EB
_GOLDEN = -7046029254386353131  # 0x9E3779B97F4A7C15


def _lsr(x: torch.Tensor, k: int) -> torch.Tensor:
    """Logical (unsigned) right shift of int64 tensor by k."""
    mask = (1 << (64 - k)) - 1
    return (x >> k) & mask


def splitmix64(x: torch.Tensor) -> torch.Tensor:
    z = x + _GOLDEN
    z = (z ^ _lsr(z, 30)) * _M1
    z = (z ^ _lsr(z, 27)) * _M2
    return z ^ _lsr(z, 31)


def hash_uniform(keys: torch.Tensor, lo: float, hi: float) -> torch.Tensor:
    """Deterministic uniform[lo, hi) float from int64 keys.

    Uses the top 24 bits of splitmix64 so float32 represents the value
    exactly (bit-match with the HIP kernel).
    """
    h = splitmix64(keys)
    u24 = _lsr(h, 40).to(torch.float32)  # in [0, 2^24)
    return lo + (hi - lo) * (u24 * (1.0 / 16777216.0))


def hash_init(ids: torch.Tensor, dim: int, lo: float, hi: float,
              seed: int = 0) -> torch.Tensor:
    """Deterministic per-id init vector: row i = uniform[lo,hi)^dim keyed by
    (id, column, seed).  Any shard computes the same row for the same id
    (the reference's coordination-free init property)."""
    cols = torch.arange(dim, dtype=torch.int64, device=ids.device)
    keys = ids.reshape(-1, 1) * dim + cols + seed * 1000003
    return hash_uniform(keys, lo, hi)


# ---------------------------------------------------------------------------
# Table ops
# ---------------------------------------------------------------------------

def gather(table: torch.Tensor, rows: torch.Tensor) -> torch.Tensor:
    """values[j] = table[rows[j]] (K3 ps_gather)."""
    return table.index_select(0, rows)


def apply_add(table: torch.Tensor, rows: torch.Tensor,
              deltas: torch.Tensor, unique: bool = False) -> None:
    """table[rows[j]] += deltas[j], duplicate rows accumulate (K1).
    ``unique=True`` asserts rows are distinct (enables the non-atomic
    device fast path; semantics identical)."""
    table.index_add_(0, rows, deltas.to(table.dtype))


def apply_adagrad(table: torch.Tensor, accum: torch.Tensor,
                  rows: torch.Tensor, grads: torch.Tensor,
                  lr: float, eps: float = 1e-8) -> None:
    """Adagrad apply (K2): G[r] += g^2 ; W[r] -= lr * g / sqrt(G[r] + eps).

    Duplicate rows within one batch are pre-aggregated (sum of grads) before
    the state update — this is the contract the HIP kernel implements (one
    read-modify-write per distinct row per round).
    """
    urows, inv = torch.unique(rows, return_inverse=True)
    g = torch.zeros(urows.numel(), grads.shape[1], dtype=grads.dtype,
                    device=grads.device)
    g.index_add_(0, inv, grads)
    G = accum.index_select(0, urows) + g * g
    accum.index_copy_(0, urows, G)
    W = table.index_select(0, urows) - lr * g / torch.sqrt(G + eps)
    table.index_copy_(0, urows, W)


# ---------------------------------------------------------------------------
# Matrix-factorization SGD (K4) — SGDUpdater.delta semantics:
#   e  = r - u.i
#   du = lr * (e * i - reg * u)
#   di = lr * (e * u - reg * i)
# user vectors are updated in place (worker-local table); item deltas are
# returned accumulated per *pulled slot* (the engine pushes them).
# All ratings in one call read the same item snapshot (round semantics).
# ---------------------------------------------------------------------------

def mf_sgd_step(user_table: torch.Tensor, user_rows: torch.Tensor,
                item_vecs: torch.Tensor, item_slots: torch.Tensor,
                ratings: torch.Tensor, lr: float, reg: float = 0.0,
                mean_users: bool = False) -> torch.Tensor:
    """One SGD pass over a batch of ratings.

    user_table : (U, k) worker-local user factors, updated in place
    user_rows  : (n,) int64 row of each rating's user in user_table
    item_vecs  : (m, k) pulled item factors (snapshot)
    item_slots : (n,) int64 index of each rating's item in item_vecs
    ratings    : (n,) f32
    returns    : (m, k) accumulated item deltas aligned with item_vecs
    """
    u = user_table.index_select(0, user_rows)
    v = item_vecs.index_select(0, item_slots)
    e = ratings - (u * v).sum(dim=1)
    di = lr * (e.unsqueeze(1) * u - reg * v)
    item_deltas = torch.zeros_like(item_vecs)
    item_deltas.index_add_(0, item_slots, di)
    if mean_users:
        # per-user averaged update: u = u*(1 - lr*reg*[cnt>0]) + acc/cnt
        acc = torch.zeros_like(user_table)
        acc.index_add_(0, user_rows, lr * e.unsqueeze(1) * v)
        cnt = torch.bincount(user_rows,
                             minlength=user_table.shape[0]).unsqueeze(1)
        decay = torch.where(cnt > 0, 1.0 - lr * reg, 1.0)
        user_table.mul_(decay).add_(acc / cnt.clamp(min=1))
    else:
        du = lr * (e.unsqueeze(1) * v - reg * u)
        user_table.index_add_(0, user_rows, du)
    return item_deltas


# ---------------------------------------------------------------------------
# Passive-aggressive classification (K7).
# Binary PA over sparse instances in CSR form.  Margin m = sum w_f x_f;
# hinge loss l = max(0, 1 - y*m); tau by variant:
#   PA    : tau = l / ||x||^2
#   PA-I  : tau = min(C, l / ||x||^2)
#   PA-II : tau = l / (||x||^2 + 1/(2C))
# update: delta w_f = tau * y * x_f   (only for loss > 0)
# ---------------------------------------------------------------------------

def pa_step(indptr: torch.Tensor, feat_slots: torch.Tensor,
            x_vals: torch.Tensor, labels: torch.Tensor,
            weights: torch.Tensor, variant: str = "PA-I", C: float = 1.0
            ) -> Tuple[torch.Tensor, torch.Tensor]:
    """Binary PA step over a CSR batch.

    indptr     : (B+1,) int64 instance boundaries
    feat_slots : (nnz,) int64 index into `weights` (pulled unique features)
    x_vals     : (nnz,) f32 feature values
    labels     : (B,) f32 in {-1, +1}; NaN = unlabeled (predict only)
    weights    : (m,) f32 pulled feature weights (snapshot)
    returns    : (margins (B,), weight_deltas (m,))
    """
    B = indptr.numel() - 1
    seg = torch.repeat_interleave(
        torch.arange(B, dtype=torch.int64, device=x_vals.device),
        indptr[1:] - indptr[:-1])
    w = weights.index_select(0, feat_slots)
    margins = torch.zeros(B, dtype=torch.float32, device=x_vals.device)
    margins.index_add_(0, seg, w * x_vals)
    xnorm2 = torch.zeros_like(margins)
    xnorm2.index_add_(0, seg, x_vals * x_vals)

    labeled = ~torch.isnan(labels)
    y = torch.where(labeled, labels, torch.zeros_like(labels))
    loss = torch.clamp(1.0 - y * margins, min=0.0)
    safe_norm = torch.clamp(xnorm2, min=1e-12)
    if variant == "PA":
        tau = loss / safe_norm
    elif variant == "PA-I":
        tau = torch.clamp(loss / safe_norm, max=C)
    elif variant == "PA-II":
        tau = loss / (safe_norm + 1.0 / (2.0 * C))
    else:
        raise ValueError(f"unknown PA variant {variant!r}")
    tau = torch.where(labeled & (loss > 0), tau, torch.zeros_like(tau))

    coeff = (tau * y).index_select(0, seg) * x_vals
    deltas = torch.zeros_like(weights)
    deltas.index_add_(0, feat_slots, coeff)
    return margins, deltas


def pa_multiclass_step(indptr: torch.Tensor, feat_slots: torch.Tensor,
                       x_vals: torch.Tensor, labels: torch.Tensor,
                       weights: torch.Tensor, variant: str = "PA-I",
                       C: float = 1.0
                       ) -> Tuple[torch.Tensor, torch.Tensor]:
    """Multiclass PA (max-margin multiclass update).

    weights : (m, n_classes) pulled per-feature class-weight rows
    labels  : (B,) int64 class index; -1 = unlabeled
    returns : (scores (B, n_classes), weight_deltas (m, n_classes))

    Update (single-prototype multiclass PA): with s = scores, r = argmax
    over wrong classes, loss = max(0, 1 - s[y] + s[r]),
    ||x||^2 doubles (the update touches two prototype rows):
    delta w_f[y] += tau x_f ; delta w_f[r] -= tau x_f.
    """
    B = indptr.numel() - 1
    n_classes = weights.shape[1]
    seg = torch.repeat_interleave(
        torch.arange(B, dtype=torch.int64, device=x_vals.device),
        indptr[1:] - indptr[:-1])
    w = weights.index_select(0, feat_slots)          # (nnz, n_classes)
    scores = torch.zeros(B, n_classes, dtype=torch.float32,
                         device=x_vals.device)
    scores.index_add_(0, seg, w * x_vals.unsqueeze(1))
    xnorm2 = torch.zeros(B, dtype=torch.float32, device=x_vals.device)
    xnorm2.index_add_(0, seg, x_vals * x_vals)

    labeled = labels >= 0
    y = torch.clamp(labels, min=0)
    s_y = scores.gather(1, y.unsqueeze(1)).squeeze(1)
    masked = scores.clone()
    masked.scatter_(1, y.unsqueeze(1), float("-inf"))
    s_r, r = masked.max(dim=1)
    loss = torch.clamp(1.0 - s_y + s_r, min=0.0)
    safe_norm = torch.clamp(2.0 * xnorm2, min=1e-12)
    if variant == "PA":
        tau = loss / safe_norm
    elif variant == "PA-I":
        tau = torch.clamp(loss / safe_norm, max=C)
    elif variant == "PA-II":
        tau = loss / (safe_norm + 1.0 / (2.0 * C))
    else:
        raise ValueError(f"unknown PA variant {variant!r}")
    tau = torch.where(labeled & (loss > 0), tau, torch.zeros_like(tau))

    tau_e = tau.index_select(0, seg) * x_vals        # (nnz,)
    deltas = torch.zeros_like(weights)
    row_idx = feat_slots
    deltas.index_put_((row_idx, y.index_select(0, seg)),
                      tau_e, accumulate=True)
    deltas.index_put_((row_idx, r.index_select(0, seg)),
                      -tau_e, accumulate=True)
    return scores, deltas


# ---------------------------------------------------------------------------
# Sparse logistic regression step (K9) — new capability (configs[3]).
# p = sigmoid(m); grad_f = (p - y01) * x_f  (pushed; server applies Adagrad)
# ---------------------------------------------------------------------------

def lr_step(indptr: torch.Tensor, feat_slots: torch.Tensor,
            x_vals: torch.Tensor, labels: torch.Tensor,
            weights: torch.Tensor
            ) -> Tuple[torch.Tensor, torch.Tensor]:
    """Returns (probs (B,), weight_grads (m,)).  labels in {0,1}; NaN =
    predict only."""
    B = indptr.numel() - 1
    seg = torch.repeat_interleave(
        torch.arange(B, dtype=torch.int64, device=x_vals.device),
        indptr[1:] - indptr[:-1])
    w = weights.index_select(0, feat_slots)
    margins = torch.zeros(B, dtype=torch.float32, device=x_vals.device)
    margins.index_add_(0, seg, w * x_vals)
    probs = torch.sigmoid(margins)
    labeled = ~torch.isnan(labels)
    y = torch.where(labeled, labels, torch.zeros_like(labels))
    g = torch.where(labeled, probs - y, torch.zeros_like(probs))
    coeff = g.index_select(0, seg) * x_vals
    grads = torch.zeros_like(weights)
    grads.index_add_(0, feat_slots, coeff)
    return probs, grads


# ---------------------------------------------------------------------------
# Word2vec skip-gram negative sampling step (K8) — new capability
# (configs[2]).  Center embeddings v, context/output embeddings u.
#   for one (center c, [pos, neg_1..neg_K]) sample with labels [1,0,..,0]:
#     s_j   = sigmoid(v_c . u_j)
#     g_j   = lr * (label_j - s_j)
#     dv_c += sum_j g_j * u_j            (accumulated per center slot)
#     du_j  = g_j * v_c                  (accumulated per context slot)
# All dots read the pulled snapshot (round semantics).
# ---------------------------------------------------------------------------

def sgns_step(vecs: torch.Tensor, center_slots: torch.Tensor,
              ctx_slots: torch.Tensor, labels: torch.Tensor, lr: float
              ) -> torch.Tensor:
    """center_slots/ctx_slots: (n,) pair lists (one row per (c, o) pair,
    positives and negatives flattened) indexing DISJOINT rows of the one
    shared buffer ``vecs`` (the model maps context ids to [V, 2V), so
    center and context rows never collide).  Returns the combined deltas
    buffer aligned with ``vecs``."""
    v = vecs.index_select(0, center_slots)
    u = vecs.index_select(0, ctx_slots)
    s = torch.sigmoid((v * u).sum(dim=1))
    g = (lr * (labels - s)).unsqueeze(1)
    deltas = torch.zeros_like(vecs)
    deltas.index_add_(0, center_slots, g * u)
    deltas.index_add_(0, ctx_slots, g * v)
    return deltas
