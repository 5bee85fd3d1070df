"""Algorithm entry points with the reference's call shapes (L4).

Reference entries:

* ``PSOnlineMatrixFactorization.psOnlineMF(ratings, numFactors, rangeMin,
  rangeMax, learningRate, negativeSampleRate, userMemory, pullLimit,
  workerParallelism, psParallelism, iterationWaitTime)``
* ``PSOfflineMatrixFactorization.psOfflineMF(...)`` (epoch sweeps)
* ``PassiveAggressiveParameterServer.transformBinary(model?)(trainingData,
  workerParallelism, psParallelism, paMethod, pullLimit, featureCount,
  rangePartitioning, iterationWaitTime)`` and ``transformMulticlass(...)``

These build the whole job (partitioner, HBM table, worker/server logics)
from the reference's parameter list and run it in-process over the
loopback transport with ``worker_parallelism`` lockstep ranks — the
analog of the reference's Flink local-mode execution.  Production
multi-GPU runs use ``transform.transform_distributed`` (one call per
rank under torch.distributed.run) or the lower-level building blocks
directly (see bench.py, examples/distributed_mf.py).

Parameter-mapping notes:

* ``worker_parallelism`` = number of ranks; each rank hosts a worker AND
  (for ``rank < ps_parallelism``) a server shard.  ``ps_parallelism``
  may be smaller than ``worker_parallelism`` (extra ranks host an empty
  shard), matching the reference where the two parallelisms are
  independent.
* ``user_memory`` (MF worker-local LRU bound) is accepted but inert: a
  worker's user partition lives dense and resident in 288 GB HBM3E —
  documented deviation, strictly more capable.
* ``iteration_wait_time`` (the reference's quiescence-timeout
  termination — an inherited wart) is accepted but
  inert: termination is an explicit all-workers-drained consensus.
"""
from __future__ import annotations

from typing import Any, Iterable, List, Optional, Tuple

import torch

from .config import PSConfig
from .models.mf import MFBatch, OnlineMFWorkerLogic, epoch_iterator
from .models.pa import PAWorkerLogic, SparseBatch
from .parallel.partitioner import make_partitioner
from .table import ShardedTable, TableServerLogic
from .transform import transform_local


# ---------------------------------------------------------------------
# Input normalization
# ---------------------------------------------------------------------

def _ratings_to_tensors(ratings: Any) -> Tuple[torch.Tensor, torch.Tensor,
                                               torch.Tensor]:
    """Accept a (users, items, ratings) tensor triple or an iterable of
    Rating-like (user, item, rating) records."""
    if (isinstance(ratings, (tuple, list)) and len(ratings) == 3
            and torch.is_tensor(ratings[0])):
        u, i, r = ratings
        return u.long(), i.long(), r.float()
    rows = [(int(u), int(i), float(r)) for u, i, r in ratings]
    if not rows:
        z = torch.zeros(0, dtype=torch.int64)
        return z, z.clone(), torch.zeros(0)
    t = torch.tensor(rows, dtype=torch.float64)
    return t[:, 0].long(), t[:, 1].long(), t[:, 2].float()


def _mf_rank_batches(users: torch.Tensor, items: torch.Tensor,
                     vals: torch.Tensor, world: int, batch: int,
                     ) -> List[List[MFBatch]]:
    """Partition a rating stream over workers by user (user vectors are
    worker-local state, so a user's ratings must stay on one worker —
    the reference achieves the same by keying its user map per subtask).
    Local user row = user // world for worker (user % world)."""
    out: List[List[MFBatch]] = []
    for r in range(world):
        mine = users % world == r
        u, i, v = users[mine] // world, items[mine], vals[mine]
        out.append([MFBatch(u[s:s + batch], i[s:s + batch], v[s:s + batch])
                    for s in range(0, int(u.numel()), batch)])
    return out


def _merged(results: List[Tuple[List[Any], List[Any]]]
            ) -> Tuple[List[Any], List[Any]]:
    wout: List[Any] = []
    sout: List[Any] = []
    for w, s in results:
        wout.extend(w)
        sout.extend(s)
    return wout, sout


# ---------------------------------------------------------------------
# Matrix factorization
# ---------------------------------------------------------------------

def _ps_mf(ratings, n_users, n_items, num_factors, range_min, range_max,
           learning_rate, negative_sample_rate, pull_limit,
           worker_parallelism, ps_parallelism, batch_size, reg, seed,
           epochs: Optional[int], output_user_factors: bool):
    world = int(worker_parallelism)
    shards = int(ps_parallelism or world)
    if shards > world:
        raise ValueError("ps_parallelism > worker_parallelism needs one "
                         "rank per shard; raise worker_parallelism")
    users, items, vals = _ratings_to_tensors(ratings)
    per_rank = _mf_rank_batches(users, items, vals, world, batch_size)
    users_local = (n_users + world - 1) // world
    cfg = PSConfig(n_workers=world, n_shards=shards, num_ids=n_items,
                   dim=num_factors, pull_limit=pull_limit)
    part = make_partitioner("mod", n_items, shards)
    workers: List[OnlineMFWorkerLogic] = [None] * world  # type: ignore

    def make_worker(rank: int) -> OnlineMFWorkerLogic:
        wl = OnlineMFWorkerLogic(
            users_local, num_factors, learning_rate=learning_rate,
            reg=reg, range_min=range_min, range_max=range_max,
            negative_sample_rate=negative_sample_rate, n_items=n_items,
            pull_limit=pull_limit, output_user_factors=output_user_factors,
            output_user_factors_on_close=True,
            seed=seed + rank)
        workers[rank] = wl
        return wl

    def make_server(rank: int) -> TableServerLogic:
        table = ShardedTable(part, rank, num_factors,
                             torch.device("cpu"),
                             init=(range_min, range_max), seed=seed)
        return TableServerLogic(table, pushes_unique=True)

    def make_data(rank: int) -> Iterable[MFBatch]:
        if epochs is None:
            return iter(per_rank[rank])
        return epoch_iterator(per_rank[rank], epochs)

    results = transform_local(world, make_data, make_worker, make_server,
                              cfg, part_factory=lambda rank: part)
    # map worker outputs (local user rows, vectors) back to global ids
    merged_w: List[Tuple[torch.Tensor, torch.Tensor]] = []
    for rank, (wouts, _) in enumerate(results):
        for rows, vecs in wouts:
            merged_w.append((rows * world + rank, vecs))
    merged_s = [o for _, souts in results for o in souts]
    return merged_w, merged_s


def ps_online_mf(ratings, n_users: int, n_items: int,
                 num_factors: int = 10, range_min: float = -0.01,
                 range_max: float = 0.01, learning_rate: float = 0.01,
                 negative_sample_rate: int = 0, user_memory: int = 0,
                 pull_limit: Optional[int] = 1600,
                 worker_parallelism: int = 1,
                 ps_parallelism: Optional[int] = None,
                 iteration_wait_time: Optional[float] = None,
                 batch_size: int = 1024, reg: float = 0.0, seed: int = 1,
                 output_user_factors: bool = True):
    """Streaming SGD matrix factorization (reference: ``psOnlineMF``).

    Returns ``(user_factors, item_dumps)``: ``user_factors`` is a list of
    ``(global_user_ids, vectors)`` pairs streamed by the workers;
    ``item_dumps`` the per-shard ``(item_ids, vectors)`` final dumps —
    together the reference's ``Either[(userId, vec), (itemId, vec)]``
    output stream, split by side."""
    del user_memory, iteration_wait_time  # inert — see module docstring
    return _ps_mf(ratings, n_users, n_items, num_factors, range_min,
                  range_max, learning_rate, negative_sample_rate,
                  pull_limit, worker_parallelism, ps_parallelism,
                  batch_size, reg, seed, epochs=None,
                  output_user_factors=output_user_factors)


def ps_offline_mf(ratings, n_users: int, n_items: int,
                  num_factors: int = 10, range_min: float = -0.01,
                  range_max: float = 0.01, learning_rate: float = 0.01,
                  negative_sample_rate: int = 0,
                  pull_limit: Optional[int] = 1600,
                  iterations: int = 5, worker_parallelism: int = 1,
                  ps_parallelism: Optional[int] = None,
                  iteration_wait_time: Optional[float] = None,
                  batch_size: int = 1024, reg: float = 0.0, seed: int = 1):
    """Offline (epoch-sweep) MF (reference: ``psOfflineMF``): workers
    buffer their rating partition and sweep it ``iterations`` times."""
    del iteration_wait_time
    return _ps_mf(ratings, n_users, n_items, num_factors, range_min,
                  range_max, learning_rate, negative_sample_rate,
                  pull_limit, worker_parallelism, ps_parallelism,
                  batch_size, reg, seed, epochs=int(iterations),
                  output_user_factors=False)


# ---------------------------------------------------------------------
# Passive-aggressive classification
# ---------------------------------------------------------------------

def _instances_to_batches(data, feature_count: int, world: int,
                          batch: int, multiclass: bool
                          ) -> List[List[SparseBatch]]:
    """data: iterable of (features, label); features = dict{fid: val} or
    sequence of (fid, val); label None = unlabeled (predict).  Instances
    are round-robined over workers (the reference's rebalance) and carry
    global instance ids for prediction outputs."""
    insts = list(data)
    out: List[List[SparseBatch]] = []
    for r in range(world):
        idxs = list(range(r, len(insts), world))
        bs: List[SparseBatch] = []
        for s in range(0, len(idxs), batch):
            chunk = idxs[s:s + batch]
            indptr = [0]
            fids: List[int] = []
            fvals: List[float] = []
            labels: List[float] = []
            for gi in chunk:
                feats, label = insts[gi]
                pairs = feats.items() if hasattr(feats, "items") else feats
                for f, v in pairs:
                    if not 0 <= f < feature_count:
                        raise ValueError(f"feature id {f} outside "
                                         f"[0, {feature_count})")
                    fids.append(int(f))
                    fvals.append(float(v))
                indptr.append(len(fids))
                if multiclass:
                    labels.append(-1 if label is None else int(label))
                else:
                    labels.append(float("nan") if label is None
                                  else float(label))
            bs.append(SparseBatch(
                torch.tensor(indptr, dtype=torch.int64),
                torch.tensor(fids, dtype=torch.int64),
                torch.tensor(fvals, dtype=torch.float32),
                torch.tensor(labels, dtype=(torch.int64 if multiclass
                                            else torch.float32)),
                ids=torch.tensor(chunk, dtype=torch.int64)))
        out.append(bs)
    return out


def _transform_pa(data, feature_count, model, worker_parallelism,
                  ps_parallelism, pa_method, C, pull_limit,
                  range_partitioning, batch_size, n_classes):
    world = int(worker_parallelism)
    shards = int(ps_parallelism or world)
    if shards > world:
        raise ValueError("ps_parallelism > worker_parallelism needs one "
                         "rank per shard; raise worker_parallelism")
    multiclass = n_classes > 0
    dim = max(1, n_classes)
    kind = "range" if range_partitioning else "mod"
    part = make_partitioner(kind, feature_count, shards)
    per_rank = _instances_to_batches(data, feature_count, world,
                                     batch_size, multiclass)
    cfg = PSConfig(n_workers=world, n_shards=shards, num_ids=feature_count,
                   dim=dim, pull_limit=pull_limit, partitioner=kind)

    def make_worker(rank: int) -> PAWorkerLogic:
        return PAWorkerLogic(variant=pa_method, C=C, n_classes=n_classes,
                             pull_limit=pull_limit,
                             n_features=feature_count)

    def make_server(rank: int) -> TableServerLogic:
        table = ShardedTable(part, rank, dim, torch.device("cpu"))
        if model is not None:
            ids, values = model
            ids = ids.long()
            mine = (part.shard_of(ids) == rank)
            if bool(mine.any()):
                table.load(ids[mine],
                           values[mine].reshape(int(mine.sum()), dim))
        return TableServerLogic(table, pushes_unique=True)

    results = transform_local(world, lambda r: iter(per_rank[r]),
                              make_worker, make_server, cfg,
                              part_factory=lambda rank: part)
    return _merged(results)


def transform_binary(data, feature_count: int,
                     model: Optional[Tuple[torch.Tensor,
                                           torch.Tensor]] = None,
                     worker_parallelism: int = 1,
                     ps_parallelism: Optional[int] = None,
                     pa_method: str = "PA-I", C: float = 1.0,
                     pull_limit: Optional[int] = None,
                     range_partitioning: bool = False,
                     iteration_wait_time: Optional[float] = None,
                     batch_size: int = 256):
    """Online binary PA classification (reference: ``transformBinary``).

    ``data``: iterable of ``(features, label)`` with label in {-1, +1} or
    None (unlabeled -> prediction).  Returns ``(predictions, dumps)``:
    predictions are ``(instance_ids, sign)`` tensor pairs; dumps the
    per-shard ``(feature_ids, weights)`` final model."""
    del iteration_wait_time
    return _transform_pa(data, feature_count, model, worker_parallelism,
                         ps_parallelism, pa_method, C, pull_limit,
                         range_partitioning, batch_size, n_classes=0)


def transform_multiclass(data, feature_count: int, n_classes: int,
                         model: Optional[Tuple[torch.Tensor,
                                               torch.Tensor]] = None,
                         worker_parallelism: int = 1,
                         ps_parallelism: Optional[int] = None,
                         pa_method: str = "PA-I", C: float = 1.0,
                         pull_limit: Optional[int] = None,
                         range_partitioning: bool = False,
                         iteration_wait_time: Optional[float] = None,
                         batch_size: int = 256):
    """Online multiclass PA (reference: ``transformMulticlass``); labels
    are class indices in [0, n_classes) or None."""
    del iteration_wait_time
    if n_classes < 2:
        raise ValueError("n_classes must be >= 2")
    return _transform_pa(data, feature_count, model, worker_parallelism,
                         ps_parallelism, pa_method, C, pull_limit,
                         range_partitioning, batch_size,
                         n_classes=n_classes)


# ---------------------------------------------------------------------
# New-capability entry points (no reference counterpart; same in-process
# execution shape as the parity entries above)
# ---------------------------------------------------------------------

def train_sgns(pairs, vocab_size: int, dim: int = 300,
               learning_rate: float = 0.025, negatives: int = 5,
               worker_parallelism: int = 1,
               ps_parallelism: Optional[int] = None,
               pull_limit: Optional[int] = None,
               batch_size: int = 1024, seed: int = 11):
    """Word2vec SGNS over the PS: positive skip-gram ``pairs`` (a
    (centers, contexts) tensor pair or an iterable of (center, context)
    ids), one 2V-row embedding table sharded like any parameter.
    Returns the per-shard ``(row_ids, vectors)`` dumps — rows [0, V) are
    center embeddings, [V, 2V) context embeddings."""
    from .models.sgns import SGNSBatch, SGNSWorkerLogic

    world = int(worker_parallelism)
    shards = int(ps_parallelism or world)
    if shards > world:
        raise ValueError("ps_parallelism > worker_parallelism needs one "
                         "rank per shard; raise worker_parallelism")
    if (isinstance(pairs, (tuple, list)) and len(pairs) == 2
            and torch.is_tensor(pairs[0])):
        centers, contexts = pairs[0].long(), pairs[1].long()
    else:
        t = torch.tensor([(int(c), int(o)) for c, o in pairs],
                         dtype=torch.int64).reshape(-1, 2)
        centers, contexts = t[:, 0], t[:, 1]
    per_rank: List[List[SGNSBatch]] = []
    for r in range(world):
        c, o = centers[r::world], contexts[r::world]
        per_rank.append([SGNSBatch(c[s:s + batch_size],
                                   o[s:s + batch_size])
                         for s in range(0, int(c.numel()), batch_size)])
    part = make_partitioner("mod", 2 * vocab_size, shards)
    cfg = PSConfig(n_workers=world, n_shards=shards,
                   num_ids=2 * vocab_size, dim=dim,
                   pull_limit=pull_limit)

    def make_server(rank: int) -> TableServerLogic:
        table = ShardedTable(part, rank, dim, torch.device("cpu"),
                             init=(-0.5 / dim, 0.5 / dim), seed=seed)
        return TableServerLogic(table, pushes_unique=True)

    results = transform_local(
        world, lambda r: iter(per_rank[r]),
        lambda r: SGNSWorkerLogic(vocab_size, dim, learning_rate,
                                  negatives=negatives,
                                  pull_limit=pull_limit, seed=seed + r),
        make_server, cfg, part_factory=lambda rank: part)
    return [o for _, souts in results for o in souts]


def train_lr(data, feature_count: int, learning_rate: float = 0.05,
             worker_parallelism: int = 1,
             ps_parallelism: Optional[int] = None,
             pull_limit: Optional[int] = None,
             batch_size: int = 256, eps: float = 1e-8):
    """Sparse logistic regression with server-side Adagrad (configs[3]).
    ``data``: iterable of ``(features, label)`` with label in {0, 1} or
    None (unlabeled -> probability prediction).  Returns
    ``(predictions, weight_dumps)``."""
    from .models.lr import LRWorkerLogic

    world = int(worker_parallelism)
    shards = int(ps_parallelism or world)
    if shards > world:
        raise ValueError("ps_parallelism > worker_parallelism needs one "
                         "rank per shard; raise worker_parallelism")
    per_rank = _instances_to_batches(data, feature_count, world,
                                     batch_size, multiclass=False)
    part = make_partitioner("mod", feature_count, shards)
    cfg = PSConfig(n_workers=world, n_shards=shards,
                   num_ids=feature_count, dim=1, pull_limit=pull_limit)

    def make_server(rank: int) -> TableServerLogic:
        table = ShardedTable(part, rank, 1, torch.device("cpu"),
                             optimizer="adagrad", lr=learning_rate,
                             eps=eps)
        return TableServerLogic(table, pushes_unique=True)

    results = transform_local(
        world, lambda r: iter(per_rank[r]),
        lambda r: LRWorkerLogic(pull_limit=pull_limit,
                                emit_predictions=True,
                                n_features=feature_count),
        make_server, cfg, part_factory=lambda rank: part)
    return _merged(results)


def run_count_min(id_stream, depth: int, width: int,
                  worker_parallelism: int = 1,
                  ps_parallelism: Optional[int] = None,
                  batch_size: int = 4096, seed: int = 17):
    """Count-min sketch on the PS (reference ``sketch/**`` analog —
    SURVEY.md §2.2, built defensively; see models/sketch.py).

    ``id_stream``: iterable of int ids (or id tensors/lists).  Returns
    ``(CountMinWorkerLogic, merged_table)`` — query with
    ``logic.estimate(merged_table, ids)`` (one-sided: never an
    under-estimate)."""
    from .models.sketch import CountMinWorkerLogic, merge_shard_dumps

    world = int(worker_parallelism)
    shards = int(ps_parallelism or world)
    if shards > world:
        raise ValueError("ps_parallelism > worker_parallelism needs one "
                         "rank per shard; raise worker_parallelism")
    ids = torch.as_tensor(list(id_stream), dtype=torch.int64).reshape(-1)
    batches = list(ids.split(batch_size))
    per_rank = [batches[r::world] for r in range(world)]
    probe = CountMinWorkerLogic(depth, width, seed=seed)
    part = make_partitioner("mod", probe.num_ids, shards)
    cfg = PSConfig(n_workers=world, n_shards=shards,
                   num_ids=probe.num_ids, dim=1)
    logics: List[Any] = [None] * world

    def make_worker(rank: int):
        logics[rank] = CountMinWorkerLogic(depth, width, seed=seed)
        return logics[rank]

    def make_server(rank: int) -> TableServerLogic:
        table = ShardedTable(part, rank, 1, torch.device("cpu"))
        return TableServerLogic(table, dump_on_close=True,
                                pushes_unique=True)

    results = transform_local(
        world, lambda r: iter(per_rank[r]), make_worker, make_server,
        cfg, part_factory=lambda rank: part)
    dumps = []
    for _, souts in results:
        dumps.extend(souts)
    return logics[0], merge_shard_dumps(dumps, depth, width)
