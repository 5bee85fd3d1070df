"""Job-level entry points — the L1 API of the reference.

Reference: ``FlinkParameterServer.transform(trainingData, workerLogic,
psLogic, ...)`` and its overloads (``transformWithModelLoad``,
``transformWithDoubleArray``): wraps the logics into
worker/server operators, wires the partitioned edges and the iteration
feedback edge, and returns the job's merged output stream.

Here the same call shapes build a :class:`PSEngine` per rank and run the
round loop:

* ``transform(...)`` — single-process (1 rank, the reference's local
  mode), or pass ``transport=`` for a distributed run (one call per rank,
  e.g. under torchrun).
* ``transform_local(...)`` — the reference's MiniCluster analog: N ranks
  as lockstep threads in one process (worker_parallelism == ps_parallelism
  == N; the reference lets them differ, which round-lockstep collectives
  do not — each rank hosts one worker AND one shard).
* ``transform_with_model_load(...)`` — warm start: load ``(id, value)``
  pairs into the shards BEFORE training (a strict two-phase version of
  the reference's eventual warm start — strictly stronger, SURVEY.md
  §3.5).

Returns ``(worker_outputs, server_outputs)`` — the two halves of the
reference's ``DataStream[Either[WOut, PSOut]]``.
"""
from __future__ import annotations

from typing import Any, Callable, Iterable, List, Optional, Tuple

import torch

from .api import ServerLogic, SimpleServerLogic, WorkerLogic
from .config import PSConfig
from .parallel.partitioner import Partitioner, make_partitioner
from .parallel.runtime import PSEngine
from .parallel.transport import Transport, run_loopback
from .table import ShardedTable, TableServerLogic


def transform(data: Iterable[Any], worker_logic: WorkerLogic,
              server_logic: ServerLogic, config: Optional[PSConfig] = None,
              transport: Optional[Transport] = None,
              part: Optional[Partitioner] = None,
              device: Optional[torch.device] = None,
              max_rounds: Optional[int] = None,
              flush: Optional[Any] = None,
              checkpointer: Optional[Any] = None
              ) -> Tuple[List[Any], List[Any]]:
    """The base ``FlinkParameterServer.transform`` shape: run
    ``worker_logic`` against ``server_logic`` over ``data``; returns
    (worker outputs, server outputs).  ``flush``/``checkpointer`` as in
    :class:`PSEngine`."""
    cfg = config or PSConfig()
    eng = PSEngine(cfg, worker_logic, server_logic, transport, part,
                   device=device, flush=flush, checkpointer=checkpointer)
    return eng.run(data, max_rounds=max_rounds)


def transform_simple(data: Iterable[Any], worker_logic: WorkerLogic,
                     param_init: Callable[[int], torch.Tensor],
                     param_update: Callable[..., torch.Tensor],
                     config: Optional[PSConfig] = None,
                     **kw) -> Tuple[List[Any], List[Any]]:
    """The ``transform(paramInit, paramUpdate)`` overload: server side is
    a ``SimplePSLogic(paramInit, paramUpdate)``."""
    return transform(data, worker_logic,
                     SimpleServerLogic(param_init, param_update),
                     config, **kw)


def transform_with_model_load(model: Tuple[torch.Tensor, torch.Tensor],
                              data: Iterable[Any],
                              worker_logic: WorkerLogic,
                              table: ShardedTable,
                              config: Optional[PSConfig] = None,
                              transport: Optional[Transport] = None,
                              dump_on_close: bool = True,
                              **kw) -> Tuple[List[Any], List[Any]]:
    """``transformWithModelLoad``: feed ``model = (ids, values)`` into the
    shard table (this rank keeps the ids it owns), barrier, then train."""
    ids, values = model
    mine = table.part.shard_of(ids) == table.shard
    if bool(mine.any()):
        table.load(ids[mine], values[mine])
    if transport is not None:
        transport.barrier()  # strict load -> train phase boundary
    return transform(data, worker_logic,
                     TableServerLogic(table, dump_on_close=dump_on_close),
                     config, transport=transport, part=table.part, **kw)


def transform_local(world: int, make_data: Callable[[int], Iterable[Any]],
                    make_worker: Callable[[int], WorkerLogic],
                    make_server: Callable[[int], ServerLogic],
                    config: PSConfig,
                    part_factory: Optional[Callable[[int], Partitioner]] = None
                    ) -> List[Tuple[List[Any], List[Any]]]:
    """Reference MiniCluster analog: N lockstep ranks in one process.
    Factories are called with the rank.  Returns per-rank
    (worker_outputs, server_outputs)."""

    def job(rank: int, transport: Transport):
        part = part_factory(rank) if part_factory else make_partitioner(
            config.partitioner, config.num_ids, world)
        eng = PSEngine(config, make_worker(rank), make_server(rank),
                       transport, part)
        return eng.run(make_data(rank))

    return run_loopback(world, job)


def transform_distributed(data: Iterable[Any], worker_logic: WorkerLogic,
                          server_logic: ServerLogic,
                          config: Optional[PSConfig] = None,
                          part: Optional[Partitioner] = None,
                          device: Optional[torch.device] = None,
                          max_rounds: Optional[int] = None,
                          flush: Optional[Any] = None,
                          checkpointer: Optional[Any] = None
                          ) -> Tuple[List[Any], List[Any]]:
    """One-process-per-GPU ``transform``: call once PER RANK under
    ``torch.distributed.run`` (or any launcher that initialized
    ``torch.distributed``); wires :class:`TorchDistTransport` (nccl =
    RCCL over xGMI on GPUs, gloo on CPU) so the same logics that ran
    under ``transform``/``transform_local`` scale across ranks.

    ``data`` is THIS rank's partition of the input (the reference's
    rebalance from the source is the launcher's/data loader's job);
    returns this rank's (worker_outputs, server_outputs) — shard dumps
    stay per-rank like the reference's per-subtask PSOut streams.
    """
    import torch.distributed as dist

    from .parallel.transport import TorchDistTransport

    if not dist.is_initialized():
        raise RuntimeError(
            "transform_distributed needs torch.distributed initialized "
            "(launch with torch.distributed.run and call "
            "dist.init_process_group first)")
    transport = TorchDistTransport(device=device)
    cfg = config or PSConfig()
    if cfg.feedback == "fresh" and transport.world > 1:
        # multi-rank default: the overlapped schedule (see PSConfig)
        cfg.feedback = "pipelined"
    eng = PSEngine(cfg, worker_logic, server_logic, transport, part,
                   device=device if device is not None
                   else transport.device,
                   flush=flush, checkpointer=checkpointer)
    return eng.run(data, max_rounds=max_rounds)
