"""Messaging layer: wire entities + send-batching strategies.

Reference: message case classes under ``entities/``
(``Pull(paramId)``, ``Push(paramId, delta)``, ``PullAnswer(paramId,
param)``, ``WorkerToPS(workerPartitionIndex, Either[Push, Pull])``,
``PSToWorker(workerPartitionIndex, PullAnswer)``), the pluggable
``WorkerSender``/``WorkerReceiver``/``PSSender``/``PSReceiver`` traits
under ``client/``/``server/``, and the generic "collect messages, flush
when a condition fires" combination logic under ``common/``
(``CombinationLogic`` with count/time flush conditions) that the batching
senders use to coalesce many logical pulls/pushes into one Flink record.

MI355X mapping: one "Flink record" is one alltoallv round segment, so
batching is structural — the engine always ships whole
tensors.  What remains meaningful from the reference's sender strategies
is *when* a worker's accumulated ops are released into a round:

* :class:`FlushAlways` — every round (the simple senders).
* :class:`CountFlush` — hold ops until at least ``n`` are pending (the
  count-triggered combination sender): small `on_recv` bursts coalesce
  across rounds into fewer, larger segments.
* :class:`TimeFlush` — hold ops at most ``seconds`` (the time-triggered
  sender); with round cadence replacing record cadence this bounds the
  extra latency a held op can suffer.

The wire entities are used by :class:`SerializingTransport`, which
round-trips every segment through encode/decode — the semantic twin of
the reference's receiver/sender codec pair, used by the tests to pin the
wire contract (and by any future non-collective transport).
"""
from __future__ import annotations

import time
from dataclasses import dataclass
from typing import Optional, Tuple

import torch

from .parallel.transport import Transport

# ---------------------------------------------------------------------
# Wire entities (reference: entities/*.scala).  Batched: one entity
# carries a tensor of ids, not a single id.
# ---------------------------------------------------------------------


@dataclass
class Pull:
    """Reference: ``case class Pull(paramId: Int)`` — batched to a tensor."""
    param_ids: torch.Tensor


@dataclass
class Push:
    """Reference: ``case class Push[P](paramId: Int, delta: P)``."""
    param_ids: torch.Tensor
    deltas: torch.Tensor


@dataclass
class PullAnswer:
    """Reference: ``case class PullAnswer[P](paramId: Int, param: P)``."""
    param_ids: torch.Tensor
    values: torch.Tensor


@dataclass
class WorkerToPS:
    """Reference: ``case class WorkerToPS[P](workerPartitionIndex: Int,
    msg: Either[Push[P], Pull])`` — batched: one envelope per round
    carrying this worker's pulls AND pushes for one destination shard."""
    worker: int
    pull: Optional[Pull]
    push: Optional[Push]


@dataclass
class PSToWorker:
    """Reference: ``case class PSToWorker[P](workerPartitionIndex: Int,
    msg: PullAnswer[P])``."""
    worker: int
    answer: PullAnswer


# ---------------------------------------------------------------------
# Codec (reference: SimpleWorkerSender/SimplePSReceiver and the reverse
# pair).  The engine's native wire format is the 5 alltoallv channels
# (pull ids / push ids / push deltas / answer ids / answer values); the
# codec maps them to/from entity envelopes.
# ---------------------------------------------------------------------


def encode_worker_to_ps(worker: int, pull_ids: Optional[torch.Tensor],
                        push_ids: Optional[torch.Tensor],
                        push_deltas: Optional[torch.Tensor]) -> WorkerToPS:
    return WorkerToPS(
        worker,
        Pull(pull_ids) if pull_ids is not None and pull_ids.numel() else None,
        Push(push_ids, push_deltas)
        if push_ids is not None and push_ids.numel() else None)


def decode_worker_to_ps(msg: WorkerToPS) -> Tuple[
        Optional[torch.Tensor], Optional[torch.Tensor],
        Optional[torch.Tensor]]:
    pull = msg.pull.param_ids if msg.pull is not None else None
    if msg.push is not None:
        return pull, msg.push.param_ids, msg.push.deltas
    return pull, None, None


def encode_ps_to_worker(worker: int, ids: Optional[torch.Tensor],
                        values: Optional[torch.Tensor]
                        ) -> Optional[PSToWorker]:
    if ids is None or not ids.numel():
        return None
    return PSToWorker(worker, PullAnswer(ids, values))


def decode_ps_to_worker(msg: Optional[PSToWorker]) -> Tuple[
        Optional[torch.Tensor], Optional[torch.Tensor]]:
    if msg is None:
        return None, None
    return msg.answer.param_ids, msg.answer.values


class SerializingTransport(Transport):
    """Wraps a transport so every worker->server segment round-trips
    through :class:`WorkerToPS` and every answer through
    :class:`PSToWorker` — pins the entity-layer contract in tests.

    The engine's channel layout (enforced here):
      3 channels = worker->server (pull ids, push ids, push deltas)
      2 channels = server->worker (answer ids, answer values)
    """

    def __init__(self, inner: Transport):
        self.inner = inner
        self.rank = inner.rank
        self.world = inner.world

    def _codec(self, parts):
        C = len(parts)
        wire = [list(chan) for chan in parts]
        if C >= 3:  # worker -> server channels (always the first 3)
            for d in range(self.world):
                env = encode_worker_to_ps(self.rank, parts[0][d],
                                          parts[1][d], parts[2][d])
                wire[0][d], wire[1][d], wire[2][d] = \
                    decode_worker_to_ps(env)
        if C == 2 or C == 5:  # id-carrying feedback channels (last 2)
            for d in range(self.world):
                env = encode_ps_to_worker(d, parts[C - 2][d],
                                          parts[C - 1][d])
                wire[C - 2][d], wire[C - 1][d] = decode_ps_to_worker(env)
        # C == 1 / C == 4: in-order feedback — answer ids intentionally
        # elided from the wire (PullAnswer.param_ids reconstructed by
        # the puller from its own sent segments); values pass as-is.
        return wire

    def alltoallv(self, parts, specs, extra=None, recv_counts=None):
        return self.inner.alltoallv(self._codec(parts), specs,
                                    extra=extra, recv_counts=recv_counts)

    # split-phase form (the pipelined schedule): the codec applies at
    # issue time, completion passes straight through
    def alltoallv_begin(self, parts, specs, extra=None, recv_counts=None):
        return self.inner.alltoallv_begin(self._codec(parts), specs,
                                          extra=extra,
                                          recv_counts=recv_counts)

    def alltoallv_end(self, handle):
        return self.inner.alltoallv_end(handle)

    def ctrl_of(self, handle):
        return self.inner.ctrl_of(handle)

    def allgather_int(self, vec):
        return self.inner.allgather_int(vec)

    def barrier(self):
        self.inner.barrier()


# ---------------------------------------------------------------------
# Flush strategies (reference: common/CombinationLogic + the count/time
# condition traits driving CountClientSender/TimerClientSender).
# ---------------------------------------------------------------------


class FlushStrategy:
    """Decides each round whether the worker's accumulated pulls/pushes
    are released into the round's alltoallv or held locally."""

    def should_flush(self, pending_ops: int, force: bool) -> bool:
        raise NotImplementedError


class FlushAlways(FlushStrategy):
    """The reference's simple (non-combining) senders: ship every round."""

    def should_flush(self, pending_ops: int, force: bool) -> bool:
        return True


class CountFlush(FlushStrategy):
    """Count-triggered combination sender: hold until >= n ops pending
    (or the engine forces a flush at drain/termination)."""

    def __init__(self, n: int):
        if n < 1:
            raise ValueError("count threshold must be >= 1")
        self.n = int(n)

    def should_flush(self, pending_ops: int, force: bool) -> bool:
        return force or pending_ops >= self.n


class TimeFlush(FlushStrategy):
    """Time-triggered combination sender: hold at most ``seconds`` since
    the first held op."""

    def __init__(self, seconds: float):
        self.seconds = float(seconds)
        self._since: Optional[float] = None

    def should_flush(self, pending_ops: int, force: bool) -> bool:
        if force or pending_ops == 0:
            self._since = None
            return True
        now = time.monotonic()
        if self._since is None:
            self._since = now
        if now - self._since >= self.seconds:
            self._since = None
            return True
        return False
