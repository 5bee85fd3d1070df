"""Core parameter-server API.

This is the MI355X-native re-design of the pull/push contract of
``FlinkML/flink-parameter-server`` (reference: ``hu/sztaki/ilab/ps/``,
``WorkerLogic.scala`` / ``ParameterServerLogic.scala`` /
``ParameterServerClient.scala`` / ``ParameterServer.scala``; the reference
mount was empty this round, so citations are at file/symbol granularity —
see SURVEY.md §0/§2.1).

Differences from the reference, by design:

* **Batched callbacks.** The reference delivers one record / one pull answer
  per callback (per-record JVM code); the honest GPU idiom is a *tensor* of
  ids/values per callback, so ``on_recv`` receives a batch of training
  records and ``on_pull_recv`` receives ``(ids, values)`` tensors for every
  parameter answered this round.  A record-at-a-time shim for API-exact
  parity tests lives in :class:`RecordModeAdapter`.
* **int64 parameter ids.** The reference uses Scala ``Int`` (an artifact of
  ``hashCode``); we widen to int64 for 1e9-feature configs.
* **Loose (PullT != PushT) by default.** The reference's
  ``LooseWorkerLogic``/``LooseParameterServerLogic`` split the pulled type
  from the pushed type; here pull values and push deltas are independent
  tensors throughout, so the "loose" shape is the only shape.
"""
from __future__ import annotations

import collections
from abc import ABC, abstractmethod
from typing import Any, Callable, Optional, TypeVar

import torch

T = TypeVar("T")       # training-data batch type
WOut = TypeVar("WOut")  # worker output type
SOut = TypeVar("SOut")  # server output type


class ParameterServerClient(ABC):
    """A worker's handle to the parameter server + job output.

    Reference: ``ParameterServerClient`` trait —
    ``pull(id)`` / ``push(id, deltaUpdate)`` / ``output(out)``.
    Batched: ``ids`` is an int64 tensor, ``deltas`` a ``(len(ids), dim)``
    tensor.  Both calls are asynchronous: ``pull`` returns immediately and
    the answers arrive via ``WorkerLogic.on_pull_recv`` in a later round.
    """

    @abstractmethod
    def pull(self, ids: torch.Tensor, unique: bool = False) -> None:
        """Request current values for ``ids`` (async, answered later).
        ``unique=True`` asserts the caller already de-duplicated ``ids``
        (sorted or not), letting the engine skip its per-round coalescing
        pass when this is the round's only pull.  The tensor must not be
        mutated after the call (async send; re-sending one cached id
        buffer every round is encouraged — the engine memoizes its
        shard bucketing by buffer identity)."""

    @abstractmethod
    def push(self, ids: torch.Tensor, deltas: torch.Tensor) -> None:
        """Send delta updates for ``ids`` (async, no reply; the tensors
        must not be mutated after the call)."""

    @abstractmethod
    def output(self, out: Any) -> None:
        """Emit a value to the job's worker-output stream."""


class WorkerLogic(ABC):
    """User-implemented training logic, one instance per worker (= rank).

    Reference: ``WorkerLogic`` trait — ``onRecv`` / ``onPullRecv`` /
    ``close``.
    """

    #: Max number of distinct parameter ids with an un-answered pull in
    #: flight.  ``None`` = unlimited.  Reference:
    #: ``WorkerLogic.addPullLimiter`` — there it exists to stop the Flink
    #: iteration from deadlocking under backpressure; here it bounds the
    #: pull/answer ring buffers (memory backpressure).  Size it for the
    #: schedule's pipeline depth: a pull split into ceil(ids/limit)
    #: chunks progresses at ~1 chunk per answer round-trip (2-3 rounds
    #: under merged/pipelined), so a tiny limit throttles throughput
    #: long before memory matters (the reference's default was 1600).
    pull_limit: Optional[int] = None

    @abstractmethod
    def on_recv(self, data: Any, client: ParameterServerClient) -> None:
        """Called with the next batch of training data."""

    @abstractmethod
    def on_pull_recv(self, ids: torch.Tensor, values: torch.Tensor,
                     client: ParameterServerClient) -> None:
        """Called with answered pulls: ``values[i]`` is the parameter row
        for ``ids[i]``.  Ids within one call are unique (the engine
        coalesces duplicate pulls per round, mirroring the reference's
        per-id pending-pull queueing in ``PSOnlineMatrixFactorization``)."""

    def close(self, client: ParameterServerClient) -> None:  # noqa: B027
        """Called once when the job drains; may emit final outputs."""


def add_pull_limiter(logic: WorkerLogic, pull_limit: int) -> WorkerLogic:
    """Parity helper for ``WorkerLogic.addPullLimiter(logic, pullLimit)``.

    The limiting itself is engine-enforced (excess pulls queue locally and
    are released as answers drain); this sets the knob on the logic.
    """
    if pull_limit <= 0:
        raise ValueError("pull_limit must be positive")
    logic.pull_limit = pull_limit
    return logic


class ParameterServer(ABC):
    """A server shard's handle for answering pulls and emitting output.

    Reference: ``ParameterServer`` trait — ``answerPull(id, value,
    workerPartitionIndex)`` / ``output(out)``.
    """

    @abstractmethod
    def answer_pull(self, ids: torch.Tensor, values: torch.Tensor,
                    worker: int) -> None:
        """Send ``values`` for ``ids`` back to worker ``worker``."""

    @abstractmethod
    def output(self, out: Any) -> None:
        """Emit a value to the job's server-output stream."""


class ServerLogic(ABC):
    """User- (or library-) implemented server-shard logic.

    Reference: ``ParameterServerLogic`` trait — ``onPullRecv`` /
    ``onPushRecv`` / ``close``.  One instance per shard (= rank); the
    engine guarantees single-owner serial execution per shard, which is
    the reference's core correctness invariant (each Flink operator
    instance processes its message stream serially — SURVEY.md §5.2).
    """

    @abstractmethod
    def on_pull_recv(self, ids: torch.Tensor, worker: int,
                     ps: ParameterServer) -> None:
        """Answer a batch of pulls from ``worker`` via ``ps.answer_pull``."""

    @abstractmethod
    def on_push_recv(self, ids: torch.Tensor, deltas: torch.Tensor,
                     ps: ParameterServer) -> None:
        """Apply a batch of delta updates."""

    def close(self, ps: ParameterServer) -> None:  # noqa: B027
        """Called once when the job drains; typically dumps the shard."""


class SimpleServerLogic(ServerLogic):
    """Default server logic over a dict — the record-mode parity twin of
    the HBM-resident :class:`~flink_parameter_server_amd.table.ShardedTable`.

    Reference: ``server/SimplePSLogic.scala`` —
    ``SimplePSLogic[P](paramInit: Int => P, paramUpdate: (P, P) => P)``
    backed by a ``mutable.HashMap[Int, P]``; ``onPullRecv`` answers
    ``params.getOrElseUpdate(id, paramInit(id))``, ``onPushRecv`` does
    ``params(id) = paramUpdate(params(id), delta)``, ``close`` outputs every
    ``(id, param)`` pair.

    This implementation keeps those exact semantics on CPU tensors and is
    used by the semantics tests; production paths use
    :class:`~flink_parameter_server_amd.table.TableServerLogic`.
    """

    #: answers every pull batch in full, in received order, same round
    #: (lets the engine elide answer ids from the feedback edge).
    answers_in_order = True

    def __init__(self, param_init: Callable[[int], torch.Tensor],
                 param_update: Callable[[torch.Tensor, torch.Tensor], torch.Tensor]):
        self.param_init = param_init
        self.param_update = param_update
        self.params: "collections.OrderedDict[int, torch.Tensor]" = collections.OrderedDict()

    def on_pull_recv(self, ids: torch.Tensor, worker: int,
                     ps: ParameterServer) -> None:
        vals = []
        for i in ids.tolist():
            if i not in self.params:
                self.params[i] = self.param_init(i)
            vals.append(self.params[i])
        if vals:
            ps.answer_pull(ids, torch.stack(vals), worker)

    def on_push_recv(self, ids: torch.Tensor, deltas: torch.Tensor,
                     ps: ParameterServer) -> None:
        for n, i in enumerate(ids.tolist()):
            if i not in self.params:
                self.params[i] = self.param_init(i)
            self.params[i] = self.param_update(self.params[i], deltas[n])

    def close(self, ps: ParameterServer) -> None:
        for i, v in self.params.items():
            ps.output((i, v))


class RecordModeAdapter(WorkerLogic):
    """Wraps a record-at-a-time logic (the reference's exact call shapes:
    one data record / one ``(paramId, paramValue)`` per callback) into the
    batched :class:`WorkerLogic` interface, for API-parity tests on CPU.

    The wrapped object must provide ``on_recv(record, client)`` and
    ``on_pull_recv(param_id: int, value: Tensor, client)``.
    """

    def __init__(self, record_logic: Any):
        self.record_logic = record_logic
        self.pull_limit = getattr(record_logic, "pull_limit", None)

    def on_recv(self, data: Any, client: ParameterServerClient) -> None:
        for record in data:
            self.record_logic.on_recv(record, client)

    def on_pull_recv(self, ids: torch.Tensor, values: torch.Tensor,
                     client: ParameterServerClient) -> None:
        for n, i in enumerate(ids.tolist()):
            self.record_logic.on_pull_recv(i, values[n], client)

    def close(self, client: ParameterServerClient) -> None:
        close = getattr(self.record_logic, "close", None)
        if close is not None:
            close(client)
