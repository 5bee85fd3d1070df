"""The round-based async engine — the MI355X-native replacement for the
reference's Flink job graph.

Reference topology (``FlinkParameterServer.transform``):
worker operator --(shuffle by hash(paramId))--> server operator
--(iteration feedback edge by workerPartitionIndex)--> worker operator,
free-running record flow, terminated by the ``iterationWaitTime``
quiescence timeout.

MI355X design: one process per GPU; each rank hosts one
worker role AND one server-shard role.  The free-running record flow
becomes *rounds* with three feedback-edge schedules (``PSConfig.feedback``):

``"fresh"`` (default) — per round: deliver last round's answers ->
consume a batch -> bucket this round's pulls/pushes -> exchange them
(grouped send/recv alltoallv; the SSP clock + drain control rides the
counts all_gather) -> server phase (apply pushes, gather answers) ->
exchange the answers back (the reference's feedback edge; in-order
servers elide the answer ids AND their counts — the puller reconstructs
both from its own sent segments).  One all_gather + two grouped p2p
phases per round; a pull is answered with the freshest possible values,
one round after it was requested.

``"merged"`` — the answers ride the NEXT round's single exchange, so a
round is exactly one counts+control all_gather + one grouped p2p phase.
+1 round of answer latency / staleness; the throughput option for large
multi-GPU jobs where per-round collective latency is the scaling tax.

``"pipelined"`` — the merged layout, split-phase: each round ISSUES its
exchange, trains on the answers held from the previous round's exchange
while the payloads cross xGMI (RCCL p2p progresses on its own HIP
stream, the fused kernels on the compute stream), then completes the
exchange and runs the server phase.  +1 further round of latency;
comm/compute overlap — the multi-GPU default.

Asynchrony (both modes): the worker never blocks on the server — the
reference's async pull/push contract; staleness comes from round/pipeline
depth (the SSP knob bounds it on top).  ``pull_limit`` bounds distinct
outstanding pull ids exactly like ``WorkerLogic.addPullLimiter`` (excess
pulls queue locally).  Termination is an explicit all-drained consensus
carried by the control lane (replacing the reference's
``iterationWaitTime`` quiescence timeout).
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any, Dict, Iterable, Iterator, List, Optional, Tuple

import torch

from ..api import ParameterServer, ParameterServerClient, ServerLogic, WorkerLogic
from ..config import PSConfig
from .partitioner import Partitioner, bucket_by_shard, make_partitioner
from .transport import Transport, LocalTransport


@dataclass
class Metrics:
    rounds: int = 0
    samples: int = 0
    pull_ids_sent: int = 0
    push_ids_sent: int = 0
    ssp_stalls: int = 0
    bytes_sent: int = 0
    held_rounds: int = 0
    # pipelined-mode host phase timers (seconds): time issuing the
    # exchange, computing under it, waiting for its completion, and in
    # the server phase.  t_comm_wait ~ 0 means the overlapped compute
    # fully hid the transfer (on nccl the wait is a stream dependency,
    # so these measure host-side enqueue/wait, not kernel execution).
    t_issue: float = 0.0
    t_overlap_compute: float = 0.0
    t_comm_wait: float = 0.0
    t_server: float = 0.0
    # staleness distribution (clock - min cross-worker clock, per round) —
    # the SSP staleness histogram BASELINE configs[4] reports.
    staleness_hist: Dict[int, int] = field(default_factory=dict)

    def as_dict(self) -> Dict[str, Any]:
        return dict(self.__dict__)


class _Client(ParameterServerClient):
    def __init__(self, engine: "PSEngine"):
        self._e = engine

    def pull(self, ids: torch.Tensor, unique: bool = False) -> None:
        if ids.numel():
            self._e._pull_buf.append((ids.reshape(-1), unique))

    def push(self, ids: torch.Tensor, deltas: torch.Tensor) -> None:
        if ids.numel():
            self._e._push_buf.append((ids.reshape(-1), deltas))

    def output(self, out: Any) -> None:
        self._e.worker_outputs.append(out)


class _PS(ParameterServer):
    def __init__(self, engine: "PSEngine"):
        self._e = engine

    def answer_pull(self, ids: torch.Tensor, values: torch.Tensor,
                    worker: int) -> None:
        self._e._answer_buf[worker].append((ids, values))

    def output(self, out: Any) -> None:
        self._e.server_outputs.append(out)


def _concat(parts: List[torch.Tensor]) -> Optional[torch.Tensor]:
    if not parts:
        return None
    if len(parts) == 1:
        return parts[0]
    return torch.cat(parts)


class PSEngine:
    """Drives one rank's worker + server roles through rounds."""

    def __init__(self, config: PSConfig, worker_logic: WorkerLogic,
                 server_logic: ServerLogic, transport: Optional[Transport],
                 part: Optional[Partitioner] = None,
                 device: Optional[torch.device] = None,
                 flush: Optional[Any] = None,
                 checkpointer: Optional[Any] = None):
        """``flush``: a messaging.FlushStrategy deciding when accumulated
        pulls/pushes are released into a round (None = every round, the
        reference's simple senders).  ``checkpointer``: an object with
        ``maybe(round_idx)`` called after each round's server phase (e.g.
        utils.checkpoint.PeriodicCheckpointer)."""
        self.cfg = config
        self.flush = flush
        self.checkpointer = checkpointer
        self.worker_logic = worker_logic
        self.server_logic = server_logic
        self.transport = transport or LocalTransport()
        self.rank = self.transport.rank
        self.world = self.transport.world
        self.part = part or make_partitioner(config.partitioner,
                                             config.num_ids, self.world)
        self.device = torch.device(device) if device is not None else \
            torch.device("cpu")
        self.client = _Client(self)
        self.ps = _PS(self)

        self.worker_outputs: List[Any] = []
        self.server_outputs: List[Any] = []
        self.metrics = Metrics()

        # round state
        self._pull_buf: List[Tuple[torch.Tensor, bool]] = []
        self._push_buf: List[Tuple[torch.Tensor, torch.Tensor]] = []
        self._answer_buf: List[List[Tuple[torch.Tensor, torch.Tensor]]] = \
            [[] for _ in range(self.world)]
        self._pull_queue: Optional[torch.Tensor] = None  # limiter overflow
        self._outstanding = 0
        # fresh mode: answers received at round end, delivered next round
        self._pending_answers: List[Tuple[torch.Tensor, torch.Tensor]] = []
        # merged mode: retained sent pull segments (answer ids of the
        # next round's in-order feedback)
        self._prev_sent_pulls: List[Optional[torch.Tensor]] = \
            [None] * self.world
        self._stalled = False
        self._bucket_cache: Dict[Tuple[int, int], Any] = {}
        self._clock = 0
        self._min_clock = 0

        limit = worker_logic.pull_limit or config.pull_limit
        self._pull_limit = int(limit) if limit else None

        # pipelined mode: answers received last round, delivered (and
        # trained on) while THIS round's exchange is in flight.
        self._held: List[Tuple[torch.Tensor, torch.Tensor]] = []

        # A server logic that answers every pull batch in full, in
        # received order, within the same round lets the engine drop
        # answer ids (and their counts exchange) from the feedback edge:
        # the puller reconstructs them from its own sent segments.
        # NOTE: the flag shapes the collective channel layout, so every
        # rank must run the same server-logic type (all bundled jobs do);
        # the first round's protocol-homogeneity preflight fails fast on
        # a mismatch instead of hanging in a malformed collective.
        self._answers_in_order = bool(getattr(server_logic,
                                              "answers_in_order", False))
        self._proto_checked = False

        # A flush strategy may coalesce several logical pushes into one
        # round, so the concatenated push batch can repeat ids — which a
        # pushes_unique server logic (non-atomic add / sort-free adagrad)
        # would corrupt silently.  Reject the combination outright.
        if (flush is not None and
                getattr(server_logic, "pushes_unique", False)):
            raise ValueError(
                "a flush strategy coalesces pushes across rounds, so "
                "push ids may repeat — incompatible with a server logic "
                "constructed with pushes_unique=True (use "
                "pushes_unique=False, whose apply path pre-aggregates "
                "duplicates)")

    # ------------------------------------------------------------------
    def run(self, data: Iterable[Any], max_rounds: Optional[int] = None
            ) -> Tuple[List[Any], List[Any]]:
        """Run to drain (or ``max_rounds``); returns (worker_outputs,
        server_outputs)."""
        it = iter(data)
        exhausted = False
        r = 0
        while True:
            r += 1
            if max_rounds is not None and r > max_rounds:
                break
            exhausted, done = self._round(it, exhausted)
            if done:
                break
        self.worker_logic.close(self.client)
        self.server_logic.close(self.ps)
        return self.worker_outputs, self.server_outputs

    # ------------------------------------------------------------------
    def _round(self, it: Iterator[Any], exhausted: bool
               ) -> Tuple[bool, bool]:
        if not self._proto_checked:
            self._validate_protocol()
        if self.cfg.feedback == "merged":
            return self._round_merged(it, exhausted)
        if self.cfg.feedback == "pipelined":
            return self._round_pipelined(it, exhausted)
        return self._round_fresh(it, exhausted)

    # ------------------------------------------------------------------
    def _validate_protocol(self) -> None:
        """Fail fast (instead of hanging in a malformed collective) when
        ranks disagree on anything that shapes the wire protocol: the
        answers_in_order elision, wire dtype, feedback schedule, row dim
        or partitioner.  One tiny allgather before the first round."""
        self._proto_checked = True
        if self.world == 1:
            return
        import zlib
        desc = "|".join([
            str(int(self._answers_in_order)), self.cfg.wire_dtype,
            self.cfg.feedback, str(self.cfg.dim), self.cfg.partitioner,
            str(self.cfg.num_ids)])
        word = zlib.crc32(desc.encode())
        g = self.transport.allgather_int([word])
        if int(g.min()) != int(g.max()):
            raise RuntimeError(
                f"wire-protocol mismatch across ranks: rank {self.rank} "
                f"has ({desc}); protocol words per rank = "
                f"{g.reshape(-1).tolist()}.  All ranks must agree on "
                "server-logic answers_in_order, wire_dtype, feedback "
                "mode, dim and partitioner.")

    # ------------------------------------------------------------------
    def _round_fresh(self, it: Iterator[Any], exhausted: bool
                     ) -> Tuple[bool, bool]:
        """feedback="fresh": answers return on a second grouped
        send/recv within the SAME round (one counts+control all_gather +
        two p2p phases) — a pull requested by a callback in round r-1
        ships in r and is delivered in r+1, and pushes ship in the round
        after the callback that queued them.  Freshest values; the
        default."""
        m = self.metrics
        m.rounds += 1

        # 1. deliver last round's answers
        if self._pending_answers:
            ids = _concat([a for a, _ in self._pending_answers])
            vals = _concat([v for _, v in self._pending_answers])
            self._pending_answers = []
            self._outstanding -= int(ids.numel())
            self.worker_logic.on_pull_recv(ids, vals, self.client)

        # 2. consume data (unless SSP-stalled / exhausted / over limit)
        lag = self._clock - self._min_clock
        m.staleness_hist[lag] = m.staleness_hist.get(lag, 0) + 1
        stalled = (self.cfg.staleness is not None and
                   lag > self.cfg.staleness)
        backpressure = (self._pull_limit is not None and
                        self._queued_pulls() >= self._pull_limit)
        if stalled:
            m.ssp_stalls += 1
        if not exhausted and not stalled and not backpressure:
            try:
                batch = next(it)
            except StopIteration:
                exhausted = True
            else:
                # a streaming source may yield None = "no batch ready this
                # round"; the round proceeds without consuming.
                if batch is not None:
                    self._clock += 1
                    m.samples += _batch_size(self.worker_logic, batch)
                    self.worker_logic.on_recv(batch, self.client)

        # 3. coalesce + bucket (unless a combination-sender strategy holds
        # this round's ops back — messaging.FlushStrategy)
        if self.flush is not None and not self.flush.should_flush(
                self._pending_ops(),
                force=(exhausted or stalled or
                       self._pull_queue is not None)):
            m.held_rounds += 1
            pull_ids = push_ids = push_deltas = None
        else:
            pull_ids = self._take_pulls()
            push_ids, push_deltas = self._take_pushes()
        send = self._bucket(pull_ids, push_ids, push_deltas)
        sent_pulls = send[0]
        m.pull_ids_sent += int(pull_ids.numel()) if pull_ids is not None else 0
        m.push_ids_sent += int(push_ids.numel()) if push_ids is not None else 0

        # 4. exchange worker->server segments.  The SSP clock + drain
        # control rides the counts exchange (one collective, not two);
        # its values are fully determined before the exchange, so this
        # is the same information the reference's quiescence detection
        # needs, one leg earlier.
        worker_done = int(exhausted and self._outstanding == 0 and
                          not self._pending_answers and
                          self._queued_pulls() == 0 and
                          not self._pull_buf and not self._push_buf and
                          push_ids is None)
        dim = self.cfg.dim
        wire = torch.bfloat16 if self.cfg.wire_dtype == "bf16" \
            else torch.float32
        self._wire_out(send[2], wire)
        m.bytes_sent += self._xgmi_bytes(send)
        recv, ctrl = self.transport.alltoallv(
            send, [(torch.int64, ()), (torch.int64, ()), (wire, (dim,))],
            extra=[self._clock, worker_done, int(exhausted)])
        recv = self._to_device(recv)
        self._unwire(recv[2])
        r_pull, r_push_ids, r_push_deltas = recv

        # 5. server role: pushes first (fresher answers), then pulls
        for s in range(self.world):
            ids, deltas = r_push_ids[s], r_push_deltas[s]
            if ids is not None and ids.numel():
                self.server_logic.on_push_recv(ids, deltas, self.ps)
        for s in range(self.world):
            ids = r_pull[s]
            if ids is not None and ids.numel():
                self.server_logic.on_pull_recv(ids, s, self.ps)

        # 6. exchange answers back (the feedback edge).  When every
        # server answers in order (our table logics do), the answer IDS
        # never travel: the puller already knows them — its own pull
        # segments — and their counts, so the counts exchange is skipped
        # too and only the value payloads cross xGMI.
        ans_vals = [_concat([v for _, v in self._answer_buf[w]])
                    for w in range(self.world)]
        if self._answers_in_order:
            self._answer_buf = [[] for _ in range(self.world)]
            self._wire_out(ans_vals, wire)
            m.bytes_sent += self._xgmi_bytes([ans_vals])
            counts = [0 if t is None else int(t.numel())
                      for t in sent_pulls]
            back_vals = self._to_device(self.transport.alltoallv(
                [ans_vals], [(wire, (dim,))], recv_counts=[counts]))[0]
            self._unwire(back_vals)
            for s in range(self.world):
                if back_vals[s] is not None and back_vals[s].numel():
                    self._pending_answers.append(
                        (sent_pulls[s].to(back_vals[s].device),
                         back_vals[s]))
        else:
            ans_ids = [_concat([a for a, _ in self._answer_buf[w]])
                       for w in range(self.world)]
            self._answer_buf = [[] for _ in range(self.world)]
            self._wire_out(ans_vals, wire)
            m.bytes_sent += self._xgmi_bytes([ans_ids, ans_vals])
            back = self._to_device(self.transport.alltoallv(
                [ans_ids, ans_vals],
                [(torch.int64, ()), (wire, (dim,))]))
            self._unwire(back[1])
            for s in range(self.world):
                if back[0][s] is not None and back[0][s].numel():
                    self._pending_answers.append((back[0][s], back[1][s]))

        if self.checkpointer is not None:
            self.checkpointer.maybe(m.rounds)

        # 7. act on the control matrix gathered in step 4: [clock,
        # worker_done, exhausted].  An exhausted worker's clock freezes,
        # so it must not hold the SSP min-clock back (it would deadlock
        # the still-running workers).
        active = ctrl[:, 2] == 0
        self._min_clock = int(ctrl[active, 0].min()) if bool(active.any()) \
            else int(ctrl[:, 0].max())
        done = bool(ctrl[:, 1].all())
        return exhausted, done

    # ------------------------------------------------------------------
    def _round_merged(self, it: Iterator[Any], exhausted: bool
                      ) -> Tuple[bool, bool]:
        """feedback="merged": ONE merged exchange carries this rank's
        bucketed pulls/pushes AND the answers its shard produced LAST
        round — the reference's feedback edge rides the next round's
        shuffle, so each round costs exactly one counts+control
        all_gather and one grouped send/recv phase (RCCL: grouped
        ncclSend/ncclRecv over xGMI).

        Timeline of a pull: requested by a callback in round r-1,
        shipped in r, served in r, shipped back in r+1, delivered to
        ``on_pull_recv`` in r+1 — the worker never blocks on the server
        (the reference's async contract); a push queued by a callback in
        round r-1 ships and applies in round r.  Staleness comes from
        this pipeline depth; the SSP knob bounds it on top.
        """
        m = self.metrics
        m.rounds += 1

        # 1. bucket the ops queued by last round's callbacks (unless a
        # combination-sender strategy holds them — messaging.FlushStrategy)
        if self.flush is not None and not self.flush.should_flush(
                self._pending_ops(),
                force=(exhausted or self._stalled or
                       self._pull_queue is not None)):
            m.held_rounds += 1
            pull_ids = push_ids = push_deltas = None
        else:
            pull_ids = self._take_pulls()
            push_ids, push_deltas = self._take_pushes()
        send = self._bucket(pull_ids, push_ids, push_deltas)
        sent_pulls = send[0]
        m.pull_ids_sent += int(pull_ids.numel()) if pull_ids is not None else 0
        m.push_ids_sent += int(push_ids.numel()) if push_ids is not None else 0

        # 2. ONE merged exchange.  Channels: pull ids / push ids / push
        # deltas / [answer ids when the server may answer out of order] /
        # answer values; the SSP clock + drain control rides the counts
        # all_gather.  In-order servers (all bundled ones) never put
        # answer ids on the wire — the puller retained its sent segments.
        worker_done = int(exhausted and self._outstanding == 0 and
                          self._queued_pulls() == 0 and
                          not self._pull_buf and not self._push_buf and
                          push_ids is None and
                          not any(self._answer_buf))
        dim = self.cfg.dim
        wire = torch.bfloat16 if self.cfg.wire_dtype == "bf16" \
            else torch.float32
        self._wire_out(send[2], wire)
        ans_vals = [_concat([v for _, v in self._answer_buf[w]])
                    for w in range(self.world)]
        self._wire_out(ans_vals, wire)
        chans = [send[0], send[1], send[2]]
        specs = [(torch.int64, ()), (torch.int64, ()), (wire, (dim,))]
        if not self._answers_in_order:
            chans.append([_concat([a for a, _ in self._answer_buf[w]])
                          for w in range(self.world)])
            specs.append((torch.int64, ()))
        chans.append(ans_vals)
        specs.append((wire, (dim,)))
        self._answer_buf = [[] for _ in range(self.world)]
        m.bytes_sent += self._xgmi_bytes(chans)
        recv, ctrl = self.transport.alltoallv(
            chans, specs,
            extra=[self._clock, worker_done, int(exhausted)])
        recv = self._to_device(recv)
        self._unwire(recv[2])
        self._unwire(recv[-1])
        r_pull, r_push_ids, r_push_deltas = recv[0], recv[1], recv[2]
        back_vals = recv[-1]
        back_ids = self._prev_sent_pulls if self._answers_in_order \
            else recv[3]

        # 3. deliver the answers — they answer the pulls sent LAST round
        pend_ids: List[torch.Tensor] = []
        pend_vals: List[torch.Tensor] = []
        for s in range(self.world):
            v = back_vals[s]
            if v is not None and v.numel():
                i = back_ids[s]
                pend_ids.append(i if i.device == v.device else
                                i.to(v.device))
                pend_vals.append(v)
        if pend_ids:
            ids = _concat(pend_ids)
            self._outstanding -= int(ids.numel())
            self.worker_logic.on_pull_recv(ids, _concat(pend_vals),
                                           self.client)
        self._prev_sent_pulls = sent_pulls

        # 4. control from THIS exchange (fresh min clock), then consume
        # data unless SSP-stalled / exhausted / over the pull limit.  An
        # exhausted worker's clock freezes, so it must not hold the SSP
        # min-clock back (it would deadlock the still-running workers).
        active = ctrl[:, 2] == 0
        self._min_clock = int(ctrl[active, 0].min()) if bool(active.any()) \
            else int(ctrl[:, 0].max())
        lag = self._clock - self._min_clock
        m.staleness_hist[lag] = m.staleness_hist.get(lag, 0) + 1
        stalled = (self.cfg.staleness is not None and
                   lag > self.cfg.staleness)
        self._stalled = stalled
        backpressure = (self._pull_limit is not None and
                        self._queued_pulls() >= self._pull_limit)
        if stalled:
            m.ssp_stalls += 1
        if not exhausted and not stalled and not backpressure:
            try:
                batch = next(it)
            except StopIteration:
                exhausted = True
            else:
                # a streaming source may yield None = "no batch ready
                # this round"; the round proceeds without consuming.
                if batch is not None:
                    self._clock += 1
                    m.samples += _batch_size(self.worker_logic, batch)
                    self.worker_logic.on_recv(batch, self.client)

        # 5. server phase: pushes first (fresher answers), then pulls;
        # produced answers are stashed for the NEXT round's exchange
        for s in range(self.world):
            ids, deltas = r_push_ids[s], r_push_deltas[s]
            if ids is not None and ids.numel():
                self.server_logic.on_push_recv(ids, deltas, self.ps)
        for s in range(self.world):
            ids = r_pull[s]
            if ids is not None and ids.numel():
                self.server_logic.on_pull_recv(ids, s, self.ps)

        if self.checkpointer is not None:
            self.checkpointer.maybe(m.rounds)

        done = bool(ctrl[:, 1].all())
        return exhausted, done

    # ------------------------------------------------------------------
    def _round_pipelined(self, it: Iterator[Any], exhausted: bool
                         ) -> Tuple[bool, bool]:
        """feedback="pipelined": the merged single-exchange layout,
        split-phase — the exchange is ISSUED (counts+control resolve,
        payloads stay on the wire), the round's fused training step runs
        on the answers HELD from last round's exchange while RCCL's
        grouped send/recv progresses on its own HIP stream, then the
        exchange COMPLETES and the server phase applies what arrived.
        This is the north star's double-buffered schedule: gradient
        compute overlaps the xGMI all-to-all.

        Timeline of a pull: requested by a callback in round r-1, ships
        in r, served in r's server phase, ships back in r+1, received at
        r+1's completion, HELD, delivered to ``on_pull_recv`` during
        r+2's flight.  One round more answer latency than "merged"
        (bounded staleness still holds: the SSP lag test uses this
        round's fresh control, resolved at issue time)."""
        m = self.metrics
        m.rounds += 1

        # 1. bucket the ops queued by last round's callbacks
        if self.flush is not None and not self.flush.should_flush(
                self._pending_ops(),
                force=(exhausted or self._stalled or
                       self._pull_queue is not None)):
            m.held_rounds += 1
            pull_ids = push_ids = push_deltas = None
        else:
            pull_ids = self._take_pulls()
            push_ids, push_deltas = self._take_pushes()
        send = self._bucket(pull_ids, push_ids, push_deltas)
        sent_pulls = send[0]
        m.pull_ids_sent += int(pull_ids.numel()) if pull_ids is not None else 0
        m.push_ids_sent += int(push_ids.numel()) if push_ids is not None else 0

        worker_done = int(exhausted and self._outstanding == 0 and
                          self._queued_pulls() == 0 and
                          not self._pull_buf and not self._push_buf and
                          push_ids is None and
                          not any(self._answer_buf) and
                          not self._held)
        dim = self.cfg.dim
        wire = torch.bfloat16 if self.cfg.wire_dtype == "bf16" \
            else torch.float32
        self._wire_out(send[2], wire)
        ans_vals = [_concat([v for _, v in self._answer_buf[w]])
                    for w in range(self.world)]
        self._wire_out(ans_vals, wire)
        chans = [send[0], send[1], send[2]]
        specs = [(torch.int64, ()), (torch.int64, ()), (wire, (dim,))]
        if not self._answers_in_order:
            chans.append([_concat([a for a, _ in self._answer_buf[w]])
                          for w in range(self.world)])
            specs.append((torch.int64, ()))
        chans.append(ans_vals)
        specs.append((wire, (dim,)))
        self._answer_buf = [[] for _ in range(self.world)]
        m.bytes_sent += self._xgmi_bytes(chans)

        # 2. ISSUE the exchange.  Control rides the counts phase, so it
        # is fully resolved when begin returns; only payloads remain in
        # flight.
        import time as _time
        t0 = _time.perf_counter()
        h = self.transport.alltoallv_begin(
            chans, specs, extra=[self._clock, worker_done, int(exhausted)])
        ctrl = self.transport.ctrl_of(h)
        t1 = _time.perf_counter()
        m.t_issue += t1 - t0

        # 3. OVERLAPPED compute: deliver the held answers (the fused
        # training step launches on the compute stream while the p2p
        # payloads cross xGMI), then consume the next batch unless
        # SSP-stalled / exhausted / over the pull limit.
        held, self._held = self._held, []
        if held:
            ids = _concat([i for i, _ in held])
            self._outstanding -= int(ids.numel())
            self.worker_logic.on_pull_recv(ids,
                                           _concat([v for _, v in held]),
                                           self.client)
        active = ctrl[:, 2] == 0
        self._min_clock = int(ctrl[active, 0].min()) if bool(active.any()) \
            else int(ctrl[:, 0].max())
        lag = self._clock - self._min_clock
        m.staleness_hist[lag] = m.staleness_hist.get(lag, 0) + 1
        stalled = (self.cfg.staleness is not None and
                   lag > self.cfg.staleness)
        self._stalled = stalled
        backpressure = (self._pull_limit is not None and
                        self._queued_pulls() >= self._pull_limit)
        if stalled:
            m.ssp_stalls += 1
        if not exhausted and not stalled and not backpressure:
            try:
                batch = next(it)
            except StopIteration:
                exhausted = True
            else:
                if batch is not None:
                    self._clock += 1
                    m.samples += _batch_size(self.worker_logic, batch)
                    self.worker_logic.on_recv(batch, self.client)

        # 4. COMPLETE the exchange; hold the received answers for the
        # next round's overlapped delivery.
        t2 = _time.perf_counter()
        m.t_overlap_compute += t2 - t1
        recv, _ = self.transport.alltoallv_end(h)
        t3 = _time.perf_counter()
        m.t_comm_wait += t3 - t2
        recv = self._to_device(recv)
        self._unwire(recv[2])
        self._unwire(recv[-1])
        r_pull, r_push_ids, r_push_deltas = recv[0], recv[1], recv[2]
        back_vals = recv[-1]
        back_ids = self._prev_sent_pulls if self._answers_in_order \
            else recv[3]
        for s in range(self.world):
            v = back_vals[s]
            if v is not None and v.numel():
                i = back_ids[s]
                self._held.append(
                    (i if i.device == v.device else i.to(v.device), v))
        self._prev_sent_pulls = sent_pulls

        # 5. server phase: pushes first (fresher answers), then pulls;
        # produced answers ship in the NEXT round's exchange.
        for s in range(self.world):
            ids, deltas = r_push_ids[s], r_push_deltas[s]
            if ids is not None and ids.numel():
                self.server_logic.on_push_recv(ids, deltas, self.ps)
        for s in range(self.world):
            ids = r_pull[s]
            if ids is not None and ids.numel():
                self.server_logic.on_pull_recv(ids, s, self.ps)

        if self.checkpointer is not None:
            self.checkpointer.maybe(m.rounds)
        m.t_server += _time.perf_counter() - t3

        done = bool(ctrl[:, 1].all())
        return exhausted, done

    # ------------------------------------------------------------------
    def _to_device(self, recv):
        """Move received segments onto the engine device (no-op when the
        transport already delivered them there, e.g. RCCL; a CPU-side
        transport like gloo hands back host tensors while compute runs on
        the GPU)."""
        dev = self.device
        for chan in recv:
            for i, t in enumerate(chan):
                if t is not None and t.device != dev:
                    chan[i] = t.to(dev, non_blocking=True)
        return recv

    # ------------------------------------------------------------------
    def _pending_ops(self) -> int:
        """Accumulated pull+push ops a flush strategy weighs against its
        threshold (the reference's CombinationLogic count condition)."""
        n = sum(int(t.numel()) for t, _ in self._pull_buf)
        n += sum(int(t.numel()) for t, _ in self._push_buf)
        if self._pull_queue is not None:
            n += int(self._pull_queue.numel())
        return n

    def _xgmi_bytes(self, channels) -> int:
        """Bytes this rank puts on the wire this round (self-addressed
        segments short-circuit in every transport and are not counted)."""
        total = 0
        for chan in channels:
            for d, t in enumerate(chan):
                if d != self.rank and t is not None:
                    total += t.numel() * t.element_size()
        return total

    def _wire_out(self, chan: List[Optional[torch.Tensor]],
                  wire: torch.dtype) -> None:
        """Cast a payload channel to the wire dtype in place (bf16 halves
        xGMI bytes; tables keep f32 masters, so only the wire narrows).
        The self-addressed segment short-circuits in every transport and
        stays f32 — full precision for the 1/world local fraction."""
        if wire is torch.float32:
            return
        for d, t in enumerate(chan):
            if t is not None and d != self.rank:
                chan[d] = t.to(wire)

    @staticmethod
    def _unwire(chan: List[Optional[torch.Tensor]]) -> None:
        for s, t in enumerate(chan):
            if t is not None and t.dtype is not torch.float32:
                chan[s] = t.float()

    # ------------------------------------------------------------------
    def _queued_pulls(self) -> int:
        n = self._pull_queue.numel() if self._pull_queue is not None else 0
        return self._outstanding + n

    # NOTE: pulls that overflowed the pull-limit budget into
    # _pull_queue were already released by a flush decision; the flush
    # gate must not hold them again (force includes "queue non-empty"),
    # otherwise backpressure stops new ops, the count threshold can
    # never be reached, and the rank deadlocks — with SSP on it then
    # stalls every other rank too (found by the pipelined kitchen-sink
    # test; the hazard existed in all three schedules).
    def _take_pulls(self) -> Optional[torch.Tensor]:
        parts = self._pull_buf
        self._pull_buf = []
        if self._pull_queue is not None:
            parts = [(self._pull_queue, False)] + parts
            self._pull_queue = None
        if not parts:
            return None
        if len(parts) == 1 and parts[0][1]:
            ids = parts[0][0]  # caller-guaranteed unique: skip coalescing
        else:
            ids = torch.unique(_concat([t for t, _ in parts]))
        if self._pull_limit is not None:
            budget = self._pull_limit - self._outstanding
            if budget <= 0:
                self._pull_queue = ids
                return None
            if ids.numel() > budget:
                self._pull_queue = ids[budget:]
                ids = ids[:budget]
        self._outstanding += int(ids.numel())
        return ids

    def _take_pushes(self) -> Tuple[Optional[torch.Tensor],
                                    Optional[torch.Tensor]]:
        parts = self._push_buf
        self._push_buf = []
        if not parts:
            return None, None
        ids = _concat([i for i, _ in parts])
        deltas = _concat([d for _, d in parts])
        return ids, deltas

    def _bucket(self, pull_ids, push_ids, push_deltas):
        """K10 bucket_pack: split each channel per destination shard."""
        W = self.world
        pulls: List[Optional[torch.Tensor]] = [None] * W
        pids: List[Optional[torch.Tensor]] = [None] * W
        pdel: List[Optional[torch.Tensor]] = [None] * W
        if pull_ids is not None and pull_ids.numel():
            if W == 1:
                pulls[0] = pull_ids
            else:
                order, counts, _ = self._shard_counts(pull_ids)
                sorted_ids = pull_ids[order]
                off = 0
                for d in range(W):
                    n = int(counts[d])
                    if n:
                        pulls[d] = sorted_ids[off:off + n]
                    off += n
        if push_ids is not None and push_ids.numel():
            if W == 1:
                pids[0], pdel[0] = push_ids, push_deltas
            else:
                order, counts, _ = self._shard_counts(push_ids)
                sorted_ids = push_ids[order]
                sorted_del = push_deltas[order]
                off = 0
                for d in range(W):
                    n = int(counts[d])
                    if n:
                        pids[d] = sorted_ids[off:off + n]
                        pdel[d] = sorted_del[off:off + n]
                    off += n
        return [pulls, pids, pdel]

    def _shard_counts(self, ids: torch.Tensor):
        """bucket_by_shard with counts padded to world size (the
        partitioner may define fewer shards than ranks when
        ps_parallelism < worker_parallelism — algorithms.py).

        Memoized by buffer identity: steady-state workers re-send one
        cached id tensor every round (dense-pull MF sends the same
        arange; pushes reuse the pulled uids), so the argsort/bincount
        bucketing runs once, not every round.  Held references keep the
        cached storages alive, so a pointer can never alias a different
        live tensor.  The client.pull/push contract forbids mutating an
        id buffer in place after handing it over; with
        ``cfg.debug_validate`` the cache keeps a content snapshot and
        re-verifies it on every hit, turning such a mutation into a
        loud error instead of silent mis-routing."""
        key = (ids.data_ptr(), ids.numel())
        hit = self._bucket_cache.get(key)
        if hit is not None:
            held, order, counts = hit
            if (self.cfg.debug_validate and
                    not torch.equal(held[1], ids)):
                raise RuntimeError(
                    "an id buffer was mutated in place after being "
                    "handed to client.pull/push (see the API contract); "
                    "shard bucketing would silently mis-route ids")
            return order, counts, None
        order, counts, shards = bucket_by_shard(ids, self.part)
        if counts.numel() < self.world:
            counts = torch.cat([counts, counts.new_zeros(
                self.world - counts.numel())])
        if len(self._bucket_cache) >= 8:
            self._bucket_cache.clear()
        snapshot = ids.clone() if self.cfg.debug_validate else ids
        self._bucket_cache[key] = ((ids, snapshot), order, counts)
        return order, counts, shards


def _batch_size(logic: WorkerLogic, batch: Any) -> int:
    fn = getattr(logic, "batch_size", None)
    if fn is not None:
        return int(fn(batch))
    try:
        return len(batch)
    except TypeError:
        return 0
