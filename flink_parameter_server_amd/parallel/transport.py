"""Transports: how pull/push segments move between ranks.

The reference's transport is Flink's streaming network runtime (Netty TCP
record shuffle worker->server, the iteration feedback edge server->worker —
SURVEY.md §5.8).  Here the same two edges are variable-length all-to-all
exchanges:

* ``TorchDistTransport`` — one process per GPU over ``torch.distributed``.
  Counts are exchanged with an all_gather of an int64 matrix, payloads with
  ``batch_isend_irecv`` (on the "nccl" backend = RCCL this lowers to grouped
  ncclSend/ncclRecv over xGMI — the alltoallv shape SURVEY.md §5.8 calls
  for; a ring collective would be bound by ONE of the 7 p2p links).  The
  same code path runs on "gloo" for CPU multi-process tests.
* ``LoopbackTransport`` — N ranks as N threads in one process, lockstep
  rounds through shared mailboxes: the analog of the reference's Flink
  MiniCluster local-mode tests, so every distributed test
  can run single-process.
* ``LocalTransport`` — world size 1; everything short-circuits.

Self-addressed segments never touch the backend in any transport
(same-device short-circuit — the 1/world fraction of traffic the reference
gets for free when Flink co-locates subtasks).
"""
from __future__ import annotations

import threading
from typing import List, Optional, Sequence, Tuple

import torch


class Transport:
    rank: int = 0
    world: int = 1

    def alltoallv(self, parts: Sequence[Sequence[Optional[torch.Tensor]]],
                  specs: Sequence[Tuple[torch.dtype, Tuple[int, ...]]],
                  extra: Optional[Sequence[int]] = None,
                  recv_counts: Optional[Sequence[Sequence[int]]] = None):
        """parts[c][d] = tensor for channel c to rank d (dim 0 = items;
        may be None = empty).  specs[c] = (dtype, trailing_shape) of
        channel c — every rank must pass the same specs (the receiver
        sizes its buffers from them, never from its own outgoing data).
        Returns recv[c][s] = tensor received on channel c from rank s.

        ``extra``: small host int vector folded into the counts exchange
        (one collective instead of two — the engine's SSP clock +
        termination control rides here); when given the return value is
        ``(recv, extras)`` with extras int64 (world, len(extra)).
        ``recv_counts[c][s]``: receive sizes already known to the caller
        (e.g. in-order pull answers: the receiver sized them when it sent
        the pulls) — the counts exchange is skipped entirely.
        """
        h = self.alltoallv_begin(parts, specs, extra, recv_counts)
        return self.alltoallv_end(h)

    # -- split-phase form: the engine's pipelined feedback mode issues
    # the exchange, runs the round's fused compute while the payloads
    # are in flight (RCCL p2p progresses on its own HIP stream), then
    # completes it — the north star's "gradient compute overlaps the
    # xGMI all-to-all".
    def alltoallv_begin(self, parts, specs, extra=None, recv_counts=None):
        """Issue the exchange; returns an opaque handle.  Counts (and the
        folded control vector) ARE resolved here — only the payload
        transfer remains in flight when this returns.  Every rank must
        call begin at the same point of the same round (collective)."""
        raise NotImplementedError

    def alltoallv_end(self, handle):
        """Complete the exchange issued by :meth:`alltoallv_begin`;
        returns what :meth:`alltoallv` would have returned."""
        raise NotImplementedError

    def ctrl_of(self, handle) -> Optional[torch.Tensor]:
        """The control matrix (world, n_extra) of an in-flight handle —
        available right after ``alltoallv_begin`` (it rides the counts
        phase), so SSP/termination decisions can be made while the
        payloads are still on the wire."""
        return handle[-1]

    def allgather_int(self, vec: Sequence[int]) -> torch.Tensor:
        """Gather a small int64 host vector from every rank -> (world, len)."""
        raise NotImplementedError

    def barrier(self) -> None:
        pass


class LocalTransport(Transport):
    def __init__(self) -> None:
        self.rank, self.world = 0, 1

    def alltoallv(self, parts, specs, extra=None, recv_counts=None):
        recv = [[p[0]] for p in parts]
        if extra is None:
            return recv
        return recv, torch.tensor([list(extra)], dtype=torch.int64)

    def alltoallv_begin(self, parts, specs, extra=None, recv_counts=None):
        recv = [[p[0]] for p in parts]
        ctrl = (torch.tensor([list(extra)], dtype=torch.int64)
                if extra is not None else None)
        return (recv, extra is not None, ctrl)

    def alltoallv_end(self, handle):
        recv, has_extra, ctrl = handle
        return (recv, ctrl) if has_extra else recv

    def allgather_int(self, vec):
        return torch.tensor([list(vec)], dtype=torch.int64)


class TorchDistTransport(Transport):
    """torch.distributed-backed transport (nccl=RCCL on ROCm, or gloo)."""

    def __init__(self, device: Optional[torch.device] = None) -> None:
        import torch.distributed as dist
        if not dist.is_initialized():
            raise RuntimeError("torch.distributed is not initialized")
        self.dist = dist
        self.rank = dist.get_rank()
        self.world = dist.get_world_size()
        self.device = torch.device(device) if device is not None else (
            torch.device("cuda", torch.cuda.current_device())
            if dist.get_backend() == "nccl" else torch.device("cpu"))
        # gloo cannot move device tensors: stage through CPU on the way
        # out and deliver on self.device on the way in — the same
        # delivery contract as nccl/RCCL (recv segments arrive
        # on-device), so the engine's device paths are exercised by the
        # CPU-backend integration tests too.
        self.stage_via_cpu = (self.device.type == "cuda" and
                              dist.get_backend() != "nccl")
        # Counts + control lane: a per-round device all_gather + .cpu()
        # on the nccl(=RCCL) backend costs a GPU round-trip per round
        # and tangles the control word with the compute stream.  A
        # dedicated host-side gloo group carries the few-hundred-byte
        # counts/control vector instead: pure host exchange, zero GPU
        # involvement, posted async so its latency overlaps the
        # send-side staging; the payload p2p posts the moment the host
        # has the counts.  Created for EVERY multi-rank backend (gloo
        # included) so the CPU multi-process tests execute exactly the
        # code path the 8-GPU RCCL run takes.  (Collective: every rank
        # constructs the transport, so every rank reaches new_group.)
        self._ctrl_pg = None
        if self.world > 1:
            self._ctrl_pg = dist.new_group(backend="gloo")

    def alltoallv_begin(self, parts, specs, extra=None, recv_counts=None):
        dist = self.dist
        C, W, me = len(parts), self.world, self.rank
        dev = torch.device("cpu") if self.stage_via_cpu else self.device

        # 1) counts: one all_gather of int64[C*W (+extra)] covers every
        # channel — skipped entirely when the caller knows its receive
        # sizes (in-order answers) and has no control payload to fold in.
        counts = extras = None
        counts_work = gathered = None
        if recv_counts is None or extra is not None:
            ne = len(extra) if extra is not None else 0
            send_counts = torch.zeros(C * W + ne, dtype=torch.int64)
            for c, chan in enumerate(parts):
                for d, t in enumerate(chan):
                    send_counts[c * W + d] = 0 if t is None else t.shape[0]
            if ne:
                send_counts[C * W:] = torch.tensor(list(extra),
                                                   dtype=torch.int64)
            if self._ctrl_pg is not None:
                # host-side lane: no GPU round-trip for counts/control.
                # Posted ASYNC so its latency overlaps the send-side
                # staging below; waited before recv buffers are sized.
                gathered = [torch.empty_like(send_counts)
                            for _ in range(W)]
                counts_work = dist.all_gather(gathered, send_counts,
                                              group=self._ctrl_pg,
                                              async_op=True)
            else:
                gdev = send_counts.to(dev)
                gathered = [torch.empty_like(gdev) for _ in range(W)]
                dist.all_gather(gathered, gdev)
                full = torch.stack([g.cpu() for g in gathered])
                counts, extras = full[:, :C * W], full[:, C * W:]

        # 2) payloads: grouped isend/irecv (RCCL: grouped ncclSend/Recv
        # over xGMI, progressing on RCCL's own HIP stream).  Sends are
        # staged (contiguous/device moves) while the counts all_gather
        # is still in flight; the p2p group posts after it resolves.
        sends: List[torch.Tensor] = []
        send_ops = []  # (tensor, dest)
        recv: List[List[Optional[torch.Tensor]]] = [
            [None] * W for _ in range(C)]
        for c, chan in enumerate(parts):
            recv[c][me] = chan[me] if chan[me] is not None else None
            for d, t in enumerate(chan):
                if d == me or t is None or t.shape[0] == 0:
                    continue
                t = t.contiguous()
                if t.device != dev:
                    t = t.to(dev)
                sends.append(t)  # keep alive until end()
                send_ops.append((t, d))

        if counts_work is not None:
            counts_work.wait()
            full = torch.stack(gathered)  # (W, C*W+ne)
            counts, extras = full[:, :C * W], full[:, C * W:]

        def n_from(s, c):
            if recv_counts is not None:
                return int(recv_counts[c][s])
            return int(counts[s, c * W + me])

        p2p = []
        for c, chan in enumerate(parts):
            dtype, trailing = specs[c]
            for s in range(W):
                if s == me:
                    continue
                n = n_from(s, c)
                if n > 0:
                    buf = torch.empty((n, *trailing), dtype=dtype, device=dev)
                    recv[c][s] = buf
                    p2p.append(dist.P2POp(dist.irecv, buf, s))
        for t, d in send_ops:
            p2p.append(dist.P2POp(dist.isend, t, d))
        works = dist.batch_isend_irecv(p2p) if p2p else []
        return (works, sends, recv, extra is not None, extras)

    def alltoallv_end(self, handle):
        works, _sends, recv, has_extra, extras = handle
        for work in works:
            # nccl(=RCCL): wait() only makes the current HIP stream wait
            # on the comm stream (no host block); gloo: joins the
            # transfer threads.
            work.wait()
        if self.stage_via_cpu:
            # deliver on-device, matching the nccl contract (self-
            # addressed segments short-circuited and are already there)
            for chan in recv:
                for i, t in enumerate(chan):
                    if t is not None and not t.is_cuda:
                        chan[i] = t.to(self.device, non_blocking=True)
        return (recv, extras) if has_extra else recv

    def allgather_int(self, vec):
        if self._ctrl_pg is not None:
            t = torch.tensor(list(vec), dtype=torch.int64)
            out = [torch.empty_like(t) for _ in range(self.world)]
            self.dist.all_gather(out, t, group=self._ctrl_pg)
            return torch.stack(out)
        dev = torch.device("cpu") if self.stage_via_cpu else self.device
        t = torch.tensor(list(vec), dtype=torch.int64, device=dev)
        out = [torch.empty_like(t) for _ in range(self.world)]
        self.dist.all_gather(out, t)
        return torch.stack([o.cpu() for o in out])

    def barrier(self):
        self.dist.barrier()


def run_loopback(world: int, fn, timeout: Optional[float] = 120.0) -> list:
    """Run ``fn(rank, transport)`` on ``world`` lockstep threads (the
    in-process MiniCluster).  Returns the per-rank results; re-raises the
    first rank exception."""
    transports = LoopbackTransport.make_group(world)
    results: List = [None] * world
    errors: List = [None] * world

    def target(r: int) -> None:
        try:
            results[r] = fn(r, transports[r])
        except BaseException as e:  # noqa: BLE001
            errors[r] = e
            transports[r].shared.barrier.abort()

    threads = [threading.Thread(target=target, args=(r,), daemon=True)
               for r in range(world)]
    for t in threads:
        t.start()
    for t in threads:
        t.join(timeout)
    for e in errors:
        if e is not None and not isinstance(e, threading.BrokenBarrierError):
            raise e
    for t in threads:
        if t.is_alive():
            raise TimeoutError("loopback job did not finish")
    return results


class _LoopbackShared:
    def __init__(self, world: int):
        self.world = world
        self.barrier = threading.Barrier(world)
        self.mailbox: List[Optional[list]] = [None] * world
        self.intbox: List[Optional[torch.Tensor]] = [None] * world


class LoopbackTransport(Transport):
    """In-process multi-rank transport (threads + lockstep barriers)."""

    @staticmethod
    def make_group(world: int) -> List["LoopbackTransport"]:
        shared = _LoopbackShared(world)
        return [LoopbackTransport(shared, r) for r in range(world)]

    def __init__(self, shared: _LoopbackShared, rank: int):
        self.shared = shared
        self.rank = rank
        self.world = shared.world

    def alltoallv(self, parts, specs, extra=None, recv_counts=None):
        sh = self.shared
        # Cross-rank segments are COPIED at send time, exactly like the
        # real backends (nccl/gloo isend serialize the buffer; only the
        # self-addressed short-circuit may alias).  Passing references
        # across threads let a peer's later in-place mutation (e.g. the
        # full-range gather alias mutated by its shard's next push
        # apply) race with the receiver's single-barrier-per-round
        # merged/pipelined consumption — a loopback-only hazard that
        # does not exist in the process-per-rank deployment.
        self_rank = self.rank
        sh.mailbox[self_rank] = [
            [t if (d == self_rank or t is None) else t.clone()
             for d, t in enumerate(chan)] for chan in parts]
        if extra is not None:
            sh.intbox[self.rank] = torch.tensor(list(extra),
                                                dtype=torch.int64)
        sh.barrier.wait()
        C = len(parts)
        recv: List[List[Optional[torch.Tensor]]] = [
            [None] * self.world for _ in range(C)]
        for s in range(self.world):
            box = sh.mailbox[s]
            for c in range(C):
                recv[c][s] = box[c][self.rank]
        extras = torch.stack(list(sh.intbox)) if extra is not None else None
        sh.barrier.wait()  # everyone has read before boxes are overwritten
        if extra is None:
            return recv
        return recv, extras

    # split-phase form: lockstep threads all reach begin at the same
    # round point, so the full exchange happens in begin (the "flight"
    # is instantaneous in-process) and end just hands the result back.
    def alltoallv_begin(self, parts, specs, extra=None, recv_counts=None):
        out = self.alltoallv(parts, specs, extra, recv_counts)
        if extra is not None:
            recv, extras = out
            return (recv, True, extras)
        return (out, False, None)

    def alltoallv_end(self, handle):
        recv, has_extra, extras = handle
        return (recv, extras) if has_extra else recv

    def allgather_int(self, vec):
        sh = self.shared
        sh.intbox[self.rank] = torch.tensor(list(vec), dtype=torch.int64)
        sh.barrier.wait()
        out = torch.stack(list(sh.intbox))
        sh.barrier.wait()
        return out

    def barrier(self):
        self.shared.barrier.wait()
