"""Job configuration.

The reference has no config system — every knob is a plain function
parameter on the entry points (``FlinkParameterServer.transform(...)``,
``psOnlineMF(...)``).  We mirror that with one dataclass
whose field names map 1:1 onto the reference's parameters:

==========================  =====================================
reference parameter          PSConfig field
==========================  =====================================
``workerParallelism``        ``n_workers``  (= world size; one worker
                             role per GPU/rank)
``psParallelism``            ``n_shards``   (= world size; one shard
                             role per GPU/rank)
``pullLimit``                ``pull_limit``
``iterationWaitTime``        (gone — quiescence timeout replaced by an
                             explicit all-drained allreduce)
==========================  =====================================

New capabilities (not in the reference): ``staleness`` (SSP bounded
staleness; the reference is unbounded-async, which is ``staleness=None``).
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Optional


@dataclass
class PSConfig:
    # Parallelism: one process per GPU; each rank hosts one worker role AND
    # one server-shard role (the reference co-locates worker and server
    # subtasks on shared TaskManagers the same way — SURVEY.md §5.8).
    n_workers: int = 1
    n_shards: int = 1

    # Max distinct parameter ids with an un-answered pull outstanding per
    # worker.  Reference default in psOnlineMF is 1600; batched rounds want
    # far more.
    pull_limit: Optional[int] = None

    # SSP bounded staleness in rounds; None = unbounded async (reference
    # parity).  A worker stalls data consumption when its clock runs more
    # than `staleness` rounds ahead of the slowest worker.
    staleness: Optional[int] = None

    # Wire dtype for pull answers / push deltas ("f32" or "bf16").
    # Tables always keep f32 master values.
    wire_dtype: str = "f32"

    # Feedback-edge scheduling:
    #   "fresh"  (default) — answers return on a second grouped
    #       send/recv in the SAME round (1 all_gather + 2 p2p phases per
    #       round; freshest values, 1-round answer latency).
    #   "merged" — answers ride the NEXT round's single exchange
    #       (1 all_gather + 1 p2p phase per round; +1 round of answer
    #       latency / staleness — the throughput option for large
    #       multi-GPU jobs where per-round collective latency dominates).
    #   "pipelined" — merged layout, split-phase: the exchange is ISSUED,
    #       the round's fused compute (training on the answers received
    #       last round) runs while the payloads are on the wire (RCCL
    #       p2p progresses on its own HIP stream), then the exchange
    #       completes and the server phase applies what arrived.  +1
    #       further round of answer latency; comm/compute overlap — the
    #       north star's double-buffered schedule and the multi-GPU
    #       default.
    feedback: str = "fresh"

    # Debug: re-verify content-derived caches (shard bucketing) every
    # round instead of trusting buffer identity — catches worker logics
    # that mutate an id buffer in place after handing it to the engine
    # (forbidden by the client.pull/push contract).
    debug_validate: bool = False

    # Partitioning of parameter ids over shards: "mod" (hash partitioner —
    # the reference's hash(paramId) % psParallelism; Scala Int hashCode is
    # the identity) or "range" (contiguous blocks, the PA path option).
    partitioner: str = "mod"

    # Total number of parameter ids (rows) the job may touch; used to size
    # the dense per-shard table slabs.
    num_ids: int = 0
    dim: int = 1

    def __post_init__(self) -> None:
        if self.n_workers < 1 or self.n_shards < 1:
            raise ValueError("n_workers and n_shards must be >= 1")
        if self.partitioner not in ("mod", "range"):
            raise ValueError(f"unknown partitioner {self.partitioner!r}")
        if self.wire_dtype not in ("f32", "bf16"):
            raise ValueError(f"unknown wire_dtype {self.wire_dtype!r}")
        if self.feedback not in ("fresh", "merged", "pipelined"):
            raise ValueError(f"unknown feedback {self.feedback!r}")
