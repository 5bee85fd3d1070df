"""Property-based engine tests (hypothesis): random combinations of
world size, feedback schedule, flush strategy, pull limit, SSP bound,
wire dtype and traffic shape must always TERMINATE, answer every pull
exactly once, and produce the global push totals.  This is the net that
catches scheduling livelocks (e.g. the queued-pull/flush-gate deadlock
fixed in round 2) that example-based tests only hit by luck."""
import pytest
import torch
from hypothesis import given, settings, strategies as st

from flink_parameter_server_amd import (PSConfig, PSEngine, ShardedTable,
                                        TableServerLogic, WorkerLogic)
from flink_parameter_server_amd.messaging import CountFlush
from flink_parameter_server_amd.parallel.partitioner import ModPartitioner
from flink_parameter_server_amd.parallel.transport import run_loopback


@pytest.mark.timeout(600)
# derandomized: a fixed example set keeps the driver's CI runs
# deterministic; run deeper random soaks manually by flipping this off
@settings(max_examples=25, deadline=None, derandomize=True)
@given(
    world=st.integers(1, 4),
    feedback=st.sampled_from(["fresh", "merged", "pipelined"]),
    flush_count=st.one_of(st.none(), st.integers(1, 20)),
    pull_limit=st.one_of(st.none(), st.integers(1, 12)),
    staleness=st.one_of(st.none(), st.integers(0, 4)),
    wire=st.sampled_from(["f32", "bf16"]),
    n_ids=st.integers(3, 40),
    ops_per_round=st.integers(1, 9),
    n_batches=st.integers(1, 12),
    seed=st.integers(0, 10**6),
)
def test_engine_always_terminates_and_accounts(
        world, feedback, flush_count, pull_limit, staleness, wire,
        n_ids, ops_per_round, n_batches, seed):
    dim = 2

    class Traffic(WorkerLogic):
        def __init__(self, rank):
            self.g = torch.Generator().manual_seed(seed * 7 + rank)
            self.pushed = torch.zeros(n_ids, dim)
            self.requested = 0
            self.answered = 0

        def on_recv(self, data, client):
            k = int(torch.randint(1, ops_per_round + 1, (1,),
                                  generator=self.g))
            ids = torch.unique(torch.randint(0, n_ids, (k,),
                                             generator=self.g))
            deltas = torch.randn(ids.numel(), dim, generator=self.g)
            client.push(ids, deltas)
            self.pushed[ids] += deltas
            pull = torch.unique(torch.randint(0, n_ids, (k,),
                                              generator=self.g))
            client.pull(pull, unique=True)
            self.requested += int(pull.numel())

        def on_pull_recv(self, ids, values, client):
            assert values.shape == (ids.numel(), dim)
            self.answered += int(ids.numel())

    def job(rank, transport):
        part = ModPartitioner(n_ids, world)
        table = ShardedTable(part, rank, dim, "cpu")
        logic = Traffic(rank)
        # a worker-logic pull_limit overrides the config one
        eng = PSEngine(
            PSConfig(num_ids=n_ids, dim=dim, feedback=feedback,
                     wire_dtype=wire, staleness=staleness,
                     pull_limit=pull_limit),
            logic, TableServerLogic(table), transport, part,
            flush=CountFlush(flush_count) if flush_count else None)
        # max_rounds as a LIVELOCK BACKSTOP far above any legitimate
        # need; hitting it (done stays False) is the failure signal
        it = iter(range(n_batches))
        exhausted = done = False
        rounds = 0
        while not done and rounds < 600:
            rounds += 1
            exhausted, done = eng._round(it, exhausted)
        assert done, (
            f"engine did not terminate in 600 rounds: rank={rank} "
            f"outstanding={eng._outstanding} queued={eng._queued_pulls()} "
            f"held={len(eng._held)} stalls={eng.metrics.ssp_stalls}")
        eng.worker_logic.close(eng.client)
        eng.server_logic.close(eng.ps)
        # exactly-once per SENT id; requests may exceed sends because
        # the engine coalesces duplicate ids pulled within one flush
        # window (documented at on_pull_recv; one answer serves every
        # pending buffer)
        assert logic.answered == eng.metrics.pull_ids_sent, \
            (logic.answered, eng.metrics.pull_ids_sent)
        assert logic.answered <= logic.requested
        gids, vals = table.dump()
        return logic.pushed, dict(
            zip(gids.tolist(), [v.clone() for v in vals]))

    results = run_loopback(world, job, timeout=120)
    expected = sum(r[0] for r in results)
    merged = {}
    for _, d in results:
        merged.update(d)
    got = torch.stack([merged[i] for i in range(n_ids)])
    tol = 0.2 if wire == "bf16" and world > 1 else 1e-4
    assert torch.allclose(got, expected, atol=tol, rtol=0.05), \
        (got - expected).abs().max()
