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


@pytest.mark.timeout(600)
@settings(max_examples=20, deadline=None, derandomize=True)
@given(
    world=st.integers(1, 3),
    feedback=st.sampled_from(["fresh", "merged", "pipelined"]),
    pull_limit=st.one_of(st.none(), st.integers(4, 40)),
    n_items=st.integers(6, 60),
    n_users=st.integers(5, 40),
    n_batches=st.integers(2, 10),
    batch_sz=st.integers(8, 200),
    seed=st.integers(0, 10**6),
)
def test_mf_pullbuffer_reassembly_stays_finite(
        world, feedback, pull_limit, n_items, n_users, n_batches,
        batch_sz, seed):
    """The MODEL-layer net for the round-2 answer-misrouting bug class:
    the real MF worker (PullBuffer reassembly, overlapping pulls across
    in-flight rounds, answers split under pull limits) must train every
    batch on COMPLETE, correctly-addressed values — tables stay finite
    and bounded under every schedule/limit/shape combination.  (The
    engine-level property test above drives raw ids and never touches
    PullBuffer, which is exactly where the corruption lived.)"""
    from flink_parameter_server_amd.models.mf import (MFBatch,
                                                      OnlineMFWorkerLogic)

    dim = 4
    g = torch.Generator().manual_seed(seed)
    per_rank_batches = []
    for r in range(world):
        bs = []
        for _ in range(n_batches):
            users = torch.randint(0, n_users, (batch_sz,), generator=g)
            items = torch.randint(0, n_items, (batch_sz,), generator=g)
            ratings = torch.randn(batch_sz, generator=g) * 0.3
            bs.append(MFBatch(users, items, ratings))
        per_rank_batches.append(bs)

    def job(rank, transport):
        part = ModPartitioner(n_items, world)
        table = ShardedTable(part, rank, dim, "cpu", init=(-0.1, 0.1))
        wl = OnlineMFWorkerLogic(n_users, dim, learning_rate=0.01,
                                 range_min=-0.1, range_max=0.1,
                                 pull_limit=pull_limit)
        eng = PSEngine(PSConfig(num_ids=n_items, dim=dim,
                                feedback=feedback), wl,
                       TableServerLogic(table), transport, part)
        it = iter(per_rank_batches[rank] * 2)
        exhausted = done = False
        rounds = 0
        # livelock backstop sized from the workload: a tiny pull limit
        # legitimately splits each batch's pull into ceil(ids/limit)
        # chunks at ~3 rounds per chunk under the pipelined schedule
        # (budget is re-checked before the overlapped delivery frees it)
        chunks = -(-min(n_items, batch_sz) // (pull_limit or n_items))
        bound = 100 + 2 * n_batches * (chunks + 4) * 3
        while not done and rounds < bound:
            rounds += 1
            exhausted, done = eng._round(it, exhausted)
        assert done, f"no termination: rank={rank} rounds={rounds}"
        assert not wl._pending, "batches left untrained at drain"
        assert torch.isfinite(wl.user_table).all()
        assert float(wl.user_table.abs().max()) < 50.0
        assert torch.isfinite(table.data).all()
        assert float(table.data.abs().max()) < 50.0
        return True

    assert all(run_loopback(world, job, timeout=120))


@pytest.mark.timeout(300)
@settings(max_examples=30, deadline=None, derandomize=True)
@given(
    num_ids=st.integers(1, 200),
    n_write=st.integers(1, 6),
    n_read=st.integers(1, 6),
    dim=st.integers(1, 9),
    adagrad=st.booleans(),
    kind=st.sampled_from(["mod", "range"]),
    seed=st.integers(0, 10**6),
)
def test_checkpoint_roundtrip_any_shard_counts(tmp_path_factory, num_ids,
                                               n_write, n_read, dim,
                                               adagrad, kind, seed):
    """Checkpoint layout contract: a dump written with M shards loads
    exactly onto N shards for any (M, N, partitioner, dim, optimizer)."""
    from flink_parameter_server_amd.parallel.partitioner import \
        make_partitioner
    from flink_parameter_server_amd.utils import checkpoint as ckpt

    tmp = tmp_path_factory.mktemp("ck")
    g = torch.Generator().manual_seed(seed)
    ref = torch.randn(num_ids, dim, generator=g)
    ref_acc = torch.rand(num_ids, dim, generator=g) if adagrad else None

    opt = dict(optimizer="adagrad", lr=0.1) if adagrad else {}
    pw = make_partitioner(kind, num_ids, n_write)
    d = str(tmp / "round_00000001")
    for s_ in range(n_write):
        t = ShardedTable(pw, s_, dim, "cpu", **opt)
        rows = torch.arange(num_ids)
        mine = pw.shard_of(rows) == s_
        if bool(mine.any()):
            t.load(rows[mine], ref[mine])
            if adagrad:
                t.accum[pw.local_row(rows[mine])] = ref_acc[mine]
        ckpt.save_shard(d, t)
    latest = ckpt.latest_checkpoint(str(tmp))
    assert latest == d

    pr = make_partitioner(kind, num_ids, n_read)
    got = torch.zeros(num_ids, dim)
    got_acc = torch.zeros(num_ids, dim)
    for s_ in range(n_read):
        t = ShardedTable(pr, s_, dim, "cpu", **opt)
        ckpt.load_shard(latest, t)
        gids, vals = t.dump()
        got[gids] = vals
        if adagrad:
            got_acc[gids] = t.accum
    assert torch.equal(got, ref)
    if adagrad:
        assert torch.equal(got_acc, ref_acc)


@pytest.mark.timeout(120)
@settings(max_examples=60, deadline=None, derandomize=True)
@given(
    num_ids=st.integers(1, 10**9),
    n_shards=st.integers(1, 16),
    kind=st.sampled_from(["mod", "range"]),
    seed=st.integers(0, 10**6),
)
def test_partitioner_inverse_invariants(num_ids, n_shards, kind, seed):
    """shard_of/local_row/global_id are mutually inverse and rows are
    contiguous per shard for any (space, shard count)."""
    from flink_parameter_server_amd.parallel.partitioner import \
        make_partitioner

    part = make_partitioner(kind, num_ids, n_shards)
    g = torch.Generator().manual_seed(seed)
    ids = torch.randint(0, num_ids, (256,), generator=g)
    sh = part.shard_of(ids)
    lr = part.local_row(ids)
    assert ((sh >= 0) & (sh < n_shards)).all()
    for s_ in range(n_shards):
        m = sh == s_
        if bool(m.any()):
            assert int(lr[m].max()) < part.shard_rows(s_)
            back = part.global_id(s_, lr[m])
            assert torch.equal(back, ids[m])
    assert sum(part.shard_rows(s_) for s_ in range(n_shards)) == num_ids
