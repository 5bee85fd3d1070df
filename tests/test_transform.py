"""The L1 entry points (FlinkParameterServer.transform parity)."""
import torch

from flink_parameter_server_amd import (PSConfig, ShardedTable,
                                        WorkerLogic)
from flink_parameter_server_amd.parallel.partitioner import ModPartitioner
from flink_parameter_server_amd.transform import (transform,
                                                  transform_local,
                                                  transform_simple,
                                                  transform_with_model_load)


class Inc(WorkerLogic):
    def on_recv(self, data, client):
        ids = torch.as_tensor(data, dtype=torch.int64)
        client.push(ids, torch.ones(ids.numel(), 1))

    def on_pull_recv(self, ids, values, client):
        pass


class PullOnce(WorkerLogic):
    def __init__(self):
        self.got = {}

    def on_recv(self, data, client):
        client.pull(torch.as_tensor(data, dtype=torch.int64))

    def on_pull_recv(self, ids, values, client):
        for i, v in zip(ids.tolist(), values.reshape(-1).tolist()):
            self.got[i] = v
        client.output(dict(self.got))


def test_transform_simple_counts():
    wouts, souts = transform_simple(
        [[1, 2, 2]], Inc(),
        param_init=lambda i: torch.zeros(1),
        param_update=lambda o, d: o + d,
        config=PSConfig(num_ids=5, dim=1))
    dump = {i: float(v) for i, v in souts}
    assert dump == {1: 1.0, 2: 2.0}


def test_transform_with_model_load_warm_start():
    part = ModPartitioner(10, 1)
    table = ShardedTable(part, 0, 1, "cpu")
    ids = torch.arange(10, dtype=torch.int64)
    values = torch.arange(10, dtype=torch.float32).reshape(-1, 1) * 2
    logic = PullOnce()
    wouts, souts = transform_with_model_load(
        (ids, values), [[3, 7]], logic, table,
        config=PSConfig(num_ids=10, dim=1))
    assert logic.got == {3: 6.0, 7: 14.0}
    # dump on close carries the loaded model through
    gids, vals = souts[-1]
    assert float(vals[3]) == 6.0


def test_transform_local_multi_rank():
    n_ids, world = 12, 3
    cfg = PSConfig(num_ids=n_ids, dim=1)

    def make_server(rank):
        part = ModPartitioner(n_ids, world)
        return __import__(
            "flink_parameter_server_amd").TableServerLogic(
            ShardedTable(part, rank, 1, "cpu"))

    results = transform_local(
        world,
        make_data=lambda r: [list(range(n_ids))],
        make_worker=lambda r: Inc(),
        make_server=make_server,
        config=cfg)
    merged = {}
    for _, souts in results:
        gids, vals = souts[-1]
        for i, v in zip(gids.tolist(), vals.reshape(-1).tolist()):
            merged[i] = v
    assert len(merged) == n_ids
    assert all(v == world for v in merged.values())


def test_transform_max_rounds_bounds_an_endless_stream():
    import itertools

    import torch

    from flink_parameter_server_amd import (PSConfig, SimpleServerLogic,
                                            WorkerLogic, transform)

    class Puller(WorkerLogic):
        def on_recv(self, data, client):
            client.push(data, torch.ones(1, 1))

        def on_pull_recv(self, ids, values, client):
            pass

    endless = (torch.tensor([0]) for _ in itertools.count())
    server = SimpleServerLogic(lambda i: torch.zeros(1),
                               lambda old, d: old + d)
    wout, souts = transform(endless, Puller(), server,
                            PSConfig(num_ids=1, dim=1), max_rounds=5)
    # 5 rounds consumed 5 batches; close() still dumped the table
    total = sum(float(v) for _, v in souts)
    assert 0 < total <= 5.0


def test_model_load_then_pipelined_train():
    """Warm start (strict two-phase transformWithModelLoad) composes
    with the pipelined schedule: loaded rows are visible to the first
    pulls and training proceeds from them."""
    import torch

    from flink_parameter_server_amd import (PSConfig, PSEngine,
                                            ShardedTable,
                                            TableServerLogic, WorkerLogic)
    from flink_parameter_server_amd.parallel.partitioner import \
        ModPartitioner
    from flink_parameter_server_amd.parallel.transport import run_loopback

    world, n_ids = 2, 12
    warm_ids = torch.arange(n_ids, dtype=torch.int64)
    warm_vals = torch.arange(n_ids, dtype=torch.float32).reshape(-1, 1) * 10

    class PullOnce(WorkerLogic):
        def __init__(self):
            self.seen = {}

        def on_recv(self, data, client):
            client.pull(warm_ids, unique=True)
            client.push(warm_ids, torch.ones(n_ids, 1))

        def on_pull_recv(self, ids, values, client):
            for i, v in zip(ids.tolist(), values.reshape(-1).tolist()):
                self.seen.setdefault(i, v)

    def job(rank, transport):
        part = ModPartitioner(n_ids, world)
        table = ShardedTable(part, rank, 1, "cpu")
        # two-phase warm start: every rank loads its own rows
        mine = part.shard_of(warm_ids) == rank
        table.load(warm_ids[mine], warm_vals[mine])
        transport.barrier()
        logic = PullOnce()
        eng = PSEngine(PSConfig(num_ids=n_ids, dim=1,
                                feedback="pipelined"), logic,
                       TableServerLogic(table, dump_on_close=False),
                       transport, part)
        eng.run([[0]])
        return logic.seen, table

    results = run_loopback(world, job)
    for seen, _ in results:
        # first pull sees warm values (possibly + concurrent +1 pushes)
        assert len(seen) == n_ids
        for i, v in seen.items():
            assert abs(v - 10 * i) <= 2.0, (i, v)
