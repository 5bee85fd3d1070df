"""Multi-process distributed tests over torch.distributed with the gloo
backend (world_size=2 on CPU) — the same TorchDistTransport code path that
runs over nccl/RCCL on the GPU node."""
import os
import tempfile

import pytest
import torch
import torch.multiprocessing as mp


def _job(rank, world, port, outdir):
    import torch.distributed as dist
    dist.init_process_group(
        "gloo", init_method=f"tcp://127.0.0.1:{port}",
        rank=rank, world_size=world)
    try:
        from flink_parameter_server_amd import (PSConfig, PSEngine,
                                                ShardedTable,
                                                TableServerLogic,
                                                WorkerLogic)
        from flink_parameter_server_amd.parallel.partitioner import \
            ModPartitioner
        from flink_parameter_server_amd.parallel.transport import \
            TorchDistTransport

        n_ids = 25

        class PushPull(WorkerLogic):
            def __init__(self):
                self.got = {}

            def on_recv(self, data, client):
                ids = torch.as_tensor(data, dtype=torch.int64)
                client.push(ids, torch.ones(ids.numel(), 1) * (rank + 1))
                client.pull(torch.unique(ids))

            def on_pull_recv(self, ids, values, client):
                for i, v in zip(ids.tolist(),
                                values.reshape(-1).tolist()):
                    self.got[i] = v

        part = ModPartitioner(n_ids, world)
        table = ShardedTable(part, rank, 1, "cpu")
        logic = PushPull()
        transport = TorchDistTransport(device=torch.device("cpu"))
        eng = PSEngine(PSConfig(num_ids=n_ids, dim=1), logic,
                       TableServerLogic(table), transport, part)
        # rank 0 sees ids 0..24 twice, rank 1 once
        data = [list(range(n_ids))] * (2 - rank)
        eng.run(data)
        gids, vals = table.dump()
        torch.save({"dump": dict(zip(gids.tolist(),
                                     vals.reshape(-1).tolist())),
                    "got": logic.got,
                    "metrics": eng.metrics.as_dict()},
                   os.path.join(outdir, f"rank{rank}.pt"))
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(180)
def test_gloo_world2_push_pull_routing():
    world = 2
    from conftest import free_port
    port = free_port()
    with tempfile.TemporaryDirectory() as d:
        mp.spawn(_job, args=(world, port, d), nprocs=world, join=True)
        r0 = torch.load(os.path.join(d, "rank0.pt"), weights_only=False)
        r1 = torch.load(os.path.join(d, "rank1.pt"), weights_only=False)
        merged = {}
        merged.update(r0["dump"])
        merged.update(r1["dump"])
        # totals: rank0 pushes 1.0 twice, rank1 pushes 2.0 once -> 4.0
        assert len(merged) == 25
        assert all(v == 4.0 for v in merged.values())
        # rank0's last pull (round 2) sees its second push: total 4.0;
        # rank1 pulled only in round 1, when totals were 1.0+2.0=3.0
        assert all(v == 4.0 for v in r0["got"].values())
        assert all(v == 3.0 for v in r1["got"].values())


def _soak_job(rank, world, port, feedback, outdir):
    """World-8 soak body: randomized pull/push traffic + SSP on, the
    exact process count the driver's 8-GPU scale run uses.
    Every pull must be answered exactly once, totals must match, and
    the observed staleness must respect the SSP bound (+1 round of
    control lag in pipelined mode)."""
    import torch.distributed as dist
    dist.init_process_group(
        "gloo", init_method=f"tcp://127.0.0.1:{port}",
        rank=rank, world_size=world)
    try:
        from flink_parameter_server_amd import (PSConfig, PSEngine,
                                                ShardedTable,
                                                TableServerLogic,
                                                WorkerLogic)
        from flink_parameter_server_amd.parallel.partitioner import \
            ModPartitioner
        from flink_parameter_server_amd.parallel.transport import \
            TorchDistTransport

        n_ids, dim, rounds, staleness = 97, 3, 30, 2

        class RandomTraffic(WorkerLogic):
            def __init__(self):
                self.g = torch.Generator().manual_seed(700 + rank)
                self.pushed = torch.zeros(n_ids, dim)
                self.pending = 0
                self.answered = 0

            def on_recv(self, data, client):
                k = int(torch.randint(1, 16, (1,), generator=self.g))
                ids = torch.unique(torch.randint(0, n_ids, (k,),
                                                 generator=self.g))
                deltas = torch.randn(ids.numel(), dim, generator=self.g)
                client.push(ids, deltas)
                self.pushed[ids] += deltas
                pull = torch.unique(torch.randint(0, n_ids, (k,),
                                                  generator=self.g))
                client.pull(pull, unique=True)
                self.pending += int(pull.numel())

            def on_pull_recv(self, ids, values, client):
                assert values.shape == (ids.numel(), dim)
                self.answered += int(ids.numel())
                self.pending -= int(ids.numel())

        part = ModPartitioner(n_ids, world)
        table = ShardedTable(part, rank, dim, "cpu")
        logic = RandomTraffic()
        transport = TorchDistTransport(device=torch.device("cpu"))
        eng = PSEngine(PSConfig(num_ids=n_ids, dim=dim,
                                feedback=feedback, staleness=staleness),
                       logic, TableServerLogic(table), transport, part)
        # rank-skewed workload so SSP actually bites: rank 0 gets 3x
        # the batches of rank 7
        n_batches = rounds * (world - rank + world) // world
        eng.run([[r] for r in range(n_batches)])
        assert logic.pending == 0, "unanswered pulls at drain"
        max_lag = max(eng.metrics.staleness_hist)
        assert max_lag <= staleness + 1, eng.metrics.staleness_hist
        gids, vals = table.dump()
        torch.save({"pushed": logic.pushed,
                    "dump": {i: v.clone() for i, v in
                             zip(gids.tolist(), vals)}},
                   os.path.join(outdir, f"rank{rank}.pt"))
    finally:
        dist.destroy_process_group()


@pytest.mark.slow
@pytest.mark.timeout(600)
@pytest.mark.parametrize("feedback", ["fresh", "merged", "pipelined"])
def test_gloo_world8_soak(feedback):
    world = 8
    from conftest import free_port
    port = free_port()
    with tempfile.TemporaryDirectory() as d:
        mp.spawn(_soak_job, args=(world, port, feedback, d),
                 nprocs=world, join=True)
        expected = torch.zeros(97, 3)
        merged = {}
        for r in range(world):
            blob = torch.load(os.path.join(d, f"rank{r}.pt"),
                              weights_only=False)
            expected += blob["pushed"]
            merged.update(blob["dump"])
        got = torch.stack([merged[i] for i in range(97)])
        assert torch.allclose(got, expected, atol=1e-4), \
            (got - expected).abs().max()


def _transform_dist_job(rank, world, port, outdir):
    import torch.distributed as dist
    dist.init_process_group(
        "gloo", init_method=f"tcp://127.0.0.1:{port}",
        rank=rank, world_size=world)
    try:
        from flink_parameter_server_amd import (PSConfig, ShardedTable,
                                                TableServerLogic,
                                                WorkerLogic)
        from flink_parameter_server_amd.parallel.partitioner import \
            ModPartitioner
        from flink_parameter_server_amd.transform import \
            transform_distributed

        n_ids = 17

        class Push(WorkerLogic):
            def on_recv(self, data, client):
                ids = torch.as_tensor(data, dtype=torch.int64)
                client.push(ids, torch.ones(ids.numel(), 1))

            def on_pull_recv(self, ids, values, client):
                pass

        part = ModPartitioner(n_ids, world)
        table = ShardedTable(part, rank, 1, "cpu")
        _, souts = transform_distributed(
            [list(range(n_ids))] * (rank + 1), Push(),
            TableServerLogic(table),
            PSConfig(num_ids=n_ids, dim=1), part)
        gids, vals = souts[0]
        torch.save(dict(zip(gids.tolist(), vals.reshape(-1).tolist())),
                   os.path.join(outdir, f"rank{rank}.pt"))
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(180)
def test_transform_distributed_gloo_world2():
    """The one-call-per-rank job entry wires the dist transport and the
    pipelined default; shard dumps carry the right totals."""
    world = 2
    from conftest import free_port
    port = free_port()
    with tempfile.TemporaryDirectory() as d:
        mp.spawn(_transform_dist_job, args=(world, port, d),
                 nprocs=world, join=True)
        merged = {}
        for r in range(world):
            merged.update(torch.load(os.path.join(d, f"rank{r}.pt"),
                                     weights_only=False))
        assert len(merged) == 17
        assert all(v == 3.0 for v in merged.values())  # 1+2 pushes
