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
    port = 29511
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
