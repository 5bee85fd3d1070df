"""Checkpoint layout contract: per-shard (id, value)
record files, unordered, shard-count-independent on load."""
import torch

from flink_parameter_server_amd import ShardedTable
from flink_parameter_server_amd.parallel.partitioner import ModPartitioner
from flink_parameter_server_amd.utils import checkpoint as ckpt


def _fill(table):
    table.data.copy_(torch.arange(table.data.numel(),
                                  dtype=torch.float32).reshape(
        table.data.shape) + table.shard * 1000)


def test_roundtrip_same_shard_count(tmp_path):
    part = ModPartitioner(20, 2)
    for s in range(2):
        t = ShardedTable(part, s, 3, "cpu")
        _fill(t)
        ckpt.save_shard(str(tmp_path), t)
    for s in range(2):
        t2 = ShardedTable(part, s, 3, "cpu")
        ckpt.load_shard(str(tmp_path), t2)
        expect = ShardedTable(part, s, 3, "cpu")
        _fill(expect)
        assert torch.equal(t2.data, expect.data)


def test_reshard_2_to_3(tmp_path):
    """A checkpoint written with 2 shards loads onto 3 shards with every
    id keeping its value."""
    n_ids, dim = 17, 2
    part2 = ModPartitioner(n_ids, 2)
    truth = {}
    for s in range(2):
        t = ShardedTable(part2, s, dim, "cpu")
        gids, _ = t.dump()
        vals = torch.randn(t.rows, dim)
        t.data.copy_(vals)
        for i, gid in enumerate(gids.tolist()):
            truth[gid] = vals[i]
        ckpt.save_shard(str(tmp_path), t)

    part3 = ModPartitioner(n_ids, 3)
    seen = {}
    for s in range(3):
        t3 = ShardedTable(part3, s, dim, "cpu")
        ckpt.load_shard(str(tmp_path), t3)
        gids, vals = t3.dump()
        for i, gid in enumerate(gids.tolist()):
            seen[gid] = vals[i]
    assert set(seen) == set(truth)
    for gid in truth:
        assert torch.equal(seen[gid], truth[gid])


def test_adagrad_accum_roundtrip(tmp_path):
    part = ModPartitioner(10, 1)
    t = ShardedTable(part, 0, 2, "cpu", optimizer="adagrad", lr=0.1)
    t.data.normal_()
    t.accum.uniform_(0, 5)
    ckpt.save_shard(str(tmp_path), t)
    t2 = ShardedTable(part, 0, 2, "cpu", optimizer="adagrad", lr=0.1)
    ckpt.load_shard(str(tmp_path), t2)
    assert torch.equal(t.data, t2.data)
    assert torch.equal(t.accum, t2.accum)
