import torch

from flink_parameter_server_amd.parallel.partitioner import (
    ModPartitioner, RangePartitioner, bucket_by_shard)


def test_mod_partitioner_bijective():
    p = ModPartitioner(100, 4)
    ids = torch.arange(100, dtype=torch.int64)
    shard, row = p.shard_of(ids), p.local_row(ids)
    back = torch.empty_like(ids)
    for s in range(4):
        m = shard == s
        back[m] = p.global_id(s, row[m])
    assert torch.equal(back, ids)
    assert sum(p.shard_rows(s) for s in range(4)) == 100
    for s in range(4):
        assert int(row[shard == s].max()) < p.shard_rows(s)


def test_range_partitioner_bijective_uneven():
    p = RangePartitioner(103, 4)
    ids = torch.arange(103, dtype=torch.int64)
    shard, row = p.shard_of(ids), p.local_row(ids)
    assert int(shard.max()) == 3
    back = torch.empty_like(ids)
    for s in range(4):
        m = shard == s
        if m.any():
            back[m] = p.global_id(s, row[m])
    assert torch.equal(back, ids)
    assert sum(p.shard_rows(s) for s in range(4)) == 103
    # contiguity: shard ids are non-decreasing over sorted ids
    assert torch.all(shard[1:] >= shard[:-1])


def test_bucket_by_shard_groups_and_counts():
    p = ModPartitioner(1000, 3)
    ids = torch.tensor([5, 3, 9, 1, 6, 2], dtype=torch.int64)
    order, counts, shards = bucket_by_shard(ids, p)
    grouped = shards[order]
    assert torch.all(grouped[1:] >= grouped[:-1])
    assert int(counts.sum()) == 6
    for s in range(3):
        assert int(counts[s]) == int((shards == s).sum())
