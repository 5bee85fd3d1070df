"""Sketch-on-the-PS family (reference sketch/** — SURVEY §2.2 [LOW],
built defensively): count-min, counting Bloom, AMS tug-of-war, all
riding the ordinary '+' sharded table through the full engine."""
import pytest
import torch

from flink_parameter_server_amd import (PSConfig, PSEngine, ShardedTable,
                                        TableServerLogic)
from flink_parameter_server_amd.models.sketch import (
    AMSSketchWorkerLogic, BloomFilterWorkerLogic, CountMinWorkerLogic,
    merge_shard_dumps)
from flink_parameter_server_amd.parallel.partitioner import ModPartitioner
from flink_parameter_server_amd.parallel.transport import run_loopback


def zipf_stream(n, space, seed):
    g = torch.Generator().manual_seed(seed)
    u = torch.rand(n, generator=g)
    return (space * u.pow(3.0)).long().clamp_(0, space - 1)


def run_sketch(logic_cls, depth, width, batches, world=1, **kw):
    def job(rank, transport):
        wl = logic_cls(depth, width, **kw)
        part = ModPartitioner(wl.num_ids, world)
        table = ShardedTable(part, rank, 1, "cpu")
        eng = PSEngine(PSConfig(num_ids=wl.num_ids, dim=1), wl,
                       TableServerLogic(table, dump_on_close=False,
                                        pushes_unique=True),
                       transport, part)
        eng.run(batches[rank::world] if world > 1 else batches)
        return wl, table.dump()

    if world == 1:
        wl, dump = job(0, None)
        return wl, merge_shard_dumps([dump], depth, width)
    results = run_loopback(world, job)
    wl = results[0][0]
    return wl, merge_shard_dumps([r[1] for r in results], depth, width)


def test_count_min_one_sided_and_tight():
    depth, width, n = 4, 2048, 60000
    stream = zipf_stream(n, 5000, seed=3)
    batches = list(stream.chunk(20))
    wl, table = run_sketch(CountMinWorkerLogic, depth, width, batches)
    true = torch.bincount(stream, minlength=5000).float()
    q = torch.arange(5000)
    est = wl.estimate(table, q)
    # never under-estimates
    assert (est >= true - 1e-5).all()
    # eps*N bound with e/width ~ 0.0013: allow 3x slack
    assert float((est - true).max()) <= 3 * 2.718 / width * n


def test_bloom_no_false_negatives_low_fp():
    depth, width = 6, 1 << 15
    present = torch.unique(zipf_stream(4000, 10**6, seed=5))
    wl, table = run_sketch(BloomFilterWorkerLogic, depth, width,
                           [present])
    assert wl.contains(table, present).all()
    absent = torch.arange(10**6, 10**6 + 5000)
    fp = float(wl.contains(table, absent).float().mean())
    assert fp < 0.05, fp


def test_ams_f2_within_tolerance():
    depth, width, n = 11, 4096, 50000
    stream = zipf_stream(n, 3000, seed=7)
    wl, table = run_sketch(AMSSketchWorkerLogic, depth, width,
                           list(stream.chunk(10)))
    true_f2 = float((torch.bincount(stream).float() ** 2).sum())
    est = wl.f2_estimate(table)
    assert abs(est - true_f2) / true_f2 < 0.15, (est, true_f2)


@pytest.mark.parametrize("world", [2, 3])
def test_count_min_sharded_across_ranks(world):
    """The sketch cells shard over ranks like any parameter table; the
    merged dump must give the same one-sided estimates."""
    depth, width, n = 4, 1024, 30000
    stream = zipf_stream(n, 2000, seed=11)
    batches = list(stream.chunk(12))
    wl, table = run_sketch(CountMinWorkerLogic, depth, width, batches,
                           world=world)
    true = torch.bincount(stream, minlength=2000).float()
    est = wl.estimate(table, torch.arange(2000))
    assert (est >= true - 1e-5).all()
    assert float((est - true).max()) <= 6 * 2.718 / width * n


def test_run_count_min_entry():
    """algorithms.run_count_min entry: 2 workers x 2 shards through
    transform_local, merged dump queryable."""
    from flink_parameter_server_amd.algorithms import run_count_min

    stream = zipf_stream(20000, 1500, seed=9)
    logic, table = run_count_min(stream.tolist(), depth=4, width=1024,
                                 worker_parallelism=2)
    true = torch.bincount(stream, minlength=1500).float()
    est = logic.estimate(table, torch.arange(1500))
    assert (est >= true - 1e-5).all()
    assert float((est - true).max()) <= 6 * 2.718 / 1024 * 20000


def test_sketch_state_checkpoints_and_reshards(tmp_path):
    """A sketch is an ordinary parameter table: its state survives the
    crash-safe checkpoint path and re-shards 2 -> 3 on load."""
    from flink_parameter_server_amd.utils import checkpoint as ckpt

    depth, width = 4, 512
    stream = zipf_stream(15000, 900, seed=21)
    wl, merged = run_sketch(CountMinWorkerLogic, depth, width,
                            list(stream.chunk(6)), world=2)

    # rebuild the 2-shard tables from the merged dump and checkpoint them
    part2 = ModPartitioner(depth * width, 2)
    d = str(tmp_path / "round_00000001")
    for r in range(2):
        t = ShardedTable(part2, r, 1, "cpu")
        rows = torch.arange(depth * width)
        mine = part2.shard_of(rows) == r
        t.load(rows[mine], merged[mine])
        ckpt.save_shard(d, t)
    latest = ckpt.latest_checkpoint(str(tmp_path))
    assert latest == d

    # load onto 3 shards; the reassembled sketch answers identically
    part3 = ModPartitioner(depth * width, 3)
    dumps = []
    for r in range(3):
        t3 = ShardedTable(part3, r, 1, "cpu")
        ckpt.load_shard(latest, t3)
        dumps.append(t3.dump())
    merged3 = merge_shard_dumps(dumps, depth, width)
    q = torch.arange(900)
    assert torch.equal(wl.estimate(merged, q), wl.estimate(merged3, q))
