"""Sketches on the parameter server (reference ``sketch/**`` analog).

Count-min frequency estimation over a zipfian id stream, sharded across
2 loopback ranks — the sketch cells are ordinary parameter ids, so the
whole engine (bucketing, alltoallv, scatter-apply, dump/reshard)
carries them unchanged.  Run:  python examples/sketches.py
"""
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from flink_parameter_server_amd.algorithms import run_count_min
from flink_parameter_server_amd.models.sketch import (AMSSketchWorkerLogic,
                                                      merge_shard_dumps)
from flink_parameter_server_amd import (PSConfig, PSEngine, ShardedTable,
                                        TableServerLogic)
from flink_parameter_server_amd.parallel.partitioner import ModPartitioner
from flink_parameter_server_amd.parallel.transport import run_loopback


def main():
    g = torch.Generator().manual_seed(7)
    stream = (5000 * torch.rand(80_000, generator=g).pow(3.0)).long()

    # --- count-min via the algorithms entry (2 workers x 2 shards) ---
    logic, table = run_count_min(stream.tolist(), depth=4, width=4096,
                                 worker_parallelism=2)
    true = torch.bincount(stream, minlength=5000).float()
    est = logic.estimate(table, torch.arange(5000))
    head = torch.topk(true, 5).indices
    print("count-min (never under-estimates):")
    for i in head.tolist():
        print(f"  id {i}: true {int(true[i])}  est {int(est[i])}")
    assert (est >= true - 1e-5).all()

    # --- AMS tug-of-war F2, assembled by hand over the engine ---
    depth, width = 11, 4096
    world = 2
    batches = list(stream.chunk(16))

    def job(rank, transport):
        wl = AMSSketchWorkerLogic(depth, width)
        part = ModPartitioner(wl.num_ids, world)
        t = ShardedTable(part, rank, 1, "cpu")
        eng = PSEngine(PSConfig(num_ids=wl.num_ids, dim=1), wl,
                       TableServerLogic(t, dump_on_close=False,
                                        pushes_unique=True),
                       transport, part)
        eng.run(batches[rank::world])
        return wl, t.dump()

    results = run_loopback(world, job)
    full = merge_shard_dumps([r[1] for r in results], depth, width)
    est_f2 = results[0][0].f2_estimate(full)
    true_f2 = float((true ** 2).sum())
    print(f"AMS F2: true {true_f2:.3e}  est {est_f2:.3e}  "
          f"err {(abs(est_f2 - true_f2) / true_f2):.1%}")


if __name__ == "__main__":
    main()
