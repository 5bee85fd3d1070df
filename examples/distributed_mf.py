"""One-process-per-GPU matrix factorization via ``transform_distributed``.

Launch (N = GPUs on the node; runs on CPU over gloo when no GPU):

    python -m torch.distributed.run --nnodes=1 --nproc-per-node N \\
        --master-addr 127.0.0.1 examples/distributed_mf.py

Each rank hosts one MF worker (its rating partition + user factors) and
one item shard; the engine's pipelined schedule overlaps the fused SGD
step with the RCCL/xGMI exchange.
"""
import os
import sys

import torch
import torch.distributed as dist

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from flink_parameter_server_amd import (PSConfig, ShardedTable,
                                        TableServerLogic,
                                        transform_distributed)
from flink_parameter_server_amd.models.mf import OnlineMFWorkerLogic, epoch_iterator
from flink_parameter_server_amd.parallel.partitioner import ModPartitioner
from flink_parameter_server_amd.utils import synth


def main():
    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    on_gpu = torch.cuda.is_available()
    backend = "nccl" if on_gpu else "gloo"
    dist.init_process_group(backend)
    dev = torch.device("cuda", int(os.environ.get("LOCAL_RANK", "0"))) \
        if on_gpu else torch.device("cpu")
    if on_gpu:
        torch.cuda.set_device(dev)

    n_users, n_items, k = 2000, 1500, 16
    # one planted teacher (same seed everywhere); each rank trains on
    # its slice of the ratings
    all_batches, _, _ = synth.planted_mf_ratings(
        n_users, n_items, k, 60_000, 2048, seed=5, device=dev)
    my_batches = all_batches[rank::world]

    part = ModPartitioner(n_items, world)
    table = ShardedTable(part, rank, k, dev, init=(-0.05, 0.05))
    wl = OnlineMFWorkerLogic(n_users, k, learning_rate=0.02,
                             range_min=-0.05, range_max=0.05,
                             n_items=n_items, device=dev)

    transform_distributed(
        epoch_iterator(my_batches, 6), wl,
        TableServerLogic(table, dump_on_close=False),
        PSConfig(num_ids=n_items, dim=k), part, device=dev)

    # reassemble the item table and report this rank's fit
    gids, vals = table.dump()
    full = torch.zeros(n_items, k, device=dev)
    full[gids.to(dev)] = vals
    dist.all_reduce(full)
    tot, n = 0.0, 0
    for b in my_batches:
        pred = (wl.user_table[b.user_rows.long()] *
                full[b.item_ids.long()]).sum(1)
        tot += float(((pred - b.ratings) ** 2).sum())
        n += len(b)
    print(f"rank {rank}/{world}: local MSE {tot / n:.4f} "
          f"({'GPU' if on_gpu else 'CPU'}, backend {backend})")
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
