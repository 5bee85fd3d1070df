#!/usr/bin/env python3
"""Online matrix factorization with the reference's entry-point shape
(psOnlineMF — SURVEY.md §3.3), runnable on CPU:

    python examples/online_mf.py

Trains rank-8 factors on a planted low-rank rating matrix with 2 workers
and 2 parameter shards (in-process lockstep ranks, the Flink local-mode
analog) and prints the fit.
"""
import math
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from flink_parameter_server_amd import ps_offline_mf, ps_online_mf


def main():
    n_users, n_items, k = 200, 150, 8
    g = torch.Generator().manual_seed(0)
    U = torch.randn(n_users, k, generator=g) / math.sqrt(k)
    V = torch.randn(n_items, k, generator=g) / math.sqrt(k)
    users = torch.randint(0, n_users, (60000,), generator=g)
    items = torch.randint(0, n_items, (60000,), generator=g)
    ratings = (U[users] * V[items]).sum(1)

    # one streaming pass (psOnlineMF) warms the factors; the offline
    # entry (psOfflineMF) then sweeps the buffered ratings in epochs
    user_out, _ = ps_online_mf(
        (users, items, ratings), n_users, n_items, num_factors=k,
        learning_rate=0.05, range_min=-0.1, range_max=0.1,
        worker_parallelism=2, ps_parallelism=2, batch_size=512)
    user_out, item_dumps = ps_offline_mf(
        (users, items, ratings), n_users, n_items, num_factors=k,
        learning_rate=0.05, range_min=-0.1, range_max=0.1,
        iterations=12, worker_parallelism=2, ps_parallelism=2,
        batch_size=512)

    Ufit = torch.zeros(n_users, k)
    for gids, vecs in user_out:
        Ufit[gids] = vecs
    Vfit = torch.zeros(n_items, k)
    for gids, vecs in item_dumps:
        if gids.numel():
            Vfit[gids] = vecs
    mse = float(((Ufit[users] * Vfit[items]).sum(1) - ratings)
                .pow(2).mean())
    print(f"rating variance ~{float(ratings.var()):.4f}  "
          f"trained MSE {mse:.4f}")
    assert mse < 0.3 * float(ratings.var())


if __name__ == "__main__":
    main()
