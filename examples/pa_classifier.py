#!/usr/bin/env python3
"""Streaming passive-aggressive classification (the reference's
transformBinary shape — SURVEY.md §3.4), runnable on CPU:

    python examples/pa_classifier.py

Labeled sparse instances train the sharded weight vector; unlabeled ones
come back as predictions on the worker-output stream.
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from flink_parameter_server_amd import transform_binary


def main():
    n, n_feat, nnz = 3000, 100, 10
    g = torch.Generator().manual_seed(1)
    w_true = torch.randn(n_feat, generator=g)
    insts, truth = [], {}
    for j in range(n):
        fids = torch.randperm(n_feat, generator=g)[:nnz]
        vals = torch.randn(nnz, generator=g)
        y = 1.0 if float(w_true[fids] @ vals) > 0 else -1.0
        labeled = j < n - 300  # last 300 are streamed predictions
        insts.append((list(zip(fids.tolist(), vals.tolist())),
                      y if labeled else None))
        truth[j] = y

    preds, dumps = transform_binary(
        insts, n_feat, worker_parallelism=2, pa_method="PA-I", C=1.0,
        range_partitioning=True, batch_size=128)

    got = {}
    for ids, p in preds:
        got.update(zip(ids.tolist(), p.tolist()))
    acc = sum(got[i] == truth[i] for i in got) / len(got)
    print(f"predicted {len(got)} unlabeled instances, accuracy {acc:.2%}")
    assert acc > 0.7


if __name__ == "__main__":
    main()
