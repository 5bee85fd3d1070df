#!/usr/bin/env python3
"""A custom WorkerLogic/ServerLogic pair on the raw engine API — the
reference's canonical "pull a counter, push an increment" example
(SURVEY.md §4), here with 3 in-process ranks:

    python examples/custom_logic.py
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from flink_parameter_server_amd import (PSConfig, PSEngine,
                                        SimpleServerLogic, WorkerLogic)
from flink_parameter_server_amd.parallel.transport import run_loopback


class CountingLogic(WorkerLogic):
    def on_recv(self, ids, client):
        client.pull(ids)                      # async: answered next round

    def on_pull_recv(self, ids, values, client):
        client.push(ids, torch.ones(ids.numel(), 1))  # increment each


def main():
    world, n_ids = 3, 12

    def job(rank, transport):
        server = SimpleServerLogic(lambda i: torch.zeros(1),
                                   lambda old, d: old + d)
        eng = PSEngine(PSConfig(num_ids=n_ids, dim=1), CountingLogic(),
                       server, transport)
        data = [torch.arange(n_ids, dtype=torch.int64)] * (rank + 1)
        return eng.run(data)

    results = run_loopback(world, job)
    totals = {}
    for _, souts in results:
        totals.update({i: float(v) for i, v in souts})
    print("per-id totals:", totals)
    # ranks pulled each id 1+2+3 = 6 times -> 6 increments per id
    assert all(v == 6.0 for v in totals.values())


if __name__ == "__main__":
    main()
