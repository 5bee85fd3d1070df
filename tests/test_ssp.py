"""SSP bounded-staleness clock (new capability — BASELINE configs[4]; the
reference is unbounded-async, which remains the default)."""
import torch

from flink_parameter_server_amd import (PSConfig, PSEngine, ShardedTable,
                                        TableServerLogic, WorkerLogic)
from flink_parameter_server_amd.parallel.partitioner import ModPartitioner
from flink_parameter_server_amd.parallel.transport import run_loopback


class ClockProbe(WorkerLogic):
    def __init__(self):
        self.consumed = 0
        self.max_lead = 0

    def on_recv(self, data, client):
        self.consumed += 1
        client.pull(torch.tensor([0], dtype=torch.int64))

    def on_pull_recv(self, ids, values, client):
        pass


class PacedIter:
    """Yields a batch every `every` rounds (None in between = dry source),
    n batches total."""

    def __init__(self, n, every):
        self.n, self.every, self.i, self.tick = n, every, 0, 0

    def __iter__(self):
        return self

    def __next__(self):
        if self.i >= self.n:
            raise StopIteration
        self.tick += 1
        if self.tick % self.every:
            return None
        self.i += 1
        return [0]


def _run(staleness, fast_n=30, slow_n=10, slow_every=3,
         feedback="fresh"):
    world = 2

    def job(rank, transport):
        part = ModPartitioner(4, world)
        table = ShardedTable(part, rank, 1, "cpu")
        logic = ClockProbe()
        eng = PSEngine(PSConfig(num_ids=4, dim=1, staleness=staleness,
                                feedback=feedback),
                       logic, TableServerLogic(table), transport, part)
        # track max clock lead over the other rank while both active
        leads = []
        orig_round = eng._round

        def tracked(it, exhausted):
            res = orig_round(it, exhausted)
            leads.append(eng._clock - eng._min_clock)
            return res

        eng._round = tracked
        it = (PacedIter(fast_n, 1) if rank == 0
              else PacedIter(slow_n, slow_every))
        eng.run(it)
        return logic.consumed, max(leads), eng.metrics.as_dict()

    return run_loopback(world, job)


def test_ssp_bounds_clock_skew():
    (fast_c, fast_lead, fast_m), (slow_c, _, _) = _run(staleness=2)
    assert fast_c == 30 and slow_c == 10
    assert fast_m["ssp_stalls"] > 0
    # while the slow worker was active, the fast one never ran more than
    # staleness+1 clocks ahead (the +1 is the in-round consume)
    assert fast_lead <= 2 + 1


def test_async_default_races_ahead():
    (fast_c, fast_lead, fast_m), _ = _run(staleness=None)
    assert fast_c == 30
    assert fast_m["ssp_stalls"] == 0
    assert fast_lead > 3  # unbounded async really does race ahead


def test_exhausted_worker_releases_the_clock():
    """Once the slow worker's data is exhausted its frozen clock must not
    stall the fast worker forever (engine excludes exhausted ranks from
    min-clock)."""
    (fast_c, _, _), (slow_c, _, _) = _run(staleness=1, fast_n=40,
                                          slow_n=3, slow_every=2)
    assert fast_c == 40 and slow_c == 3


def test_staleness_histogram_recorded():
    (fast_c, _, fast_m), _ = _run(staleness=2)
    hist = fast_m["staleness_hist"]
    assert sum(hist.values()) == fast_m["rounds"]
    assert max(hist) <= 2 + 1  # bounded lag distribution
    assert min(hist) >= 0


def test_ssp_bounds_clock_skew_merged_feedback():
    (fast_c, fast_lead, fast_m), (slow_c, _, _) = _run(staleness=2,
                                                       feedback="merged")
    assert fast_c == 30 and slow_c == 10
    assert fast_m["ssp_stalls"] > 0
    assert fast_lead <= 2 + 1
