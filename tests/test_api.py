"""Semantics tests for the core API layer (reference parity: SURVEY.md
§2.1 — WorkerLogic / ServerLogic / client / SimplePSLogic contracts)."""
import torch

from flink_parameter_server_amd import (PSConfig, PSEngine,
                                        SimpleServerLogic, WorkerLogic,
                                        add_pull_limiter)
from flink_parameter_server_amd.api import RecordModeAdapter


class CountingLogic(WorkerLogic):
    """Pulls a counter id, pushes +1 for every record seen (the reference's
    canonical core test: final server dump equals expected counts)."""

    def __init__(self):
        self.answers = []

    def on_recv(self, data, client):
        ids = torch.as_tensor(data, dtype=torch.int64)
        client.push(ids, torch.ones(len(data), 1))
        client.pull(torch.unique(ids))

    def on_pull_recv(self, ids, values, client):
        self.answers.append((ids.clone(), values.clone()))

    def close(self, client):
        client.output("closed")


def make_simple_logic():
    return SimpleServerLogic(
        param_init=lambda i: torch.zeros(1),
        param_update=lambda old, delta: old + delta)


def test_counting_end_to_end():
    server = make_simple_logic()
    eng = PSEngine(PSConfig(num_ids=10, dim=1), CountingLogic(), server,
                   None)
    data = [[1, 2, 2, 3], [3, 3, 7]]
    wouts, souts = eng.run(data)
    # final dump = per-id record counts
    final = {i: v.item() for i, v in server.params.items()}
    assert final == {1: 1.0, 2: 2.0, 3: 3.0, 7: 1.0}
    assert "closed" in wouts
    # close() dumped every (id, param) pair
    dumped = {i: v.item() for i, v in souts}
    assert dumped == final


def test_pull_answers_reflect_pushes_applied_before():
    """Within one round the engine applies pushes before answering pulls
    (fresher answers), and the answer arrives the NEXT round (async)."""
    logic = CountingLogic()
    eng = PSEngine(PSConfig(num_ids=10, dim=1), logic, make_simple_logic(),
                   None)
    eng.run([[5, 5]])
    (ids, values), = logic.answers
    assert ids.tolist() == [5]
    assert values.reshape(-1).tolist() == [2.0]


class PullStorm(WorkerLogic):
    """Pulls many distinct ids at once to exercise the pull limiter."""

    def __init__(self, n):
        self.n = n
        self.got = 0
        self.max_in_round = 0

    def on_recv(self, data, client):
        client.pull(torch.arange(self.n, dtype=torch.int64))

    def on_pull_recv(self, ids, values, client):
        self.got += ids.numel()
        self.max_in_round = max(self.max_in_round, int(ids.numel()))


def test_pull_limiter_bounds_outstanding_and_delivers_all():
    logic = add_pull_limiter(PullStorm(100), 16)
    assert logic.pull_limit == 16
    eng = PSEngine(PSConfig(num_ids=100, dim=1), logic, make_simple_logic(),
                   None)
    eng.run([[0]])
    assert logic.got == 100          # every pull eventually answered
    assert logic.max_in_round <= 16  # never more than the limit in flight


class RecordLogic:
    """Reference-exact record-at-a-time shapes (one record / one
    (paramId, value) per callback)."""

    def __init__(self):
        self.seen = []

    def on_recv(self, record, client):
        client.push(torch.tensor([record]), torch.tensor([[1.0]]))
        client.pull(torch.tensor([record]))

    def on_pull_recv(self, param_id, value, client):
        assert isinstance(param_id, int)
        self.seen.append((param_id, float(value.reshape(-1)[0])))


def test_record_mode_adapter():
    inner = RecordLogic()
    server = make_simple_logic()
    eng = PSEngine(PSConfig(num_ids=10, dim=1), RecordModeAdapter(inner),
                   server, None)
    eng.run([[4, 4, 9]])
    assert server.params[4].item() == 2.0
    assert server.params[9].item() == 1.0
    assert dict(inner.seen)[4] == 2.0


def test_record_mode_adapter_under_pipelined():
    """The reference-exact record-at-a-time shapes (RecordModeAdapter)
    run unchanged under the pipelined schedule."""
    import torch

    from flink_parameter_server_amd import (PSConfig, PSEngine,
                                            SimpleServerLogic)
    from flink_parameter_server_amd.api import RecordModeAdapter

    class RecordLogic:
        def __init__(self):
            self.got = {}

        def on_recv(self, record, client):
            client.push(torch.tensor([record]),
                        torch.tensor([[1.0]]))
            client.pull(torch.tensor([record]))

        def on_pull_recv(self, param_id, value, client):
            self.got[param_id] = float(value)

    inner = RecordLogic()
    logic = RecordModeAdapter(inner)
    server = SimpleServerLogic(
        param_init=lambda i: torch.zeros(1),
        param_update=lambda p, d: p + d)
    eng = PSEngine(PSConfig(num_ids=8, dim=1, feedback="pipelined"),
                   logic, server, None)
    eng.run([[3, 5], [3]])
    # 3 pushed twice, 5 once; pulls answered per request
    assert float(server.params[3].sum()) == 2.0
    assert float(server.params[5].sum()) == 1.0
    assert set(inner.got) == {3, 5}


def test_config_and_api_validation_errors():
    """Bad knob values fail loudly at construction (the reference's
    plain-parameter style has no validation layer; ours does)."""
    import pytest as _pytest
    import torch

    from flink_parameter_server_amd import PSConfig, add_pull_limiter
    from flink_parameter_server_amd.api import WorkerLogic

    with _pytest.raises(ValueError, match="feedback"):
        PSConfig(feedback="bogus")
    with _pytest.raises(ValueError, match="wire_dtype"):
        PSConfig(wire_dtype="fp8")
    with _pytest.raises(ValueError, match="partitioner"):
        PSConfig(partitioner="hash")
    with _pytest.raises(ValueError, match="n_workers"):
        PSConfig(n_workers=0)

    class L(WorkerLogic):
        def on_recv(self, d, c):
            pass

        def on_pull_recv(self, i, v, c):
            pass

    with _pytest.raises(ValueError, match="positive"):
        add_pull_limiter(L(), 0)
    logic = add_pull_limiter(L(), 7)
    assert logic.pull_limit == 7

    from flink_parameter_server_amd import ShardedTable
    from flink_parameter_server_amd.parallel.partitioner import \
        ModPartitioner
    with _pytest.raises(ValueError, match="optimizer"):
        ShardedTable(ModPartitioner(4, 1), 0, 1, "cpu",
                     optimizer="adam")

    from flink_parameter_server_amd.models.mf import OnlineMFWorkerLogic
    with _pytest.raises(ValueError, match="aggregate"):
        OnlineMFWorkerLogic(4, 2, 0.1, aggregate="median")

    from flink_parameter_server_amd.models.sketch import CountMinWorkerLogic
    with _pytest.raises(ValueError, match="depth"):
        CountMinWorkerLogic(0, 16)


def test_metrics_schema_stable():
    """Metrics.as_dict carries the fields bench.py and downstream
    consumers read — pin the schema."""
    from flink_parameter_server_amd.parallel.runtime import Metrics

    d = Metrics().as_dict()
    for k in ("rounds", "samples", "pull_ids_sent", "push_ids_sent",
              "ssp_stalls", "bytes_sent", "held_rounds", "t_issue",
              "t_overlap_compute", "t_comm_wait", "t_server",
              "staleness_hist"):
        assert k in d, k
