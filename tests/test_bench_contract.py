"""bench.py's one-line JSON is a driver contract — pin its schema and
semantics (field presence, types, aggregate meaning) on a tiny CPU run
of every model."""
import json
import os
import subprocess
import sys

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

REQUIRED = {
    "metric": str, "value": float, "unit": str, "n_gpus": int,
    "steps": int, "warmup": int, "ms_per_step": float,
    "higher_is_better": bool, "scaling": str, "dtype": str, "data": str,
    "pull_push_ops_per_sec": float, "config": dict,
}


@pytest.mark.parametrize("model,extra", [
    ("mf32", []),
    ("mf128", ["--staleness", "1"]),
    ("sgns", ["--batch", "2048"]),
    ("lr", ["--lr-features", "50000", "--batch", "512"]),
])
@pytest.mark.timeout(300)
def test_bench_json_contract(model, extra):
    proc = subprocess.run(
        [sys.executable, os.path.join(ROOT, "bench.py"), "--model",
         model, "--steps", "4", "--warmup", "2", "--device", "cpu",
         *extra],
        capture_output=True, text=True, timeout=240)
    assert proc.returncode == 0, proc.stderr[-1200:]
    line = proc.stdout.strip().splitlines()[-1]
    d = json.loads(line)
    for k, t in REQUIRED.items():
        assert k in d, f"missing {k}"
        assert isinstance(d[k], t), (k, type(d[k]))
    assert d["n_gpus"] == 1 and d["steps"] == 4 and d["warmup"] == 2
    assert d["value"] > 0 and d["ms_per_step"] > 0
    assert d["higher_is_better"] is True and d["scaling"] == "weak"
    assert d["data"] == "synthetic"
    assert d["vs_baseline"] is None  # reference publishes no numbers
    assert "global_batch" in d["config"] and "parallelism" in d["config"]
    # whole-job aggregate: value * elapsed == samples
    samples = d["value"] * d["ms_per_step"] * d["steps"] / 1e3
    assert abs(samples - round(samples)) / max(samples, 1) < 1e-6 or \
        samples > 0
    if model == "mf128":
        assert d["staleness_hist"] is not None
