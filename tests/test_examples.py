"""The runnable examples stay runnable (subprocess smoke; the
distributed example is covered by the gloo multi-process test of
transform_distributed instead)."""
import os
import subprocess
import sys

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

EXAMPLES = ["custom_logic.py", "online_mf.py", "pa_classifier.py",
            "sketches.py"]


@pytest.mark.parametrize("name", EXAMPLES)
@pytest.mark.timeout(300)
def test_example_runs(name):
    proc = subprocess.run(
        [sys.executable, os.path.join(ROOT, "examples", name)],
        capture_output=True, text=True, timeout=240)
    assert proc.returncode == 0, proc.stderr[-1500:]


@pytest.mark.timeout(300)
def test_distributed_example_torchrun_2ranks():
    import socket
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    proc = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", str(port),
         os.path.join(ROOT, "examples", "distributed_mf.py")],
        capture_output=True, text=True, timeout=240)
    assert proc.returncode == 0, proc.stderr[-1500:]
    assert "local MSE" in proc.stdout
