"""Example-script bitrot guards: each reference-parity example must run end
to end with 2 local processes (reference: examples/ were the de-facto
integration tests)."""
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.mark.parametrize("script,expect", [
    ("examples/dp_training.py", "epoch 0"),
    ("examples/downpour.py", "downpour done"),
    ("examples/easgd.py", "easgd done"),
])
def test_example_runs(script, expect):
    env = dict(os.environ, TORCHMPI_EXAMPLE_FAST="1")
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, script), "--world", "2"],
        capture_output=True, text=True, timeout=600, cwd=REPO, env=env)
    assert out.returncode == 0, out.stderr[-2000:]
    assert expect in out.stdout, out.stdout[-2000:]
