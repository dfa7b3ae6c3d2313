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


@pytest.mark.parametrize("script,args,expect", [
    ("benchmarks/llama_dp.py", ["--config", "debug", "--steps", "2",
                                "--warmup", "1"], '"bench": "llama_dp"'),
    ("benchmarks/allreduce_bw.py", ["--sizes", "4096,65536",
                                    "--dtypes", "f32"], '"results"'),
])
def test_benchmark_scripts_two_ranks(script, args, expect):
    """The scripts that will produce the 8-GPU config-#4 / headline
    evidence must keep running under torchrun at W=2 on CPU (bitrot
    guard; the GPU path shares all code above the transport)."""
    from .util import find_free_port
    port = find_free_port()
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", str(port), os.path.join(REPO, script)] + args,
        capture_output=True, text=True, timeout=600, cwd=REPO)
    assert out.returncode == 0, (out.stdout[-1500:], out.stderr[-1500:])
    assert expect in out.stdout, out.stdout[-2000:]
