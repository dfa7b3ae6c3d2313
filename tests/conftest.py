import os
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: test requires an AMD GPU (MI355X)")


def pytest_sessionstart(session):
    """GPU-box preflight, run before ANY test:

    1. A plain-PyTorch canary (randn/matmul/sync). If the box's GPU stack is
       broken (stale driver state, bad lease), this faults HERE with an
       unambiguous marker line — distinguishing an environment fault from a
       torchmpi_amd kernel bug in the driver's log.
    2. Load the in-tree native extension and print its path, so the native
       .so load is observable even if a later test crashes the process.
    """
    try:
        import torch
    except Exception:
        return
    if not torch.cuda.is_available():
        return
    sys.stderr.write("[conftest] PLAIN-TORCH GPU CANARY: start\n")
    sys.stderr.flush()
    a = torch.randn(256, 256, device="cuda")
    b = (a @ a).sum()
    torch.cuda.synchronize()
    sys.stderr.write(
        f"[conftest] PLAIN-TORCH GPU CANARY: ok (sum={float(b):.3f}, "
        f"device={torch.cuda.get_device_name(0)})\n")
    import torchmpi_amd
    sys.stderr.write(
        f"[conftest] native extension: {torchmpi_amd._C.__file__}\n")
    # Exercise one of our kernels under synchronize so a broken extension
    # fails loudly at session start, not mid-suite.
    x = torch.randn(1 << 16, device="cuda", dtype=torch.bfloat16)
    y = torch.randn(1 << 16, device="cuda", dtype=torch.bfloat16)
    ref = x.float() + y.float()
    torchmpi_amd.ops.reduce_into(x, y, "sum")
    torch.cuda.synchronize()
    assert torch.allclose(x.float(), ref, rtol=1e-2, atol=1e-2), \
        "torchmpi_amd reduce kernel canary mismatch"
    sys.stderr.write("[conftest] NATIVE KERNEL CANARY: ok\n")
    sys.stderr.flush()
