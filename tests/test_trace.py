"""Chrome-trace debug flag (SURVEY.md section 5 observability)."""
import json
import os

import torch

from .util import run_distributed


def _trace_body(tm, prefix):
    t = torch.ones(1000)
    tm.allreduce(t)
    h = tm.allreduce(t, async_op=True)
    h.wait()
    tm.broadcast(t, root=0)
    # dump happens in tm.stop() (called by the harness)


def test_trace_writes_chrome_json(tmp_path):
    prefix = str(tmp_path / "trace")
    os.environ["TORCHMPI_TRACE"] = prefix
    try:
        run_distributed(_trace_body, world=2, args=(prefix,))
    finally:
        del os.environ["TORCHMPI_TRACE"]
    for rank in range(2):
        path = f"{prefix}.rank{rank}.json"
        assert os.path.exists(path), path
        data = json.load(open(path))
        names = [e["name"] for e in data["traceEvents"]]
        assert "allreduce" in names and "broadcast" in names
        assert all(e["dur"] >= 0 for e in data["traceEvents"])
