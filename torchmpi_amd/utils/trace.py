"""Per-collective chrome-trace dump (debug observability; SURVEY.md section
5: the reference had only timed benchmark loops — this framework adds a
`chrome://tracing` / Perfetto-compatible event dump behind an env flag).

Enable with  TORCHMPI_TRACE=/path/prefix  ; each rank writes
/path/prefix.rank<N>.json on tm.stop() (or trace_dump()).

Events record the host-side duration of each collective call (for sync ops:
enqueue + completion wait; for async ops: the submit, plus a separate event
for the handle wait).
"""
import json
import os
import time
from typing import Optional

_events = []
_t0 = None
_path: Optional[str] = None
_enabled = False


def _init():
    global _t0, _path, _enabled
    if _t0 is not None:
        return
    _t0 = time.perf_counter()
    p = os.environ.get("TORCHMPI_TRACE")
    if p:
        _path = f"{p}.rank{os.environ.get('RANK', '0')}.json"
        _enabled = True


def trace_enabled() -> bool:
    _init()
    return _enabled


class trace_op:
    """Context manager recording one collective call."""

    __slots__ = ("name", "nbytes", "begin")

    def __init__(self, name: str, nbytes: int = 0):
        self.name = name
        self.nbytes = nbytes

    def __enter__(self):
        self.begin = time.perf_counter()
        return self

    def __exit__(self, *exc):
        end = time.perf_counter()
        _events.append({
            "name": self.name,
            "ph": "X",
            "ts": (self.begin - _t0) * 1e6,
            "dur": (end - self.begin) * 1e6,
            "pid": int(os.environ.get("RANK", 0)),
            "tid": 0,
            "args": {"bytes": self.nbytes},
        })
        return False


def trace_dump():
    """Write the chrome trace file (called from tm.stop())."""
    if not _enabled or not _events:
        return
    with open(_path, "w") as f:
        json.dump({"traceEvents": _events}, f)
    _events.clear()
