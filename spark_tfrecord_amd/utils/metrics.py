"""Observability: per-stage wall-clock + byte counters and structured logs.

The reference has no metrics of its own (SURVEY.md §5 — Spark's UI covers
task timing externally); the MI355X engine replaces that with a lightweight
in-process report: every read/write records stage timings (encode, decode,
file IO, shuffle) and byte/row counts, queryable via `last_metrics()` and
logged at DEBUG level. Kernel-level profiling is rocprofv3's job, not ours
(profiles/ in the repo root holds captured summaries).
"""

from __future__ import annotations

import logging
import threading
import time
from typing import Dict, Optional

_log = logging.getLogger("spark_tfrecord_amd")
_tls = threading.local()


def get_logger() -> logging.Logger:
    return _log


class IOMetrics:
    """Counters for one logical read or write job."""

    def __init__(self, op: str):
        self.op = op
        self.rows = 0
        self.bytes = 0
        self.files = 0
        self.stages: Dict[str, float] = {}
        self.t0 = time.perf_counter()
        self.elapsed: Optional[float] = None

    def add(self, rows: int = 0, nbytes: int = 0, files: int = 0):
        self.rows += rows
        self.bytes += nbytes
        self.files += files

    def stage(self, name: str, seconds: float):
        self.stages[name] = self.stages.get(name, 0.0) + seconds

    def finish(self) -> "IOMetrics":
        self.elapsed = time.perf_counter() - self.t0
        _tls.last = self
        if _log.isEnabledFor(logging.DEBUG):
            _log.debug("%s", self.report())
        return self

    # -- reporting --------------------------------------------------------
    @property
    def rows_per_sec(self) -> float:
        e = self.elapsed or (time.perf_counter() - self.t0)
        return self.rows / e if e > 0 else 0.0

    @property
    def mb_per_sec(self) -> float:
        e = self.elapsed or (time.perf_counter() - self.t0)
        return self.bytes / e / 1e6 if e > 0 else 0.0

    def report(self) -> dict:
        return {
            "op": self.op,
            "rows": self.rows,
            "bytes": self.bytes,
            "files": self.files,
            "elapsed_s": round(self.elapsed or 0.0, 6),
            "rows_per_sec": round(self.rows_per_sec, 1),
            "mb_per_sec": round(self.mb_per_sec, 2),
            "stages_ms": {k: round(v * 1000, 3) for k, v in self.stages.items()},
        }

    def __repr__(self):
        return f"IOMetrics({self.report()})"


def last_metrics() -> Optional[IOMetrics]:
    """Metrics of the most recent read/write on this thread."""
    return getattr(_tls, "last", None)


class StageTimer:
    """`with StageTimer(metrics, "encode"):` — accumulates wall time."""

    def __init__(self, metrics: Optional[IOMetrics], name: str):
        self.m = metrics
        self.name = name

    def __enter__(self):
        self.t = time.perf_counter()
        return self

    def __exit__(self, *exc):
        dt = time.perf_counter() - self.t
        if self.m is not None:
            self.m.stage(self.name, dt)
        _trace_event(self.name, self.t, dt)
        return False


# ---------------------------------------------------------------------------
# Chrome-trace event log: TFREC_TRACE=/path/trace.json records every stage as
# a chrome://tracing / Perfetto "X" event. Kernel-level timelines come from
# rocprofv3; this covers the host orchestration above them.
# ---------------------------------------------------------------------------

_trace_path = None
_trace_lock = threading.Lock()
_trace_t0 = time.perf_counter()


def _trace_event(name: str, start: float, dur: float):
    import os

    global _trace_path
    path = os.environ.get("TFREC_TRACE")
    if not path:
        return
    ev = {"name": name, "ph": "X", "pid": os.getpid(),
          "tid": threading.get_ident() % 1_000_000,
          "ts": (start - _trace_t0) * 1e6, "dur": dur * 1e6}
    import json
    with _trace_lock:
        first = _trace_path != path
        if first:
            _trace_path = path
            with open(path, "w") as f:
                f.write("[\n" + json.dumps(ev))
        else:
            with open(path, "a") as f:
                f.write(",\n" + json.dumps(ev))
