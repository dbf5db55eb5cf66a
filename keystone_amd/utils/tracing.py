"""Structured timing/trace records (aux subsystem).

KeystoneML had no always-on tracer (Spark UI + the sampling Profiler fed
its AutoCacheRule).  Here the executor emits one record per operator
execution; enable JSONL output with ``enable_trace(path)`` or env
``KEYSTONE_TRACE=/path/trace.jsonl``.  GPU kernel-level profiling is done
externally with rocprofv3 (see profiles/).
"""

from __future__ import annotations

import json
import os
import time
from typing import Optional, TextIO

_trace_file: Optional[TextIO] = None
_enabled = False


def enable_trace(path: Optional[str] = None) -> None:
    global _trace_file, _enabled
    path = path or os.environ.get("KEYSTONE_TRACE")
    if path:
        _trace_file = open(path, "a")
    _enabled = True


def disable_trace() -> None:
    global _trace_file, _enabled
    if _trace_file:
        _trace_file.close()
    _trace_file = None
    _enabled = False


def trace_event(kind: str, **fields) -> None:
    if not _enabled:
        return
    rec = {"t": time.time(), "kind": kind, **fields}
    if _trace_file:
        _trace_file.write(json.dumps(rec) + "\n")
        _trace_file.flush()


def export_chrome_trace(jsonl_path: str, out_path: str) -> None:
    """Convert a JSONL op trace into chrome://tracing format."""
    events = []
    with open(jsonl_path) as f:
        for line in f:
            rec = json.loads(line)
            if rec.get("kind") != "op":
                continue
            events.append({
                "name": rec.get("name", "op"),
                "ph": "X",
                "ts": rec["t"] * 1e6,
                "dur": rec.get("seconds", 0) * 1e6,
                "pid": 0,
                "tid": 0,
                "args": {"bytes": rec.get("bytes", 0)},
            })
    with open(out_path, "w") as f:
        json.dump({"traceEvents": events}, f)


# auto-enable from env
if os.environ.get("KEYSTONE_TRACE"):
    enable_trace()
