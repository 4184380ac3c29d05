"""Lightweight structured tracing.

The reference's observability is a pool progress meter plus wall-clock
t_train/t_test fields.  This adds a structured span log the
GPU engine can afford everywhere: each span is one JSON line
{name, t_start, dur_s, rank, ...meta} appended to the file named by the
FLAKE16_TRACE environment variable (or set_trace_file()).  Disabled (zero
cost) when no file is configured.
"""

import json
import os
import threading
import time
from contextlib import contextmanager

_lock = threading.Lock()
_file = None
_explicit = False


def set_trace_file(path):
    global _file, _explicit
    _file = path
    _explicit = True


def _trace_path():
    if _explicit:
        return _file
    return os.environ.get("FLAKE16_TRACE") or None


def trace_event(name, t_start, dur_s, **meta):
    path = _trace_path()
    if not path:
        return
    rec = {"name": name, "t_start": t_start, "dur_s": dur_s}
    try:
        import torch.distributed as dist
        if dist.is_initialized():
            rec["rank"] = dist.get_rank()
    except Exception:
        pass
    rec.update(meta)
    line = json.dumps(rec)
    with _lock:
        with open(path, "a") as fd:
            fd.write(line + "\n")


@contextmanager
def trace_span(name, **meta):
    t0 = time.time()
    try:
        yield
    finally:
        trace_event(name, t0, time.time() - t0, **meta)
