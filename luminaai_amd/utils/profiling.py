"""Lightweight function-level profiling.

Keeps the reference's decorator API (@profile_function, profiling_context,
get_profiling_stats — /root/reference/Src/Main_Scripts/core/model.py:142-221)
but times GPU work with HIP events instead of host clocks, so decorated
forwards do not force device syncs in the hot loop. Deeper profiling belongs
to rocprofv3 (see profiles/)."""

from __future__ import annotations

import contextlib
import functools
import threading
import time
from collections import defaultdict
from typing import Dict

import torch

_lock = threading.Lock()
_stats: Dict[str, Dict] = defaultdict(
    lambda: {"calls": 0, "total_s": 0.0, "max_s": 0.0})
_ENABLED = False
_pending = []          # (name, start_event, end_event)


def enable_profiling(on: bool = True):
    global _ENABLED
    _ENABLED = on


def _record(name: str, dt: float):
    with _lock:
        s = _stats[name]
        s["calls"] += 1
        s["total_s"] += dt
        s["max_s"] = max(s["max_s"], dt)


def profile_function(name: str = None):
    """Decorator; no-op unless enable_profiling(True) was called."""
    def deco(fn):
        label = name or fn.__qualname__

        @functools.wraps(fn)
        def wrapper(*a, **kw):
            if not _ENABLED:
                return fn(*a, **kw)
            if torch.cuda.is_available():
                ev0 = torch.cuda.Event(enable_timing=True)
                ev1 = torch.cuda.Event(enable_timing=True)
                ev0.record()
                out = fn(*a, **kw)
                ev1.record()
                with _lock:
                    _pending.append((label, ev0, ev1))
                return out
            t0 = time.perf_counter()
            out = fn(*a, **kw)
            _record(label, time.perf_counter() - t0)
            return out
        return wrapper
    return deco


@contextlib.contextmanager
def profiling_context(name: str):
    if not _ENABLED:
        yield
        return
    if torch.cuda.is_available():
        ev0 = torch.cuda.Event(enable_timing=True)
        ev1 = torch.cuda.Event(enable_timing=True)
        ev0.record()
        yield
        ev1.record()
        with _lock:
            _pending.append((name, ev0, ev1))
    else:
        t0 = time.perf_counter()
        yield
        _record(name, time.perf_counter() - t0)


def _drain_pending():
    """Resolve queued HIP event pairs (one sync at read time, not per call)."""
    with _lock:
        pending, _pending[:] = list(_pending), []
    if pending:
        torch.cuda.synchronize()
        for name, ev0, ev1 in pending:
            _record(name, ev0.elapsed_time(ev1) / 1000.0)


def get_profiling_stats() -> Dict[str, Dict]:
    if torch.cuda.is_available():
        _drain_pending()
    with _lock:
        return {k: dict(v, mean_s=v["total_s"] / max(v["calls"], 1))
                for k, v in _stats.items()}


def reset_profiling_stats():
    with _lock:
        _stats.clear()
        _pending.clear()
