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
_pending = []          # (name, start_event, end_event, tid)
_TRACE = False
_trace_events = []     # chrome-trace slices: (name, ts_us, dur_us, tid)
_TRACE_CAP = 200_000
_epoch_cpu = 0.0
_epoch_ev = None       # HIP event anchoring device timestamps


def enable_profiling(on: bool = True, trace: bool = False):
    """trace=True additionally collects per-call slices for
    export_chrome_trace (bounded to 200k events)."""
    global _ENABLED, _TRACE, _epoch_cpu, _epoch_ev
    _ENABLED = on
    _TRACE = on and trace
    if _TRACE:
        _epoch_cpu = time.perf_counter()
        if torch.cuda.is_available():
            _epoch_ev = torch.cuda.Event(enable_timing=True)
            _epoch_ev.record()


def _record(name: str, dt: float, ts_s: float = None):
    with _lock:
        s = _stats[name]
        s["calls"] += 1
        s["total_s"] += dt
        s["max_s"] = max(s["max_s"], dt)
        if _TRACE and ts_s is not None and len(_trace_events) < _TRACE_CAP:
            _trace_events.append(
                (name, (ts_s - _epoch_cpu) * 1e6, dt * 1e6,
                 threading.get_ident()))


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
            _record(label, time.perf_counter() - t0, ts_s=t0)
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
        _record(name, time.perf_counter() - t0, ts_s=t0)


def _drain_pending():
    """Resolve queued HIP event pairs (one sync at read time, not per call)."""
    with _lock:
        pending, _pending[:] = list(_pending), []
    if pending:
        torch.cuda.synchronize()
        for name, ev0, ev1 in pending:
            dt = ev0.elapsed_time(ev1) / 1000.0
            _record(name, dt)
            if _TRACE and _epoch_ev is not None \
                    and len(_trace_events) < _TRACE_CAP:
                with _lock:
                    _trace_events.append(
                        (name, _epoch_ev.elapsed_time(ev0) * 1e3, dt * 1e6,
                         0))


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
        _trace_events.clear()


def attach_module_profilers(model, prefix: str = "layer"):
    """Per-module forward timing (the reference decorated every module
    forward, reference model.py:142-221; here it is opt-in so the hot loop
    pays nothing by default). Returns hook handles; remove() them to
    detach. Works with enable_profiling(True[, trace=True]) — each module
    forward becomes a stat entry and (with trace) a chrome-trace slice."""
    handles = []

    def make_pre(name):
        def pre(module, args):
            if not _ENABLED:
                return
            if torch.cuda.is_available():
                ev = torch.cuda.Event(enable_timing=True)
                ev.record()
                module.__dict__["_prof_start"] = ev
            else:
                module.__dict__["_prof_start"] = time.perf_counter()
        return pre

    def make_post(name):
        def post(module, args, output):
            start = module.__dict__.pop("_prof_start", None)
            if not _ENABLED or start is None:
                return
            if torch.cuda.is_available():
                ev = torch.cuda.Event(enable_timing=True)
                ev.record()
                with _lock:
                    _pending.append((name, start, ev))
            else:
                t0 = start
                _record(name, time.perf_counter() - t0, ts_s=t0)
        return post

    for i, layer in enumerate(getattr(model, "layers", [])):
        name = f"{prefix}{i}"
        handles.append(layer.register_forward_pre_hook(make_pre(name)))
        handles.append(layer.register_forward_hook(make_post(name)))
    return handles


def export_chrome_trace(path: str) -> int:
    """Write collected slices as a chrome://tracing / Perfetto JSON file
    (ROADMAP: trace export for the decorator API). Requires
    enable_profiling(True, trace=True). Returns the event count."""
    import json
    if torch.cuda.is_available():
        _drain_pending()
    with _lock:
        events = [{"name": n, "ph": "X", "ts": ts, "dur": dur,
                   "pid": 0, "tid": tid, "cat": "lumina"}
                  for n, ts, dur, tid in _trace_events]
    with open(path, "w") as f:
        json.dump({"traceEvents": events,
                   "displayTimeUnit": "ms"}, f)
    return len(events)
