"""Profiler (parity with python/mxnet/profiler.py + src/profiler/ —
chrome://tracing JSON dump + aggregate per-op stats).

MI355X-native: op/kernel timestamps come from the HIP runtime via
torch.profiler (roctracer under the hood — the same source rocprofv3 uses);
scopes (Task/Frame/Event/Counter/Marker) are emitted into the same trace.
"""
from __future__ import annotations

import json
import time
import threading

import torch

__all__ = ["set_config", "start", "stop", "pause", "resume", "dump",
           "dumps", "Task", "Frame", "Event", "Counter", "Marker",
           "scope"]

_config = {"filename": "profile_output.json", "aggregate_stats": False}
_prof = None
_custom_events = []
_running = False
_agg_lock = threading.Lock()


def set_config(**kwargs):
    _config.update(kwargs)


def start():
    global _prof, _running
    acts = [torch.profiler.ProfilerActivity.CPU]
    if torch.cuda.is_available():
        acts.append(torch.profiler.ProfilerActivity.CUDA)
    _prof = torch.profiler.profile(activities=acts, record_shapes=False)
    _prof.__enter__()
    _running = True


def stop():
    global _prof, _running
    if _prof is not None:
        _prof.__exit__(None, None, None)
        _running = False


def pause():
    pass  # stream-ordered profiling has no engine pause; parity no-op


def resume():
    pass


def dump(finished=True):
    if _prof is not None:
        _prof.export_chrome_trace(_config["filename"])
        if _custom_events:
            _merge_custom_events(_config["filename"])


def dumps(reset=False):
    """Aggregate per-op stats table (reference MXAggregateProfileStatsPrint)."""
    if _prof is None:
        return ""
    return _prof.key_averages().table(sort_by="self_cpu_time_total",
                                      row_limit=50)


def _merge_custom_events(filename):
    try:
        with open(filename) as f:
            trace = json.load(f)
        events = trace.get("traceEvents", trace if isinstance(trace, list)
                           else [])
        events.extend(_custom_events)
        if isinstance(trace, dict):
            trace["traceEvents"] = events
        with open(filename, "w") as f:
            json.dump(trace, f)
    except Exception:
        pass


class _Scope:
    """Named duration scope emitted into the chrome trace."""

    _cat = "scope"

    def __init__(self, name):
        self.name = name
        self._t0 = None

    def start(self):
        self._t0 = time.perf_counter_ns()

    def stop(self):
        if self._t0 is None:
            return
        t1 = time.perf_counter_ns()
        _custom_events.append({
            "name": self.name, "cat": self._cat, "ph": "X",
            "ts": self._t0 / 1000.0, "dur": (t1 - self._t0) / 1000.0,
            "pid": 0, "tid": threading.get_ident() % 100000,
        })
        self._t0 = None

    def __enter__(self):
        self.start()
        return self

    def __exit__(self, *a):
        self.stop()


class Task(_Scope):
    _cat = "task"

    def __init__(self, name, domain=None):
        super().__init__(name)


class Frame(_Scope):
    _cat = "frame"

    def __init__(self, name, domain=None):
        super().__init__(name)


class Event(_Scope):
    _cat = "event"


class Counter:
    def __init__(self, name, domain=None, value=0):
        self.name = name
        self.value = value

    def set_value(self, value):
        self.value = value
        _custom_events.append({
            "name": self.name, "cat": "counter", "ph": "C",
            "ts": time.perf_counter_ns() / 1000.0, "pid": 0,
            "args": {self.name: value}})

    def increment(self, delta=1):
        self.set_value(self.value + delta)

    def decrement(self, delta=1):
        self.set_value(self.value - delta)


class Marker:
    def __init__(self, name, domain=None):
        self.name = name

    def mark(self, scope="process"):
        _custom_events.append({
            "name": self.name, "cat": "marker", "ph": "i",
            "ts": time.perf_counter_ns() / 1000.0, "pid": 0, "tid": 0,
            "s": "p" if scope == "process" else "t"})


def scope(name="<unk>"):
    return _Scope(name)
