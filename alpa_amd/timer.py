"""Named timers + activity tracer with chrome-trace export.

Mirrors the reference's ``alpa/timer.py`` (timers:76, Tracer:81) and the
pipeshard worker's per-instruction trace dump
(``pipeshard_executable.py:592`` dump_stage_execution_trace_internal):
timers optionally synchronize the device; the tracer logs (name, start,
end) events and writes the standard chrome://tracing JSON.
"""
from __future__ import annotations

import json
import time
from typing import Callable, Dict, List, Optional

import torch


class _Timer:

    def __init__(self, name: str, sync_fn: Optional[Callable] = None):
        self.name = name
        self.sync_fn = sync_fn
        self.costs: List[float] = []
        self._start: Optional[float] = None

    def start(self):
        if self.sync_fn:
            self.sync_fn()
        self._start = time.perf_counter()

    def stop(self):
        if self.sync_fn:
            self.sync_fn()
        assert self._start is not None, f"timer {self.name} not started"
        self.costs.append(time.perf_counter() - self._start)
        self._start = None

    def reset(self):
        self.costs = []
        self._start = None

    @property
    def elapsed(self) -> float:
        return sum(self.costs)

    def mean(self, warmup: int = 0) -> float:
        c = self.costs[warmup:]
        return sum(c) / len(c) if c else 0.0

    def __enter__(self):
        self.start()
        return self

    def __exit__(self, *a):
        self.stop()


def _cuda_sync():
    if torch.cuda.is_available():
        torch.cuda.synchronize()


class Timers:
    """Registry of named timers (reference timers:76)."""

    def __init__(self):
        self._timers: Dict[str, _Timer] = {}

    def __call__(self, name: str, sync: bool = False) -> _Timer:
        if name not in self._timers:
            self._timers[name] = _Timer(name,
                                        _cuda_sync if sync else None)
        return self._timers[name]

    def log(self, names=None, reset: bool = False) -> str:
        names = names or sorted(self._timers)
        lines = []
        for n in names:
            t = self._timers[n]
            lines.append(f"{n}: {t.elapsed * 1e3:.2f} ms "
                         f"({len(t.costs)} calls)")
            if reset:
                t.reset()
        return "\n".join(lines)


timers = Timers()


class Tracer:
    """Event log -> chrome://tracing JSON (reference Tracer:81)."""

    def __init__(self):
        self.events: List[dict] = []
        self._stack: List[tuple] = []

    def begin(self, name: str, cat: str = "op"):
        self._stack.append((name, cat, time.perf_counter()))

    def end(self):
        name, cat, t0 = self._stack.pop()
        self.events.append({"name": name, "cat": cat,
                            "ts": t0 * 1e6,
                            "dur": (time.perf_counter() - t0) * 1e6})

    def log_span(self, name: str, start: float, end: float, cat: str = "op",
                 tid: int = 0):
        self.events.append({"name": name, "cat": cat, "ts": start * 1e6,
                            "dur": (end - start) * 1e6, "tid": tid})

    def dump_chrome_trace(self, path: str, pid: int = 0):
        out = [{"name": e["name"], "cat": e.get("cat", "op"), "ph": "X",
                "ts": e["ts"], "dur": e["dur"], "pid": pid,
                "tid": e.get("tid", 0)} for e in self.events]
        with open(path, "w") as f:
            json.dump({"traceEvents": out}, f)

    def reset(self):
        self.events = []


tracer = Tracer()
