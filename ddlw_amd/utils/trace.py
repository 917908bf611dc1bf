"""Chrome-trace (chrome://tracing) event emitter — the Horovod Timeline
equivalent (reference option at
``Part 1 .../03_model_training_distributed.py:407-409``, enabled by setting
``DDLW_TIMELINE=<path>`` like HOROVOD_TIMELINE)."""
from __future__ import annotations

import json
import os
import threading
import time
from typing import Optional


class ChromeTracer:
    def __init__(self, path: Optional[str] = None, pid: Optional[int] = None):
        self.path = path or os.environ.get("DDLW_TIMELINE")
        self.enabled = bool(self.path)
        self.pid = pid if pid is not None else os.getpid()
        self._events = []
        self._lock = threading.Lock()
        if self.enabled:
            # one file per rank (multi-process DP), dumped at exit — like
            # HOROVOD_TIMELINE, no explicit save needed from user code
            rank = os.environ.get("RANK")
            if rank is not None:
                self.path = f"{self.path}.rank{rank}"
            import atexit

            atexit.register(self.save)

    def event(self, name: str, cat: str, t0_us: float, dur_us: float, tid: int = 0, args: Optional[dict] = None):
        if not self.enabled:
            return
        with self._lock:
            self._events.append(
                {
                    "name": name,
                    "cat": cat,
                    "ph": "X",
                    "ts": t0_us,
                    "dur": dur_us,
                    "pid": self.pid,
                    "tid": tid,
                    "args": args or {},
                }
            )

    class _Span:
        def __init__(self, tracer, name, cat, tid):
            self.tracer, self.name, self.cat, self.tid = tracer, name, cat, tid

        def __enter__(self):
            self.t0 = time.time() * 1e6
            return self

        def __exit__(self, *exc):
            self.tracer.event(self.name, self.cat, self.t0, time.time() * 1e6 - self.t0, self.tid)

    def span(self, name: str, cat: str = "op", tid: int = 0) -> "ChromeTracer._Span":
        return ChromeTracer._Span(self, name, cat, tid)

    def save(self) -> None:
        if not self.enabled or not self.path:
            return
        with self._lock:
            with open(self.path, "w") as f:
                json.dump({"traceEvents": self._events}, f)


_shared: Optional[ChromeTracer] = None


def get_tracer() -> ChromeTracer:
    """Process-wide shared tracer: every subsystem (train loop spans,
    DistributedOptimizer collective events) appends to ONE timeline file —
    separate instances on the same path would overwrite each other. Rebuilt
    when DDLW_TIMELINE changes (tests toggle it per-case)."""
    global _shared
    env = os.environ.get("DDLW_TIMELINE")
    if _shared is None or _shared._env_path != env:
        _shared = ChromeTracer()
        _shared._env_path = env
    return _shared
