"""Chrome-trace timeline (Horovod ``HOROVOD_TIMELINE`` equivalent,
``horovod/common/timeline.cc`` — SURVEY.md §5.1).

Python-side writer for engine-level phases; the native comm engine emits its
own per-bucket LAUNCH/RCCL/DONE events into the same file format.  Load the
output in chrome://tracing or Perfetto.  Enable with ``DDL_TIMELINE=path`` or
``TrainConfig.timeline``.
"""
from __future__ import annotations

import json
import threading
import time


class Timeline:
    def __init__(self, path: str, rank: int = 0):
        self.path = path
        self.rank = rank
        self._events = []
        self._lock = threading.Lock()
        self._t0 = time.perf_counter()
        self._open_spans: dict[str, float] = {}

    def _now_us(self) -> float:
        return (time.perf_counter() - self._t0) * 1e6

    def begin(self, name: str) -> None:
        with self._lock:
            self._open_spans[name] = self._now_us()

    def end(self, name: str) -> None:
        with self._lock:
            ts = self._open_spans.pop(name, None)
            if ts is None:
                return
            self._events.append({
                "name": name, "ph": "X", "ts": ts,
                "dur": self._now_us() - ts,
                "pid": self.rank, "tid": 0,
            })

    def mark(self, name: str) -> None:
        with self._lock:
            self._events.append({
                "name": name, "ph": "i", "ts": self._now_us(),
                "pid": self.rank, "tid": 0, "s": "t",
            })

    def save(self) -> None:
        with self._lock:
            events = list(self._events)
        with open(self.path, "w") as f:
            json.dump({"traceEvents": events, "displayTimeUnit": "ms"}, f)

    def __enter__(self):
        return self

    def __exit__(self, *exc):
        self.save()
