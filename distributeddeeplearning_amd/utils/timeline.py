"""Chrome-trace timeline (Horovod ``HOROVOD_TIMELINE`` equivalent,
``horovod/common/timeline.cc`` — SURVEY.md §5.1).

Python-side writer for engine-level phases; the native comm engine emits its
own per-bucket LAUNCH/RCCL/DONE events into the same file format.  Load the
output in chrome://tracing or Perfetto.  Enable with ``DDL_TIMELINE=path`` or
``TrainConfig.timeline``.

Two span families:

* host spans (``begin``/``end``, tid 0) — wall-clock of the Python loop;
* device spans (``begin_device``/``end_device``, tid 1) — hipEvent pairs
  recorded on the current stream, so an async phase (backward kernels +
  overlapped RCCL) is measured by DEVICE time, not by when the host happened
  to return (host spans under/over-state overlap).
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
        # device-event machinery (lazy: CPU-only runs never touch torch.cuda)
        self._dev_anchor = None
        self._dev_anchor_host_us = 0.0
        self._dev_open: dict[str, object] = {}
        self._dev_spans: list[tuple[str, object, object]] = []

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

    # -- device spans -------------------------------------------------------

    def _new_event(self):
        import torch
        return torch.cuda.Event(enable_timing=True)

    def begin_device(self, name: str) -> None:
        import torch
        if not torch.cuda.is_available():
            return
        if self._dev_anchor is None:
            self._dev_anchor = self._new_event()
            self._dev_anchor.record()
            self._dev_anchor_host_us = self._now_us()
        ev = self._new_event()
        ev.record()
        self._dev_open[name] = ev

    def end_device(self, name: str) -> None:
        ev0 = self._dev_open.pop(name, None)
        if ev0 is None:
            return
        ev1 = self._new_event()
        ev1.record()
        self._dev_spans.append((name, ev0, ev1))

    def _drain_device_spans(self) -> None:
        """Resolve recorded hipEvent pairs into trace events (synchronises)."""
        if not self._dev_spans:
            return
        import torch
        torch.cuda.synchronize()
        with self._lock:
            for name, e0, e1 in self._dev_spans:
                ts = (self._dev_anchor_host_us
                      + self._dev_anchor.elapsed_time(e0) * 1000.0)
                self._events.append({
                    "name": f"dev/{name}", "ph": "X", "ts": ts,
                    "dur": e0.elapsed_time(e1) * 1000.0,
                    "pid": self.rank, "tid": 1,
                })
            self._dev_spans.clear()

    def save(self) -> None:
        self._drain_device_spans()
        with self._lock:
            events = list(self._events)
        with open(self.path, "w") as f:
            json.dump({"traceEvents": events, "displayTimeUnit": "ms"}, f)

    def __enter__(self):
        return self

    def __exit__(self, *exc):
        self.save()
