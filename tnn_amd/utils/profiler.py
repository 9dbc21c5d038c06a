"""Event profiler (reference include/profiling/profiler.hpp:13,132).

Same event model as the reference — {type, start, end, name, source} with
cross-rank merge — plus a Chrome-trace exporter (replaces the reference's
``visualizers/visualize_profiler.py`` Gantt chart; chrome://tracing reads
the output directly). Kernel-level profiling is rocprofv3's job.
"""

from __future__ import annotations

import json
import threading
import time
from dataclasses import dataclass
from enum import Enum
from typing import Dict, List


class EventType(Enum):
    COMPUTE = "compute"
    COMMUNICATION = "communication"
    OTHER = "other"


@dataclass
class Event:
    type: EventType
    start: float
    end: float
    name: str
    source: str = ""


class Profiler:
    def __init__(self, source: str = ""):
        self.source = source
        self.events: List[Event] = []
        self.enabled = False
        self._lock = threading.Lock()

    def start(self):
        self.enabled = True

    def stop(self):
        self.enabled = False

    def clear(self):
        with self._lock:
            self.events.clear()

    def record(self, type_: EventType, name: str, start: float, end: float):
        if not self.enabled:
            return
        with self._lock:
            self.events.append(Event(type_, start, end, name, self.source))

    class _Span:
        def __init__(self, prof, type_, name):
            self.prof, self.type_, self.name = prof, type_, name

        def __enter__(self):
            self.t0 = time.perf_counter()
            return self

        def __exit__(self, *a):
            self.prof.record(self.type_, self.name, self.t0, time.perf_counter())

    def span(self, type_: EventType, name: str) -> "_Span":
        return self._Span(self, type_, name)

    def merge(self, other: "Profiler", clock_offset: float = 0.0):
        """Merge another rank's events, adjusting for clock skew
        (reference profiler.hpp:52-63)."""
        with self._lock:
            for e in other.events:
                self.events.append(Event(e.type, e.start + clock_offset,
                                         e.end + clock_offset, e.name,
                                         e.source or other.source))

    # -- serialization (travels over the control plane between ranks) --------
    def to_dict(self):
        return {"source": self.source,
                "events": [{"type": e.type.value, "start": e.start, "end": e.end,
                            "name": e.name, "source": e.source}
                           for e in self.events]}

    @classmethod
    def from_dict(cls, d) -> "Profiler":
        p = cls(d.get("source", ""))
        for e in d.get("events", []):
            p.events.append(Event(EventType(e["type"]), e["start"], e["end"],
                                  e["name"], e.get("source", "")))
        return p

    def export_chrome_trace(self, path: str):
        events = []
        for e in self.events:
            events.append({
                "name": e.name, "cat": e.type.value, "ph": "X",
                "ts": e.start * 1e6, "dur": (e.end - e.start) * 1e6,
                "pid": e.source or self.source or "tnn", "tid": e.type.value,
            })
        with open(path, "w") as f:
            json.dump({"traceEvents": events}, f)

    def summary(self) -> Dict[str, float]:
        total: Dict[str, float] = {}
        for e in self.events:
            total[e.name] = total.get(e.name, 0.0) + (e.end - e.start)
        return total


GlobalProfiler = Profiler("global")
