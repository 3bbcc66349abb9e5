"""Structured training events: JSON spans for goodput accounting and
postmortems.

Parity target: ref dlrover/python/training_event/ (emitter.py:37-341
DurationSpan/Process, exporter.py:51-229 async file exporter, predefined
agent/master/trainer events; design docs/design/training-event.md). One
module instead of a package: emitter + async exporter + the predefined event
factories the agent/master call.
"""

import atexit
import json
import os
import queue
import threading
import time
import uuid
from typing import Any, Dict, Optional



class AsyncExporter:
    """Background JSONL writer (ref: exporter.py:51 AsyncExporter)."""

    _instance = None
    _lock = threading.Lock()

    def __init__(self, path: Optional[str] = None):
        base = os.getenv("DLROVER_EVENT_DIR", "/tmp/dlrover_amd_events")
        os.makedirs(base, exist_ok=True)
        role = os.getenv("DLROVER_ROLE", "proc")
        self.path = path or os.path.join(base, f"events_{role}_{os.getpid()}.jsonl")
        self._q: "queue.Queue[Optional[dict]]" = queue.Queue(maxsize=10000)
        self._thread = threading.Thread(target=self._run, daemon=True, name="events")
        self._thread.start()
        atexit.register(self.close)

    @classmethod
    def get(cls) -> "AsyncExporter":
        if cls._instance is None:
            with cls._lock:
                if cls._instance is None:
                    cls._instance = cls()
        return cls._instance

    def export(self, record: dict):
        try:
            self._q.put_nowait(record)
        except queue.Full:
            pass  # drop rather than stall training

    def _run(self):
        with open(self.path, "a", buffering=1) as f:
            while True:
                rec = self._q.get()
                if rec is None:
                    return
                try:
                    f.write(json.dumps(rec) + "\n")
                except (TypeError, OSError):
                    pass

    def close(self):
        try:
            self._q.put_nowait(None)
            self._thread.join(timeout=2)
        except (queue.Full, RuntimeError):
            pass


class EventEmitter:
    """Named event source; instant events + duration spans."""

    def __init__(self, target: str, exporter: Optional[AsyncExporter] = None):
        self.target = target
        self._exporter = exporter or AsyncExporter.get()

    def instant(self, name: str, content: Optional[Dict[str, Any]] = None):
        self._exporter.export(
            {
                "event_id": uuid.uuid4().hex[:12],
                "ts": time.time(),
                "target": self.target,
                "name": name,
                "type": "instant",
                "content": content or {},
            }
        )

    def duration(self, name: str, content: Optional[Dict[str, Any]] = None
                 ) -> "DurationSpan":
        return DurationSpan(self, name, content or {})


class DurationSpan:
    """begin/end (or success/fail) span (ref: emitter.py DurationSpan)."""

    def __init__(self, emitter: EventEmitter, name: str, content: Dict[str, Any]):
        self.emitter = emitter
        self.name = name
        self.content = content
        self.span_id = uuid.uuid4().hex[:12]
        self._begin_ts: Optional[float] = None

    def begin(self) -> "DurationSpan":
        self._begin_ts = time.time()
        self.emitter._exporter.export(
            {
                "event_id": self.span_id,
                "ts": self._begin_ts,
                "target": self.emitter.target,
                "name": self.name,
                "type": "begin",
                "content": self.content,
            }
        )
        return self

    def end(self, success: bool = True, extra: Optional[dict] = None):
        now = time.time()
        rec = {
            "event_id": self.span_id,
            "ts": now,
            "target": self.emitter.target,
            "name": self.name,
            "type": "end" if success else "fail",
            "duration_s": (now - self._begin_ts) if self._begin_ts else None,
            "content": {**self.content, **(extra or {})},
        }
        self.emitter._exporter.export(rec)

    def __enter__(self):
        return self.begin()

    def __exit__(self, exc_type, exc, tb):
        self.end(success=exc_type is None,
                 extra={"error": repr(exc)} if exc else None)
        return False


# predefined event sources (ref: training_event/predefined/)
def agent_events() -> EventEmitter:
    return EventEmitter("dlrover-agent")


def master_events() -> EventEmitter:
    return EventEmitter("dlrover-master")


def trainer_events() -> EventEmitter:
    return EventEmitter("dlrover-trainer")
