"""In-process Python runtime tracing: GC pauses and dataloader wait time.

Parity target: ref xpu_timer python/py_tracing_manager.cc +
py_tracing_data.h — the reference injects a tracing lib that counts GC
events and dataloader batches into ring buffers; those counts feed the hang
metrics ("is the trainer stuck in GC / starved by input?"). Ours hooks
``gc.callbacks`` and wraps the dataloader iterator in pure Python and
exports the same style of Prometheus lines next to hiptimer's files, so the
agent's scrape endpoint serves them per rank.
"""

import gc
import os
import threading
import time
from typing import Dict, Iterable, Iterator, Optional

from dlrover_amd.common.log import logger


class GcTracer:
    """Counts collections and total stop-the-world pause per generation."""

    def __init__(self):
        self.counts: Dict[int, int] = {0: 0, 1: 0, 2: 0}
        self.pause_s: Dict[int, float] = {0: 0.0, 1: 0.0, 2: 0.0}
        self._t0: Optional[float] = None
        self._installed = False

    def _cb(self, phase: str, info: dict):
        gen = info.get("generation", 0)
        if phase == "start":
            self._t0 = time.perf_counter()
        elif phase == "stop" and self._t0 is not None:
            self.counts[gen] = self.counts.get(gen, 0) + 1
            self.pause_s[gen] = (
                self.pause_s.get(gen, 0.0) + time.perf_counter() - self._t0
            )
            self._t0 = None

    def start(self) -> "GcTracer":
        if not self._installed:
            gc.callbacks.append(self._cb)
            self._installed = True
        return self

    def stop(self):
        if self._installed:
            try:
                gc.callbacks.remove(self._cb)
            except ValueError:
                pass
            self._installed = False

    def metrics(self) -> Dict[str, float]:
        out: Dict[str, float] = {}
        for gen in sorted(self.counts):
            out[f'py_gc_collections{{gen="{gen}"}}'] = self.counts[gen]
            out[f'py_gc_pause_ms{{gen="{gen}"}}'] = round(
                self.pause_s[gen] * 1e3, 3
            )
        return out


class DataLoaderTracer:
    """Wrap any batch iterable: counts batches and the time the TRAINING
    LOOP spent waiting on next(batch) — the input-starvation signal."""

    def __init__(self, loader: Iterable):
        self._loader = loader
        self.batches = 0
        self.wait_s = 0.0

    def __iter__(self) -> Iterator:
        it = iter(self._loader)
        while True:
            t0 = time.perf_counter()
            try:
                batch = next(it)
            except StopIteration:
                return
            self.wait_s += time.perf_counter() - t0
            self.batches += 1
            yield batch

    def __len__(self):
        return len(self._loader)  # type: ignore[arg-type]

    def metrics(self) -> Dict[str, float]:
        return {
            "py_dataloader_batches": self.batches,
            "py_dataloader_wait_ms": round(self.wait_s * 1e3, 3),
        }


class PyRuntimeTracer:
    """Periodic exporter of GC + dataloader metrics in Prometheus text,
    written next to hiptimer's per-rank files so the agent endpoint serves
    them (pymetrics_<rank>.prom)."""

    def __init__(self, metrics_dir: str = "", interval: float = 5.0):
        self.metrics_dir = metrics_dir or os.getenv(
            "HIPTIMER_METRICS_DIR", "/tmp/hiptimer"
        )
        self.rank = int(os.getenv("RANK", "0"))
        self.interval = interval
        self.gc = GcTracer()
        self.loaders: list = []
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None

    def wrap_loader(self, loader: Iterable) -> DataLoaderTracer:
        t = DataLoaderTracer(loader)
        self.loaders.append(t)
        return t

    def start(self) -> "PyRuntimeTracer":
        os.makedirs(self.metrics_dir, exist_ok=True)
        self.gc.start()
        self._thread = threading.Thread(
            target=self._loop, name="py-runtime-tracer", daemon=True
        )
        self._thread.start()
        return self

    def stop(self):
        self._stop.set()
        if self._thread is not None:
            self._thread.join(timeout=3)
        self.gc.stop()
        self.dump()

    def metrics(self) -> Dict[str, float]:
        out = dict(self.gc.metrics())
        batches = sum(t.batches for t in self.loaders)
        wait = sum(t.wait_s for t in self.loaders)
        out["py_dataloader_batches"] = batches
        out["py_dataloader_wait_ms"] = round(wait * 1e3, 3)
        return out

    def dump(self):
        path = os.path.join(self.metrics_dir, f"pymetrics_{self.rank}.prom")
        tmp = path + ".tmp"
        try:
            with open(tmp, "w") as f:
                for k, v in self.metrics().items():
                    f.write(f"{k} {v}\n")
            os.replace(tmp, path)
        except OSError as e:  # pragma: no cover
            logger.warning("py tracer dump failed: %s", e)

    def _loop(self):
        while not self._stop.wait(self.interval):
            self.dump()
