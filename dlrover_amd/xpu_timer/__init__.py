"""hiptimer: the MI355X profiling/hang-detection stack (xpu_timer rebuild).

C++ side: csrc/hiptimer.cc -> libhiptimer.so, LD_PRELOADed into workers.
Python side: env setup, Prometheus-text metrics parsing, and the agent-side
collector that feeds the master's hang diagnostician.
"""

import glob
import json
import os
import threading
import time
from typing import Dict, Optional

from dlrover_amd.common.log import logger

HANG_METRIC = "XPU_TIMER_COMMON_HANG"


def library_path() -> str:
    return os.path.join(os.path.dirname(os.path.abspath(__file__)), "libhiptimer.so")


def available() -> bool:
    return os.path.exists(library_path())


def preload_env(
    metrics_dir: str = "/tmp/hiptimer",
    hang_secs: float = 60.0,
    base_env: Optional[dict] = None,
) -> dict:
    """Env additions that enable hiptimer in child processes."""
    env = dict(base_env) if base_env else {}
    prev = env.get("LD_PRELOAD", os.environ.get("LD_PRELOAD", ""))
    lib = library_path()
    if not os.path.exists(lib):
        logger.warning("libhiptimer.so not built — hiptimer disabled")
        return env
    env["LD_PRELOAD"] = f"{lib}:{prev}" if prev else lib
    env["HIPTIMER_METRICS_DIR"] = metrics_dir
    env["HIPTIMER_HANG_SECS"] = str(hang_secs)
    return env


def parse_metrics_file(path: str) -> Dict[str, float]:
    """Parse the Prometheus-text dump (labels folded into the key)."""
    out: Dict[str, float] = {}
    try:
        with open(path) as f:
            for line in f:
                line = line.strip()
                if not line or line.startswith("#"):
                    continue
                key, _, val = line.rpartition(" ")
                try:
                    out[key] = float(val)
                except ValueError:
                    continue
    except FileNotFoundError:
        pass
    return out


class HiptimerCollector:
    """Agent-side collector (ref: diagnosis/datacollector/
    xpu_timer_metric_collector.py:28-80): reads every local rank's metrics
    file and reports the hang state to the master as DiagnosisReportData."""

    def __init__(self, metrics_dir: str, client=None, interval: float = 15.0):
        self.metrics_dir = metrics_dir
        self.interval = interval
        self._client = client
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None
        self._hang_started: float = 0.0

    def start(self):
        self._thread = threading.Thread(
            target=self._loop, name="hiptimer-collector", daemon=True
        )
        self._thread.start()
        return self

    def stop(self):
        self._stop.set()
        if self._thread is not None:
            self._thread.join(timeout=3)

    def snapshot(self) -> Dict[int, Dict[str, float]]:
        out = {}
        for path in glob.glob(os.path.join(self.metrics_dir, "hiptimer_*.prom")):
            try:
                rank = int(os.path.basename(path).split("_")[1].split(".")[0])
            except (IndexError, ValueError):
                continue
            out[rank] = parse_metrics_file(path)
        return out

    def node_hang_state(self) -> dict:
        """hang iff EVERY local rank reports hang (ref: training_hang.py:160:
        a hang is global, one busy rank means the node is not hung)."""
        snap = self.snapshot()
        if not snap:
            return {"hang": False, "since": 0.0}
        all_hang = all(m.get(HANG_METRIC, 0) >= 1 for m in snap.values())
        if all_hang:
            if self._hang_started == 0.0:
                self._hang_started = time.time()
            sinces = [m.get("hiptimer_hang_since_seconds", 0) for m in snap.values()]
            return {"hang": True, "since": self._hang_started, "ranks": len(snap),
                    "device_since_monotonic": min(sinces)}
        self._hang_started = 0.0
        return {"hang": False, "since": 0.0, "ranks": len(snap)}

    def _loop(self):
        while not self._stop.wait(self.interval):
            try:
                state = self.node_hang_state()
                if self._client is not None:
                    self._client.report_diagnosis_data("hang", json.dumps(state))
                if state.get("hang"):
                    logger.warning("hiptimer: node reports GPU hang: %s", state)
                    # name the communicators involved (rccl_env introspection)
                    from dlrover_amd.xpu_timer.rccl_env import format_comm_report

                    for rank, metrics in sorted(self.snapshot().items()):
                        report = format_comm_report(metrics)
                        if report:
                            logger.warning(
                                "hiptimer: rank %s communicators:\n%s",
                                rank, report,
                            )
            except Exception:  # noqa: BLE001
                logger.exception("hiptimer collector iteration failed")


class PrometheusExporter:
    """HTTP scrape endpoint serving the node's aggregated hiptimer metrics
    (ref: xpu_timer daemon LocalPrometheusService on :18889,
    server/server.cc:33 — ours is an in-agent stdlib HTTP server over the
    per-rank .prom files, same exposition format so any Prometheus scraper
    or the master's SimpleMetricMonitor consumes it unchanged)."""

    def __init__(self, metrics_dir: str, port: int = 18889, host: str = "0.0.0.0"):
        from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer

        metrics_dir_ = metrics_dir

        class _H(BaseHTTPRequestHandler):
            protocol_version = "HTTP/1.1"

            def log_message(self, fmt, *args):  # noqa: N802
                pass

            def do_GET(self):  # noqa: N802
                if self.path not in ("/metrics", "/"):
                    self.send_response(404)
                    self.send_header("Content-Length", "0")
                    self.end_headers()
                    return
                lines = []
                paths = sorted(
                    glob.glob(os.path.join(metrics_dir_, "hiptimer_*.prom"))
                ) + sorted(
                    glob.glob(os.path.join(metrics_dir_, "pymetrics_*.prom"))
                )
                for path in paths:
                    try:
                        rank = os.path.basename(path).split("_")[1].split(".")[0]
                        for raw in open(path):
                            raw = raw.strip()
                            if not raw:
                                continue
                            key, _, val = raw.rpartition(" ")
                            if "{" in key:
                                key = key.replace("{", '{rank="%s",' % rank, 1)
                            else:
                                key = '%s{rank="%s"}' % (key, rank)
                            lines.append(f"{key} {val}")
                    except OSError:
                        continue
                body = ("\n".join(lines) + "\n").encode()
                self.send_response(200)
                self.send_header("Content-Type", "text/plain; version=0.0.4")
                self.send_header("Content-Length", str(len(body)))
                self.end_headers()
                self.wfile.write(body)

        self._server = ThreadingHTTPServer((host, port), _H)
        self.port = self._server.server_address[1]
        self._thread = threading.Thread(
            target=self._server.serve_forever, name="hiptimer-prom", daemon=True
        )

    def start(self) -> "PrometheusExporter":
        self._thread.start()
        logger.info("hiptimer Prometheus endpoint on :%s/metrics", self.port)
        return self

    def stop(self):
        self._server.shutdown()
        self._server.server_close()
