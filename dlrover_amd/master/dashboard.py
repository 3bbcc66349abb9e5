"""Master dashboard: job / node / metrics views over HTTP.

Parity target: ref dlrover/dashboard/app.py:34-260 (tornado JobInfoHandler,
NodesHandler, LogsHandler). Rebuilt on the stdlib HTTP server (no tornado in
this stack): JSON API + a minimal HTML index, enabled with
--enable_dashboard / JobMaster.start_dashboard().
"""

import html
import json
import threading
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer

from dlrover_amd.common.log import logger


class _Handler(BaseHTTPRequestHandler):
    protocol_version = "HTTP/1.1"

    def log_message(self, fmt, *args):
        pass

    def _send(self, code: int, body: bytes, ctype: str = "application/json"):
        self.send_response(code)
        self.send_header("Content-Type", ctype)
        self.send_header("Content-Length", str(len(body)))
        self.end_headers()
        self.wfile.write(body)

    def do_GET(self):
        master = self.server.master  # type: ignore[attr-defined]
        try:
            if self.path in ("/", "/index.html"):
                self._send(200, self._index(master).encode(), "text/html")
            elif self.path == "/api/job":
                self._send(200, json.dumps(self._job_info(master)).encode())
            elif self.path == "/api/nodes":
                self._send(200, json.dumps(self._nodes(master)).encode())
            elif self.path == "/api/metrics":
                self._send(200, json.dumps(self._metrics(master)).encode())
            elif self.path.startswith("/api/logs"):
                self._send(200, json.dumps(self._logs()).encode())
            elif self.path.startswith("/api/events"):
                self._send(200, json.dumps(self._events()).encode())
            else:
                self._send(404, b'{"error": "not found"}')
        except Exception as e:  # noqa: BLE001
            logger.exception("dashboard request failed")
            self._send(500, json.dumps({"error": repr(e)}).encode())

    @staticmethod
    def _job_info(master) -> dict:
        from dlrover_amd.common.constants import RendezvousName

        rdzv = master.rdzv_managers[RendezvousName.TRAINING]
        return {
            "stage": master.ctx.job_stage,
            "exit_reason": master.ctx.exit_reason,
            "rdzv_round": rdzv.rdzv_round,
            "world": rdzv.current_world(),
            "nodes_waiting": rdzv.num_nodes_waiting(),
            "global_step": master.perf_monitor.completed_global_step,
            "steps_per_sec": round(master.perf_monitor.running_speed(), 4),
            "model": (
                master.perf_monitor.model_info.__dict__
                if getattr(master.perf_monitor, "model_info", None)
                else None
            ),
        }

    @staticmethod
    def _nodes(master) -> list:
        out = []
        for node in master.ctx.job_nodes().values():
            out.append(
                {
                    "id": node.id,
                    "rank": node.rank_index,
                    "status": node.status,
                    "addr": node.host_ip,
                    "relaunches": node.relaunch_count,
                    "heartbeat_age_s": (
                        round(__import__("time").time() - node.heartbeat_time, 1)
                        if node.heartbeat_time
                        else None
                    ),
                    "exit_reason": node.exit_reason,
                }
            )
        return out

    @staticmethod
    def _logs(n: int = 100) -> dict:
        """Tail the master/agent log files (ref: dashboard LogsHandler :229)."""
        import glob
        import os

        out = {}
        log_dir = os.getenv("DLROVER_LOG_DIR", "")
        if log_dir:
            for path in sorted(glob.glob(os.path.join(log_dir, "dlrover_*.log")))[-8:]:
                try:
                    with open(path, errors="replace") as f:
                        out[os.path.basename(path)] = f.readlines()[-n:]
                except OSError:
                    continue
        return out

    @staticmethod
    def _events(n: int = 200) -> list:
        """Tail of the training-event JSONL spans (goodput/postmortem view;
        ref: training_event exporters + dashboard)."""
        import glob
        import os

        base = os.getenv("DLROVER_EVENT_DIR", "/tmp/dlrover_amd_events")
        out = []
        for path in sorted(glob.glob(os.path.join(base, "events_*.jsonl")))[-8:]:
            try:
                with open(path, errors="replace") as f:
                    for line in f.readlines()[-n:]:
                        try:
                            out.append(json.loads(line))
                        except ValueError:
                            continue
            except OSError:
                continue
        out.sort(key=lambda r: r.get("ts", 0))
        return out[-n:]

    @staticmethod
    def _metrics(master) -> dict:
        hangs = master.diagnosis_manager.data.latest_by_node("hang")
        return {
            "hang_reports": {nid: d.data_content for nid, d in hangs.items()},
        }

    def _index(self, master) -> str:
        job = self._job_info(master)
        nodes = self._nodes(master)
        rows = "".join(
            f"<tr><td>{n['id']}</td><td>{n['rank']}</td>"
            f"<td>{html.escape(str(n['status']))}</td>"
            f"<td>{n['relaunches']}</td><td>{n['heartbeat_age_s']}</td></tr>"
            for n in nodes
        )
        return f"""<html><head><title>dlrover_amd</title></head><body>
<h2>dlrover_amd job master</h2>
<p>stage: {job['stage']} | rdzv round: {job['rdzv_round']} |
world: {job['world']} | step: {job['global_step']} |
steps/s: {job['steps_per_sec']}</p>
<table border=1 cellpadding=4>
<tr><th>id</th><th>rank</th><th>status</th><th>relaunches</th><th>hb age s</th></tr>
{rows}</table>
<p>APIs: <a href=/api/job>/api/job</a> <a href=/api/nodes>/api/nodes</a>
<a href=/api/metrics>/api/metrics</a> <a href=/api/events>/api/events</a>
<a href=/api/logs>/api/logs</a></p></body></html>"""


class Dashboard:
    def __init__(self, master, port: int = 0, host: str = "0.0.0.0"):
        self._server = ThreadingHTTPServer((host, port), _Handler)
        self._server.master = master  # type: ignore[attr-defined]
        self.port = self._server.server_address[1]
        self._thread = threading.Thread(
            target=self._server.serve_forever, name="dashboard", daemon=True
        )

    def start(self) -> "Dashboard":
        self._thread.start()
        logger.info("dashboard on port %s", self.port)
        return self

    def stop(self):
        self._server.shutdown()
        self._server.server_close()
