"""Local Brain service: the cluster-level resource-optimization endpoint.

Parity target: ref dlrover/go/brain (gRPC Optimize service with per-stage
optimizers and a datastore). The reference marks Brain optional/phase-2;
this is a WORKING single-binary Python stand-in speaking the same two-call
surface the client uses (/brain.Brain/Optimize, /brain.Brain/ReportMetrics,
JSON-over-bytes), with per-stage plan logic and an in-memory metrics store —
enough to run `optimizeMode: cluster` jobs end-to-end and to integration-
test BrainClient against a live peer.

Run standalone:  python -m dlrover_amd.master.brain_service --port 50051
"""

import argparse
import json
import threading
import time
from collections import defaultdict
from typing import Dict, List, Optional

from dlrover_amd.common.log import logger


class BrainOptimizer:
    """Per-stage resource planning over reported job metrics (ref:
    pkg/optimizer/implementation/optprocessor — create/initial/running)."""

    def __init__(self):
        self._metrics: Dict[str, List[dict]] = defaultdict(list)
        # job -> (node_count, per_node_speed) at the last grow suggestion,
        # used to verify the growth actually scaled before growing again
        self._last_grow: Dict[str, tuple] = {}
        self._lock = threading.Lock()

    def report(self, job: str, metrics: dict):
        with self._lock:
            hist = self._metrics[job]
            hist.append({**metrics, "_ts": time.time()})
            del hist[:-256]  # bounded history

    def optimize(self, job: str, stage: str, stats: Optional[dict]) -> dict:
        with self._lock:
            hist = list(self._metrics[job])
        stats = stats or {}
        if stage in ("create", "job_stage_create"):
            # no history: start from the requested/declared size
            return {
                "node_count": int(stats.get("request_nodes", 1)),
                "comment": "brain:create-default",
            }
        if not hist:
            return {"node_count": int(stats.get("current_nodes", 1)),
                    "comment": "brain:no-history"}
        # running stage: scale on throughput-per-node trend — if the last
        # doubling of nodes raised per-node speed (sub-linear loss < 15%),
        # suggest growing toward max; if per-node speed collapsed, shrink.
        cur = int(stats.get("current_nodes", 1)) or 1
        speeds = [m.get("steps_per_sec", 0.0) for m in hist[-8:]]
        avg = sum(speeds) / max(len(speeds), 1)
        per_node = avg / cur
        plan = {"node_count": cur, "comment": "brain:hold"}
        max_nodes = int(stats.get("max_nodes", cur))
        prev = self._last_grow.get(job)
        # grow only while scaling stays near-linear: per-node speed after the
        # previous doubling must be within 15% of what it was before it
        scaled_ok = (
            prev is None or cur <= prev[0] or per_node >= 0.85 * prev[1]
        )
        if per_node > 0 and cur < max_nodes and scaled_ok:
            self._last_grow[job] = (cur, per_node)
            plan = {"node_count": min(cur * 2, max_nodes),
                    "comment": "brain:grow"}
        elif avg == 0 and cur > 1:
            plan = {"node_count": max(1, cur // 2), "comment": "brain:shrink"}
        return plan


class BrainService:
    """gRPC server exposing the optimizer on the same generic byte methods
    BrainClient calls (no compiled proto needed on either side)."""

    def __init__(self, port: int = 0, host: str = "0.0.0.0"):
        import grpc

        self.optimizer = BrainOptimizer()
        self._server = grpc.server(
            __import__("concurrent.futures", fromlist=["ThreadPoolExecutor"])
            .ThreadPoolExecutor(max_workers=8)
        )

        svc = self

        class Handler(grpc.GenericRpcHandler):
            def service(self, handler_call_details):
                method = handler_call_details.method
                if method == "/brain.Brain/Optimize":
                    return grpc.unary_unary_rpc_method_handler(
                        svc._handle_optimize,
                        request_deserializer=lambda b: b,
                        response_serializer=lambda b: b,
                    )
                if method == "/brain.Brain/ReportMetrics":
                    return grpc.unary_unary_rpc_method_handler(
                        svc._handle_report,
                        request_deserializer=lambda b: b,
                        response_serializer=lambda b: b,
                    )
                return None

        self._server.add_generic_rpc_handlers((Handler(),))
        self.port = self._server.add_insecure_port(f"{host}:{port}")

    def _handle_optimize(self, request: bytes, context) -> bytes:
        req = json.loads(request or b"{}")
        plan = self.optimizer.optimize(
            req.get("job", ""), req.get("stage", ""), req.get("stats")
        )
        return json.dumps(plan).encode()

    def _handle_report(self, request: bytes, context) -> bytes:
        req = json.loads(request or b"{}")
        job = req.pop("job", "")
        self.optimizer.report(job, req)
        return b"{}"

    def start(self) -> "BrainService":
        self._server.start()
        logger.info("brain service on port %s", self.port)
        return self

    def stop(self):
        self._server.stop(grace=1)


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--port", type=int, default=50051)
    args = p.parse_args()
    svc = BrainService(port=args.port).start()
    try:
        while True:
            time.sleep(3600)
    except KeyboardInterrupt:
        svc.stop()


if __name__ == "__main__":
    main()
