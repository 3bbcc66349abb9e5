"""Brain service client: cluster-level resource optimization RPC.

Parity target: ref dlrover/python/brain/client.py:1-185 (gRPC client to the
Go Brain Optimize service, MySQL-backed). The Brain service itself is out of
scope for the MI355X build (reference marks it optional phase-2 —
SURVEY.md §2.2); this client keeps the integration point: it speaks the same
two-call surface (optimize / report_metrics) over gRPC when a
DLROVER_BRAIN_ADDR endpoint exists, and degrades to the in-master
LocalResourceOptimizer otherwise.
"""

import json
import os
from typing import Optional

from dlrover_amd.common.log import logger


class BrainClient:
    def __init__(self, addr: str = ""):
        self.addr = addr or os.getenv("DLROVER_BRAIN_ADDR", "")
        self._stub = None
        if self.addr:
            try:
                import grpc

                channel = grpc.insecure_channel(self.addr)
                # generic bytes-in/bytes-out methods (no compiled proto needed)
                self._optimize = channel.unary_unary(
                    "/brain.Brain/Optimize",
                    request_serializer=lambda b: b,
                    response_deserializer=lambda b: b,
                )
                self._report = channel.unary_unary(
                    "/brain.Brain/ReportMetrics",
                    request_serializer=lambda b: b,
                    response_deserializer=lambda b: b,
                )
                self._stub = channel
            except Exception:  # noqa: BLE001
                logger.warning("brain endpoint %s unreachable", self.addr)
                self._stub = None

    @property
    def available(self) -> bool:
        return self._stub is not None

    def get_optimization_plan(self, job_name: str, stage: str,
                              stats: Optional[dict] = None) -> Optional[dict]:
        """Ask the Brain for a resource plan; None when unavailable (callers
        fall back to LocalResourceOptimizer)."""
        if not self.available:
            return None
        try:
            req = json.dumps(
                {"job": job_name, "stage": stage, "stats": stats or {}}
            ).encode()
            resp = self._optimize(req, timeout=10)
            return json.loads(resp) if resp else None
        except Exception:  # noqa: BLE001
            logger.warning("brain optimize call failed", exc_info=True)
            return None

    def report_metrics(self, job_name: str, metrics: dict) -> bool:
        if not self.available:
            return False
        try:
            self._report(json.dumps({"job": job_name, **metrics}).encode(),
                         timeout=10)
            return True
        except Exception:  # noqa: BLE001
            return False
