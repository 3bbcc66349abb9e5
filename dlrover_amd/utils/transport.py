"""Control-plane RPC transports.

The reference exposes one two-verb service — report(msg) and get(msg) — over
pluggable gRPC/HTTP/Ray transports (ref: master/servicer.py:871-1140,
master_client.py subclasses). We keep the same two-verb contract with two
stdlib transports:

  - tcp  (default): length-prefixed restricted-pickle frames over a threaded
    TCP server. No external deps, lowest latency, fine for the volumes this
    control plane moves (heartbeats, rendezvous, KV bootstrap).
  - http : the same frames as POST bodies on /get and /report — for
    environments that require L7 (ingress, probes). Uses stdlib http.server.

Both serialize dlrover_amd.common.comm.BaseRequest/BaseResponse through the
allow-listed unpickler (serialize.py).
"""

import socket
import socketserver
import struct
import threading
import time
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer
from typing import Callable, Optional
from urllib import request as urlrequest

from dlrover_amd.common import serialize
from dlrover_amd.common.comm import BaseRequest, BaseResponse
from dlrover_amd.common.constants import CommServiceType
from dlrover_amd.common.log import logger

Handler = Callable[[str, BaseRequest], BaseResponse]

_HEADER = struct.Struct("<cI")  # verb byte + payload length
_VERBS = {b"g": "get", b"r": "report", b"p": "ping"}
_VERB_BYTES = {v: k for k, v in _VERBS.items()}


# ---------------------------------------------------------------------------
# TCP
# ---------------------------------------------------------------------------


def _recv_exact(sock: socket.socket, n: int) -> bytes:
    buf = bytearray()
    while len(buf) < n:
        chunk = sock.recv(min(1 << 20, n - len(buf)))
        if not chunk:
            raise ConnectionError("peer closed")
        buf += chunk
    return bytes(buf)


class _TcpHandler(socketserver.BaseRequestHandler):
    def handle(self):
        handler: Handler = self.server.rpc_handler  # type: ignore[attr-defined]
        sock = self.request
        sock.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
        while True:
            try:
                head = _recv_exact(sock, _HEADER.size)
            except (ConnectionError, OSError):
                return
            verb_b, length = _HEADER.unpack(head)
            verb = _VERBS.get(verb_b, "")
            try:
                payload = _recv_exact(sock, length) if length else b""
                if verb == "ping":
                    resp = BaseResponse(success=True)
                else:
                    req = serialize.loads(payload)
                    resp = handler(verb, req)
            except Exception as e:  # noqa: BLE001 — report to client
                logger.exception("RPC %s failed", verb)
                resp = BaseResponse(success=False, reason=repr(e))
            try:
                out = serialize.dumps(resp)
                sock.sendall(_HEADER.pack(verb_b, len(out)) + out)
            except (ConnectionError, OSError):
                return


class _ThreadedTCPServer(socketserver.ThreadingTCPServer):
    daemon_threads = True
    allow_reuse_address = True


class TcpRpcServer:
    def __init__(self, port: int, handler: Handler, host: str = "0.0.0.0"):
        self._server = _ThreadedTCPServer((host, port), _TcpHandler)
        self._server.rpc_handler = handler  # type: ignore[attr-defined]
        self.port = self._server.server_address[1]
        self._thread = threading.Thread(
            target=self._server.serve_forever, name="dlrover-rpc", daemon=True
        )

    def start(self):
        self._thread.start()
        logger.info("TCP RPC server on port %s", self.port)
        return self

    def stop(self):
        self._server.shutdown()
        self._server.server_close()


class TcpRpcClient:
    def __init__(self, addr: str, timeout: float = 30.0):
        host, port = addr.rsplit(":", 1)
        self._host, self._port = host, int(port)
        self._timeout = timeout
        self._sock: Optional[socket.socket] = None
        self._lock = threading.Lock()

    def _connect(self):
        s = socket.create_connection((self._host, self._port), timeout=self._timeout)
        s.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
        self._sock = s

    def call(self, verb: str, req: BaseRequest) -> BaseResponse:
        payload = serialize.dumps(req) if verb != "ping" else b""
        vb = _VERB_BYTES[verb]
        with self._lock:
            for attempt in (0, 1):
                try:
                    if self._sock is None:
                        self._connect()
                    self._sock.sendall(_HEADER.pack(vb, len(payload)) + payload)
                    head = _recv_exact(self._sock, _HEADER.size)
                    _, length = _HEADER.unpack(head)
                    data = _recv_exact(self._sock, length)
                    return serialize.loads(data)
                except (ConnectionError, OSError, socket.timeout):
                    self._close_locked()
                    if attempt == 1:
                        raise
        raise ConnectionError("unreachable")

    def _close_locked(self):
        if self._sock is not None:
            try:
                self._sock.close()
            finally:
                self._sock = None

    def close(self):
        with self._lock:
            self._close_locked()


# ---------------------------------------------------------------------------
# HTTP
# ---------------------------------------------------------------------------


class _HttpHandler(BaseHTTPRequestHandler):
    protocol_version = "HTTP/1.1"

    def log_message(self, fmt, *args):  # silence stdlib request logging
        pass

    def do_POST(self):
        handler: Handler = self.server.rpc_handler  # type: ignore[attr-defined]
        verb = self.path.strip("/")
        length = int(self.headers.get("Content-Length", 0))
        body = self.rfile.read(length)
        try:
            if verb == "ping":
                resp = BaseResponse(success=True)
            elif verb in ("get", "report"):
                resp = handler(verb, serialize.loads(body))
            else:
                resp = BaseResponse(success=False, reason=f"unknown verb {verb}")
        except Exception as e:  # noqa: BLE001
            logger.exception("HTTP RPC %s failed", verb)
            resp = BaseResponse(success=False, reason=repr(e))
        out = serialize.dumps(resp)
        self.send_response(200)
        self.send_header("Content-Type", "application/octet-stream")
        self.send_header("Content-Length", str(len(out)))
        self.end_headers()
        self.wfile.write(out)


class HttpRpcServer:
    def __init__(self, port: int, handler: Handler, host: str = "0.0.0.0"):
        self._server = ThreadingHTTPServer((host, port), _HttpHandler)
        self._server.rpc_handler = handler  # type: ignore[attr-defined]
        self.port = self._server.server_address[1]
        self._thread = threading.Thread(
            target=self._server.serve_forever, name="dlrover-http", daemon=True
        )

    def start(self):
        self._thread.start()
        logger.info("HTTP RPC server on port %s", self.port)
        return self

    def stop(self):
        self._server.shutdown()
        self._server.server_close()


class HttpRpcClient:
    def __init__(self, addr: str, timeout: float = 30.0):
        self._base = f"http://{addr}"
        self._timeout = timeout

    def call(self, verb: str, req: BaseRequest) -> BaseResponse:
        payload = serialize.dumps(req) if verb != "ping" else b""
        r = urlrequest.Request(
            f"{self._base}/{verb}",
            data=payload,
            headers={"Content-Type": "application/octet-stream"},
            method="POST",
        )
        with urlrequest.urlopen(r, timeout=self._timeout) as resp:
            return serialize.loads(resp.read())

    def close(self):
        pass


# ---------------------------------------------------------------------------
# gRPC (generic bytes handlers — no protoc-generated stubs needed)
# ---------------------------------------------------------------------------


class GrpcRpcServer:
    """Two-verb service over gRPC, mirroring the reference's default
    transport (ref: servicer.py:910 GrpcMasterServicer). Methods are
    registered generically with bytes (de)serializers, so the wire payload is
    the same restricted-pickle frame the tcp/http transports use."""

    SERVICE = "dlrover.Master"

    def __init__(self, port: int, handler: Handler, host: str = "0.0.0.0"):
        import grpc
        from concurrent import futures

        self._handler = handler

        def _unary(verb):
            def call(request_bytes, context):
                try:
                    if verb == "ping":
                        resp = BaseResponse(success=True)
                    else:
                        resp = handler(verb, serialize.loads(request_bytes))
                except Exception as e:  # noqa: BLE001
                    logger.exception("gRPC %s failed", verb)
                    resp = BaseResponse(success=False, reason=repr(e))
                return serialize.dumps(resp)

            return grpc.unary_unary_rpc_method_handler(
                call,
                request_deserializer=lambda b: b,
                response_serializer=lambda b: b,
            )

        service = grpc.method_handlers_generic_handler(
            self.SERVICE,
            {"get": _unary("get"), "report": _unary("report"),
             "ping": _unary("ping")},
        )
        self._server = grpc.server(
            futures.ThreadPoolExecutor(max_workers=32),
            options=[
                ("grpc.max_send_message_length", 256 << 20),
                ("grpc.max_receive_message_length", 256 << 20),
            ],
        )
        self._server.add_generic_rpc_handlers((service,))
        self.port = self._server.add_insecure_port(f"{host}:{port}")

    def start(self):
        self._server.start()
        logger.info("gRPC RPC server on port %s", self.port)
        return self

    def stop(self):
        self._server.stop(grace=1)


class GrpcRpcClient:
    def __init__(self, addr: str, timeout: float = 30.0):
        import grpc

        self._timeout = timeout
        self._channel = grpc.insecure_channel(
            addr,
            options=[
                ("grpc.max_send_message_length", 256 << 20),
                ("grpc.max_receive_message_length", 256 << 20),
            ],
        )
        self._methods = {
            verb: self._channel.unary_unary(
                f"/{GrpcRpcServer.SERVICE}/{verb}",
                request_serializer=lambda b: b,
                response_deserializer=lambda b: b,
            )
            for verb in ("get", "report", "ping")
        }

    def call(self, verb: str, req: BaseRequest) -> BaseResponse:
        payload = serialize.dumps(req) if verb != "ping" else b""
        out = self._methods[verb](payload, timeout=self._timeout)
        return serialize.loads(out)

    def close(self):
        self._channel.close()


# ---------------------------------------------------------------------------
# factories
# ---------------------------------------------------------------------------


def create_rpc_server(service_type: str, port: int, handler: Handler):
    if service_type == CommServiceType.HTTP:
        return HttpRpcServer(port, handler)
    if service_type == CommServiceType.GRPC:
        return GrpcRpcServer(port, handler)
    return TcpRpcServer(port, handler)


def create_rpc_client(service_type: str, addr: str, timeout: float = 30.0):
    if service_type == CommServiceType.HTTP:
        return HttpRpcClient(addr, timeout)
    if service_type == CommServiceType.GRPC:
        return GrpcRpcClient(addr, timeout)
    return TcpRpcClient(addr, timeout)


def wait_for_server(addr: str, timeout: float = 60.0, service_type: str = "tcp") -> bool:
    client = create_rpc_client(service_type, addr, timeout=2.0)
    deadline = time.time() + timeout
    while time.time() < deadline:
        try:
            client.call("ping", BaseRequest())
            client.close()
            return True
        except Exception:  # noqa: BLE001 — includes grpc.RpcError
            time.sleep(0.3)
    return False
