"""FakeOtlpCollector — minimal OTLP/HTTP+JSON collector double.

Records every payload POSTed to /v1/traces and /v1/metrics so tests can
assert the daemon's span + counter export surface.
"""

from __future__ import annotations

import json
import threading
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer


class FakeOtlpCollector:
    def __init__(self, host: str = "127.0.0.1", port: int = 0):
        self._lock = threading.Lock()
        self.traces: list[dict] = []
        self.metrics: list[dict] = []
        self.traces_pb: list[bytes] = []   # raw binary-protobuf payloads
        self.metrics_pb: list[bytes] = []

        fixture = self

        class Handler(BaseHTTPRequestHandler):
            protocol_version = "HTTP/1.1"
            disable_nagle_algorithm = True

            def log_message(self, *args):
                pass

            def do_POST(self):
                length = int(self.headers.get("Content-Length", "0"))
                raw = self.rfile.read(length) or b""
                ctype = self.headers.get("Content-Type", "")
                if "protobuf" in ctype:
                    with fixture._lock:
                        if self.path.endswith("/v1/traces"):
                            fixture.traces_pb.append(raw)
                        elif self.path.endswith("/v1/metrics"):
                            fixture.metrics_pb.append(raw)
                    payload = None
                else:
                    try:
                        payload = json.loads(raw or b"{}")
                    except json.JSONDecodeError:
                        payload = {}
                if payload is not None:
                    with fixture._lock:
                        if self.path.endswith("/v1/traces"):
                            fixture.traces.append(payload)
                        elif self.path.endswith("/v1/metrics"):
                            fixture.metrics.append(payload)
                body = b"{}"
                self.send_response(200)
                self.send_header("Content-Type", "application/json")
                self.send_header("Content-Length", str(len(body)))
                self.end_headers()
                self.wfile.write(body)

        # ThreadingHTTPServer's default listen backlog is 5: the engine's
        # 32-way fan-out (plus informer reconnects) overflows it under
        # load, surfacing as spurious connection-refused/reset
        class _Server(ThreadingHTTPServer):
            request_queue_size = 128

        self._server = _Server((host, port), Handler)
        self._server.daemon_threads = True
        self._thread = threading.Thread(
            target=lambda: self._server.serve_forever(poll_interval=0.05), daemon=True)

    def start(self):
        self._thread.start()
        return self

    def stop(self):
        self._server.shutdown()
        self._server.server_close()

    @property
    def url(self) -> str:
        host, port = self._server.server_address[:2]
        return f"http://{host}:{port}"

    def __enter__(self):
        return self.start()

    def __exit__(self, *exc):
        self.stop()

    # -- helpers --------------------------------------------------------------
    def drain(self) -> tuple[int, int]:
        """Drop recorded payloads (long soaks would otherwise measure the
        fixture's own accumulation as a 'leak'); returns (n_traces, n_metrics)
        dropped."""
        with self._lock:
            n = (len(self.traces) + len(self.traces_pb),
                 len(self.metrics) + len(self.metrics_pb))
            self.traces.clear()
            self.metrics.clear()
            self.traces_pb.clear()
            self.metrics_pb.clear()
            return n

    def span_names(self) -> list[str]:
        with self._lock:
            names = []
            for payload in self.traces:
                for rs in payload.get("resourceSpans", []):
                    for ss in rs.get("scopeSpans", []):
                        names.extend(s.get("name") for s in ss.get("spans", []))
            return names

    def metric_points(self) -> dict[str, int]:
        """latest value per metric name across all exports."""
        with self._lock:
            out = {}
            for payload in self.metrics:
                for rm in payload.get("resourceMetrics", []):
                    for sm in rm.get("scopeMetrics", []):
                        for m in sm.get("metrics", []):
                            dps = (m.get("sum") or m.get("gauge") or {}).get(
                                "dataPoints", [])
                            if dps:
                                out[m["name"]] = int(dps[-1].get("asInt", 0))
            return out

    def spans(self) -> list[dict]:
        """All exported span dicts (JSON transport)."""
        with self._lock:
            out = []
            for payload in self.traces:
                for rs in payload.get("resourceSpans", []):
                    for ss in rs.get("scopeSpans", []):
                        out.extend(ss.get("spans", []))
            return out


class FakeOtlpGrpcCollector:
    """OTLP/gRPC collector double (the reference's tonic transport): a real
    grpcio server accepting TraceService/MetricsService Export as raw bytes —
    an independent HTTP/2 + gRPC implementation validating the hand-rolled
    h2c client (native/common/grpc_client.cpp)."""

    def __init__(self, host: str = "127.0.0.1", port: int = 0):
        import grpc
        from concurrent import futures

        self._lock = threading.Lock()
        self.traces_pb: list[bytes] = []
        self.metrics_pb: list[bytes] = []
        fixture = self

        class Handler(grpc.GenericRpcHandler):
            def service(self, handler_call_details):
                method = handler_call_details.method

                def unary_unary(request, context):
                    with fixture._lock:
                        if "TraceService" in method:
                            fixture.traces_pb.append(request)
                        elif "MetricsService" in method:
                            fixture.metrics_pb.append(request)
                    return b""  # empty Export*ServiceResponse

                return grpc.unary_unary_rpc_method_handler(
                    unary_unary, request_deserializer=None,
                    response_serializer=None)

        self._server = grpc.server(futures.ThreadPoolExecutor(max_workers=4),
                                   handlers=(Handler(),))
        self._port = self._server.add_insecure_port(f"{host}:{port}")
        self._host = host

    def start(self):
        self._server.start()
        return self

    def stop(self):
        self._server.stop(grace=None)

    @property
    def url(self) -> str:
        return f"http://{self._host}:{self._port}"

    def __enter__(self):
        return self.start()

    def __exit__(self, *exc):
        self.stop()
