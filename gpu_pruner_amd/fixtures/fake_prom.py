"""FakePrometheus — minimal Prometheus HTTP API double.

Serves ``/api/v1/query`` (GET and form-encoded POST, like the real API)
returning a configurable instant-vector result, recording every query it
receives. Can be told to fail N times (HTTP 500) to exercise the daemon's
consecutive-failure breaker.
"""

from __future__ import annotations

import json
import threading
import urllib.parse
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer


class FakePrometheus:
    def __init__(self, host: str = "127.0.0.1", port: int = 0,
                 certfile: str | None = None, keyfile: str | None = None):
        self._lock = threading.Lock()
        self.series: list[dict] = []
        self.queries: list[str] = []
        self.fail_next = 0  # serve this many 500s before succeeding
        self.bearer_tokens: list[str | None] = []
        # when set, served verbatim as the API `data` object (e.g. a matrix
        # result for querytest tests)
        self.data_override: dict | None = None

        fixture = self

        class Handler(BaseHTTPRequestHandler):
            protocol_version = "HTTP/1.1"
            disable_nagle_algorithm = True

            def log_message(self, *args):  # silence
                pass

            def _handle_query(self, query: str):
                with fixture._lock:
                    fixture.queries.append(query)
                    auth = self.headers.get("Authorization")
                    fixture.bearer_tokens.append(
                        auth.split(" ", 1)[1] if auth and " " in auth else None
                    )
                    if fixture.fail_next > 0:
                        fixture.fail_next -= 1
                        self._send(500, {"status": "error", "error": "induced failure"})
                        return
                    if fixture.data_override is not None:
                        data = dict(fixture.data_override)
                    else:
                        data = {"resultType": "vector",
                                "result": [dict(s) for s in fixture.series]}
                self._send(200, {"status": "success", "data": data})

            def _send(self, status: int, obj: dict):
                body = json.dumps(obj).encode()
                self.send_response(status)
                self.send_header("Content-Type", "application/json")
                self.send_header("Content-Length", str(len(body)))
                self.end_headers()
                self.wfile.write(body)

            def do_GET(self):
                parsed = urllib.parse.urlparse(self.path)
                if parsed.path.endswith("/api/v1/query"):
                    q = urllib.parse.parse_qs(parsed.query).get("query", [""])[0]
                    self._handle_query(q)
                else:
                    self._send(404, {"status": "error", "error": "not found"})

            def do_POST(self):
                parsed = urllib.parse.urlparse(self.path)
                length = int(self.headers.get("Content-Length", "0"))
                body = self.rfile.read(length).decode()
                if parsed.path.endswith("/api/v1/query"):
                    q = urllib.parse.parse_qs(body).get("query", [""])[0]
                    self._handle_query(q)
                else:
                    self._send(404, {"status": "error", "error": "not found"})

        # ThreadingHTTPServer's default listen backlog is 5: the engine's
        # 32-way fan-out (plus informer reconnects) overflows it under
        # load, surfacing as spurious connection-refused/reset
        class _Server(ThreadingHTTPServer):
            request_queue_size = 128

        self._server = _Server((host, port), Handler)
        self._server.daemon_threads = True
        self._tls = certfile is not None
        if certfile is not None:
            import ssl

            ctx = ssl.SSLContext(ssl.PROTOCOL_TLS_SERVER)
            ctx.load_cert_chain(certfile, keyfile)
            self._server.socket = ctx.wrap_socket(self._server.socket, server_side=True)
        self._thread = threading.Thread(
            target=lambda: self._server.serve_forever(poll_interval=0.05), daemon=True)

    # -- lifecycle -----------------------------------------------------------
    def start(self) -> "FakePrometheus":
        self._thread.start()
        return self

    def stop(self):
        self._server.shutdown()
        self._server.server_close()

    @property
    def url(self) -> str:
        host, port = self._server.server_address[:2]
        scheme = "https" if self._tls else "http"
        return f"{scheme}://{host}:{port}"

    def __enter__(self):
        return self.start()

    def __exit__(self, *exc):
        self.stop()

    # -- series helpers ------------------------------------------------------
    def add_idle_series(
        self,
        pod: str,
        namespace: str,
        container: str = "main",
        gpu: str = "0",
        model_name: str = "AMD Instinct MI355X",
        hostname: str = "mi355-node-0",
        node_type: str | None = "amd-mi355x",
        value: float = 0.0,
        honor_labels: bool = False,
        ts: float = 1700000000.0,
    ):
        """Append one instant-vector series shaped like the idle query output."""
        prefix = "" if honor_labels else "exported_"
        metric = {
            "Hostname": hostname,
            f"{prefix}pod": pod,
            f"{prefix}namespace": namespace,
            f"{prefix}container": container,
            "gpu": gpu,
            "modelName": model_name,
        }
        if node_type is not None:
            metric["node_type"] = node_type
        with self._lock:
            self.series.append({"metric": metric, "value": [ts, str(value)]})

    def clear(self):
        with self._lock:
            self.series.clear()
            self.queries.clear()
            self.bearer_tokens.clear()
