"""Native HTTP client edge-case tests.

Real Prometheus serves large responses with chunked Transfer-Encoding, and
apiservers may close keep-alive connections between ticks; these pin the
client's chunked decoding, close-delimited bodies, keep-alive reuse after a
server-side close, and header handling against raw-socket fixture servers
(native/common/http.cpp).
"""

import socket
import threading

import pytest


@pytest.fixture
def raw_server():
    """One-shot raw TCP server: send canned bytes per accepted connection."""
    servers = []

    def make(responses, keep_open=False):
        sock = socket.socket()
        sock.bind(("127.0.0.1", 0))
        sock.listen(8)
        state = {"requests": []}

        def loop():
            for resp in responses:
                try:
                    conn, _ = sock.accept()
                except OSError:
                    return
                data = conn.recv(65536)
                state["requests"].append(data)
                conn.sendall(resp)
                if not keep_open:
                    conn.close()
                else:
                    state.setdefault("conns", []).append(conn)

        t = threading.Thread(target=loop, daemon=True)
        t.start()
        servers.append(sock)
        return f"http://127.0.0.1:{sock.getsockname()[1]}", state

    yield make
    for s in servers:
        s.close()


def test_chunked_transfer_decoding(core, raw_server):
    body = (b"HTTP/1.1 200 OK\r\n"
            b"Content-Type: application/json\r\n"
            b"Transfer-Encoding: chunked\r\n\r\n"
            b"5\r\nhello\r\n"
            b"7\r\n, world\r\n"
            b"0\r\n\r\n")
    url, _ = raw_server([body])
    r = core._http_get(url + "/x")
    assert r["status"] == 200
    assert r["body"] == b"hello, world"


def test_chunked_large_chunks(core, raw_server):
    payload = b"A" * 70000
    resp = (b"HTTP/1.1 200 OK\r\nTransfer-Encoding: chunked\r\n\r\n" +
            hex(len(payload))[2:].encode() + b"\r\n" + payload + b"\r\n0\r\n\r\n")
    url, _ = raw_server([resp])
    r = core._http_get(url + "/big")
    assert r["body"] == payload


def test_close_delimited_body(core, raw_server):
    """No Content-Length, no chunking: body runs to connection close."""
    resp = (b"HTTP/1.0 200 OK\r\nConnection: close\r\n\r\n"
            b"until-the-end")
    url, _ = raw_server([resp])
    r = core._http_get(url + "/legacy")
    assert r["body"] == b"until-the-end"


def test_retry_after_server_close(core, raw_server):
    """A stale pooled connection (server closed keep-alive) retries once."""
    ok = (b"HTTP/1.1 200 OK\r\nContent-Length: 2\r\n\r\nok")
    # two separate accepts: connection closed after each response
    url, state = raw_server([ok, ok])
    assert core._http_get(url + "/a")["status"] == 200
    assert core._http_get(url + "/b")["status"] == 200
    assert len(state["requests"]) == 2


def test_status_and_headers_parsed(core, raw_server):
    resp = (b"HTTP/1.1 404 Not Found\r\n"
            b"Content-Length: 9\r\n"
            b"X-Custom: Value\r\n\r\n"
            b"not found")
    url, _ = raw_server([resp])
    r = core._http_get(url + "/missing")
    assert r["status"] == 404
    assert r["headers"]["x-custom"] == "Value"
    assert r["body"] == b"not found"


def test_interim_100_continue_skipped(core, raw_server):
    """An interim 100 response is skipped without re-sending the request."""
    resp = (b"HTTP/1.1 100 Continue\r\n\r\n"
            b"HTTP/1.1 200 OK\r\nContent-Length: 4\r\n\r\ndone")
    url, state = raw_server([resp])
    r = core._http_get(url + "/x")
    assert r["status"] == 200
    assert r["body"] == b"done"
    assert len(state["requests"]) == 1  # no duplicate request


def test_server_survives_malformed_requests(core):
    """The exporter's HTTP server is network-exposed in-cluster: garbage,
    partial, and oversized requests must not crash or wedge it."""
    import socket

    b = core.SyntheticBackend(n_pods=2)
    b.start()
    try:
        host, port = b.prom_url.rsplit(":", 1)[0].split("//")[1], int(b.prom_url.rsplit(":", 1)[1])
        payloads = [
            b"\x00\x01\x02\xff\xfe garbage\r\n\r\n",
            b"GET\r\n\r\n",                       # missing path
            b"PRI * HTTP/2.0\r\n\r\nSM\r\n\r\n",  # h2 preface at an h1 server
            b"GET / HTTP/1.1\r\nContent-Length: 99999999\r\n\r\nshort",
            b"A" * 70000,                          # oversized junk, no CRLF
            b"GET /api/v1/query HTTP/1.1\r\nHost",  # truncated header
        ]
        for p in payloads:
            s = socket.create_connection((host, port), timeout=5)
            try:
                s.sendall(p)
                s.settimeout(2)
                try:
                    s.recv(4096)
                except socket.timeout:
                    pass
            finally:
                s.close()
        # server still serves real requests afterwards
        import urllib.request

        r = urllib.request.urlopen(b.prom_url + "/api/v1/query?query=up", timeout=5)
        assert r.status == 200
    finally:
        b.stop()


def test_ipv6_literal_url(core):
    """http://[::1]:port/... parses and connects."""
    import http.server
    import socket
    import socketserver
    import threading

    class V6Server(socketserver.ThreadingTCPServer):
        address_family = socket.AF_INET6
        allow_reuse_address = True

    class Handler(http.server.BaseHTTPRequestHandler):
        protocol_version = "HTTP/1.1"

        def log_message(self, *args):
            pass

        def do_GET(self):
            body = b"v6-ok"
            self.send_response(200)
            self.send_header("Content-Length", str(len(body)))
            self.end_headers()
            self.wfile.write(body)

    try:
        srv = V6Server(("::1", 0), Handler)
    except OSError:
        pytest.skip("no IPv6 loopback in this environment")
    t = threading.Thread(target=srv.serve_forever, daemon=True)
    t.start()
    try:
        port = srv.server_address[1]
        r = core._http_get(f"http://[::1]:{port}/x")
        assert r["status"] == 200 and r["body"] == b"v6-ok"
    finally:
        srv.shutdown()
        srv.server_close()
