"""BodyStream chunked-transfer decoding edge cases.

The informer's watch path depends on native/common/http.cpp's incremental
chunk decoder; a byte-split bug there silently corrupts watch events. A raw
socket server replays chunked responses fragmented at every awkward
boundary (inside the size line, inside data, across the trailing CRLF).
"""

import socket
import threading

import pytest


def serve_fragments(fragments, delay_s=0.0):
    """One-shot server: accepts a connection, reads the request, writes the
    raw byte fragments in order, closes. Returns (host, port, thread)."""
    import time

    srv = socket.socket()
    srv.bind(("127.0.0.1", 0))
    srv.listen(1)

    def run():
        conn, _ = srv.accept()
        conn.recv(65536)  # request head
        for frag in fragments:
            conn.sendall(frag)
            if delay_s:
                time.sleep(delay_s)
        conn.close()
        srv.close()

    t = threading.Thread(target=run, daemon=True)
    t.start()
    return srv.getsockname()


HEAD = (b"HTTP/1.1 200 OK\r\n"
        b"Content-Type: application/json\r\n"
        b"Transfer-Encoding: chunked\r\n\r\n")


def chunked(payload: bytes) -> bytes:
    return f"{len(payload):x}\r\n".encode() + payload + b"\r\n"


def stream_lines(core, addr):
    host, port = addr
    return list(core.http_stream_lines(f"http://{host}:{port}/watch"))


def test_single_chunk_single_line(core):
    addr = serve_fragments([HEAD + chunked(b'{"a":1}\n') + b"0\r\n\r\n"])
    assert stream_lines(core, addr) == ['{"a":1}']


def test_line_split_across_chunks(core):
    addr = serve_fragments([HEAD + chunked(b'{"a"') + chunked(b':1}\n') + b"0\r\n\r\n"])
    assert stream_lines(core, addr) == ['{"a":1}']


def test_fragmented_at_every_boundary(core):
    """Each TCP segment splits the framing somewhere nasty: inside the
    chunk-size line, between size and data, inside data, and across the
    chunk's trailing CRLF."""
    body = chunked(b'{"x":1}\n') + chunked(b'{"y":2}\n') + b"0\r\n\r\n"
    whole = HEAD + body
    # split into 1-byte fragments: worst case for any incremental parser
    frags = [whole[i:i + 1] for i in range(len(whole))]
    addr = serve_fragments(frags)
    assert stream_lines(core, addr) == ['{"x":1}', '{"y":2}']


def test_multiple_lines_in_one_chunk(core):
    addr = serve_fragments([HEAD + chunked(b'{"x":1}\n{"y":2}\n{"z":3}\n') + b"0\r\n\r\n"])
    assert stream_lines(core, addr) == ['{"x":1}', '{"y":2}', '{"z":3}']


def test_crlf_line_endings_are_stripped(core):
    addr = serve_fragments([HEAD + chunked(b'{"a":1}\r\n{"b":2}\r\n') + b"0\r\n\r\n"])
    assert stream_lines(core, addr) == ['{"a":1}', '{"b":2}']


def test_unterminated_trailing_line_on_eof(core):
    # connection drops before the final newline/terminator: the partial
    # line is still surfaced (callers decide what to do with it)
    addr = serve_fragments([HEAD + chunked(b'{"a":1}\n{"partial"')])
    assert stream_lines(core, addr) == ['{"a":1}', '{"partial"']


def test_large_chunk_spanning_many_reads(core):
    line = b'{"k":"' + b"v" * 100_000 + b'"}\n'
    addr = serve_fragments([HEAD, chunked(line), b"0\r\n\r\n"])
    lines = stream_lines(core, addr)
    assert len(lines) == 1 and len(lines[0]) == len(line) - 1


def test_content_length_stream(core):
    # non-chunked streaming (Content-Length response read incrementally)
    body = b'{"a":1}\n{"b":2}\n'
    head = (b"HTTP/1.1 200 OK\r\nContent-Type: application/json\r\n"
            b"Content-Length: " + str(len(body)).encode() + b"\r\n\r\n")
    addr = serve_fragments([head + body[:5], body[5:]])
    assert stream_lines(core, addr) == ['{"a":1}', '{"b":2}']


# ---- property-based fuzz: random payloads x random chunking x random
# fragmentation must always round-trip ------------------------------------

hypothesis = pytest.importorskip("hypothesis")
from hypothesis import HealthCheck, given, settings, strategies as st  # noqa: E402


@settings(max_examples=60, deadline=None,
          suppress_health_check=[HealthCheck.function_scoped_fixture])
@given(
    lines=st.lists(st.text(alphabet=st.characters(min_codepoint=0x20,
                                                  max_codepoint=0x7E),
                           min_size=1, max_size=80),
                   min_size=1, max_size=12),
    chunk_sizes=st.lists(st.integers(min_value=1, max_value=64), min_size=1,
                         max_size=8),
    frag=st.integers(min_value=1, max_value=97),
)
def test_chunk_decoder_roundtrip_fuzz(core, lines, chunk_sizes, frag):
    body = "".join(l + "\n" for l in lines).encode()
    wire = b""
    i = 0
    k = 0
    while i < len(body):
        n = min(chunk_sizes[k % len(chunk_sizes)], len(body) - i)
        wire += chunked(body[i:i + n])
        i += n
        k += 1
    wire += b"0\r\n\r\n"
    whole = HEAD + wire
    frags = [whole[i:i + frag] for i in range(0, len(whole), frag)]
    addr = serve_fragments(frags)
    assert stream_lines(core, addr) == lines
