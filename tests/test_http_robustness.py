"""HTTP-layer robustness: the native server must survive malformed and
adversarial inputs (garbage bytes, bad content-lengths, huge headers,
half-closed connections) without crashing or wedging."""
import os
import socket
import subprocess
import sys
import time

import httpx
import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
BIN = os.path.join(REPO, "ollamamq_amd", "csrc", "dispatcher",
                   "ollamamq-server")


@pytest.fixture(scope="module")
def srv(tmp_path_factory):
    if not os.path.exists(BIN):
        subprocess.run([sys.executable, "-m", "ollamamq_amd.build"],
                       check=True)
    tmp = tmp_path_factory.mktemp("robust")
    p = subprocess.Popen(
        [BIN, "--no-tui", "-p", "0",
         "-c", os.path.join(str(tmp), "absent.yaml")],
        stderr=subprocess.PIPE, cwd=str(tmp), text=True)
    line = p.stderr.readline()
    port = int(line.rsplit(":", 1)[1].split()[0])
    yield port
    p.terminate()


def raw(port, data, expect_reply=False):
    s = socket.create_connection(("127.0.0.1", port), timeout=5)
    try:
        s.sendall(data)
        if expect_reply:
            s.settimeout(5)
            return s.recv(4096)
        return b""
    except OSError:
        return b""
    finally:
        s.close()


def alive(port):
    r = httpx.get(f"http://127.0.0.1:{port}/health", timeout=5.0)
    return r.status_code == 200


def test_garbage_bytes(srv):
    raw(srv, os.urandom(5000))
    assert alive(srv)


def test_bad_request_line(srv):
    raw(srv, b"NOT A REQUEST\r\n\r\n")
    assert alive(srv)


def test_oversized_content_length(srv):
    r = raw(srv, b"POST /api/chat HTTP/1.1\r\nHost: x\r\n"
                 b"Content-Length: 99999999999\r\n\r\n{}",
            expect_reply=True)
    assert b"400" in r or r == b""
    assert alive(srv)


def test_header_flood(srv):
    data = b"GET /health HTTP/1.1\r\n" + b"X-A: b\r\n" * 20000 + b"\r\n"
    raw(srv, data)
    assert alive(srv)


def test_half_close_mid_body(srv):
    s = socket.create_connection(("127.0.0.1", srv), timeout=5)
    s.sendall(b"POST /api/chat HTTP/1.1\r\nContent-Length: 100\r\n\r\nhalf")
    s.close()   # body never completes
    time.sleep(0.2)
    assert alive(srv)


def test_many_rapid_connections(srv):
    for _ in range(100):
        s = socket.create_connection(("127.0.0.1", srv), timeout=5)
        s.close()
    assert alive(srv)


def test_chunked_request_body(srv):
    """Transfer-Encoding: chunked request bodies are decoded (axum
    parity); an /api/chat body arriving in chunks must parse."""
    body = b'{"model":"nosuch-model"}'
    mid = len(body) // 2
    data = (b"POST /admin/models/load HTTP/1.1\r\nHost: x\r\n"
            b"Transfer-Encoding: chunked\r\n\r\n"
            + hex(mid)[2:].encode() + b"\r\n" + body[:mid] + b"\r\n"
            + hex(len(body) - mid)[2:].encode() + b"\r\n" + body[mid:]
            + b"\r\n0\r\n\r\n")
    r = raw(srv, data, expect_reply=True)
    # the chunk-decoded JSON reaches the handler: a clean 404 (unknown
    # model), not a parse error, and the server stays alive
    assert r.startswith(b"HTTP/1.1 404"), r[:80]
    assert alive(srv)


def test_chunked_request_body_garbage_size(srv):
    raw(srv, b"POST /api/chat HTTP/1.1\r\n"
             b"Transfer-Encoding: chunked\r\n\r\nZZZ\r\n")
    assert alive(srv)


def test_chunked_overflow_size_ingress(srv):
    """Chunk size ffffffffffffffff: csz+2 would wrap to 1 — must be a
    clean 400, not a corrupted parse."""
    r = raw(srv, b"POST /admin/models/load HTTP/1.1\r\nHost: x\r\n"
                 b"Transfer-Encoding: chunked\r\n\r\n"
                 b"ffffffffffffffff\r\nXX\r\n0\r\n\r\n",
            expect_reply=True)
    assert r.startswith(b"HTTP/1.1 400"), r[:80]
    assert alive(srv)


def test_rogue_backend_chunked_overflow(tmp_path):
    """A malicious/buggy BACKEND answering probes with an overflowing
    chunk size (ffffffffffffffff) must not crash the dispatcher's HTTP
    client decoder (pre-fix: out-of-bounds deliver of SIZE_MAX bytes)."""
    import threading

    hits = []

    def backend():
        ls = socket.create_server(("127.0.0.1", 0))
        ports.append(ls.getsockname()[1])
        ls.settimeout(30)
        started.set()
        while not stop.is_set():
            try:
                c, _ = ls.accept()
            except socket.timeout:
                continue
            with c:
                try:
                    c.settimeout(5)
                    c.recv(4096)
                    hits.append(1)
                    c.sendall(b"HTTP/1.1 200 OK\r\n"
                              b"Transfer-Encoding: chunked\r\n\r\n"
                              b"ffffffffffffffff\r\n"
                              b'{"models":[]}\r\n0\r\n\r\n')
                except OSError:
                    pass
        ls.close()

    ports, started, stop = [], threading.Event(), threading.Event()
    t = threading.Thread(target=backend, daemon=True)
    t.start()
    assert started.wait(10)
    p = subprocess.Popen(
        [BIN, "--no-tui", "-p", "0", "-o",
         f"http://127.0.0.1:{ports[0]}", "--probe-interval-ms", "200",
         "-c", os.path.join(str(tmp_path), "absent.yaml")],
        stderr=subprocess.PIPE, cwd=str(tmp_path), text=True)
    try:
        port = int(p.stderr.readline().rsplit(":", 1)[1].split()[0])
        deadline = time.time() + 20
        while time.time() < deadline and len(hits) < 3:
            assert p.poll() is None, "dispatcher crashed on rogue backend"
            time.sleep(0.2)
        assert len(hits) >= 3, "backend was never probed"
        assert p.poll() is None
        assert alive(port)
    finally:
        stop.set()
        p.terminate()
        p.wait(timeout=10)


def test_half_open_flood_no_wedge(tmp_path):
    """Hundreds of half-open connections (headers never finished) must
    not starve the accept loop: health stays served during an under-cap
    flood, and the server recovers immediately once sockets close
    (pre-fix: thread pile-up with a 120 s idle timeout eventually made
    connect() itself time out)."""
    p = subprocess.Popen(
        [BIN, "--no-tui", "-p", "0",
         "-c", os.path.join(str(tmp_path), "absent.yaml")],
        stderr=subprocess.PIPE, cwd=str(tmp_path), text=True)
    try:
        port = int(p.stderr.readline().rsplit(":", 1)[1].split()[0])
        held = []
        for _ in range(300):            # well under the 2048 cap
            s = socket.create_connection(("127.0.0.1", port), timeout=5)
            s.sendall(b"POST /api/chat HTTP/1.1\r\nContent-Length: 9\r\n")
            held.append(s)
        # health must answer WHILE 300 threads sit in read_headers
        for _ in range(3):
            assert alive(port)
        for s in held:
            s.close()
        time.sleep(0.5)
        assert alive(port)
        assert p.poll() is None
    finally:
        p.terminate()
        p.wait(timeout=10)


def test_deeply_nested_json_body(srv):
    """A 100k-deep JSON nesting bomb must be rejected by the parser's
    depth guard, not blow the stack."""
    bomb = b"[" * 100000 + b"]" * 100000
    data = (b"POST /admin/models/load HTTP/1.1\r\nHost: x\r\n"
            b"Content-Length: " + str(len(bomb)).encode() + b"\r\n\r\n"
            + bomb)
    r = raw(srv, data, expect_reply=True)
    assert r.startswith(b"HTTP/1.1 4"), r[:60]   # clean 4xx, no crash
    assert alive(srv)
