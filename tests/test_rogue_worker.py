"""Dispatcher robustness against a misbehaving in-process worker.

A rogue UDS worker sends garbage hellos, a zero max_concurrency, and
out-of-range response statuses.  The dispatcher must stay alive, clamp
max_concurrency to 1 (else the backend is permanently unschedulable),
and map an insane status to 502 for the client.
"""
import json
import os
import socket
import socketserver
import subprocess
import threading
import time

import httpx
import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
BIN = os.path.join(REPO, "ollamamq_amd", "csrc", "dispatcher",
                   "ollamamq-server")


class RogueWorker(socketserver.ThreadingUnixStreamServer):
    """First probe: non-JSON garbage.  Later probes: valid hello with
    max_concurrency 0.  Requests: status 9999 then junk bytes."""
    daemon_threads = True
    allow_reuse_address = True

    def __init__(self, path):
        self.probes = 0
        super().__init__(path, self._Handler)

    class _Handler(socketserver.StreamRequestHandler):
        def handle(self):
            line = self.rfile.readline()
            try:
                cmd = json.loads(line).get("cmd")
            except (json.JSONDecodeError, AttributeError):
                return
            srv = self.server
            if cmd == "probe":
                srv.probes += 1
                if srv.probes == 1:
                    self.wfile.write(b"\x00\xffnot json at all\n")
                else:
                    self.wfile.write((json.dumps({
                        "online": True, "models": ["rogue-model"],
                        "loaded": ["rogue-model"],
                        "max_concurrency": 0}) + "\n").encode())
            elif cmd == "request":
                self.wfile.write((json.dumps({
                    "status": 9999,
                    "content_type": "application/json"}) + "\n").encode())
                self.wfile.write(b'{"done": true}\n')


class SlowStreamWorker(socketserver.ThreadingUnixStreamServer):
    """Valid worker whose stream pauses 7 s mid-generation (legal under
    load: admission behind a full batch)."""
    daemon_threads = True
    allow_reuse_address = True

    def __init__(self, path):
        super().__init__(path, self._Handler)

    class _Handler(socketserver.StreamRequestHandler):
        def handle(self):
            line = self.rfile.readline()
            try:
                cmd = json.loads(line).get("cmd")
            except (json.JSONDecodeError, AttributeError):
                return
            if cmd == "probe":
                self.wfile.write((json.dumps({
                    "online": True, "models": ["slow-model"],
                    "loaded": ["slow-model"],
                    "max_concurrency": 1}) + "\n").encode())
            elif cmd == "request":
                self.wfile.write((json.dumps({
                    "status": 200,
                    "content_type": "application/x-ndjson"}) +
                    "\n").encode())
                self.wfile.write(b'{"response": "a", "done": false}\n')
                self.wfile.flush()
                time.sleep(7)   # > the old 5 s connect-time SO_RCVTIMEO
                self.wfile.write(b'{"response": "b", "done": true}\n')


def test_stream_survives_long_token_gap(tmp_path):
    """A >5 s pause between stream chunks must not truncate the response:
    the stream phase runs under the request timeout (-t), not the 5 s
    connect timeout."""
    sock = os.path.join(str(tmp_path), "slow.sock")
    sw = SlowStreamWorker(sock)
    t = threading.Thread(target=sw.serve_forever, daemon=True)
    t.start()
    server = subprocess.Popen(
        [BIN, "--no-tui", "-p", "0", "-w", sock,
         "--probe-interval-ms", "300",
         "-c", os.path.join(str(tmp_path), "absent.yaml")],
        stderr=subprocess.PIPE, cwd=str(tmp_path), text=True)
    try:
        port = int(server.stderr.readline().rsplit(":", 1)[1].split()[0])
        base = f"http://127.0.0.1:{port}"
        deadline = time.time() + 20
        while time.time() < deadline:
            try:
                b = httpx.get(base + "/admin/models", timeout=5).json()
                if b and b[0]["online"]:
                    break
            except httpx.HTTPError:
                pass
            time.sleep(0.2)
        r = httpx.post(base + "/api/generate",
                       json={"model": "slow-model", "prompt": "x"},
                       timeout=60)
        assert r.status_code == 200
        assert '"done": true' in r.text, f"stream truncated: {r.text!r}"
    finally:
        server.terminate()
        server.wait(timeout=10)
        sw.shutdown()
        sw.server_close()


@pytest.fixture
def rogue(tmp_path):
    sock = os.path.join(str(tmp_path), "rogue.sock")
    rw = RogueWorker(sock)
    t = threading.Thread(target=rw.serve_forever, daemon=True)
    t.start()
    server = subprocess.Popen(
        [BIN, "--no-tui", "-p", "0", "-w", sock, "--probe-interval-ms", "300",
         "-c", os.path.join(str(tmp_path), "absent.yaml")],
        stderr=subprocess.PIPE, cwd=str(tmp_path), text=True)
    line = server.stderr.readline()
    port = int(line.rsplit(":", 1)[1].split()[0])
    yield f"http://127.0.0.1:{port}", server
    server.terminate()
    server.wait(timeout=10)
    rw.shutdown()
    rw.server_close()


def test_rogue_worker_survived_and_clamped(rogue):
    base, server = rogue
    # garbage first probe must not kill the dispatcher
    deadline = time.time() + 30
    online = False
    while time.time() < deadline:
        assert server.poll() is None, "dispatcher died on rogue worker"
        try:
            b = httpx.get(base + "/admin/models", timeout=5).json()
            if b and b[0]["online"] and \
                    "rogue-model" in b[0]["available_models"]:
                online = True
                break
        except httpx.HTTPError:
            pass
        time.sleep(0.2)
    assert online, "backend never came online after valid probe"

    # max_concurrency 0 clamped to 1: the request must actually dispatch,
    # and status 9999 maps to 502
    r = httpx.post(base + "/api/generate",
                   json={"model": "rogue-model", "prompt": "x"},
                   timeout=30)
    assert r.status_code == 502
    assert server.poll() is None
    assert httpx.get(base + "/health", timeout=5).text == "OK"
