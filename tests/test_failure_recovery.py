"""Failure detection + elastic recovery: a dead worker goes
offline via the health loop, requests for it park/503, and a restarted
worker rejoins automatically with learned state re-probed."""
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


def _wait_socket(path, timeout=60):
    t0 = time.time()
    while time.time() - t0 < timeout:
        if os.path.exists(path):
            try:
                s = socket.socket(socket.AF_UNIX, socket.SOCK_STREAM)
                s.connect(path)
                s.close()
                return True
            except OSError:
                pass
        time.sleep(0.2)
    return False


def _spawn_worker(sock):
    return subprocess.Popen(
        [sys.executable, "-m", "ollamamq_amd.engine.worker",
         "--socket", sock, "--model", "tiny-cpu", "--max-ctx", "256",
         "--max-batch", "4"],
        cwd=REPO, stdout=subprocess.DEVNULL, stderr=subprocess.STDOUT)


def _online(base):
    try:
        r = httpx.get(base + "/admin/models", timeout=2.0).json()
        return r[0]["online"]
    except Exception:
        return None


def test_worker_death_and_rejoin(tmp_path):
    if not os.path.exists(BIN):
        subprocess.run([sys.executable, "-m", "ollamamq_amd.build"],
                       check=True)
    sock = os.path.join(str(tmp_path), "w.sock")
    worker = _spawn_worker(sock)
    assert _wait_socket(sock)
    server = subprocess.Popen(
        [BIN, "--no-tui", "-p", "0", "-w", sock, "--probe-interval-ms",
         "300", "--stuck-timeout", "2",
         "-c", os.path.join(str(tmp_path), "absent.yaml")],
        stderr=subprocess.PIPE, cwd=str(tmp_path), text=True)
    try:
        line = server.stderr.readline()
        port = int(line.rsplit(":", 1)[1].split()[0])
        base = f"http://127.0.0.1:{port}"
        deadline = time.time() + 15
        while time.time() < deadline and _online(base) is not True:
            time.sleep(0.2)
        assert _online(base) is True

        # request works
        r = httpx.post(base + "/api/generate",
                       json={"model": "tiny-cpu", "prompt": "a",
                             "stream": False,
                             "options": {"num_predict": 2}},
                       headers={"X-User-ID": "f1"}, timeout=60.0)
        assert r.status_code == 200

        # kill the worker: health loop flips it offline
        worker.terminate()
        worker.wait(timeout=10)
        os.unlink(sock)
        deadline = time.time() + 15
        while time.time() < deadline and _online(base) is not False:
            time.sleep(0.2)
        assert _online(base) is False

        # a request for the dead fleet parks then 503s (stuck timeout 2 s)
        t0 = time.time()
        r = httpx.post(base + "/api/generate",
                       json={"model": "tiny-cpu", "prompt": "a"},
                       headers={"X-User-ID": "f2"}, timeout=30.0)
        assert r.status_code == 503
        assert time.time() - t0 < 15

        # restart the worker: it rejoins automatically
        worker = _spawn_worker(sock)
        assert _wait_socket(sock)
        deadline = time.time() + 20
        while time.time() < deadline and _online(base) is not True:
            time.sleep(0.2)
        assert _online(base) is True
        r = httpx.post(base + "/api/generate",
                       json={"model": "tiny-cpu", "prompt": "b",
                             "stream": False,
                             "options": {"num_predict": 2}},
                       headers={"X-User-ID": "f3"}, timeout=60.0)
        assert r.status_code == 200
    finally:
        server.terminate()
        worker.terminate()


def test_midstream_worker_death_truncates_cleanly(tmp_path):
    """Worker dies while a response is STREAMING: the client's stream
    ends (truncated, no done marker), the dispatcher survives, and the
    health loop flips the backend offline (SURVEY §5 mid-stream
    failure semantics)."""
    if not os.path.exists(BIN):
        subprocess.run([sys.executable, "-m", "ollamamq_amd.build"],
                       check=True)
    sock = os.path.join(str(tmp_path), "w.sock")
    worker = _spawn_worker(sock)
    assert _wait_socket(sock)
    server = subprocess.Popen(
        [BIN, "--no-tui", "-p", "0", "-w", sock, "--probe-interval-ms",
         "300", "-c", os.path.join(str(tmp_path), "absent.yaml")],
        stderr=subprocess.PIPE, cwd=str(tmp_path), text=True)
    try:
        port = int(server.stderr.readline().rsplit(":", 1)[1].split()[0])
        base = f"http://127.0.0.1:{port}"
        deadline = time.time() + 15
        while time.time() < deadline and _online(base) is not True:
            time.sleep(0.2)

        chunks = []
        saw_done = False
        with httpx.stream(
                "POST", base + "/api/generate",
                json={"model": "tiny-cpu", "prompt": "a", "stream": True,
                      "options": {"num_predict": 2000}},
                timeout=60.0) as r:
            assert r.status_code == 200
            for line in r.iter_lines():
                chunks.append(line)
                if '"done": true' in line or '"done":true' in line:
                    saw_done = True
                    break
                if len(chunks) == 3:
                    worker.kill()        # mid-stream death
        assert chunks and not saw_done, \
            f"stream should truncate without done: {chunks[-1:]}"
        assert server.poll() is None, "dispatcher died with the worker"
        assert httpx.get(base + "/health", timeout=5).text == "OK"
        deadline = time.time() + 15
        while time.time() < deadline and _online(base) is not False:
            time.sleep(0.2)
        assert _online(base) is False
    finally:
        server.terminate()
        if worker.poll() is None:
            worker.terminate()
