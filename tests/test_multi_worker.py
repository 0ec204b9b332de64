"""Config-3 semantics on CPU: two engine workers as independent backends;
the scheduler least-connections-spreads users across them and both serve
(the launch.py topology, minus GPUs)."""
import json
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


@pytest.fixture(scope="module")
def node(tmp_path_factory):
    if not os.path.exists(BIN):
        subprocess.run([sys.executable, "-m", "ollamamq_amd.build"],
                       check=True)
    tmp = tmp_path_factory.mktemp("node")
    socks, workers = [], []
    for i in range(2):
        sock = os.path.join(str(tmp), f"w{i}.sock")
        socks.append(sock)
        workers.append(subprocess.Popen(
            [sys.executable, "-m", "ollamamq_amd.engine.worker",
             "--socket", sock, "--model", "tiny-cpu", "--max-ctx", "256",
             "--max-batch", "4"],
            cwd=REPO, stdout=subprocess.DEVNULL,
            stderr=subprocess.STDOUT))
    for s in socks:
        assert _wait_socket(s)
    server = subprocess.Popen(
        [BIN, "--no-tui", "-p", "0", "-w", ",".join(socks),
         "--probe-interval-ms", "500",
         "-c", os.path.join(str(tmp), "absent.yaml")],
        stderr=subprocess.PIPE, cwd=str(tmp), text=True)
    line = server.stderr.readline()
    port = int(line.rsplit(":", 1)[1].split()[0])
    base = f"http://127.0.0.1:{port}"
    deadline = time.time() + 20
    while time.time() < deadline:
        try:
            r = httpx.get(base + "/admin/models", timeout=2.0).json()
            if all(b["online"] for b in r["backends"]):
                break
        except Exception:
            pass
        time.sleep(0.2)
    yield base
    server.terminate()
    for w in workers:
        w.terminate()


def test_both_backends_online(node):
    r = httpx.get(node + "/admin/models").json()
    assert len(r["backends"]) == 2
    assert all(b["online"] for b in r["backends"])


def test_load_spreads_across_workers(node):
    import concurrent.futures as cf

    def one(i):
        r = httpx.post(node + "/api/generate",
                       json={"model": "tiny-cpu", "prompt": "x",
                             "stream": False,
                             "options": {"num_predict": 3}},
                       headers={"X-User-ID": f"mw{i}"}, timeout=120.0)
        return r.status_code

    with cf.ThreadPoolExecutor(12) as ex:
        codes = list(ex.map(one, range(24)))
    assert codes == [200] * 24
    st = httpx.get(node + "/admin/stats").json()
    counts = [b["processed_count"] for b in st["backends"]]
    assert sum(counts) >= 24
    # least-connections must have used BOTH workers substantially
    assert min(counts) >= 5, counts
