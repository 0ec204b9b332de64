"""Full stack ON the GPU: C++ dispatcher -> UDS -> engine worker running
the tiny preset on cuda:0 with the real HIP kernels (graphed decode).
This is the round-end hardware check of the serving path itself."""
import json
import os
import socket
import subprocess
import sys
import time

import httpx
import pytest

pytestmark = pytest.mark.gpu

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
BIN = os.path.join(REPO, "ollamamq_amd", "csrc", "dispatcher",
                   "ollamamq-server")


def _wait_socket(path, timeout=120):
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
def gpu_stack(tmp_path_factory):
    assert os.path.exists(BIN), "native server not built"
    tmp = tmp_path_factory.mktemp("gstack")
    sock = os.path.join(str(tmp), "wg.sock")
    worker = subprocess.Popen(
        [sys.executable, "-m", "ollamamq_amd.engine.worker",
         "--socket", sock, "--gpu", "0", "--model", "tiny",
         "--max-ctx", "512", "--max-batch", "8"],
        cwd=REPO, stdout=subprocess.PIPE, stderr=subprocess.STDOUT,
        text=True)
    assert _wait_socket(sock), "GPU worker did not come up"
    server = subprocess.Popen(
        [BIN, "--no-tui", "-p", "0", "-w", sock,
         "-c", os.path.join(str(tmp), "absent.yaml")],
        stderr=subprocess.PIPE, cwd=str(tmp), text=True)
    line = server.stderr.readline()
    port = int(line.rsplit(":", 1)[1].split()[0])
    base = f"http://127.0.0.1:{port}"
    deadline = time.time() + 30
    while time.time() < deadline:
        try:
            r = httpx.get(base + "/admin/models", timeout=2.0).json()
            if r and r[0]["online"]:
                break
        except Exception:
            pass
        time.sleep(0.2)
    yield base
    server.terminate()
    worker.terminate()


def test_gpu_worker_serves_ollama_chat(gpu_stack):
    r = httpx.post(gpu_stack + "/api/chat",
                   json={"model": "tiny",
                         "messages": [{"role": "user", "content": "hi"}],
                         "options": {"num_predict": 8}},
                   headers={"X-User-ID": "g1"}, timeout=180.0)
    assert r.status_code == 200, r.text
    lines = [json.loads(l) for l in r.text.strip().split("\n")]
    assert lines[-1]["done"] is True
    assert lines[-1]["eval_count"] >= 1
    st = httpx.get(gpu_stack + "/admin/models").json()[0]
    assert st["api"] == "both" and "tiny" in st["loaded_models"]


def test_gpu_worker_concurrent(gpu_stack):
    import concurrent.futures as cf

    def one(u):
        r = httpx.post(gpu_stack + "/api/generate",
                       json={"model": "tiny", "prompt": "abc" * 10,
                             "stream": False,
                             "options": {"num_predict": 4}},
                       headers={"X-User-ID": f"gu{u}"}, timeout=180.0)
        return r.status_code

    with cf.ThreadPoolExecutor(6) as ex:
        assert list(ex.map(one, range(6))) == [200] * 6


def test_gpu_worker_embeddings(gpu_stack):
    r = httpx.post(gpu_stack + "/api/embed",
                   json={"model": "tiny", "input": "hello world"},
                   headers={"X-User-ID": "ge"}, timeout=120.0)
    assert r.status_code == 200, r.text
    emb = r.json()["embeddings"]
    assert len(emb) == 1 and len(emb[0]) == 512  # tiny hidden size
    assert any(abs(x) > 1e-6 for x in emb[0])


def test_gpu_worker_openai_sse(gpu_stack):
    with httpx.stream(
            "POST", gpu_stack + "/v1/chat/completions",
            json={"model": "tiny", "stream": True,
                  "messages": [{"role": "user", "content": "hi"}],
                  "max_tokens": 6},
            headers={"X-User-ID": "gs"}, timeout=120.0) as r:
        assert r.status_code == 200
        lines = [l for l in r.iter_lines() if l.startswith("data: ")]
    assert lines[-1].strip() == "data: [DONE]"
    import json as j
    chunks = [j.loads(l[6:]) for l in lines[:-1]]
    assert any(c["choices"][0]["delta"].get("content") for c in chunks)
