"""E2E: the native ollamamq-server binary against fake Ollama + LM Studio.

Equivalent of the reference's tests/e2e/run.sh 17-check suite (admin
inventory, load with name resolution + exact wire bodies, instance-id
unload, error paths, "any" selector, busy/duplicate 409) plus streaming
proxy, auth, blocklist and scheduling checks — driven over real HTTP
against the compiled C++ dispatcher.
"""
import json
import os
import subprocess
import sys
import time

import httpx
import pytest

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
from mocks import MockFleet

BIN = os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
                   "ollamamq_amd", "csrc", "dispatcher", "ollamamq-server")


def wait_http(url, timeout=10.0):
    t0 = time.time()
    while time.time() - t0 < timeout:
        try:
            r = httpx.get(url, timeout=1.0)
            if r.status_code == 200:
                return True
        except Exception:
            pass
        time.sleep(0.1)
    return False


class Proxy:
    def __init__(self, backend_urls, tmpdir, env=None, extra=None):
        self.port = None
        e = dict(os.environ)
        if env:
            e.update(env)
        self.proc = subprocess.Popen(
            [BIN, "--no-tui", "-p", "0", "-o", ",".join(backend_urls),
             "-c", os.path.join(str(tmpdir), "absent.yaml")]
            + (extra or []),
            stderr=subprocess.PIPE, cwd=str(tmpdir), env=e, text=True)
        # parse the actual port from the startup line
        line = self.proc.stderr.readline()
        assert "listening on" in line, line
        self.port = int(line.rsplit(":", 1)[1].split()[0])
        self.base = f"http://127.0.0.1:{self.port}"
        assert wait_http(self.base + "/health")

    def stop(self):
        self.proc.terminate()
        try:
            self.proc.wait(timeout=5)
        except subprocess.TimeoutExpired:
            self.proc.kill()


@pytest.fixture(scope="module")
def fleet():
    if not os.path.exists(BIN):
        subprocess.run([sys.executable, "-m", "ollamamq_amd.build"],
                       check=True)
    f = MockFleet(load_delay=1.0)
    yield f
    f.stop()


@pytest.fixture(scope="module")
def proxy(fleet, tmp_path_factory):
    p = Proxy([fleet.ollama_url, fleet.lmstudio_url],
              tmp_path_factory.mktemp("proxy"))
    # give the health loop a beat to probe both backends
    deadline = time.time() + 15
    while time.time() < deadline:
        r = httpx.get(p.base + "/admin/models").json()
        if all(b["online"] for b in r) and \
                all(b["api"] != "unknown" for b in r):
            break
        time.sleep(0.3)
    yield p
    p.stop()


def test_health(proxy):
    r = httpx.get(proxy.base + "/health")
    assert r.status_code == 200 and r.text == "OK"


def test_admin_inventory(proxy):
    r = httpx.get(proxy.base + "/admin/models").json()
    assert len(r) == 2
    b0, b1 = r
    assert b0["online"] and b1["online"]
    assert b0["api"] == "ollama"
    assert "llama3:latest" in b0["available_models"]
    assert b1["api"] == "openai" and b1["lmstudio"]
    assert "mock/qwen2-7b-instruct" in b1["available_models"]
    assert "mock/qwen2-7b-instruct" in b1["loaded_models"]


def test_load_resolution_and_wire_body(proxy, fleet):
    # "llama3" resolves to "llama3:latest"; exact generate body with
    # keep_alive + num_ctx (reference run.sh check)
    r = httpx.post(proxy.base + "/admin/models/load",
                   json={"model": "llama3", "backend": 0, "num_ctx": 8192})
    assert r.status_code == 202, r.text
    assert r.json()["model"] == "llama3:latest"
    deadline = time.time() + 10
    while time.time() < deadline:
        calls = fleet.recorder.of("ollama", "/api/generate")
        if calls:
            break
        time.sleep(0.1)
    body = calls[-1]["body"]
    assert body["model"] == "llama3:latest"
    assert body["keep_alive"] == 86400
    assert body["options"]["num_ctx"] == 8192


def test_unload_ollama_keepalive_zero(proxy, fleet):
    # wait until llama3 shows loaded from the previous test's probe
    deadline = time.time() + 10
    while time.time() < deadline:
        r = httpx.get(proxy.base + "/admin/models").json()
        if "llama3:latest" in r[0]["loaded_models"]:
            break
        time.sleep(0.2)
    r = httpx.post(proxy.base + "/admin/models/unload",
                   json={"model": "llama3:latest", "backend": 0})
    assert r.status_code == 202, r.text
    deadline = time.time() + 10
    while time.time() < deadline:
        calls = [c for c in fleet.recorder.of("ollama", "/api/generate")
                 if c["body"].get("keep_alive") == 0]
        if calls:
            break
        time.sleep(0.1)
    assert calls[-1]["body"]["model"] == "llama3:latest"


def test_lmstudio_load_unload_instance_id(proxy, fleet):
    r = httpx.post(proxy.base + "/admin/models/load",
                   json={"model": "mock/llama-8b", "backend": 1,
                         "num_ctx": 4096})
    assert r.status_code == 202, r.text
    deadline = time.time() + 10
    while time.time() < deadline:
        calls = fleet.recorder.of("lmstudio", "/api/v1/models/load")
        if calls:
            break
        time.sleep(0.1)
    assert calls[-1]["body"] == {"model": "mock/llama-8b",
                                 "context_length": 4096}
    # wait for op completion + post-op probe
    time.sleep(2.0)
    r = httpx.post(proxy.base + "/admin/models/unload",
                   json={"model": "mock/qwen2-7b-instruct", "backend": 1})
    assert r.status_code == 202, r.text
    deadline = time.time() + 10
    while time.time() < deadline:
        calls = fleet.recorder.of("lmstudio", "/api/v1/models/unload")
        if calls:
            break
        time.sleep(0.1)
    assert calls[-1]["body"]["instance_id"] == "inst-qwen-1"


def test_error_paths(proxy):
    r = httpx.post(proxy.base + "/admin/models/load",
                   json={"model": "definitely-not-a-model", "backend": 0})
    assert r.status_code == 404
    r = httpx.post(proxy.base + "/admin/models/load",
                   json={"model": ""})
    assert r.status_code == 400
    r = httpx.post(proxy.base + "/admin/models/load",
                   json={"model": "llama3", "backend": 99})
    assert r.status_code == 404
    r = httpx.post(proxy.base + "/admin/models/unload",
                   json={"model": "llama3", "backend": 0})
    # not loaded anymore -> 400
    assert r.status_code == 400, r.text


def test_any_selector_picks_resolvable(proxy, fleet):
    r = httpx.post(proxy.base + "/admin/models/load",
                   json={"model": "qwen2.5", "backend": "any"})
    assert r.status_code == 202, r.text
    assert r.json()["backend_index"] == 0   # resolves on backend 0
    assert r.json()["backend"].startswith("http")   # URL (reference shape)
    time.sleep(0.5)


def test_backend_selector_semantics(proxy, fleet):
    """Reference selector rules (control.rs:1179-1240): a numeric STRING
    is an index (not a URL substring — "0" must not match "127.0.0.1"),
    substring match is case-insensitive, "ANY" works."""
    # "1" as a string indexes backend 1 (the LM Studio mock), which does
    # not have qwen2.5 — proving it was treated as an index, because as a
    # substring it would match backend 0's URL (digits in 127.0.0.1)
    r = httpx.post(proxy.base + "/admin/models/load",
                   json={"model": "qwen2.5", "backend": "1"})
    assert r.status_code == 404, r.text          # not on backend 1
    r = httpx.post(proxy.base + "/admin/models/load",
                   json={"model": "qwen2.5", "backend": "ANY"})
    assert r.status_code == 202, r.text
    assert r.json()["backend_index"] == 0
    time.sleep(0.5)
    r = httpx.post(proxy.base + "/admin/models/load",
                   json={"model": "qwen2.5", "backend": "127.0.0.1"})
    assert r.status_code in (202, 409), r.text   # substring hit backend 0
    time.sleep(0.5)
    r = httpx.post(proxy.base + "/admin/models/load",
                   json={"model": "qwen2.5", "backend": "no-such-url"})
    assert r.status_code == 404


def test_busy_backend_409(fleet, tmp_path_factory):
    """A backend mid-request refuses control ops with 409 (reference
    run.sh busy-window check via the 12 s hanging /api/chat mock)."""
    import threading
    slow = MockFleet(chat_delay=6.0)
    p = Proxy([slow.ollama_url], tmp_path_factory.mktemp("busy"))
    try:
        deadline = time.time() + 15
        while time.time() < deadline:
            b = httpx.get(p.base + "/admin/models").json()
            if b and b[0]["online"] and b[0]["api"] != "unknown":
                break
            time.sleep(0.3)
        t = threading.Thread(
            target=lambda: httpx.post(
                p.base + "/api/chat",
                json={"model": "llama3:latest", "messages": []},
                timeout=30),
            daemon=True)
        t.start()
        # wait until the request is actually dispatched onto the backend
        deadline = time.time() + 10
        busy = False
        while time.time() < deadline:
            b = httpx.get(p.base + "/admin/models").json()[0]
            if b.get("active_requests", 0) > 0:
                busy = True
                break
            time.sleep(0.1)
        assert busy, "chat never occupied the backend"
        r = httpx.post(p.base + "/admin/models/load",
                       json={"model": "llama3:latest", "backend": 0},
                       timeout=10)
        assert r.status_code == 409
        assert "busy" in r.json()["error"]
        t.join(timeout=20)
    finally:
        p.stop()
        slow.stop()


def test_duplicate_op_409(proxy, fleet):
    # the 1 s slow LM Studio load opens the race window
    r1 = httpx.post(proxy.base + "/admin/models/load",
                    json={"model": "mock/llama-8b", "backend": 1})
    assert r1.status_code == 202, r1.text
    r2 = httpx.post(proxy.base + "/admin/models/load",
                    json={"model": "mock/llama-8b", "backend": 1})
    assert r2.status_code == 409
    # op visible in admin state during the window
    r = httpx.get(proxy.base + "/admin/models").json()
    op = r[1]["operation"]
    assert op and op["action"] == "load"
    time.sleep(1.5)


def test_proxy_streams_chat(proxy):
    r = httpx.post(proxy.base + "/api/chat",
                   json={"model": "llama3", "messages": []},
                   headers={"X-User-ID": "alice"}, timeout=10.0)
    assert r.status_code == 200
    lines = [json.loads(l) for l in r.text.strip().split("\n")]
    assert lines[-1]["done"] is True
    text = "".join(l.get("message", {}).get("content", "")
                   for l in lines[:-1])
    assert text == "hello from mock"


def test_proxy_unknown_route_404(proxy):
    r = httpx.get(proxy.base + "/definitely/not/a/route")
    assert r.status_code == 404


def test_allow_all_routes_fallback_proxy(fleet, tmp_path_factory):
    """--allow-all-routes: unknown paths are PROXIED to a backend instead
    of 404ing locally (reference main.rs:294-296 fallback router)."""
    p = Proxy([fleet.ollama_url], tmp_path_factory.mktemp("allr"),
              extra=["--allow-all-routes"])
    try:
        deadline = time.time() + 15
        while time.time() < deadline:
            r = httpx.get(p.base + "/admin/models").json()
            if r and r[0]["online"]:
                break
            time.sleep(0.2)
        r = httpx.get(p.base + "/definitely/not/a/route", timeout=30)
        # the mock backend answered (its own 404 JSON), so the request
        # was proxied rather than rejected by the dispatcher's router
        assert r.status_code == 404
        assert r.json() == {"error": "not found"}
    finally:
        p.stop()


def test_model_aware_routing(proxy):
    # unroutable model parks in queue, then stuck-times-out; use a fresh
    # user so counters don't interfere
    r = httpx.post(proxy.base + "/api/chat",
                   json={"model": "qwen2.5:7b", "messages": []},
                   headers={"X-User-ID": "bob"}, timeout=10.0)
    assert r.status_code == 200


def test_auth_required_when_key_set(fleet, tmp_path_factory):
    p = Proxy([fleet.ollama_url], tmp_path_factory.mktemp("auth"),
              env={"OLLAMA_MQ_API_KEY": "sekrit"})
    try:
        assert httpx.get(p.base + "/health").status_code == 200  # exempt
        assert httpx.get(p.base + "/admin/models").status_code == 401
        r = httpx.get(p.base + "/admin/models",
                      headers={"X-API-Key": "sekrit"})
        assert r.status_code == 200
        r = httpx.get(p.base + "/admin/models",
                      headers={"Authorization": "bearer sekrit"})
        assert r.status_code == 200
        r = httpx.get(p.base + "/admin/models",
                      headers={"Authorization": "Bearer wrong"})
        assert r.status_code == 401
    finally:
        p.stop()


def test_stuck_timeout_503(fleet, tmp_path_factory):
    p = Proxy([fleet.ollama_url], tmp_path_factory.mktemp("stuck"),
              extra=["--stuck-timeout", "1"])
    try:
        t0 = time.time()
        r = httpx.post(p.base + "/api/chat",
                       json={"model": "no-such-model-anywhere"},
                       headers={"X-User-ID": "carol"}, timeout=30.0)
        assert r.status_code == 503
        assert time.time() - t0 < 10
    finally:
        p.stop()


def test_blocklist_persistence(fleet, tmp_path_factory):
    """blocked_items.json uses the REFERENCE's serde field names
    ("users"/"ips", src/dispatcher.rs:22-25) so a migrated file works
    unchanged; our earlier "blocked_*" spelling is still accepted."""
    for fields in ({"users": ["evil"], "ips": []},
                   {"blocked_users": ["evil"], "blocked_ips": []}):
        tmp = tmp_path_factory.mktemp("block")
        with open(os.path.join(str(tmp), "blocked_items.json"), "w") as f:
            json.dump(fields, f)
        p = Proxy([fleet.ollama_url], tmp)
        try:
            r = httpx.post(p.base + "/api/chat", json={"model": "llama3"},
                           headers={"X-User-ID": "evil"})
            assert r.status_code == 403, fields
            r = httpx.post(p.base + "/api/chat",
                           json={"model": "llama3", "messages": []},
                           headers={"X-User-ID": "good"}, timeout=10.0)
            assert r.status_code == 200
        finally:
            p.stop()


def test_admin_stats_surface(proxy):
    r = httpx.get(proxy.base + "/admin/stats")
    assert r.status_code == 200
    st = r.json()
    assert st["uptime_s"] > 0
    assert "queue_wait" in st and "p50_ms" in st["queue_wait"]
    assert any(u["processed"] > 0 for u in st["users"])
    assert len(st["backends"]) == 2


def test_metrics_prometheus(fleet, tmp_path_factory):
    p = Proxy([fleet.ollama_url], tmp_path_factory.mktemp("metrics"))
    try:
        # generate one request so counters move
        httpx.post(p.base + "/api/generate",
                   json={"model": "llama3:8b", "prompt": "x",
                         "stream": False},
                   headers={"X-User-ID": "m1"}, timeout=30.0)
        r = httpx.get(p.base + "/metrics", timeout=10.0)
        assert r.status_code == 200
        assert "text/plain" in r.headers["content-type"]
        body = r.text
        assert "ollamamq_requests_processed_total" in body
        assert "ollamamq_backend_online{url=" in body
        assert 'ollamamq_queue_wait_ms{quantile="0.5"}' in body
        # processed counter reflects the request above
        line = [l for l in body.splitlines()
                if l.startswith("ollamamq_requests_processed_total")][0]
        assert float(line.split()[-1]) >= 1
    finally:
        p.stop()


def test_cli_help_and_unknown_flag():
    r = subprocess.run([BIN, "--help"], capture_output=True, text=True,
                       timeout=15)
    assert r.returncode == 0
    for flag in ("-p, --port", "--workers", "--stuck-timeout",
                 "--probe-interval-ms", "--no-tui"):
        assert flag in r.stdout
    r2 = subprocess.run([BIN, "--bogus-flag"], capture_output=True,
                        text=True, timeout=15)
    assert r2.returncode == 2
    assert "unknown flag" in r2.stderr
