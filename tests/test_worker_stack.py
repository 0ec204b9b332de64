"""Full stack on CPU: C++ dispatcher -> UDS -> Python engine worker (tiny
model, reference-free wire compatibility checks for Ollama + OpenAI)."""
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
            s = socket.socket(socket.AF_UNIX, socket.SOCK_STREAM)
            try:
                s.connect(path)
                s.close()
                return True
            except OSError:
                pass
        time.sleep(0.2)
    return False


@pytest.fixture(scope="module")
def stack(tmp_path_factory):
    if not os.path.exists(BIN):
        subprocess.run([sys.executable, "-m", "ollamamq_amd.build"],
                       check=True)
    tmp = tmp_path_factory.mktemp("stack")
    sock = os.path.join(str(tmp), "w0.sock")
    worker = subprocess.Popen(
        [sys.executable, "-m", "ollamamq_amd.engine.worker",
         "--socket", sock, "--model", "tiny-cpu", "--max-ctx", "256",
         "--max-batch", "4"],
        cwd=REPO, stdout=subprocess.PIPE, stderr=subprocess.STDOUT,
        text=True)
    assert _wait_socket(sock), "worker did not come up"
    server = subprocess.Popen(
        [BIN, "--no-tui", "-p", "0", "-w", sock,
         "-c", os.path.join(str(tmp), "absent.yaml")],
        stderr=subprocess.PIPE, cwd=str(tmp), text=True)
    line = server.stderr.readline()
    port = int(line.rsplit(":", 1)[1].split()[0])
    base = f"http://127.0.0.1:{port}"
    # wait for probe
    deadline = time.time() + 20
    while time.time() < deadline:
        try:
            r = httpx.get(base + "/admin/models").json()
            if r and r[0]["online"]:
                break
        except Exception:
            pass
        time.sleep(0.2)
    yield base
    server.terminate()
    worker.terminate()


def test_worker_probed(stack):
    b = httpx.get(stack + "/admin/models").json()[0]
    assert b["online"]
    assert b["api"] == "both"
    assert "tiny-cpu" in b["available_models"]
    assert "tiny-cpu" in b["loaded_models"]


def test_ollama_chat_streams_through_stack(stack):
    r = httpx.post(stack + "/api/chat",
                   json={"model": "tiny-cpu",
                         "messages": [{"role": "user", "content": "hi"}],
                         "options": {"num_predict": 6}},
                   headers={"X-User-ID": "u1"}, timeout=120.0)
    assert r.status_code == 200, r.text
    lines = [json.loads(l) for l in r.text.strip().split("\n")]
    assert lines[-1]["done"] is True
    assert lines[-1]["eval_count"] >= 1
    assert all("message" in l for l in lines[:-1])


def test_ollama_generate_nonstream(stack):
    r = httpx.post(stack + "/api/generate",
                   json={"model": "tiny-cpu", "prompt": "abc",
                         "stream": False, "options": {"num_predict": 4}},
                   headers={"X-User-ID": "u2"}, timeout=120.0)
    assert r.status_code == 200, r.text
    obj = json.loads(r.text.strip())
    assert obj["done"] is True
    assert isinstance(obj["response"], str)
    # Ollama final-chunk schema: counts, durations, done_reason (here
    # "length": generation ran to num_predict)
    assert obj["done_reason"] == "length"
    assert obj["eval_count"] == 4 and obj["prompt_eval_count"] == 3
    assert obj["total_duration"] >= obj["eval_duration"] >= 0
    assert "prompt_eval_duration" in obj and "load_duration" in obj


def test_openai_chat_completion(stack):
    r = httpx.post(stack + "/v1/chat/completions",
                   json={"model": "tiny-cpu", "max_tokens": 5,
                         "messages": [{"role": "user", "content": "yo"}]},
                   headers={"X-User-ID": "u3"}, timeout=120.0)
    assert r.status_code == 200, r.text
    obj = r.json()
    assert obj["choices"][0]["message"]["role"] == "assistant"
    assert obj["usage"]["completion_tokens"] >= 1
    # OpenAI SDK pydantic models require these
    assert obj["id"].startswith("chatcmpl-") and obj["created"] > 0
    assert obj["choices"][0]["finish_reason"] in ("stop", "length")


def test_openai_stream_sse(stack):
    with httpx.stream(
            "POST", stack + "/v1/chat/completions",
            json={"model": "tiny-cpu", "max_tokens": 5, "stream": True,
                  "messages": [{"role": "user", "content": "yo"}]},
            headers={"X-User-ID": "u4"}, timeout=120.0) as r:
        assert r.status_code == 200
        body = "".join(r.iter_text())
    assert "data: " in body
    assert body.strip().endswith("data: [DONE]")


def test_openai_stream_include_usage(stack):
    """stream_options.include_usage: a usage-only chunk arrives before
    [DONE] (OpenAI streaming spec)."""
    with httpx.stream(
            "POST", stack + "/v1/chat/completions",
            json={"model": "tiny-cpu", "max_tokens": 4, "stream": True,
                  "stream_options": {"include_usage": True},
                  "messages": [{"role": "user", "content": "yo"}]},
            timeout=120.0) as r:
        assert r.status_code == 200
        body = "".join(r.iter_text())
    chunks = [json.loads(l[6:]) for l in body.splitlines()
              if l.startswith("data: ") and l != "data: [DONE]"]
    assert chunks[-1]["choices"] == []
    assert chunks[-1]["usage"]["completion_tokens"] == 4


def test_tags_and_ps_served_by_worker(stack):
    r = httpx.get(stack + "/api/tags", timeout=30.0)
    assert r.status_code == 200
    names = [m["name"] for m in r.json()["models"]]
    assert "tiny-cpu" in names
    r = httpx.get(stack + "/api/version", timeout=30.0)
    assert "ollamamq-amd" in r.json()["version"]


def test_concurrent_users_fair_share(stack):
    import concurrent.futures as cf

    def one(user):
        r = httpx.post(stack + "/api/generate",
                       json={"model": "tiny-cpu", "prompt": "x" * 5,
                             "stream": False,
                             "options": {"num_predict": 3}},
                       headers={"X-User-ID": user}, timeout=120.0)
        return r.status_code

    with cf.ThreadPoolExecutor(8) as ex:
        codes = list(ex.map(one, [f"user{i}" for i in range(8)]))
    assert codes == [200] * 8


def test_client_disconnect_frees_backend(stack):
    """Client hangs up mid-stream: the hangup propagates through the
    dispatcher to the worker (engine cancel, KV freed) and the backend's
    active_requests returns to 0 so new requests dispatch immediately."""
    with httpx.stream(
            "POST", stack + "/api/generate",
            json={"model": "tiny-cpu", "prompt": "c", "stream": True,
                  "options": {"num_predict": 5000}},
            headers={"X-User-ID": "canceller"}, timeout=60.0) as r:
        assert r.status_code == 200
        for i, _ in enumerate(r.iter_lines()):
            if i >= 2:
                break               # context exit closes the connection
    deadline = time.time() + 20
    freed = False
    while time.time() < deadline:
        stats = httpx.get(stack + "/admin/stats", timeout=5).json()
        if stats.get("processing", 1) == 0:
            freed = True
            break
        time.sleep(0.2)
    assert freed, f"backend never freed after disconnect: {stats}"
    r = httpx.post(stack + "/api/generate",
                   json={"model": "tiny-cpu", "prompt": "d",
                         "stream": False, "options": {"num_predict": 2}},
                   timeout=60.0)
    assert r.status_code == 200


def test_malformed_options_400(stack):
    """Garbage option types (temperature: "hot") are a client error:
    clean 400 through the whole stack, and the worker keeps serving."""
    r = httpx.post(stack + "/api/generate",
                   json={"model": "tiny-cpu", "prompt": "x",
                         "stream": False,
                         "options": {"temperature": "hot"}},
                   timeout=60.0)
    assert r.status_code == 400
    assert "invalid options" in r.json()["error"]
    r = httpx.post(stack + "/api/generate",
                   json={"model": "tiny-cpu", "prompt": "x",
                         "stream": False, "options": {"num_predict": 2}},
                   timeout=60.0)
    assert r.status_code == 200


def test_openai_error_object_format(stack):
    """/v1/* errors use the OpenAI error-object shape so SDKs can parse
    them; /api/* errors stay Ollama's string form."""
    r = httpx.post(stack + "/v1/chat/completions",
                   json={"model": "tiny-cpu", "temperature": "hot",
                         "messages": [{"role": "user", "content": "x"}]},
                   timeout=60.0)
    assert r.status_code == 400
    err = r.json()["error"]
    assert isinstance(err, dict) and "message" in err and "type" in err
    r = httpx.post(stack + "/api/generate",
                   json={"model": "tiny-cpu", "prompt": "x",
                         "options": {"temperature": "hot"}},
                   timeout=60.0)
    assert r.status_code == 400
    assert isinstance(r.json()["error"], str)


def test_oversized_prompt_truncates_not_500(stack):
    """A prompt longer than the context window degrades (front-truncated
    in the engine) instead of erroring."""
    r = httpx.post(stack + "/api/generate",
                   json={"model": "tiny-cpu", "prompt": "y" * 5000,
                         "stream": False, "options": {"num_predict": 2}},
                   timeout=120.0)
    assert r.status_code == 200


def test_num_predict_negative_bounded_by_ctx(stack):
    """num_predict: -1 (Ollama infinite) must terminate — bounded by the
    context window — and return a non-empty response."""
    r = httpx.post(stack + "/api/generate",
                   json={"model": "tiny-cpu", "prompt": "z",
                         "stream": False,
                         "options": {"num_predict": -1}},
                   timeout=180.0)
    assert r.status_code == 200
    assert r.json()["done"] is True


def test_admin_load_unload_on_worker(stack):
    r = httpx.post(stack + "/admin/models/load",
                   json={"model": "tiny", "backend": 0, "num_ctx": 256})
    assert r.status_code == 202, r.text
    deadline = time.time() + 60
    ok = False
    while time.time() < deadline:
        st = httpx.get(stack + "/admin/models").json()[0]
        if "tiny" in st["loaded_models"] and st["operation"] is None:
            ok = True
            break
        time.sleep(0.3)
    assert ok, "tiny did not load"
    r = httpx.post(stack + "/admin/models/unload",
                   json={"model": "tiny", "backend": 0})
    assert r.status_code == 202, r.text
    deadline = time.time() + 30
    while time.time() < deadline:
        st = httpx.get(stack + "/admin/models").json()[0]
        if "tiny" not in st["loaded_models"]:
            return
        time.sleep(0.3)
    assert False, "tiny did not unload"


def test_embeddings_endpoints(stack):
    r = httpx.post(stack + "/api/embed",
                   json={"model": "tiny-cpu", "input": ["hello", "world"]},
                   headers={"X-User-ID": "e1"}, timeout=120.0)
    assert r.status_code == 200, r.text
    obj = r.json()
    assert len(obj["embeddings"]) == 2
    assert len(obj["embeddings"][0]) == 256     # tiny-cpu hidden
    r = httpx.post(stack + "/v1/embeddings",
                   json={"model": "tiny-cpu", "input": "hello"},
                   headers={"X-User-ID": "e2"}, timeout=120.0)
    assert r.status_code == 200, r.text
    assert r.json()["data"][0]["object"] == "embedding"
    r = httpx.post(stack + "/api/embeddings",
                   json={"model": "tiny-cpu", "prompt": "hello"},
                   headers={"X-User-ID": "e3"}, timeout=120.0)
    assert r.status_code == 200, r.text
    assert isinstance(r.json()["embedding"], list)


def test_stop_sequences(stack):
    # with a stop string that appears immediately (any printable output),
    # generation must end early rather than exhausting num_predict
    t0 = time.time()
    r = httpx.post(stack + "/api/generate",
                   json={"model": "tiny-cpu", "prompt": "q",
                         "stream": False,
                         "options": {"num_predict": 40,
                                     "stop": ["a", "b", "c", "d", "e", "f",
                                              "0", "1", "2", "3", "\\x"]}},
                   headers={"X-User-ID": "s1"}, timeout=120.0)
    assert r.status_code == 200, r.text
    obj = json.loads(r.text.strip())
    assert obj["done"] is True


def test_multimodal_fields_tolerated(stack):
    import base64
    img = base64.b64encode(b"\x89PNG fake image bytes" * 100).decode()
    r = httpx.post(stack + "/api/generate",
                   json={"model": "tiny-cpu", "prompt": "what is this?",
                         "images": [img], "stream": False,
                         "options": {"num_predict": 3}},
                   headers={"X-User-ID": "mm1"}, timeout=120.0)
    assert r.status_code == 200, r.text
    obj = json.loads(r.text.strip())
    assert obj["done"] is True and "created_at" in obj


def test_request_num_ctx_shapes_on_demand_load(stack):
    """options.num_ctx on a generate request sets the context of an
    on-demand model load (Ollama parity; resident models keep their
    context); /api/ps reports it."""
    # ensure not resident (earlier tests may have loaded it)
    httpx.post(stack + "/admin/models/unload",
               json={"model": "tiny-cpu", "backend": "any"}, timeout=30.0)
    deadline = time.time() + 20
    while time.time() < deadline:
        ps0 = httpx.get(stack + "/api/ps", timeout=10.0).json()["models"]
        if not any(m["name"] == "tiny-cpu" for m in ps0):
            break
        time.sleep(0.3)
    r = httpx.post(stack + "/api/generate",
                   json={"model": "tiny-cpu", "prompt": "x",
                         "stream": False,
                         "options": {"num_predict": 2, "num_ctx": 128}},
                   headers={"X-User-ID": "nctx"}, timeout=120.0)
    assert r.status_code == 200, r.text
    ps = httpx.post(stack + "/api/ps", json={}, timeout=30.0)
    if ps.status_code != 200:
        ps = httpx.get(stack + "/api/ps", timeout=30.0)
    models = {m["name"]: m for m in ps.json()["models"]}
    assert models["tiny-cpu"]["context_length"] == 128


def test_request_keep_alive_zero_unloads(stack):
    """Ollama parity: "keep_alive": 0 on a generate request frees the
    model after the response completes (if idle)."""
    r = httpx.post(stack + "/api/generate",
                   json={"model": "tiny-cpu", "prompt": "y",
                         "stream": False, "keep_alive": 0,
                         "options": {"num_predict": 2}},
                   headers={"X-User-ID": "ka0"}, timeout=120.0)
    assert r.status_code == 200, r.text
    deadline = time.time() + 20
    while time.time() < deadline:
        ps = httpx.get(stack + "/api/ps", timeout=10.0).json()["models"]
        if not any(m["name"] == "tiny-cpu" for m in ps):
            return
        time.sleep(0.3)
    raise AssertionError(f"tiny-cpu still resident: {ps}")


def test_multimodal_images_accepted_and_ignored(stack):
    """Ollama multimodal shape: base64 `images` on a chat message must
    not break a text-only model (bytes accepted, ignored)."""
    import base64
    r = httpx.post(stack + "/api/chat",
                   json={"model": "tiny-cpu",
                         "messages": [{"role": "user", "content": "hi",
                                       "images": [base64.b64encode(
                                           b"\x89PNG fake").decode()]}],
                         "stream": False,
                         "options": {"num_predict": 3}},
                   headers={"X-User-ID": "img"}, timeout=120.0)
    assert r.status_code == 200, r.text
    assert r.json()["done"] is True


def test_api_show_known_and_unknown(stack):
    r = httpx.post(stack + "/api/show", json={"model": "tiny-cpu"},
                   timeout=30.0)
    assert r.status_code == 200
    obj = r.json()
    assert obj["model_info"]["llama.embedding_length"] == 256
    assert obj["details"]["family"] == "llama"
    assert "template" in obj and "capabilities" in obj
    # an UNROUTABLE model parks in the dispatcher queue (reference
    # semantics) — the worker-level 404 needs a name that routes (fuzzy
    # substring) but is ambiguous at the worker resolver ("t" matches
    # both tiny and tiny-cpu, and never-guess resolution returns none)
    r = httpx.post(stack + "/api/show", json={"model": "t"},
                   timeout=30.0)
    assert r.status_code == 404


def test_v1_models_item_route(stack):
    r = httpx.get(stack + "/v1/models/tiny-cpu", timeout=30.0)
    assert r.status_code == 200
    obj = r.json()
    assert obj["id"] == "tiny-cpu" and obj["object"] == "model"
    assert obj["created"] > 0 and obj["owned_by"]
