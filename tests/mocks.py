"""In-process fake Ollama + LM Studio backends for dispatcher E2E tests.

Same role as the reference's mock servers (reference
tests/e2e/mock_backends.py — studied for wire shapes, re-implemented):
fake Ollama speaks /api/tags, /api/ps, /api/generate (with keep_alive:0
unload semantics), streaming /api/chat; fake LM Studio speaks /v1/models,
/api/v1/models and the native load/unload.  Every control call is recorded
for exact wire-body assertions.
"""
import json
import threading
import time
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer


class _Recorder:
    def __init__(self):
        self.calls = []
        self.lock = threading.Lock()

    def add(self, server, path, body):
        with self.lock:
            self.calls.append({"server": server, "path": path, "body": body})

    def of(self, server=None, path=None):
        with self.lock:
            return [c for c in self.calls
                    if (server is None or c["server"] == server)
                    and (path is None or c["path"] == path)]


def make_ollama(recorder, chat_delay=0.0):
    class OllamaHandler(BaseHTTPRequestHandler):
        loaded = {"qwen2.5:7b": 4096}     # model -> ctx (stateful)
        available = ["llama3:latest", "qwen2.5:7b"]

        def log_message(self, *a):
            pass

        def _json(self, code, obj):
            data = json.dumps(obj).encode()
            self.send_response(code)
            self.send_header("Content-Type", "application/json")
            self.send_header("Content-Length", str(len(data)))
            self.end_headers()
            self.wfile.write(data)

        def do_GET(self):
            if self.path == "/api/tags":
                self._json(200, {"models": [{"name": m}
                                            for m in self.available]})
            elif self.path == "/api/ps":
                self._json(200, {"models": [
                    {"name": m, "context_length": c}
                    for m, c in self.loaded.items()]})
            else:
                self._json(404, {"error": "not found"})

        def do_POST(self):
            n = int(self.headers.get("Content-Length", 0))
            body = json.loads(self.rfile.read(n) or b"{}")
            if self.path == "/api/generate":
                recorder.add("ollama", self.path, body)
                model = body.get("model", "")
                if body.get("keep_alive") == 0:
                    self.loaded.pop(model, None)
                    self._json(200, {"model": model, "done": True,
                                     "done_reason": "unload"})
                else:
                    ctx = (body.get("options") or {}).get("num_ctx", 4096)
                    self.loaded[model] = ctx
                    self._json(200, {"model": model, "done": True,
                                     "response": ""})
            elif self.path == "/api/chat":
                if chat_delay:
                    time.sleep(chat_delay)
                # streamed JSON-lines answer (chunked)
                self.send_response(200)
                self.send_header("Content-Type", "application/x-ndjson")
                self.send_header("Transfer-Encoding", "chunked")
                self.end_headers()
                for word in ["hello", " from", " mock"]:
                    data = json.dumps({
                        "model": body.get("model"),
                        "message": {"role": "assistant", "content": word},
                        "done": False}).encode() + b"\n"
                    self.wfile.write(b"%x\r\n%s\r\n" % (len(data), data))
                    self.wfile.flush()
                data = json.dumps({"model": body.get("model"),
                                   "done": True}).encode() + b"\n"
                self.wfile.write(b"%x\r\n%s\r\n0\r\n\r\n" % (len(data), data))
            else:
                self._json(404, {"error": "not found"})

    return OllamaHandler


def make_lmstudio(recorder, load_delay=0.0):
    class LMHandler(BaseHTTPRequestHandler):
        instances = {"mock/qwen2-7b-instruct": "inst-qwen-1"}

        def log_message(self, *a):
            pass

        def _json(self, code, obj):
            data = json.dumps(obj).encode()
            self.send_response(code)
            self.send_header("Content-Type", "application/json")
            self.send_header("Content-Length", str(len(data)))
            self.end_headers()
            self.wfile.write(data)

        def do_GET(self):
            if self.path == "/v1/models":
                self._json(200, {"data": [{"id": "mock/qwen2-7b-instruct"},
                                          {"id": "mock/llama-8b"}]})
            elif self.path == "/api/v1/models":
                models = []
                for key in ["mock/qwen2-7b-instruct", "mock/llama-8b"]:
                    inst = self.instances.get(key)
                    models.append({
                        "key": key, "id": key,
                        "display_name": key.split("/")[1].replace("-", " "),
                        "loaded_instances":
                            [{"id": inst}] if inst else [],
                    })
                self._json(200, {"models": models})
            elif self.path == "/":
                self._json(200, {"ok": True})
            else:
                self._json(404, {"error": "not found"})

        def do_POST(self):
            n = int(self.headers.get("Content-Length", 0))
            body = json.loads(self.rfile.read(n) or b"{}")
            if self.path == "/api/v1/models/load":
                recorder.add("lmstudio", self.path, body)
                if load_delay:
                    time.sleep(load_delay)
                key = body.get("model")
                self.instances[key] = "inst-" + key.split("/")[-1]
                self._json(200, {"type": "llm",
                                 "instance_id": self.instances[key],
                                 "status": "loaded"})
            elif self.path == "/api/v1/models/unload":
                recorder.add("lmstudio", self.path, body)
                iid = body.get("instance_id")
                for k, v in list(self.instances.items()):
                    if v == iid or k == iid:
                        del self.instances[k]
                self._json(200, {"instance_id": iid})
            else:
                self._json(404, {"error": "not found"})

    return LMHandler


class MockFleet:
    """Starts a fake Ollama and a fake LM Studio on ephemeral ports."""

    def __init__(self, chat_delay=0.0, load_delay=0.0):
        self.recorder = _Recorder()
        self.ollama = ThreadingHTTPServer(
            ("127.0.0.1", 0), make_ollama(self.recorder, chat_delay))
        self.lmstudio = ThreadingHTTPServer(
            ("127.0.0.1", 0), make_lmstudio(self.recorder, load_delay))
        for srv in (self.ollama, self.lmstudio):
            threading.Thread(target=srv.serve_forever, daemon=True).start()

    @property
    def ollama_url(self):
        return f"http://127.0.0.1:{self.ollama.server_address[1]}"

    @property
    def lmstudio_url(self):
        return f"http://127.0.0.1:{self.lmstudio.server_address[1]}"

    def stop(self):
        self.ollama.shutdown()
        self.lmstudio.shutdown()
