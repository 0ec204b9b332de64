"""Per-GPU engine worker process.

One worker owns one MI355X (or a TP group of them) and serves the
dispatcher over a unix domain socket — the in-process replacement for the
reference's HTTP hop to an external Ollama server (reference
src/dispatcher.rs:742; SURVEY.md §5 "Distributed communication backend":
the hop becomes a direct submit into the engine's continuous batch).

Protocol (newline-framed JSON in, framed stream out):
  {"cmd":"probe"}                  -> one JSON line of state
  {"cmd":"load","model":M,...}     -> {"ok":true} | {"error":...}
  {"cmd":"unload","model":M}       -> same
  {"cmd":"request","method":...,"path":...,"body":"<raw http body>"}
                                   -> {"status":...,"content_type":...}\n
                                      then raw response bytes, EOF = done.
Cancellation: the dispatcher closes the socket; the worker notices the
write failure and cancels the sequence (KV pages freed) — the reference
only dropped bytes, we stop the compute (SURVEY.md §7 hard-part 4).

Run:  python -m ollamamq_amd.engine.worker --socket /tmp/omq0.sock \
          --gpu 0 [--model llama3-8b --max-ctx 4096 --max-batch 64]
"""
from __future__ import annotations

import argparse
import datetime
import json
import os
import queue
import socket
import threading
import time
from typing import Dict, Optional

import torch

from ..models import LlamaModel, PRESETS
from .engine import GenParams, LlamaEngine
from .kvcache import PagedKVCache
from .tokenizer import ByteTokenizer

VERSION = "0.1.0-ollamamq-amd"


class Worker:
    def __init__(self, gpu: int, max_batch: int = 32,
                 default_ctx: int = 4096, models: Optional[list] = None):
        self.gpu = gpu
        self.max_batch = max_batch
        self.default_ctx = default_ctx
        # optional fleet-sharding restriction: this worker only advertises
        # (and loads) these models, so the dispatcher's model routing
        # (smart/fuzzy match against available_models) steers each request
        # to the workers that carry its model (reference
        # src/dispatcher.rs:599-620; BASELINE.json config 5 mixed fleet)
        self.model_filter = list(models) if models else None
        self.device = f"cuda:{gpu}" if torch.cuda.is_available() else "cpu"
        self.dtype = (torch.bfloat16 if torch.cuda.is_available()
                      else torch.float32)
        self.engines: Dict[str, LlamaEngine] = {}
        self.tokenizers: Dict[str, ByteTokenizer] = {}
        self.loaded_ctx: Dict[str, int] = {}
        self.lock = threading.RLock()          # engine-state lock
        self.work_ev = threading.Event()
        self.started = time.time()
        threading.Thread(target=self._engine_loop, daemon=True).start()

    # ---------------------------------------------------------- control
    def available_models(self):
        if self.model_filter is not None:
            return list(self.model_filter)
        if self.device == "cpu":
            return ["tiny", "tiny-cpu"]
        return [n for n in PRESETS if n != "tiny-cpu"]

    def load(self, model: str, num_ctx: int = 0) -> Optional[str]:
        with self.lock:
            if model in self.engines:
                if num_ctx and self.loaded_ctx.get(model) != num_ctx:
                    err = self.unload(model)
                    if err:
                        return err
                else:
                    return None
            cfg = PRESETS.get(model)
            if cfg is None:
                return f"unknown model: {model}"
            ctx = min(num_ctx or self.default_ctx, cfg.max_ctx)
            try:
                if torch.cuda.is_available():
                    torch.cuda.set_device(self.gpu)
                m = LlamaModel(cfg, device=self.device, dtype=self.dtype,
                               seed=1234, fast_init=self.device != "cpu")
                pages = (self.max_batch + 2) * ((ctx + 15) // 16 + 2)
                kv = PagedKVCache.for_model(
                    cfg, n_pages=pages, max_slots=self.max_batch + 2,
                    max_ctx=ctx, device=self.device, dtype=self.dtype)
                eng = LlamaEngine(m, kv, max_batch=self.max_batch)
                eng.warm_graphs()   # pre-capture: no first-request jitter
                self.engines[model] = eng
                self.tokenizers[model] = ByteTokenizer(cfg.vocab)
                self.loaded_ctx[model] = ctx
                return None
            except torch.cuda.OutOfMemoryError:
                return "out of HBM: model + KV pool do not fit"
            except Exception as e:  # pragma: no cover
                return f"load failed: {e}"

    def unload(self, model: str, only_if_idle: bool = False) \
            -> Optional[str]:
        with self.lock:
            eng = self.engines.get(model)
            if eng is None:
                return f"model not loaded: {model}"
            if only_if_idle and eng.has_work():
                return "model busy"
            self.engines.pop(model, None)
            self.tokenizers.pop(model, None)
            self.loaded_ctx.pop(model, None)
            del eng
            if torch.cuda.is_available():
                torch.cuda.empty_cache()
            return None

    def resolve(self, requested: str) -> Optional[str]:
        with self.lock:
            if requested in self.engines:
                return requested
            for name in self.engines:
                if requested.split(":")[0].lower() == name.lower():
                    return name
            # auto-load when a known preset is requested (within this
            # worker's fleet-shard restriction, if any)
            allowed = (set(self.model_filter)
                       if self.model_filter is not None else None)
            if requested in PRESETS and \
                    (allowed is None or requested in allowed):
                return requested
            base = requested.split(":")[0].lower()
            for name in PRESETS:
                if name.lower() == base and \
                        (allowed is None or name in allowed):
                    return name
        return None

    # ----------------------------------------------------------- engine
    def _engine_loop(self):
        while True:
            busy = False
            with self.lock:
                engines = list(self.engines.values())
            for eng in engines:
                with self.lock:
                    if eng.has_work():
                        try:
                            eng.step()
                        except Exception as e:
                            # a failed step (e.g. OOM under load) must not
                            # kill the loop: fail the in-flight sequences
                            # so their streams terminate, keep serving
                            print(f"engine step failed: {e!r}",
                                  flush=True)
                            for seq in (list(eng.waiting)
                                        + list(eng.running)):
                                seq.cancelled = True
                            try:
                                eng.step()   # reap + emit done markers
                            except Exception:
                                pass
                        busy = eng.has_work() or busy
            if not busy:
                self.work_ev.wait(timeout=0.02)
                self.work_ev.clear()

    def generate(self, model: str, prompt_tokens, params: GenParams,
                 on_token, num_ctx: int = 0) -> int:
        with self.lock:
            if model not in self.engines:
                # Ollama parity: a request's options.num_ctx shapes the
                # context of an on-demand load (already-resident models
                # keep their context; /admin reload changes it)
                err = self.load(model, num_ctx)
                if err:
                    raise RuntimeError(err)
            sid = self.engines[model].submit(prompt_tokens, params, on_token)
        self.work_ev.set()
        return sid

    def cancel(self, model: str, sid: int):
        with self.lock:
            eng = self.engines.get(model)
            if eng:
                eng.cancel(sid)


# ------------------------------------------------------------------ wire
def _params_from_body(body: dict, tok: ByteTokenizer,
                      openai: bool) -> GenParams:
    opts = body.get("options") or {}
    max_t = body.get("max_tokens")
    if max_t is None:
        max_t = body.get("max_completion_tokens")
    if max_t is None:
        max_t = opts.get("num_predict")
    max_t = 64 if max_t is None else int(max_t)
    if max_t < 0:
        # Ollama semantics: -1 = infinite, -2 = fill context; both are
        # bounded by the context window in the engine
        max_t = 1 << 30
    return GenParams(
        max_tokens=max_t,
        temperature=float(body.get("temperature",
                                   opts.get("temperature", 0.0)) or 0.0),
        top_k=int(opts.get("top_k", body.get("top_k", 0)) or 0),
        top_p=float(body.get("top_p", opts.get("top_p", 1.0)) or 1.0),
        stop_token=tok.stop_token,
        # coerce here so a junk seed is caught by the caller's 400 path,
        # not deep inside engine.submit
        seed=(int(seed) if (seed := opts.get("seed", body.get("seed")))
              is not None else None),
    )


def _now_iso() -> str:
    return datetime.datetime.now(datetime.timezone.utc).isoformat()


def _prompt_text(body: dict, path: str) -> str:
    if "prompt" in body and isinstance(body["prompt"], str):
        return body["prompt"]
    parts = []
    for m in body.get("messages") or []:
        content = m.get("content", "")
        if isinstance(content, list):   # OpenAI content-part arrays
            content = " ".join(p.get("text", "") for p in content
                               if isinstance(p, dict))
        parts.append(f"<{m.get('role', 'user')}>{content}")
        # multimodal fields (images) are accepted and ignored: random-init
        # text models have no vision tower; the bytes still flow through
    return "\n".join(parts) or " "


class Conn:
    """One dispatcher request served over one accepted UDS connection."""

    def __init__(self, sock: socket.socket, worker: Worker):
        self.sock = sock
        self.worker = worker

    def run(self):
        try:
            line = self._recv_line()
            if line is None:
                return
            msg = json.loads(line)
            cmd = msg.get("cmd")
            if cmd == "probe":
                self._probe()
            elif cmd == "load":
                err = self.worker.load(msg.get("model", ""),
                                       int(msg.get("num_ctx", 0) or 0))
                self._line({"ok": err is None, **({"error": err} if err
                                                 else {})})
            elif cmd == "unload":
                err = self.worker.unload(msg.get("model", ""))
                self._line({"ok": err is None, **({"error": err} if err
                                                 else {})})
            elif cmd == "request":
                self._request(msg)
            else:
                self._line({"error": f"unknown cmd {cmd}"})
        except (BrokenPipeError, ConnectionResetError, OSError):
            pass
        except Exception as e:
            try:
                self._line({"error": str(e)})
            except OSError:
                pass
        finally:
            try:
                self.sock.close()
            except OSError:
                pass

    # -- framing helpers --
    def _recv_line(self):
        buf = b""
        while b"\n" not in buf:
            if len(buf) > (1 << 30):  # dispatcher enforces a 1 GB body cap
                return None
            chunk = self.sock.recv(65536)
            if not chunk:
                return None
            buf += chunk
        return buf.split(b"\n", 1)[0].decode(errors="replace")

    def _line(self, obj):
        self.sock.sendall((json.dumps(obj) + "\n").encode())

    def _send_err(self, status, msg, openai=False):
        # OpenAI-family clients expect {"error": {"message", "type"}};
        # Ollama-family expects {"error": "<string>"}
        self._line({"status": status, "content_type": "application/json"})
        payload = ({"error": {"message": msg,
                              "type": "invalid_request_error"}}
                   if openai else {"error": msg})
        self.sock.sendall(json.dumps(payload).encode())

    def _probe(self):
        w = self.worker
        with w.lock:
            loaded = list(w.engines.keys())
            ctx = dict(w.loaded_ctx)
            tokens = sum(e.tokens_out for e in w.engines.values())
            steps = sum(e.steps for e in w.engines.values())
            active = sum(e.n_active() for e in w.engines.values())
        self._line({
            "online": True,
            "models": w.available_models(),
            "loaded": loaded,
            "ctx": ctx,
            "max_concurrency": w.max_batch,
            "device": w.device,
            "stats": {"tokens_out": tokens, "engine_steps": steps,
                      "active_seqs": active,
                      "uptime_s": round(time.time() - w.started, 1)},
        })

    # -- the request path --
    def _request(self, msg):
        path = msg.get("path", "/")
        body_raw = msg.get("body") or "{}"
        try:
            body = json.loads(body_raw) if body_raw.strip() else {}
        except json.JSONDecodeError:
            body = {}
        w = self.worker

        if path in ("/api/tags", "/v1/models", "/api/ps", "/api/version",
                    "/", "/api/show") or path.startswith("/v1/models/"):
            return self._meta(path, body)
        if path in ("/api/embed", "/api/embeddings", "/v1/embeddings"):
            return self._embed(path, body)
        if path in ("/api/create", "/api/copy", "/api/delete", "/api/pull",
                    "/api/push") or path.startswith("/api/blobs"):
            self._send_err(
                501, f"{path} is not supported by the in-process GPU "
                     "worker (models are resident presets; use "
                     "/admin/models/load)")
            return

        openai = path.startswith("/v1/")
        model_req = body.get("model") or ""
        model = w.resolve(model_req) if model_req else \
            (next(iter(w.engines), None) or
             ("tiny" if w.device == "cpu" else "llama3-8b"))
        if model is None:
            self._send_err(404, f"model not found: {model_req}", openai)
            return

        tok = w.tokenizers.get(model) or ByteTokenizer(
            PRESETS[model].vocab if model in PRESETS else 512)
        try:
            params = _params_from_body(body, tok, openai)
            prompt = tok.encode(_prompt_text(body, path))
        except (TypeError, ValueError) as e:
            # malformed option types (e.g. temperature: "hot") are a
            # client error, not a worker fault: clean 400
            self._send_err(400, f"invalid options: {e}", openai)
            return
        stream = body.get("stream", not openai)

        stops = body.get("stop") or (body.get("options") or {}).get("stop")
        if isinstance(stops, str):
            stops = [stops]
        stops = [s for s in (stops or []) if s]

        q: "queue.Queue" = queue.Queue()
        t0 = time.time()
        # OpenAI SDKs validate id/created on every chunk
        self._oai_id = ("cmpl-" if path == "/v1/completions"
                        else "chatcmpl-") + f"{int(t0 * 1e6):x}"
        self._oai_created = int(t0)
        sid = w.generate(model, prompt, params,
                         lambda t, done: q.put((t, done)),
                         num_ctx=int((body.get("options") or {})
                                     .get("num_ctx", 0) or 0))

        if openai:
            ct = "text/event-stream" if stream else "application/json"
        else:
            ct = "application/x-ndjson" if stream else "application/json"
        self._line({"status": 200, "content_type": ct})

        pieces = []
        n_out = 0
        t_first = None
        try:
            stopped = False
            while True:
                t, done = q.get(timeout=600)
                if done:
                    break
                if t_first is None:
                    t_first = time.time()
                n_out += 1
                piece = tok.decode_one(t)
                pieces.append(piece)
                # stop sequences (Ollama `stop` option): cancel generation
                # when the decoded tail matches any stop string
                if stops and not stopped:
                    tail = "".join(pieces[-8:])
                    if any(sp in tail for sp in stops):
                        stopped = True
                        w.cancel(model, sid)
                if stream and not stopped:
                    self._stream_piece(path, model, piece, openai)
            reason = "stop" if (stopped or n_out < params.max_tokens) \
                else "length"
            want_usage = bool((body.get("stream_options") or {})
                              .get("include_usage"))
            self._final(path, model, pieces, n_out, t0, openai, stream,
                        len(prompt), reason, t_first, want_usage)
            # Ollama parity: "keep_alive": 0 on the request frees the
            # model after the response (only if the engine is idle —
            # other users' in-flight sequences always win)
            if body.get("keep_alive") == 0 or \
                    (body.get("options") or {}).get("keep_alive") == 0:
                w.unload(model, only_if_idle=True)
        except (BrokenPipeError, ConnectionResetError, OSError):
            w.cancel(model, sid)
        except queue.Empty:
            w.cancel(model, sid)

    def _stream_piece(self, path, model, piece, openai):
        if openai:
            if path == "/v1/completions":
                obj = {"id": self._oai_id, "created": self._oai_created,
                       "object": "text_completion", "model": model,
                       "choices": [{"index": 0, "text": piece,
                                    "finish_reason": None}]}
            else:
                obj = {"id": self._oai_id, "created": self._oai_created,
                       "object": "chat.completion.chunk", "model": model,
                       "choices": [{"index": 0,
                                    "delta": {"content": piece},
                                    "finish_reason": None}]}
            self.sock.sendall(f"data: {json.dumps(obj)}\n\n".encode())
        else:
            if path == "/api/generate":
                obj = {"model": model, "created_at": _now_iso(),
                       "response": piece, "done": False}
            else:
                obj = {"model": model, "created_at": _now_iso(),
                       "message": {"role": "assistant", "content": piece},
                       "done": False}
            self.sock.sendall((json.dumps(obj) + "\n").encode())

    def _final(self, path, model, pieces, n_out, t0, openai, stream,
               n_prompt, reason="stop", t_first=None, want_usage=False):
        t_end = time.time()
        dur_ns = int((t_end - t0) * 1e9)
        if t_first is None:
            t_first = t_end
        prompt_ns = int((t_first - t0) * 1e9)
        eval_ns = max(0, dur_ns - prompt_ns)
        text = "".join(pieces)
        if openai:
            if stream:
                obj = {"id": self._oai_id,
                       "created": self._oai_created,
                       "object": "chat.completion.chunk", "model": model,
                       "choices": [{"index": 0, "delta": {},
                                    "finish_reason": reason}]}
                self.sock.sendall(f"data: {json.dumps(obj)}\n\n".encode())
                if want_usage:
                    # OpenAI stream_options.include_usage: one final
                    # usage-only chunk before [DONE]
                    u = {"id": self._oai_id,
                         "created": self._oai_created,
                         "object": "chat.completion.chunk",
                         "model": model, "choices": [],
                         "usage": {"prompt_tokens": n_prompt,
                                   "completion_tokens": n_out,
                                   "total_tokens": n_prompt + n_out}}
                    self.sock.sendall(
                        f"data: {json.dumps(u)}\n\n".encode())
                self.sock.sendall(b"data: [DONE]\n\n")
            else:
                key = ("text" if path == "/v1/completions" else "message")
                choice = {"index": 0, "finish_reason": reason}
                if key == "text":
                    choice["text"] = text
                else:
                    choice["message"] = {"role": "assistant",
                                         "content": text}
                obj = {"id": self._oai_id,
                       "created": self._oai_created,
                       "object": ("text_completion"
                                  if path == "/v1/completions"
                                  else "chat.completion"),
                       "model": model, "choices": [choice],
                       "usage": {"prompt_tokens": n_prompt,
                                 "completion_tokens": n_out,
                                 "total_tokens": n_prompt + n_out}}
                self.sock.sendall(json.dumps(obj).encode())
        else:
            obj = {"model": model, "created_at": _now_iso(),
                   "done": True, "done_reason": reason,
                   "total_duration": dur_ns,
                   "load_duration": 0,
                   "prompt_eval_count": n_prompt,
                   "prompt_eval_duration": prompt_ns,
                   "eval_count": n_out,
                   "eval_duration": eval_ns}
            if not stream:
                if path == "/api/generate":
                    obj["response"] = text
                else:
                    obj["message"] = {"role": "assistant", "content": text}
            self.sock.sendall((json.dumps(obj) + "\n").encode())

    def _embed(self, path, body):
        w = self.worker
        model_req = body.get("model") or ""
        model = w.resolve(model_req) if model_req else \
            next(iter(w.engines), "tiny" if w.device == "cpu"
                 else "llama3-8b")
        if model not in w.engines:
            err = w.load(model)
            if err:
                self._line({"status": 404,
                            "content_type": "application/json"})
                self.sock.sendall(json.dumps({"error": err}).encode())
                return
        raw = body.get("input", body.get("prompt", ""))
        inputs = raw if isinstance(raw, list) else [raw]
        tok = w.tokenizers[model]
        t0 = time.time()
        vecs = []
        n_tok = 0
        with w.lock:
            eng = w.engines[model]
            for text in inputs:
                ids = tok.encode(str(text))
                n_tok += len(ids)
                vecs.append(eng.embed(ids))
        self._line({"status": 200, "content_type": "application/json"})
        if path == "/v1/embeddings":
            obj = {"object": "list", "model": model,
                   "data": [{"object": "embedding", "index": i,
                             "embedding": v} for i, v in enumerate(vecs)],
                   "usage": {"prompt_tokens": n_tok,
                             "total_tokens": n_tok}}
        elif path == "/api/embed":
            obj = {"model": model, "embeddings": vecs,
                   "total_duration": int((time.time() - t0) * 1e9),
                   "load_duration": 0,
                   "prompt_eval_count": n_tok}
        else:  # legacy /api/embeddings
            obj = {"model": model, "embedding": vecs[0] if vecs else []}
        self.sock.sendall(json.dumps(obj).encode())

    def _meta(self, path, body):
        w = self.worker
        with w.lock:
            loaded = list(w.engines.keys())
            ctx = dict(w.loaded_ctx)
        def entry(m, res=False):
            # Ollama tags/ps entry schema (name/model/size/digest/details)
            import hashlib
            cfg = PRESETS.get(m)
            qd = cfg.n_heads * cfg.head_dim if cfg else 0
            kvd = cfg.n_kv_heads * cfg.head_dim if cfg else 0
            size = 2 * (cfg.vocab * cfg.hidden * 2 + cfg.n_layers *
                        (cfg.hidden * (qd + 2 * kvd) + qd * cfg.hidden +
                         3 * cfg.hidden * cfg.ffn)) if cfg else 0
            e = {"name": m, "model": m, "size": size,
                 "digest": hashlib.sha256(m.encode()).hexdigest(),
                 "modified_at": _now_iso(),
                 "details": {"format": "safetensors", "family": "llama",
                             "families": ["llama"], "parameter_size": m,
                             "quantization_level": "BF16"}}
            if res:
                e["size_vram"] = size
                e["expires_at"] = _now_iso()
                e["context_length"] = ctx.get(m, 0)
            return e

        if path == "/api/tags":
            obj = {"models": [entry(m) for m in w.available_models()]}
        elif path == "/v1/models":
            obj = {"object": "list",
                   "data": [{"id": m, "object": "model",
                             "created": int(self.worker.started),
                             "owned_by": "ollamamq-amd"}
                            for m in w.available_models()]}
        elif path == "/api/ps":
            obj = {"models": [entry(m, res=True) for m in loaded]}
        elif path == "/api/version":
            obj = {"version": VERSION}
        elif path.startswith("/v1/models/"):
            m = path[len("/v1/models/"):]
            r = w.resolve(m)
            if r is None:
                self._line({"status": 404,
                            "content_type": "application/json"})
                self.sock.sendall(json.dumps(
                    {"error": f"model not found: {m}"}).encode())
                return
            obj = {"id": r, "object": "model",
                   "created": int(self.worker.started),
                   "owned_by": "ollamamq-amd"}
        elif path == "/api/show":
            m = body.get("model", "")
            cfg = PRESETS.get(m) or PRESETS.get(w.resolve(m) or "")
            if cfg is None:
                # Ollama returns 404 for unknown models on /api/show
                self._line({"status": 404,
                            "content_type": "application/json"})
                self.sock.sendall(json.dumps(
                    {"error": f"model not found: {m}"}).encode())
                return
            # Ollama show schema: SDKs read details.* and the
            # "<arch>.<key>" model_info namespace
            obj = {
                "modelfile": "",
                "parameters": f"num_ctx {cfg.max_ctx}",
                "template": "{{ .Prompt }}",
                "details": {"format": "safetensors", "family": "llama",
                            "families": ["llama"],
                            "parameter_size": cfg.name,
                            "quantization_level": "BF16"},
                "model_info": {
                    "general.architecture": "llama",
                    "llama.block_count": cfg.n_layers,
                    "llama.embedding_length": cfg.hidden,
                    "llama.context_length": cfg.max_ctx,
                    "llama.feed_forward_length": cfg.ffn,
                    "llama.attention.head_count": cfg.n_heads,
                    "llama.attention.head_count_kv": cfg.n_kv_heads,
                    "llama.vocab_size": cfg.vocab,
                },
                "capabilities": ["completion"],
            }
        else:
            obj = {"status": "ollamamq-amd worker",
                   "device": w.device, "loaded": loaded}
        self._line({"status": 200, "content_type": "application/json"})
        self.sock.sendall(json.dumps(obj).encode())


def serve(sock_path: str, worker: Worker):
    try:
        os.unlink(sock_path)
    except FileNotFoundError:
        pass
    srv = socket.socket(socket.AF_UNIX, socket.SOCK_STREAM)
    srv.bind(sock_path)
    srv.listen(64)
    print(f"worker ready on {sock_path} ({worker.device})", flush=True)
    while True:
        conn, _ = srv.accept()
        threading.Thread(target=Conn(conn, worker).run,
                         daemon=True).start()


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--socket", required=True)
    ap.add_argument("--gpu", type=int, default=0)
    ap.add_argument("--model", type=str, default=None,
                    help="preload this model")
    ap.add_argument("--max-ctx", type=int, default=4096)
    ap.add_argument("--max-batch", type=int, default=32)
    ap.add_argument("--models", type=str, default=None,
                    help="comma list restricting this worker's advertised "
                         "models (fleet sharding for multi-model routing)")
    args = ap.parse_args()

    w = Worker(args.gpu, max_batch=args.max_batch, default_ctx=args.max_ctx,
               models=args.models.split(",") if args.models else None)
    if args.model:
        err = w.load(args.model, args.max_ctx)
        if err:
            raise SystemExit(f"preload failed: {err}")
    serve(args.socket, w)


if __name__ == "__main__":
    main()
