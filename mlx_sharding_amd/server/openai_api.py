"""OpenAI-compatible HTTP server (`mlx-sharding-api`).

Behavior-parity with /root/reference/shard/openai_api.py: endpoints
(/v1/completions, /v1/chat/completions, /chat/completions — :182-186),
sampling params + validation (:206-294), stop-sequence machinery with
stream-side buffering so stop strings never leak (:30-43, :448-490),
SSE framing `data: ...` + `data: [DONE]` (:486-505), non-stream
responses with usage counts and token_logprobs/top_logprobs (:296-355),
ModelProvider hot-swap with path-escape guard (:70-127), chat template
fallback (:46-67), static web-UI serving (:157-176) and CORS (:137-140).

Upgrades over the reference: threaded HTTP server (the reference is
single-threaded, :543-563) and per-request session state (no global
CACHE race).
"""

from __future__ import annotations

import argparse
import json
import logging
import os
import threading
import time
import uuid
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer
from pathlib import Path
from typing import List, Optional, Tuple

import torch

from ..parallel.engine import SamplingParams, generate_step
from ..parallel.grpc_transport import make_clients
from ..utils import metrics
from ..utils.detokenizer import StreamingDetokenizer
from ..utils.loading import load_model

log = logging.getLogger(__name__)

STATIC_DIR = Path(__file__).parent / "static"


# ---------------------------------------------------------------------------
# helpers (parity: openai_api.py:30-67)
# ---------------------------------------------------------------------------

def stopping_criteria(tokens: List[int], stop_id_sequences: List[List[int]],
                      eos_token_id: Optional[int]) -> Tuple[bool, int]:
    """Return (stop_met, trim_length): trim_length tokens to drop from the
    end when a stop sequence matched (reference openai_api.py:30-43)."""
    if tokens and eos_token_id is not None and tokens[-1] == eos_token_id:
        return True, 1
    for stop_ids in stop_id_sequences:
        if len(tokens) >= len(stop_ids) and tokens[-len(stop_ids):] == stop_ids:
            return True, len(stop_ids)
    return False, 0


def sequence_overlap(s1: List[int], s2: List[int]) -> bool:
    """True if a suffix of s1 is a prefix of s2 (stream-side buffering)."""
    for i in range(1, min(len(s1), len(s2)) + 1):
        if s1[-i:] == s2[:i]:
            return True
    return False


def convert_chat(messages: List[dict], role_mapping: Optional[dict] = None) -> str:
    """Fallback chat templating (reference openai_api.py:46-67)."""
    default_role_mapping = {
        "system_prompt": ("A chat between a curious user and an artificial "
                          "intelligence assistant. The assistant follows the "
                          "given rules no matter what."),
        "system": "ASSISTANT's RULE: ",
        "user": "USER: ",
        "assistant": "ASSISTANT: ",
        "stop": "\n",
    }
    role_mapping = role_mapping if role_mapping is not None else default_role_mapping
    prompt = ""
    for line in messages:
        role_prefix = role_mapping.get(line["role"], "")
        stop = role_mapping.get("stop", "")
        content = line.get("content", "")
        prompt += f"{role_prefix}{content}{stop}"
    prompt += role_mapping.get("assistant", "")
    return prompt.rstrip()


# ---------------------------------------------------------------------------
# ModelProvider (parity: openai_api.py:70-127)
# ---------------------------------------------------------------------------

class ModelProvider:
    def __init__(self, cli_args):
        self.args = cli_args
        self.model_key: Optional[str] = None
        self.model = None
        self.tokenizer = None
        self.remotes = make_clients(cli_args.llm_shard_addresses.split(",")) \
            if cli_args.llm_shard_addresses else []
        self.lock = threading.Lock()
        # opt-in prefix cache (MLXS_PREFIX_CACHE=1): the previous
        # generation's (token ids, KV caches); a follow-up prompt that
        # shares a prefix re-uses those K/V rows (take/store under the
        # lock — a concurrent second request simply runs fresh)
        self._prefix_state: Optional[Tuple[List[int], list]] = None
        if cli_args.model is not None:
            self.load("default_model")

    def take_prefix_state(self) -> Optional[Tuple[List[int], list]]:
        with self.lock:
            s, self._prefix_state = self._prefix_state, None
            return s

    def store_prefix_state(self, tokens: List[int], cache: list):
        with self.lock:
            self._prefix_state = (tokens, cache)

    def _validate_model_path(self, model_path: str):
        model_path = Path(model_path)
        if model_path.exists() and not model_path.is_relative_to(Path.cwd()):
            raise RuntimeError("Local models must be relative to the current working dir.")

    def load(self, model_path: str):
        with self.lock:
            if self.model_key == model_path:
                return self.model, self.tokenizer
            self.model = None
            self.tokenizer = None
            self.model_key = None
            self._prefix_state = None  # caches belong to the old model
            if model_path in ("default_model", None):
                path = self.args.model
            else:
                self._validate_model_path(model_path)
                path = model_path
            from transformers import AutoTokenizer

            from ..utils.loading import get_model_path
            path = get_model_path(path)  # local dir or HF repo id
            tokenizer = AutoTokenizer.from_pretrained(
                str(path),
                trust_remote_code=getattr(self.args, "trust_remote_code", False))
            tpl = getattr(self.args, "chat_template", "")
            if tpl:
                tokenizer.chat_template = tpl
            elif getattr(self.args, "use_default_chat_template", False):
                if getattr(tokenizer, "default_chat_template", None):
                    tokenizer.chat_template = tokenizer.default_chat_template
            device = "cuda" if torch.cuda.is_available() else "cpu"
            model, _config = load_model(path, self.args.start_layer,
                                        self.args.end_layer, device=device)
            self.model_key = model_path
            self.model = model
            self.tokenizer = tokenizer
            return self.model, self.tokenizer


# ---------------------------------------------------------------------------
# HTTP handler
# ---------------------------------------------------------------------------

class APIHandler(BaseHTTPRequestHandler):
    provider: ModelProvider = None  # set by run()

    protocol_version = "HTTP/1.1"

    def _set_cors(self):
        self.send_header("Access-Control-Allow-Origin", "*")
        self.send_header("Access-Control-Allow-Methods", "GET, POST, OPTIONS")
        self.send_header("Access-Control-Allow-Headers", "Content-Type, Authorization")

    def do_OPTIONS(self):
        self.send_response(204)
        self._set_cors()
        self.end_headers()

    def log_message(self, fmt, *args_):  # quiet by default; use logging
        log.debug("%s - %s", self.address_string(), fmt % args_)

    # -- static files (web UI) + observability ---------------------------
    def do_GET(self):
        path = self.path.split("?")[0]
        if path == "/health":
            # liveness + readiness (model loaded?) — SURVEY.md §5.3: the
            # reference has no health checks at all
            ready = getattr(self.provider, "model", None) is not None or \
                hasattr(self.provider, "generate")
            data = json.dumps({
                "status": "ok" if ready else "loading",
                "model": getattr(self.provider, "model_key", None)}).encode()
            self.send_response(200 if ready else 503)
            self._set_cors()
            self.send_header("Content-Type", "application/json")
            self.send_header("Content-Length", str(len(data)))
            self.end_headers()
            self.wfile.write(data)
            return
        if path == "/metrics":
            data = metrics.REGISTRY.render().encode()
            self.send_response(200)
            self._set_cors()
            self.send_header("Content-Type", "text/plain; version=0.0.4")
            self.send_header("Content-Length", str(len(data)))
            self.end_headers()
            self.wfile.write(data)
            return
        if path in ("/", "/index.html"):
            path = "/index.html"
        f = (STATIC_DIR / path.lstrip("/")).resolve()
        if not str(f).startswith(str(STATIC_DIR.resolve())) or not f.is_file():
            self._error(404, "not found")
            return
        ctype = {"html": "text/html", "js": "application/javascript",
                 "css": "text/css", "json": "application/json"}.get(
            f.suffix.lstrip("."), "application/octet-stream")
        data = f.read_bytes()
        self.send_response(200)
        self._set_cors()
        self.send_header("Content-Type", ctype)
        self.send_header("Content-Length", str(len(data)))
        self.end_headers()
        self.wfile.write(data)

    # -- POST -------------------------------------------------------------
    def do_POST(self):
        raw = self.rfile.read(int(self.headers.get("Content-Length", 0)))
        if self.path not in ("/v1/completions", "/v1/chat/completions",
                             "/chat/completions"):
            self._error(404, "Not Found")
            return
        try:
            body = json.loads(raw or b"{}")
        except json.JSONDecodeError:
            self._error(400, "invalid JSON body")
            return
        try:
            params = self._parse_params(body)
        except ValueError as e:
            self._error(400, str(e))
            return
        try:
            model, tokenizer = self.provider.load(params["model"])
        except Exception as e:  # noqa: BLE001
            self._error(400, f"failed to load model: {e}")
            return

        is_chat = self.path.endswith("chat/completions")
        if is_chat:
            prompt = self._chat_prompt(body, tokenizer)
            object_type = "chat.completion.chunk" if params["stream"] else "chat.completion"
        else:
            prompt = body.get("prompt", "")
            if not isinstance(prompt, str):
                self._error(400, "prompt must be a string")
                return
            object_type = "text_completion"

        prompt_ids = tokenizer.encode(prompt)
        stop_words = params["stop"]
        stop_id_sequences = [tokenizer.encode(sw, add_special_tokens=False)
                             for sw in stop_words]
        rid = f"{'chatcmpl' if is_chat else 'cmpl'}-{uuid.uuid4().hex}"
        if params["stream"]:
            self._handle_stream(rid, object_type, is_chat, model, tokenizer,
                                prompt_ids, stop_id_sequences, params)
        else:
            self._handle_completion(rid, object_type, is_chat, model, tokenizer,
                                    prompt_ids, stop_id_sequences, params)

    # -- param parsing (parity: openai_api.py:206-294) --------------------
    def _parse_params(self, body: dict) -> dict:
        p = {
            "stream": bool(body.get("stream", False)),
            "model": body.get("model", "default_model"),
            "max_tokens": body.get("max_tokens", 100),
            "temperature": body.get("temperature", 1.0),
            "top_p": body.get("top_p", 1.0),
            "repetition_penalty": body.get("repetition_penalty"),
            "repetition_context_size": body.get("repetition_context_size", 20),
            "logit_bias": body.get("logit_bias"),
            "logprobs": body.get("logprobs", -1),
            "stop": body.get("stop") or [],
            "seed": body.get("seed"),
        }
        if isinstance(p["stop"], str):
            p["stop"] = [p["stop"]]
        if not isinstance(p["max_tokens"], int) or p["max_tokens"] < 0:
            raise ValueError("max_tokens must be a non-negative integer")
        if not isinstance(p["temperature"], (int, float)) or p["temperature"] < 0:
            raise ValueError("temperature must be a non-negative float")
        if not isinstance(p["top_p"], (int, float)) or not 0 <= p["top_p"] <= 1:
            raise ValueError("top_p must be a float between 0 and 1")
        if p["repetition_penalty"] is not None and (
                not isinstance(p["repetition_penalty"], (int, float))
                or p["repetition_penalty"] < 0):
            raise ValueError("repetition_penalty must be a non-negative float")
        if p["logprobs"] != -1 and not 0 < p["logprobs"] <= 10:
            raise ValueError(f"logprobs must be between 1 and 10 but got {p['logprobs']}")
        if p["logit_bias"] is not None:
            if not isinstance(p["logit_bias"], dict):
                raise ValueError("logit_bias must be a dict of int to float")
            try:
                p["logit_bias"] = {int(k): float(v) for k, v in p["logit_bias"].items()}
            except ValueError:
                raise ValueError("logit_bias must be a dict of int to float") from None
        return p

    def _chat_prompt(self, body: dict, tokenizer) -> str:
        messages = body.get("messages", [])
        if getattr(tokenizer, "chat_template", None):
            return tokenizer.apply_chat_template(
                messages, tokenize=False, add_generation_prompt=True)
        return convert_chat(messages, body.get("role_mapping"))

    # -- generation -------------------------------------------------------
    def _gen(self, model, prompt_ids, params):
        sp = SamplingParams(
            temperature=float(params["temperature"]),
            top_p=float(params["top_p"]),
            repetition_penalty=params["repetition_penalty"],
            repetition_context_size=params["repetition_context_size"],
            logit_bias=params["logit_bias"],
            seed=params["seed"],
            max_tokens=params["max_tokens"],
        )
        if hasattr(self.provider, "generate"):  # e.g. the RCCL pipeline
            return self.provider.generate(list(prompt_ids), sp)
        device = next(model.parameters()).device
        ids = torch.tensor([prompt_ids], device=device)
        chunk = int(os.environ.get("MLXS_PREFILL_CHUNK", "0"))
        use_prefix = (os.environ.get("MLXS_PREFIX_CACHE") == "1"
                      and not self.provider.remotes)
        cache = None
        prefill_from = 0
        if use_prefix:
            state = self.provider.take_prefix_state()
            if state is not None:
                prev, prev_cache = state
                p = 0
                limit = min(len(prev), len(prompt_ids), prev_cache[0].offset)
                while p < limit and prev[p] == prompt_ids[p]:
                    p += 1
                p = min(p, len(prompt_ids) - 1)  # must prefill >= 1 token
                if p > 0:
                    for c in prev_cache:
                        c.trim(p)
                    cache = prev_cache
                    prefill_from = p
                    metrics.REGISTRY.counter(
                        "mlxs_prefix_cache_hits_total",
                        "requests that reused a cached prefix").inc()
                    metrics.REGISTRY.counter(
                        "mlxs_prefix_tokens_saved_total",
                        "prompt tokens not re-prefilled").inc(p)
        if cache is None:
            cache = model.make_cache(batch_size=1)
        gen = generate_step(ids, model, cache, self.provider.remotes, sp,
                            prefill_chunk=chunk, prefill_from=prefill_from)
        if not use_prefix:
            return gen

        def _tracked():
            toks: List[int] = []
            try:
                for tid, lp in gen:
                    toks.append(tid)
                    yield tid, lp
            finally:
                # cache now holds prompt + yielded tokens (the one-step
                # lookahead appended each yielded token's K/V already)
                self.provider.store_prefix_state(list(prompt_ids) + toks,
                                                 cache)
        return _tracked()

    def _record_metrics(self, n_prompt: int, n_gen: int, ttft_ms: float,
                        gen_tps: float):
        m = metrics.serving_metrics()
        m["requests"].inc()
        m["prompt_tokens"].inc(n_prompt)
        m["gen_tokens"].inc(n_gen)
        m["ttft_ms"].observe(ttft_ms)
        if gen_tps > 0:
            m["decode_tps"].observe(gen_tps)

    def _top_logprobs(self, tokenizer, logprobs_t: torch.Tensor, k: int) -> dict:
        vals, idx = torch.topk(logprobs_t.float(), k)
        return {tokenizer.decode([int(i)]): float(v)
                for i, v in zip(idx.tolist(), vals.tolist())}

    def _handle_completion(self, rid, object_type, is_chat, model, tokenizer,
                           prompt_ids, stop_id_sequences, params):
        created = int(time.time())
        eos = tokenizer.eos_token_id
        tokens: List[int] = []
        token_logprobs: List[float] = []
        top_logprobs: List[dict] = []
        t_start = time.perf_counter()
        t_first = t_start
        # a shard death mid-generation (ShardUnavailable / RuntimeError)
        # becomes a clean 502 — no headers are sent until the loop ends
        finish_reason = "length"
        try:
            for (tid, logprobs) in self._gen(model, prompt_ids, params):
                if not tokens:
                    t_first = time.perf_counter()
                tokens.append(tid)
                if params["logprobs"] > 0:
                    token_logprobs.append(float(logprobs[tid]))
                    top_logprobs.append(self._top_logprobs(
                        tokenizer, logprobs, params["logprobs"]))
                stop, trim = stopping_criteria(tokens, stop_id_sequences, eos)
                if stop:
                    tokens = tokens[: len(tokens) - trim]
                    token_logprobs = token_logprobs[: len(tokens)]
                    top_logprobs = top_logprobs[: len(tokens)]
                    finish_reason = "stop"
                    break
                if len(tokens) >= params["max_tokens"]:
                    break
        except Exception as e:  # noqa: BLE001
            log.error("generation failed: %s", e)
            self._error(502, f"generation failed: {e}")
            return
        t_end = time.perf_counter()
        ttft_ms = (t_first - t_start) * 1e3
        gen_tps = (len(tokens) - 1) / max(t_end - t_first, 1e-9) \
            if len(tokens) > 1 else 0.0
        self._record_metrics(len(prompt_ids), len(tokens), ttft_ms, gen_tps)
        text = tokenizer.decode(tokens)
        logprobs_block = None
        if params["logprobs"] > 0:
            logprobs_block = {"token_logprobs": token_logprobs,
                              "top_logprobs": top_logprobs,
                              "tokens": tokens}
        if is_chat:
            choice = {"index": 0,
                      "message": {"role": "assistant", "content": text},
                      "logprobs": logprobs_block,
                      "finish_reason": finish_reason}
        else:
            choice = {"index": 0, "text": text, "logprobs": logprobs_block,
                      "finish_reason": finish_reason}
        resp = {
            "id": rid, "object": object_type, "created": created,
            "model": params["model"],
            "system_fingerprint": f"fp_{uuid.uuid4().hex[:10]}",
            "choices": [choice],
            # ttft_ms / generation_tps extend the reference's usage block
            # (per-token TTFT/TPS surfaced in `usage`)
            "usage": {"prompt_tokens": len(prompt_ids),
                      "completion_tokens": len(tokens),
                      "total_tokens": len(prompt_ids) + len(tokens),
                      "ttft_ms": round(ttft_ms, 3),
                      "generation_tps": round(gen_tps, 3)},
        }
        data = json.dumps(resp).encode()
        self.send_response(200)
        self._set_cors()
        self.send_header("Content-Type", "application/json")
        self.send_header("Content-Length", str(len(data)))
        self.end_headers()
        self.wfile.write(data)

    def _handle_stream(self, rid, object_type, is_chat, model, tokenizer,
                       prompt_ids, stop_id_sequences, params):
        created = int(time.time())
        self.send_response(200)
        self._set_cors()
        self.send_header("Content-Type", "text/event-stream")
        self.send_header("Cache-Control", "no-cache")
        self.send_header("Connection", "close")
        self.close_connection = True
        self.end_headers()
        eos = tokenizer.eos_token_id
        detok = StreamingDetokenizer(tokenizer)
        tokens: List[int] = []
        pending: List[int] = []  # buffered while overlapping a stop sequence
        finish_reason = "length"

        def emit(delta_text: str, fin: Optional[str] = None):
            if is_chat:
                choice = {"index": 0, "delta": {"content": delta_text} if delta_text else {},
                          "finish_reason": fin}
            else:
                choice = {"index": 0, "text": delta_text, "finish_reason": fin}
            chunk = {"id": rid, "object": object_type, "created": created,
                     "model": params["model"], "choices": [choice]}
            self.wfile.write(f"data: {json.dumps(chunk)}\n\n".encode())
            self.wfile.flush()

        t_start = time.perf_counter()
        t_first = t_start
        # headers are already on the wire — a shard death mid-stream is
        # reported as an SSE error event, then the stream closes cleanly
        try:
            for (tid, _logprobs) in self._gen(model, prompt_ids, params):
                if not tokens:
                    t_first = time.perf_counter()
                tokens.append(tid)
                pending.append(tid)
                stop, trim = stopping_criteria(tokens, stop_id_sequences, eos)
                if stop:
                    pending = pending[: len(pending) - trim]
                    finish_reason = "stop"
                    break
                hit_length = len(tokens) >= params["max_tokens"]
                # the overlap hold-back must not skip the length check: a
                # max_tokens cut mid-overlap finishes with reason "length"
                # and flushes the held-back tokens below (reference flushes
                # its buffer post-loop, openai_api.py:492-503)
                if not hit_length and any(sequence_overlap(tokens, s)
                                          for s in stop_id_sequences):
                    continue  # hold back until the overlap resolves
                for t in pending:
                    txt = detok.add_token(t)
                    if txt:
                        emit(txt)
                pending = []
                if hit_length:
                    break
        except Exception as e:  # noqa: BLE001
            log.error("generation failed mid-stream: %s", e)
            metrics.serving_metrics()["errors"].inc()
            err = {"error": {"message": f"generation failed: {e}",
                             "type": "shard_unavailable"}}
            self.wfile.write(f"data: {json.dumps(err)}\n\n".encode())
            self.wfile.write(b"data: [DONE]\n\n")
            self.wfile.flush()
            return
        # post-loop flush: tokens still held back when the loop exited
        # for ANY reason (stop trim already removed the stop ids)
        for t in pending:
            txt = detok.add_token(t)
            if txt:
                emit(txt)
        tail = detok.finalize()
        if tail:
            emit(tail)
        emit("", finish_reason)
        self.wfile.write(b"data: [DONE]\n\n")
        self.wfile.flush()
        t_end = time.perf_counter()
        gen_tps = (len(tokens) - 1) / max(t_end - t_first, 1e-9) \
            if len(tokens) > 1 else 0.0
        self._record_metrics(len(prompt_ids), len(tokens),
                             (t_first - t_start) * 1e3, gen_tps)

    def _error(self, code: int, message: str):
        if code >= 400:
            metrics.serving_metrics()["errors"].inc()
        data = json.dumps({"error": message}).encode()
        self.send_response(code)
        self._set_cors()
        self.send_header("Content-Type", "application/json")
        self.send_header("Content-Length", str(len(data)))
        self.end_headers()
        self.wfile.write(data)


def run(host: str, port: int, provider: ModelProvider) -> ThreadingHTTPServer:
    # bind the provider on a per-server subclass, not on APIHandler
    # itself — two servers in one process must not share state
    handler = type("BoundAPIHandler", (APIHandler,), {"provider": provider})
    server = ThreadingHTTPServer((host, port), handler)
    return server


def main(argv=None):
    p = argparse.ArgumentParser(description="OpenAI-compatible API server")
    p.add_argument("--model", type=str, default=None,
                   help="default model checkpoint directory")
    p.add_argument("--adapter-path", type=str, default=None,
                   help="(accepted for reference-CLI parity; adapters are "
                        "not supported yet)")
    p.add_argument("--host", type=str, default="127.0.0.1")
    p.add_argument("--port", type=int, default=8080)
    p.add_argument("--trust-remote-code", action="store_true",
                   help="trust remote code for the tokenizer")
    p.add_argument("-s", "--llm-shard-addresses", type=str, default="",
                   help="comma-separated remote shard servers")
    p.add_argument("-sl", "--start-layer", type=int, default=None)
    p.add_argument("-el", "--end-layer", type=int, default=None)
    p.add_argument("--cache-limit-gb", type=int, default=None,
                   help="per-process GPU memory cap (fraction of device)")
    p.add_argument("--chat-template", type=str, default="",
                   help="override the tokenizer's chat template")
    p.add_argument("--use-default-chat-template", action="store_true")
    p.add_argument("--static-dir", type=str, default=None,
                   help="directory for web-UI static files")
    p.add_argument("--log-level", type=str, default="INFO",
                   choices=["DEBUG", "INFO", "WARNING", "ERROR", "CRITICAL"])
    args = p.parse_args(argv)
    logging.basicConfig(level=getattr(logging, args.log_level.upper(), logging.INFO))
    if args.adapter_path:
        log.warning("--adapter-path is accepted for CLI parity but adapters "
                    "are not applied")
    if args.static_dir:
        global STATIC_DIR
        STATIC_DIR = Path(args.static_dir)
    if args.cache_limit_gb is not None and torch.cuda.is_available():
        total = torch.cuda.get_device_properties(0).total_memory
        torch.cuda.set_per_process_memory_fraction(
            min(1.0, args.cache_limit_gb * (1 << 30) / total))
    provider = ModelProvider(args)
    server = run(args.host, args.port, provider)
    print(f"API server listening on {args.host}:{server.server_address[1]}", flush=True)
    server.serve_forever()


if __name__ == "__main__":
    main()
