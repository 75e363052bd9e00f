"""MegaScope inference server: WebSocket (+ Flask REST fallback).

Reference: megatron/inference/text_generation_server.py (InferenceWSServer
:273, MegatronGenerate :300, wire protocol SURVEY.md §2.6).

Rank (pp0, tp0) serves; other ranks spin in a broadcast loop
(reference tools/run_text_generation_server.py:232-251) receiving
(choice, configs, prompts, params) each request so every rank enters the
model forward together.
"""

from __future__ import annotations

import json
import threading
from typing import List, Optional

import torch
import torch.distributed as dist

from ..core import parallel_state
from ..core.inference.sampling_params import SamplingParams
from ..core.inference.static_engine import StaticInferenceEngine, run_mcore_engine
from ..core.tensor_disturbance import get_disturbance
from ..core.tensor_tracer import FlagType, enable_tensor_tracers
from ..utils.ws import WebSocketConnection, WebSocketServer

GENERATE_NUM = 0
PING_NUM = 1

LOCK = threading.Lock()


def _apply_configs(configs: dict, num_layers: int, tokenizer, report_func):
    tt = enable_tensor_tracers()
    tt.set_num_layers(num_layers)
    tt.tokenizer = tokenizer
    tt.tt_flags.set_by_configs(configs.get("visualization_flags") or {})
    tt.set_compressor_configs(configs.get("compressor_config") or {})
    get_disturbance().set_by_configs(configs.get("disturbance_configs") or {})
    tt.set_report(report_func)


def _broadcast_request(payload):
    """Rank 0 -> all: (choice, request dict)."""
    holder = [payload]
    dist.broadcast_object_list(holder, src=0)
    return holder[0]


class InferenceGenerate:
    """Executes one generate request on every rank (reference
    InferenceGenerate.query :182-267)."""

    def __init__(self, engine: StaticInferenceEngine, num_layers: int):
        self.engine = engine
        self.num_layers = num_layers

    def run(self, request: dict, websocket: Optional[WebSocketConnection]):
        tokenizer = self.engine.controller.tokenizer
        prompts = request.get("prompts") or []
        # request-size guard (reference --max-tokens-to-oom): refuse
        # generations that would exceed the configured token budget
        try:
            from ..training.global_vars import get_args
            args = get_args()
            budget = getattr(args, "max_tokens_to_oom", None)
            max_seq = getattr(args, "inference_max_seq_length", None)
        except Exception:
            budget = max_seq = None
        n_new = int(request.get("tokens_to_generate", 64))
        total = sum(len(tokenizer.tokenize(p)) + n_new for p in prompts)
        if budget and total > budget:
            err = {"type": "error",
                   "message": f"request of {total} tokens exceeds "
                              f"--max-tokens-to-oom {budget}"}
            if websocket is not None:
                websocket.send(err)
            return err
        if max_seq:
            request = dict(request)
            request["tokens_to_generate"] = min(n_new, max_seq)
        report = (lambda msg: websocket.send(msg)) if websocket else (lambda msg: None)
        _apply_configs(request, self.num_layers, tokenizer, report)

        if websocket is not None:
            prompt_tokens = [
                [{"id": int(t), "token": tokenizer.detokenize([int(t)])}
                 for t in tokenizer.tokenize(p)] for p in prompts]
            websocket.send({"type": "start", "prompts": prompt_tokens,
                            "num_layers": self.num_layers})

        tt = enable_tensor_tracers()

        def report_step(step, logits, sampled):
            tt.tik_result(logits, sampled_token=sampled)

        result = run_mcore_engine(
            self.engine, prompts,
            temperature=float(request.get("temperature", 1.0)),
            top_k=int(request.get("top_k", 0)),
            top_p=float(request.get("top_p", 0.0)),
            logprobs=bool(request.get("logprobs", False)),
            tokens_to_generate=int(request.get("tokens_to_generate", 64)),
            report_step=report_step if websocket is not None else None,
            beam_width=int(request.get("beam_width", 0)),
            length_penalty=float(request.get("length_penalty", 1.0)))
        tt.tik_end()
        if websocket is not None:
            websocket.send({"type": "finish", **result})
        return result


class InferenceWSServer:
    def __init__(self, engine: StaticInferenceEngine, port: int,
                 num_layers: int):
        self.port = port
        self.generate = InferenceGenerate(engine, num_layers)
        self._server = WebSocketServer(port=port)

    def parser(self, conn: WebSocketConnection):
        while conn.open:
            msg = conn.recv_message()
            if msg is None:
                return
            try:
                data = json.loads(msg)
            except json.JSONDecodeError:
                conn.send({"type": "error", "message": "bad json"})
                continue
            t = data.get("type")
            if t == "ping":
                conn.send({"type": "pong"})
            elif t == "generate":
                with LOCK:
                    if dist.is_initialized() and dist.get_world_size() > 1:
                        _broadcast_request((GENERATE_NUM, data))
                    try:
                        self.generate.run(data, conn)
                    except Exception as e:  # noqa: BLE001
                        conn.send({"type": "error", "message": str(e)})
            else:
                conn.send({"type": "error", "message": f"unknown type {t}"})

    def run(self):
        print(f"[MegaScope] inference WS server on port {self.port}", flush=True)
        self._server.serve_forever(self.parser)


class MegatronServer:
    """Flask REST fallback (PUT /api with {"prompts": [...], ...})."""

    def __init__(self, engine: StaticInferenceEngine, num_layers: int,
                 port: int = 5000):
        from flask import Flask, jsonify, request as freq
        self.app = Flask(__name__)
        self.port = port
        gen = InferenceGenerate(engine, num_layers)

        @self.app.route("/api", methods=["PUT"])
        def api():
            data = freq.get_json(force=True)
            with LOCK:
                if dist.is_initialized() and dist.get_world_size() > 1:
                    _broadcast_request((GENERATE_NUM, data))
                result = gen.run(data, None)
            return jsonify(result)

    def run(self, host="0.0.0.0"):
        self.app.run(host=host, port=self.port, threaded=True)


def follower_loop(engine: StaticInferenceEngine, num_layers: int):
    """Non-rank-0: receive broadcast requests and run them in lockstep."""
    gen = InferenceGenerate(engine, num_layers)
    while True:
        choice, data = _broadcast_request(None)
        if choice == GENERATE_NUM:
            try:
                gen.run(data, None)
            except Exception:  # noqa: BLE001
                pass
        elif choice == PING_NUM:
            continue
