"""Continuous-batching inference engine (reference
core/inference/engines/dynamic_engine.py:182).

Requests join and leave the decode batch independently: a waiting
request is admitted by prefilling its prompt into a free KV slot; every
``step()`` then runs ONE decode forward for all running slots (per-row
positions + padding masks via DynamicInferenceContext), samples, and
retires finished rows.  ``generate`` drives steps until a set of
prompts drains, but ``add_request``/``step`` are the real API —
requests can be added while others are mid-generation.
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Dict, List, Optional

import torch

from .sampling_params import SamplingParams
from .dynamic_context import DynamicInferenceContext
from .static_engine import InferenceRequest
from .text_generation_controller import TextGenerationController
from .. import parallel_state
from ..tensor_parallel.mappings import gather_from_tensor_model_parallel_region


@dataclass
class _Running:
    request_id: int
    prompt: str
    prompt_tokens: List[int]
    sampling: SamplingParams
    slot: int
    generated: List[int] = field(default_factory=list)
    pending_token: Optional[int] = None   # sampled, not yet decoded


class DynamicInferenceEngine:
    def __init__(self, controller: TextGenerationController,
                 max_batch_size: int = 8,
                 max_sequence_length: int = 2048):
        self.controller = controller
        self.max_sequence_length = max_sequence_length
        self.context = DynamicInferenceContext(max_batch_size,
                                               max_sequence_length)
        self.waiting: List[_Running] = []
        self.running: List[_Running] = []
        self.finished: Dict[int, InferenceRequest] = {}
        self._next_id = 0
        self._device = ("cuda" if torch.cuda.is_available() else "cpu")

    # ------------------------------------------------------------- intake
    def add_request(self, prompt: str,
                    sampling: Optional[SamplingParams] = None) -> int:
        sampling = sampling or SamplingParams()
        toks = self.controller.tokenize_prompts([prompt])[0]
        rid = self._next_id
        self._next_id += 1
        self.waiting.append(_Running(rid, prompt, toks, sampling, slot=-1))
        return rid

    def has_unfinished_requests(self) -> bool:
        return bool(self.waiting or self.running)

    # -------------------------------------------------------------- logic
    def _logits_last(self, logits):
        last = logits[:, -1, :]
        if parallel_state.get_tensor_model_parallel_world_size() > 1:
            last = gather_from_tensor_model_parallel_region(last)
        return last

    def _admit(self):
        ctx = self.context
        while self.waiting:
            slot = ctx.free_slot()
            if slot is None:
                return
            req = self.waiting.pop(0)
            req.slot = slot
            plen = len(req.prompt_tokens)
            ctx.set_active([slot], prefill=True)
            toks = torch.tensor([req.prompt_tokens], dtype=torch.long,
                                device=self._device)
            pos = torch.arange(plen, device=self._device).unsqueeze(0)
            with torch.no_grad():
                logits = self.controller.model(toks, pos,
                                               inference_context=ctx)
            ctx.advance(plen)
            tok = int(self.controller.sample(
                self._logits_last(logits), req.sampling))
            req.pending_token = tok
            self.running.append(req)

    def _retire(self, req: _Running, include_pending: bool):
        if include_pending and req.pending_token is not None:
            req.generated.append(req.pending_token)
        self.context.release(req.slot)
        gen = torch.tensor(req.generated, dtype=torch.long)
        out = InferenceRequest(
            prompt=req.prompt, prompt_tokens=req.prompt_tokens,
            generated_tokens=gen, generated_log_probs=None)
        out.generated_text = self.controller.detokenize(gen)
        out.request_id = req.request_id
        self.finished[req.request_id] = out

    def step(self):
        """Admit waiting requests, run one decode step, retire rows."""
        self._admit()
        if not self.running:
            return
        # retire rows whose pending token ends them BEFORE decoding it
        still = []
        for req in self.running:
            t = req.pending_token
            ends = (t is not None and
                    ((req.sampling.termination_id >= 0 and
                      t == req.sampling.termination_id) or
                     len(req.generated) + 1 >=
                     req.sampling.num_tokens_to_generate))
            if ends:
                self._retire(req, include_pending=True)
            else:
                still.append(req)
        self.running = still
        if not self.running:
            return

        ctx = self.context
        ctx.set_active([r.slot for r in self.running], prefill=False)
        toks = torch.tensor([[r.pending_token] for r in self.running],
                            dtype=torch.long, device=self._device)
        pos = ctx.row_positions().unsqueeze(1)
        with torch.no_grad():
            logits = self.controller.model(toks, pos, inference_context=ctx)
        ctx.advance(1)
        new = self.controller.sample(self._logits_last(logits),
                                     self.running[0].sampling)
        for i, req in enumerate(self.running):
            req.generated.append(req.pending_token)
            req.pending_token = int(new[i])

    # ---------------------------------------------------------- frontend
    @torch.no_grad()
    def generate(self, prompts: List[str],
                 sampling_params: Optional[SamplingParams] = None
                 ) -> List[InferenceRequest]:
        ids = [self.add_request(p, sampling_params) for p in prompts]
        while self.has_unfinished_requests():
            self.step()
        return [self.finished[i] for i in ids]


def get_dynamic_inference_engine(model, tokenizer, max_batch_size=8,
                                 max_sequence_length=2048):
    return DynamicInferenceEngine(
        TextGenerationController(model, tokenizer, use_hip_graphs=False),
        max_batch_size, max_sequence_length)
