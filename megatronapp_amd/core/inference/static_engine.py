"""Static-batch inference engine (reference
core/inference/engines/static_engine.py:129 generate)."""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import List, Optional

import torch

from .sampling_params import SamplingParams
from .text_generation_controller import TextGenerationController


@dataclass
class InferenceRequest:
    prompt: str
    prompt_tokens: List[int]
    generated_text: str = ""
    generated_tokens: Optional[torch.Tensor] = None
    generated_log_probs: Optional[torch.Tensor] = None


class StaticInferenceEngine:
    def __init__(self, text_generation_controller: TextGenerationController,
                 max_batch_size: int = 8):
        self.controller = text_generation_controller
        self.max_batch_size = max_batch_size

    @torch.no_grad()
    def generate(self, prompts: List[str],
                 sampling_params: Optional[SamplingParams] = None,
                 report_step=None) -> List[InferenceRequest]:
        sampling_params = sampling_params or SamplingParams()
        requests: List[InferenceRequest] = []
        if sampling_params.beam_width > 0:
            # beam decoding runs one prompt at a time (hypotheses take
            # the batch dim)
            for prompt in prompts:
                toks = self.controller.tokenize_prompts(
                    [prompt], sampling_params.add_BOS)[0]
                best, score, _ = self.controller.generate_beam_search(
                    toks, sampling_params)
                gen = torch.tensor(best[len(toks):])
                req = InferenceRequest(
                    prompt=prompt, prompt_tokens=toks,
                    generated_tokens=gen, generated_log_probs=None)
                req.generated_text = self.controller.detokenize(gen)
                req.score = score
                requests.append(req)
            return requests
        for i in range(0, len(prompts), self.max_batch_size):
            chunk = prompts[i:i + self.max_batch_size]
            toks = self.controller.tokenize_prompts(chunk,
                                                    sampling_params.add_BOS)
            out_tokens, logprobs = \
                self.controller.generate_all_output_tokens_static_batch(
                    toks, sampling_params, report_step=report_step)
            for j, prompt in enumerate(chunk):
                row = out_tokens[j]
                gen = row[len(toks[j]):].cpu()
                if sampling_params.termination_id >= 0:
                    hits = (gen == sampling_params.termination_id).nonzero()
                    if hits.numel():
                        gen = gen[:int(hits[0]) + 1]   # stop token inclusive
                req = InferenceRequest(
                    prompt=prompt, prompt_tokens=toks[j],
                    generated_tokens=gen,
                    generated_log_probs=(logprobs[j].cpu()
                                         if logprobs is not None else None))
                req.generated_text = self.controller.detokenize(
                    req.generated_tokens)
                requests.append(req)
        return requests


def get_inference_engine(model, tokenizer, max_batch_size=8):
    return StaticInferenceEngine(
        TextGenerationController(model, tokenizer), max_batch_size)


def run_mcore_engine(engine: StaticInferenceEngine, prompts: List[str],
                     temperature=1.0, top_k=0, top_p=0.0,
                     logprobs=False, tokens_to_generate=64, report_step=None,
                     beam_width=0, length_penalty=1.0):
    """REST/WS-facing wrapper (reference mcore_engine_server.py)."""
    sp = SamplingParams(num_tokens_to_generate=tokens_to_generate,
                        temperature=temperature, top_k=top_k, top_p=top_p,
                        return_log_probs=logprobs, beam_width=beam_width,
                        length_penalty=length_penalty)
    reqs = engine.generate(prompts, sp, report_step=report_step)
    return {
        "text": [r.prompt + r.generated_text for r in reqs],
        "segments": [[engine.controller.tokenizer.detokenize([int(t)])
                      for t in r.generated_tokens] for r in reqs],
        "logprobs": [r.generated_log_probs.tolist()
                     if r.generated_log_probs is not None else []
                     for r in reqs],
    }
