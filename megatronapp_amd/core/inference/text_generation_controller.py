"""Batched prefill/decode text generation controller.

Reference: core/inference/text_generation_controllers/
text_generation_controller.py:26 — tokenize -> padded batch prefill ->
token-by-token decode with top-k/top-p/temperature sampling; MegaScope
taps fire from inside the model forward, tik_result reports per-step
top-k candidates.

PP>1: the last stage samples and broadcasts the chosen token ids over the
pipeline group so every stage feeds the same next token.
"""

from __future__ import annotations

from typing import List, Optional

import torch
import torch.distributed as dist

from .. import parallel_state
from ..inference_params import InferenceParams
from ..tensor_parallel.mappings import gather_from_tensor_model_parallel_region
from .sampling_params import SamplingParams


class TextGenerationController:
    def __init__(self, inference_wrapped_model, tokenizer,
                 use_hip_graphs: bool = True):
        self.model = inference_wrapped_model
        self.tokenizer = tokenizer
        self.use_hip_graphs = use_hip_graphs

    # -------------------------------------------------------------- sampling
    @staticmethod
    def sample(logits: torch.Tensor, sampling: SamplingParams) -> torch.Tensor:
        """logits [b, v] -> sampled token ids [b]."""
        if sampling.temperature == 0.0 or (
                sampling.top_k in (0, 1) and sampling.top_p == 0.0
                and sampling.temperature == 1.0 and sampling.top_k == 1):
            pass
        if sampling.top_k == 1 or sampling.temperature == 0.0:
            return logits.argmax(dim=-1)
        logits = logits / max(sampling.temperature, 1e-6)
        if sampling.top_k > 1:
            kth = torch.topk(logits, sampling.top_k, dim=-1)[0][..., -1, None]
            logits = logits.masked_fill(logits < kth, float("-inf"))
        if sampling.top_p > 0.0:
            sorted_logits, sorted_idx = torch.sort(logits, descending=True,
                                                   dim=-1)
            probs = torch.softmax(sorted_logits, dim=-1)
            cum = torch.cumsum(probs, dim=-1)
            mask = cum - probs > sampling.top_p
            sorted_logits = sorted_logits.masked_fill(mask, float("-inf"))
            logits = torch.full_like(logits, float("-inf")).scatter(
                -1, sorted_idx, sorted_logits)
        probs = torch.softmax(logits.float(), dim=-1)
        return torch.multinomial(probs, 1).squeeze(-1)

    # ------------------------------------------------------------- generation
    def tokenize_prompts(self, prompts: List[str], add_BOS=False):
        tokens = [self.tokenizer.tokenize(p) for p in prompts]
        return tokens

    def generate_all_output_tokens_static_batch(
            self, prompts_tokens: List[List[int]], sampling: SamplingParams,
            report_step=None):
        """Returns (tokens incl. prompt, per-step logprobs or None)."""
        device = "cuda" if torch.cuda.is_available() else "cpu"
        b = len(prompts_tokens)
        lengths = [len(t) for t in prompts_tokens]
        max_prompt = max(lengths)
        total = max_prompt + sampling.num_tokens_to_generate
        pad = getattr(self.tokenizer, "eod", 0)
        batch = torch.full((b, total), pad, dtype=torch.long, device=device)
        for i, toks in enumerate(prompts_tokens):
            batch[i, :len(toks)] = torch.tensor(toks, device=device)

        pp_group = parallel_state.get_pipeline_model_parallel_group()
        pp_world = parallel_state.get_pipeline_model_parallel_world_size()
        is_last = parallel_state.is_pipeline_last_stage()
        # hipGraph the single-token decode step (launch-bound) when shapes
        # allow: one process, CUDA, replayable cache appends
        use_graphs = (self.use_hip_graphs and device == "cuda"
                      and pp_world == 1)
        if use_graphs:
            from ..hip_graphs import GraphDecodeContext
            inference_params = GraphDecodeContext(b, total)
        else:
            inference_params = InferenceParams(b, total)
        graph_step = None
        logprobs = [] if sampling.return_log_probs else None

        pos = 0
        step_tokens = batch[:, :max_prompt]
        for step in range(sampling.num_tokens_to_generate + 1):
            cur_len = step_tokens.shape[1]
            position_ids = torch.arange(
                pos, pos + cur_len, device=device).unsqueeze(0).expand(b, -1)
            if use_graphs and cur_len == 1:
                if graph_step is None:
                    from ..hip_graphs import BucketedGraphedDecodeStep
                    graph_step = BucketedGraphedDecodeStep(
                        self.model, inference_params, b)
                logits = graph_step(step_tokens, position_ids)
            else:
                logits = self.model(step_tokens, position_ids,
                                    inference_context=inference_params)
                inference_params.increment_sequence_len_offset(cur_len)
            pos += cur_len
            if pos >= total:
                break

            if is_last:
                last_logits = logits[:, -1, :]
                # un-shard vocab across TP for sampling
                if parallel_state.get_tensor_model_parallel_world_size() > 1:
                    last_logits = gather_from_tensor_model_parallel_region(
                        last_logits)
                new_tokens = self.sample(last_logits, sampling)
                if logprobs is not None:
                    lp = torch.log_softmax(last_logits.float(), dim=-1)
                    logprobs.append(lp.gather(
                        -1, new_tokens.unsqueeze(-1)).squeeze(-1))
                if report_step is not None:
                    report_step(step, last_logits, new_tokens)
            else:
                new_tokens = torch.zeros(b, dtype=torch.long, device=device)
            if pp_world > 1:
                src = parallel_state.get_pipeline_model_parallel_last_rank()
                dist.broadcast(new_tokens, src, group=pp_group)

            # only fill positions still inside the generation window; keep
            # prompt tokens where prompts are longer than current position
            write_pos = pos
            if write_pos < total:
                mask = torch.tensor(
                    [write_pos >= lengths[i] for i in range(b)], device=device)
                batch[:, write_pos] = torch.where(mask, new_tokens,
                                                  batch[:, write_pos])
            step_tokens = batch[:, pos:pos + 1]

            if sampling.termination_id >= 0 and bool(
                    (new_tokens == sampling.termination_id).all()):
                break

        if logprobs is not None and logprobs:
            logprobs = torch.stack(logprobs, dim=1)
        return batch, logprobs

    # ------------------------------------------------------------ beam search
    @torch.no_grad()
    def generate_beam_search(self, prompt_tokens: List[int],
                             sampling: SamplingParams):
        """Single-prompt beam search (reference
        text_generation_server.py:243-267 / beam_search API): maintains
        ``beam_width`` hypotheses in the batch dim, reorders the KV cache
        on every step, scores finished hypotheses with a GNMT-style
        length penalty, and returns (best_tokens, best_score,
        all_beams)."""
        device = "cuda" if torch.cuda.is_available() else "cpu"
        beams = sampling.beam_width
        assert beams > 0
        eos = sampling.termination_id if sampling.termination_id >= 0 \
            else getattr(self.tokenizer, "eod", -1)
        plen = len(prompt_tokens)
        total = plen + sampling.num_tokens_to_generate
        inference_params = InferenceParams(beams, total)

        tokens = torch.tensor(prompt_tokens, device=device).unsqueeze(0) \
            .expand(beams, plen).contiguous()
        scores = torch.full((beams,), float("-inf"), device=device)
        scores[0] = 0.0          # all beams identical at start: keep one
        finished: list = []      # (score, tokens list)

        def lp(length: int) -> float:
            return ((5.0 + length) / 6.0) ** sampling.length_penalty

        step_tokens = tokens
        pos = 0
        for step in range(sampling.num_tokens_to_generate):
            cur_len = step_tokens.shape[1]
            position_ids = torch.arange(
                pos, pos + cur_len, device=device).unsqueeze(0).expand(
                    beams, -1)
            logits = self.model(step_tokens, position_ids,
                                inference_context=inference_params)
            inference_params.increment_sequence_len_offset(cur_len)
            pos += cur_len
            last = logits[:, -1, :]
            if parallel_state.get_tensor_model_parallel_world_size() > 1:
                last = gather_from_tensor_model_parallel_region(last)
            logprobs = torch.log_softmax(last.float(), dim=-1)   # [beams, v]
            v = logprobs.shape[-1]
            cand = scores.unsqueeze(-1) + logprobs               # [beams, v]
            top_scores, top_idx = cand.view(-1).topk(beams)
            beam_idx = top_idx // v
            tok_idx = top_idx % v
            inference_params.reorder_batch(beam_idx)
            tokens = torch.cat(
                [tokens.index_select(0, beam_idx),
                 tok_idx.unsqueeze(-1)], dim=1)
            scores = top_scores
            # retire finished hypotheses, keep the rest going
            if eos >= 0:
                done = tok_idx == eos
                for i in torch.nonzero(done).flatten().tolist():
                    gen_len = tokens.shape[1] - plen
                    finished.append((float(scores[i]) / lp(gen_len),
                                     tokens[i].tolist()))
                    scores[i] = float("-inf")
                if len(finished) >= beams or bool(
                        torch.isinf(scores).all()):
                    break
            step_tokens = tok_idx.unsqueeze(-1)

        for i in range(beams):
            if not torch.isinf(scores[i]):
                gen_len = tokens.shape[1] - plen
                finished.append((float(scores[i]) / lp(gen_len),
                                 tokens[i].tolist()))
        finished.sort(key=lambda x: -x[0])
        best_score, best_tokens = finished[0]
        return best_tokens, best_score, finished

    def detokenize(self, tokens_row: torch.Tensor) -> str:
        return self.tokenizer.detokenize([int(t) for t in tokens_row])
