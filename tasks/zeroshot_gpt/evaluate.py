"""Zero-shot GPT evaluation (reference tasks/zeroshot_gpt/evaluate.py):

* WIKITEXT103 — perplexity with an overlapping sliding window: each
  window of ``seq_length`` advances by ``overlapping_eval`` tokens and
  only the new tokens contribute to the log-likelihood sum (reference
  ``_LMDataset`` + loss accumulation semantics)
* LAMBADA — cloze accuracy: the model must greedily predict every token
  of the final word of each passage (reference ``_LambadaDataset``)
"""

from __future__ import annotations

import json
import math

import torch
import torch.nn.functional as F

from megatronapp_amd.core.models.gpt import GPTModel
from megatronapp_amd.core.models.gpt.gpt_layer_specs import (
    get_gpt_layer_local_spec,
)
from megatronapp_amd.training.arguments import (
    core_transformer_config_from_args,
)
from megatronapp_amd.training.global_vars import get_args


def build_gpt(args):
    config = core_transformer_config_from_args(args)
    model = GPTModel(
        config=config,
        transformer_layer_spec=get_gpt_layer_local_spec(
            normalization=args.normalization, use_flash=False),
        vocab_size=args.padded_vocab_size,
        max_sequence_length=args.max_position_embeddings,
        position_embedding_type=args.position_embedding_type)
    device = "cuda" if torch.cuda.is_available() else "cpu"
    model = model.to(device).eval()
    if args.load:
        from megatronapp_amd.training.checkpointing import load_checkpoint
        load_checkpoint([model], None, None)
    return model, device


@torch.no_grad()
def _window_logprobs(model, device, tokens, targets):
    """Sum log p(target) and argmax-correct count for one window."""
    ids = torch.tensor(tokens, device=device).unsqueeze(0)
    pos = torch.arange(ids.shape[1], device=device).unsqueeze(0)
    logits = model(ids, pos)                 # [1, s, v]
    logp = F.log_softmax(logits.float(), dim=-1)[0]
    tgt = torch.tensor(targets, device=device)
    picked = logp[torch.arange(len(targets)), tgt]
    correct = (logp.argmax(-1) == tgt)
    return picked, correct


def evaluate_wikitext(model, device, token_ids, seq_length,
                      overlapping_eval):
    """Overlapped-window ppl (reference evaluate.py wikitext path)."""
    total_lp = 0.0
    total_tokens = 0
    start = 0
    first = True
    n = len(token_ids)
    while start + 1 < n:
        end = min(start + seq_length + 1, n)
        window = token_ids[start:end]
        inputs, targets = window[:-1], window[1:]
        picked, _ = _window_logprobs(model, device, inputs, targets)
        if first:
            count = len(targets)
            total_lp += picked.sum().item()
            first = False
        else:
            count = min(overlapping_eval, len(targets))
            total_lp += picked[-count:].sum().item()
        total_tokens += count
        if end == n:
            break
        start += overlapping_eval
    ppl = math.exp(min(-total_lp / max(total_tokens, 1), 20.0))
    return {"ppl": ppl, "tokens": total_tokens,
            "avg_logprob": total_lp / max(total_tokens, 1)}


def evaluate_lambada(model, device, samples, tokenizer):
    """Last-word cloze accuracy (reference evaluate.py lambada path):
    every token of the final word must be the argmax."""
    correct = 0
    for text in samples:
        words = text.rstrip().split(" ")
        context = " ".join(words[:-1])
        last = " " + words[-1]
        ctx_ids = tokenizer.tokenize(context)
        last_ids = tokenizer.tokenize(last)
        if not ctx_ids or not last_ids:
            continue
        tokens = ctx_ids + last_ids
        _, hit = _window_logprobs(model, device, tokens[:-1], tokens[1:])
        if bool(hit[-len(last_ids):].all()):
            correct += 1
    acc = correct / max(len(samples), 1)
    return {"accuracy": acc, "num_samples": len(samples)}


def main(tokenizer):
    args = get_args()
    model, device = build_gpt(args)
    task = args.task.upper()
    path = (args.valid_data or args.train_data)[0]
    if task == "WIKITEXT103":
        with open(path, encoding="utf-8") as f:
            token_ids = tokenizer.tokenize(f.read())
        out = evaluate_wikitext(model, device, token_ids,
                                args.seq_length, args.overlapping_eval)
        print(f"wikitext results: ppl {out['ppl']:.4f} on "
              f"{out['tokens']} tokens", flush=True)
    elif task == "LAMBADA":
        with open(path, encoding="utf-8") as f:
            samples = [json.loads(line)["text"] for line in f if
                       line.strip()]
        out = evaluate_lambada(model, device, samples, tokenizer)
        print(f"lambada results: accuracy {out['accuracy'] * 100:.2f}% "
              f"on {out['num_samples']} samples", flush=True)
    else:
        raise ValueError(task)
    return out
