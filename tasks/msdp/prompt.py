"""Multi-stage dialogue prompting (reference tasks/msdp/prompt.py):
stage 1 prompts the LM to generate grounding knowledge for a dialogue
turn; stage 2 prompts it to generate the response conditioned on that
knowledge.  Generation is greedy and device-agnostic (plain model
forward), so the harness runs on CPU and GPU alike."""

from __future__ import annotations

import torch


@torch.no_grad()
def greedy_generate(model, tokenizer, prompt: str, max_new_tokens: int,
                    device, stop_token: int = None) -> str:
    ids = tokenizer.tokenize(prompt)
    out = []
    for _ in range(max_new_tokens):
        inp = torch.tensor(ids + out, device=device).unsqueeze(0)
        pos = torch.arange(inp.shape[1], device=device).unsqueeze(0)
        logits = model(inp, pos)
        nxt = int(logits[0, -1].argmax())
        if stop_token is not None and nxt == stop_token:
            break
        out.append(nxt)
    return tokenizer.detokenize(out)


def _format_knowledge_prompt(turns, topic):
    ctx = " ".join(turns[-3:])
    return f"Topic: {topic}. Dialogue: {ctx} Knowledge: "


def _format_response_prompt(turns, knowledge):
    ctx = " ".join(turns[-3:])
    return f"Knowledge: {knowledge} Dialogue: {ctx} Response: "


def run_prompting(model, tokenizer, samples, prompt_type: str,
                  max_new_tokens: int, device, out_path: str):
    """samples: list of dicts {turns: [...], topic, knowledge?}.
    Writes one generation per line (reference output format)."""
    with open(out_path, "w", encoding="utf-8") as f:
        for s in samples:
            if prompt_type == "knowledge":
                prompt = _format_knowledge_prompt(
                    s["turns"], s.get("topic", ""))
            else:
                prompt = _format_response_prompt(
                    s["turns"], s.get("knowledge", ""))
            text = greedy_generate(model, tokenizer, prompt,
                                   max_new_tokens, device)
            f.write(text.replace("\n", " ") + "\n")
