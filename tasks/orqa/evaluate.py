"""Open-retrieval QA retriever evaluation (reference
tasks/orqa/evaluate_orqa.py + evaluate_utils.py).

Embeds questions with the biencoder's query tower and evidence blocks
with its context tower, retrieves top-k blocks by exact MIPS, and
scores top-k retrieval accuracy: a question counts as a hit when any of
its gold answer strings appears in a retrieved block (the reference's
NQ string-match protocol).
"""

from __future__ import annotations

import json
from typing import List

import torch

from tools.retro.preprocess import BruteForceMIPSIndex


def load_qa_file(path: str):
    """JSON-lines {question, answers:[...]}, or TSV question\tanswers."""
    questions, answers = [], []
    with open(path, encoding="utf-8") as f:
        for line in f:
            line = line.strip()
            if not line:
                continue
            if line.startswith("{"):
                d = json.loads(line)
                questions.append(d["question"])
                answers.append(d["answers"])
            else:
                q, a = line.split("\t", 1)
                questions.append(q)
                answers.append(json.loads(a) if a.startswith("[")
                               else [a])
    return questions, answers


def load_evidence_file(path: str) -> List[str]:
    """JSON-lines {text} or plain text, one block per line."""
    blocks = []
    with open(path, encoding="utf-8") as f:
        for line in f:
            line = line.strip()
            if not line:
                continue
            blocks.append(json.loads(line)["text"]
                          if line.startswith("{") else line)
    return blocks


@torch.no_grad()
def evaluate_retriever(model, tokenizer, questions: List[str],
                       answers: List[List[str]], evidence: List[str],
                       seq_length: int, topk_list=(1, 5, 20)):
    """Returns {f"top{k}_accuracy": float}."""
    from tasks.data_utils import build_sample
    device = next(model.parameters()).device

    def embed(texts, tower):
        embs = []
        for i in range(0, len(texts), 32):
            ids, types, masks = [], [], []
            for t in texts[i:i + 32]:
                a, ty, ms = build_sample(tokenizer, t, None, seq_length)
                ids.append(a)
                types.append(ty)
                masks.append(ms)
            embs.append(tower(
                torch.stack(ids).to(device),
                torch.stack(masks).to(device),
                torch.stack(types).to(device)).float().cpu())
        return torch.cat(embs).numpy()

    q_emb = embed(questions, model.embed_query)
    e_emb = embed(evidence, model.embed_context)
    index = BruteForceMIPSIndex(e_emb, device="cpu")
    kmax = min(max(topk_list), len(evidence))
    top = index.search(q_emb, kmax)

    results = {}
    for k in topk_list:
        k_eff = min(k, kmax)
        hits = 0
        for qi in range(len(questions)):
            blocks = [evidence[b].lower() for b in top[qi, :k_eff]]
            if any(any(a.lower() in blk for blk in blocks)
                   for a in answers[qi]):
                hits += 1
        results[f"top{k}_accuracy"] = hits / max(len(questions), 1)
    return results


def main(tokenizer):
    from megatronapp_amd.training.arguments import (
        core_transformer_config_from_args)
    from megatronapp_amd.training.global_vars import get_args
    from megatronapp_amd.core.models.bert.bert_layer_specs import (
        get_bert_layer_local_spec)
    from megatronapp_amd.core.models.biencoder import (
        biencoder_model_provider)

    args = get_args()
    config = core_transformer_config_from_args(args)
    device = "cuda" if torch.cuda.is_available() else "cpu"
    model = biencoder_model_provider(
        config=config,
        transformer_layer_spec=get_bert_layer_local_spec(),
        vocab_size=args.padded_vocab_size,
        max_sequence_length=args.max_position_embeddings,
        projection_dim=getattr(args, "biencoder_projection_dim", 0) or 0,
    ).to(device).eval()
    if args.load:
        sd = torch.load(args.load, map_location="cpu",
                        weights_only=False)
        model.load_state_dict(sd.get("model", sd), strict=False)

    questions, answers = load_qa_file(args.qa_data)
    evidence = load_evidence_file(args.evidence_data)
    out = evaluate_retriever(model, tokenizer, questions, answers,
                             evidence, args.seq_length)
    for k, v in sorted(out.items()):
        print(f"orqa {k}: {v * 100:.2f}%", flush=True)
    return out
