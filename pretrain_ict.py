#!/usr/bin/env python3
"""Inverse Cloze Task (ICT) biencoder pretraining (reference
pretrain_ict.py).

Trains a query tower to retrieve the context block a sentence was
removed from: in-batch softmax over query·context inner products,
all-gathered across data-parallel ranks.  The corpus here is synthetic
(each context block shares a token prefix with its query, so retrieval
is learnable); plug a real ICT dataset in via
``train_valid_test_datasets_provider``.

  torchrun --nproc-per-node 1 --master-addr 127.0.0.1 pretrain_ict.py \
      --num-layers 4 --hidden-size 256 --num-attention-heads 8 \
      --seq-length 64 --micro-batch-size 8 --global-batch-size 8 \
      --train-iters 100 --lr 1e-4
"""

import functools
import math
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

import torch
import torch.nn.functional as F

from megatronapp_amd.core import parallel_state
from megatronapp_amd.core.enums import ModelType
from megatronapp_amd.core.models.bert.bert_layer_specs import (
    get_bert_layer_local_spec,
)
from megatronapp_amd.core.models.biencoder import (
    AllgatherFromDataParallelRegion,
    biencoder_model_provider,
)
from megatronapp_amd.training.arguments import (
    core_transformer_config_from_args,
)
from megatronapp_amd.training.global_vars import get_args
from megatronapp_amd.training.training import pretrain


def add_ict_extra_args(parser):
    g = parser.add_argument_group("biencoder")
    g.add_argument("--biencoder-projection-dim", type=int, default=0)
    g.add_argument("--biencoder-shared-query-context-model",
                   action="store_true")
    g.add_argument("--retriever-score-scaling", action="store_true")
    g.add_argument("--retriever-report-topk-accuracies", nargs="+",
                   type=int, default=[1])
    return parser


def model_provider(pre_process=True, post_process=True, vp_stage=None):
    args = get_args()
    config = core_transformer_config_from_args(args)
    return biencoder_model_provider(
        config=config,
        transformer_layer_spec=get_bert_layer_local_spec(),
        vocab_size=args.padded_vocab_size,
        max_sequence_length=args.max_position_embeddings,
        projection_dim=args.biencoder_projection_dim,
        shared_query_context_model=args.biencoder_shared_query_context_model)


class MockICTDataset(torch.utils.data.Dataset):
    """Query = sentence span; context = block sharing its prefix."""

    def __init__(self, n, seq_length, vocab_size, seed=1234):
        self.n = n
        self.seq_length = seq_length
        self.vocab_size = vocab_size
        self.seed = seed

    def __len__(self):
        return self.n

    def __getitem__(self, idx):
        g = torch.Generator().manual_seed(self.seed + idx)
        s = self.seq_length
        half = s // 2
        topic = torch.randint(0, self.vocab_size, (half,), generator=g)
        q = torch.cat([topic,
                       torch.randint(0, self.vocab_size, (s - half,),
                                     generator=g)])
        c = torch.cat([topic,
                       torch.randint(0, self.vocab_size, (s - half,),
                                     generator=g)])
        return {"query_tokens": q, "query_mask": torch.ones(s),
                "context_tokens": c, "context_mask": torch.ones(s)}


def train_valid_test_datasets_provider(train_val_test_num_samples):
    args = get_args()
    mk = lambda n, seed: MockICTDataset(
        max(n or 0, 1), args.seq_length, args.padded_vocab_size, seed)
    return (mk(train_val_test_num_samples[0], 1234),
            mk(train_val_test_num_samples[1], 4321),
            mk(train_val_test_num_samples[2], 5678))


def loss_func(output_tensor):
    """In-batch retrieval NLL over the DP-global batch
    (reference pretrain_ict.py:72-115)."""
    args = get_args()
    query_logits, context_logits = output_tensor
    all_q = AllgatherFromDataParallelRegion.apply(query_logits)
    all_c = AllgatherFromDataParallelRegion.apply(context_logits)
    scores = all_q @ all_c.t()
    if args.retriever_score_scaling:
        scores = scores / math.sqrt(args.hidden_size)
    logprobs = F.log_softmax(scores.float(), dim=1)
    n = logprobs.shape[0]
    labels = torch.arange(n, device=logprobs.device)
    loss = F.nll_loss(logprobs, labels, reduction="mean")
    stats = {"lm loss": loss.detach().clone()}
    with torch.no_grad():
        ranks = logprobs.argsort(dim=1, descending=True)
        for k in args.retriever_report_topk_accuracies:
            acc = (ranks[:, :k] == labels.unsqueeze(1)).any(1).float().mean()
            stats[f"top{k}_acc"] = acc * 100
    dp = parallel_state.get_data_parallel_world_size()
    return loss * dp, stats


def forward_step(data_iterator, model):
    data = next(data_iterator)
    device = "cuda" if torch.cuda.is_available() else "cpu"
    b = {k: v.to(device, non_blocking=True) for k, v in data.items()}
    qt = torch.zeros_like(b["query_tokens"])
    ct = torch.zeros_like(b["context_tokens"])
    output_tensor = model(b["query_tokens"], b["query_mask"], qt,
                          b["context_tokens"], b["context_mask"], ct)
    return output_tensor, functools.partial(loss_func)


if __name__ == "__main__":
    pretrain(train_valid_test_datasets_provider, model_provider,
             ModelType.encoder_or_decoder, forward_step,
             extra_args_provider=add_ict_extra_args,
             args_defaults={"tokenizer_type": "NullTokenizer"})
