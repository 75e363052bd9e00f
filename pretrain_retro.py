#!/usr/bin/env python3
"""Retro pretraining entry point (reference pretrain_retro.py).

Retrieval-augmented GPT: each sequence chunk is paired with
``retro_num_neighbors`` retrieved neighbor chunks (plus continuations),
encoded by a small transformer and attended to via chunked
cross-attention.  With ``--retro-project-dir`` the neighbors come from a
preprocessed retrieval database (tools/retro/preprocess.py: chunk db ->
BERT embeddings -> exact MIPS neighbor search); without it a synthetic
neighbor stream exercises the same training path.

  torchrun --nproc-per-node 1 --master-addr 127.0.0.1 pretrain_retro.py \
      --num-layers 12 --hidden-size 512 --num-attention-heads 8 \
      --seq-length 512 --retro-chunk-length 64 --micro-batch-size 2 \
      --global-batch-size 4 --train-iters 100 --lr 1e-4
"""

import functools
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

import numpy as np
import torch

from megatronapp_amd.core import parallel_state
from megatronapp_amd.core.enums import ModelType
from megatronapp_amd.core.models.retro import (
    RetroConfig,
    RetroModel,
    get_retro_decoder_block_spec,
)
from megatronapp_amd.training.arguments import (
    core_transformer_config_from_args,
)
from megatronapp_amd.training.global_vars import get_args
from megatronapp_amd.training.training import pretrain


def add_retro_extra_args(parser):
    g = parser.add_argument_group("retro")
    g.add_argument("--retro-chunk-length", type=int, default=64)
    g.add_argument("--retro-num-neighbors", type=int, default=2)
    g.add_argument("--retro-num-retrieved-chunks", type=int, default=2)
    g.add_argument("--retro-encoder-num-layers", type=int, default=2)
    g.add_argument("--retro-encoder-hidden-dropout", type=float, default=0.1)
    g.add_argument("--retro-encoder-attention-dropout", type=float,
                   default=0.1)
    g.add_argument("--retro-project-dir", default=None)
    return parser


def retro_config_from_args(args):
    base = core_transformer_config_from_args(args)
    import dataclasses
    kw = dataclasses.asdict(base)
    kw = {f.name: getattr(base, f.name)
          for f in dataclasses.fields(base) if f.init}
    return RetroConfig(
        retro_chunk_length=args.retro_chunk_length,
        retro_num_neighbors=args.retro_num_neighbors,
        retro_num_retrieved_chunks=args.retro_num_retrieved_chunks,
        retro_encoder_num_layers=args.retro_encoder_num_layers,
        retro_encoder_hidden_dropout=args.retro_encoder_hidden_dropout,
        retro_encoder_attention_dropout=args.retro_encoder_attention_dropout,
        retro_project_dir=args.retro_project_dir,
        **kw)


def model_provider(pre_process=True, post_process=True, vp_stage=None):
    args = get_args()
    config = retro_config_from_args(args)
    spec = get_retro_decoder_block_spec(config)
    return RetroModel(
        config=config, transformer_layer_spec=spec,
        vocab_size=args.padded_vocab_size,
        max_sequence_length=args.max_position_embeddings,
        pre_process=pre_process, post_process=post_process,
        share_embeddings_and_output_weights=not args.untie_embeddings_and_output_weights,
    )


class MockRetroDataset(torch.utils.data.Dataset):
    """GPT-style token stream + synthetic retrieved neighbors."""

    def __init__(self, n, seq_length, vocab_size, chunk, k, retrieved,
                 seed=1234):
        self.n = n
        self.seq_length = seq_length
        self.vocab_size = vocab_size
        self.chunk = chunk
        self.k = k
        self.retrieved = retrieved
        self.seed = seed

    def __len__(self):
        return self.n

    def __getitem__(self, idx):
        g = torch.Generator().manual_seed(self.seed + idx)
        s = self.seq_length
        l = s // self.chunk
        tokens = torch.randint(0, self.vocab_size, (s,), generator=g)
        labels = torch.roll(tokens, -1)
        labels[-1] = 0
        return {
            "tokens": tokens,
            "labels": labels,
            "loss_mask": torch.ones(s),
            "position_ids": torch.arange(s),
            # [k*l, r] neighbor chunks for this sample
            "neighbor_tokens": torch.randint(
                0, self.vocab_size, (self.k * l, self.retrieved),
                generator=g),
        }


class RetroProjectDataset(torch.utils.data.Dataset):
    """Samples from a preprocessed retro project directory
    (tools/retro/preprocess.py): each sample is ``l`` consecutive chunks
    with their retrieved neighbors, ordered (l, k) so the batch
    flattens to the (b, l, k) layout the neighbor encoder reshapes."""

    def __init__(self, project_dir, seq_length, num_neighbors,
                 num_retrieved_chunks):
        from tools.retro.preprocess import (
            load_retro_project, load_neighbor_tokens)
        self._load_neighbor_tokens = load_neighbor_tokens
        self.chunks, self.doc_ids, self.neighbors, self.meta = \
            load_retro_project(project_dir)
        self.m = self.meta["chunk_length"]
        assert seq_length % self.m == 0
        self.l = seq_length // self.m
        self.k = min(num_neighbors, self.neighbors.shape[1])
        self.num_retrieved_chunks = num_retrieved_chunks
        self.n = max(len(self.chunks) // self.l, 1)

    def __len__(self):
        return self.n

    def __getitem__(self, idx):
        c0 = idx * self.l
        sel = [min(c0 + j, len(self.chunks) - 1) for j in range(self.l)]
        tokens = torch.as_tensor(
            np.concatenate([self.chunks[c] for c in sel]),
            dtype=torch.long)
        labels = torch.roll(tokens, -1)
        labels[-1] = 0
        nts = [self._load_neighbor_tokens(
                   self.chunks, self.doc_ids, self.neighbors[c, :self.k],
                   self.meta["pad_id"], self.num_retrieved_chunks)
               for c in sel]                     # l × [k, r]
        neighbor_tokens = torch.as_tensor(
            np.concatenate(nts, axis=0), dtype=torch.long)  # [l*k, r]
        return {"tokens": tokens, "labels": labels,
                "loss_mask": torch.ones(len(tokens)),
                "position_ids": torch.arange(len(tokens)),
                "neighbor_tokens": neighbor_tokens}


def train_valid_test_datasets_provider(train_val_test_num_samples):
    args = get_args()
    if args.retro_project_dir:
        import numpy  # noqa: F401 - RetroProjectDataset uses np
        ds = RetroProjectDataset(
            args.retro_project_dir, args.seq_length,
            args.retro_num_neighbors, args.retro_num_retrieved_chunks)
        return ds, ds, ds
    r = args.retro_num_retrieved_chunks * args.retro_chunk_length
    mk = lambda n, seed: MockRetroDataset(
        max(n or 0, 1), args.seq_length, args.padded_vocab_size,
        args.retro_chunk_length, args.retro_num_neighbors, r, seed)
    return (mk(train_val_test_num_samples[0], 1234),
            mk(train_val_test_num_samples[1], 4321),
            mk(train_val_test_num_samples[2], 5678))


def loss_func(loss_mask, output_tensor):
    losses = output_tensor.float()
    loss_mask = loss_mask.reshape(-1).float()
    loss = torch.sum(losses.reshape(-1) * loss_mask) / loss_mask.sum()
    averaged = loss.detach().clone()
    if parallel_state.get_data_parallel_world_size() > 1:
        torch.distributed.all_reduce(
            averaged, group=parallel_state.get_data_parallel_group())
        averaged /= parallel_state.get_data_parallel_world_size()
    return loss, {"lm loss": averaged}


def forward_step(data_iterator, model):
    data = next(data_iterator)
    device = "cuda" if torch.cuda.is_available() else "cpu"
    b = {k: v.to(device, non_blocking=True) for k, v in data.items()}
    tokens = b["tokens"]
    bs, ns = tokens.shape
    # neighbor_tokens: [bs, k*l, r] -> [k*bs*l, r] (neighbor-major
    # grouping expected by the encoder reshape)
    nt = b["neighbor_tokens"]
    k_l, r = nt.shape[1], nt.shape[2]
    context_ids = nt.reshape(bs * k_l, r)
    context_pos = torch.arange(
        r, device=device).expand(context_ids.shape[0], -1)
    output_tensor = model(
        tokens, b["position_ids"], context_input_ids=context_ids,
        context_position_ids=context_pos, labels=b["labels"])
    return output_tensor, functools.partial(loss_func, b["loss_mask"])


if __name__ == "__main__":
    pretrain(train_valid_test_datasets_provider, model_provider,
             ModelType.retro_decoder, forward_step,
             extra_args_provider=add_retro_extra_args,
             args_defaults={"tokenizer_type": "NullTokenizer"})
