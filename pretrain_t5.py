#!/usr/bin/env python3
"""T5 pretraining entry (reference pretrain_t5.py): span-corruption-style
objective over the encoder-decoder (mock data: the decoder reconstructs
a shifted window of the encoder input)."""

import functools
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

import numpy as np
import torch

from megatronapp_amd.core.enums import ModelType
from megatronapp_amd.core.models.t5 import T5Model
from megatronapp_amd.core.models.t5.t5_spec import (
    get_t5_decoder_layer_spec,
    get_t5_encoder_layer_spec,
)
from megatronapp_amd.training.arguments import core_transformer_config_from_args
from megatronapp_amd.training.global_vars import get_args
from megatronapp_amd.training.training import pretrain


class MockT5Dataset(torch.utils.data.Dataset):
    def __init__(self, seed, seq, dec_seq, vocab, num_samples):
        self.seed, self.seq, self.dec_seq = seed, seq, dec_seq
        self.vocab, self.num_samples = vocab, num_samples

    def __len__(self):
        return self.num_samples

    def __getitem__(self, idx):
        rng = np.random.default_rng(self.seed + int(idx))
        enc = rng.integers(0, self.vocab, self.seq, dtype=np.int64)
        tgt = enc[: self.dec_seq + 1].copy()
        return {
            "text_enc": torch.from_numpy(enc),
            "text_dec": torch.from_numpy(tgt[:-1]),
            "labels": torch.from_numpy(tgt[1:]),
            "loss_mask": torch.ones(self.dec_seq),
            "enc_mask": torch.ones(self.seq, dtype=torch.int64),
        }


def model_provider(pre_process=True, post_process=True, vp_stage=None):
    args = get_args()
    config = core_transformer_config_from_args(args)
    return T5Model(
        config=config,
        encoder_layer_spec=get_t5_encoder_layer_spec(args.normalization),
        decoder_layer_spec=get_t5_decoder_layer_spec(args.normalization),
        vocab_size=args.padded_vocab_size,
        max_sequence_length=args.max_position_embeddings)


def train_valid_test_datasets_provider(sizes):
    args = get_args()
    return [MockT5Dataset(args.seed, args.seq_length,
                          max(args.seq_length // 2, 8),
                          args.padded_vocab_size, n or 1) if n else None
            for n in sizes]


def loss_func(loss_mask, output_tensor):
    losses = output_tensor.float()
    loss_mask = loss_mask.reshape(-1).float()
    loss = torch.sum(losses.reshape(-1) * loss_mask) / loss_mask.sum()
    return loss, {"lm loss": loss.detach()}


def forward_step(data_iterator, model):
    data = next(data_iterator)
    device = "cuda" if torch.cuda.is_available() else "cpu"
    data = {k: v.to(device) for k, v in data.items()}
    out = model(data["text_enc"], data["text_dec"], data["enc_mask"],
                lm_labels=data["labels"])
    return out, functools.partial(loss_func, data["loss_mask"])


if __name__ == "__main__":
    pretrain(train_valid_test_datasets_provider, model_provider,
             ModelType.encoder_and_decoder, forward_step,
             args_defaults={"tokenizer_type": "NullTokenizer"})
