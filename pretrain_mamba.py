#!/usr/bin/env python3
"""Mamba pretraining entry point (reference pretrain_mamba.py).

  torchrun --nproc-per-node 1 --master-addr 127.0.0.1 pretrain_mamba.py \
      --num-layers 12 --hidden-size 768 --num-attention-heads 12 \
      --seq-length 1024 --micro-batch-size 4 --global-batch-size 16 \
      --bf16 --mock-data --train-iters 20 --lr 1e-4

``--hybrid-pattern`` takes a per-layer string of 'M' (mamba mixer) and
'*' (attention layer), e.g. MMM*MMM* — the reference's hybrid stacks.
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

import torch

from megatronapp_amd.core import parallel_state
from megatronapp_amd.core.datasets import (
    BlendedMegatronDatasetBuilder,
    GPTDataset,
    GPTDatasetConfig,
)
from megatronapp_amd.core.enums import ModelType
from megatronapp_amd.core.models.mamba import MambaModel
from megatronapp_amd.core.models.gpt.gpt_layer_specs import (
    get_gpt_layer_local_spec,
)
from megatronapp_amd.training.arguments import core_transformer_config_from_args
from megatronapp_amd.training.global_vars import get_args
from megatronapp_amd.training.training import pretrain
from pretrain_gpt import get_batch, loss_func, train_valid_test_datasets_provider


def model_provider(pre_process=True, post_process=True, vp_stage=None):
    args = get_args()
    config = core_transformer_config_from_args(args)
    pattern = getattr(args, "hybrid_pattern", None)
    attn_spec = None
    if pattern and "*" in pattern:
        attn_spec = get_gpt_layer_local_spec(
            normalization="RMSNorm",
            use_flash=args.attention_backend in ("auto", "flash"))
    return MambaModel(
        config=config, vocab_size=args.padded_vocab_size,
        max_sequence_length=args.max_position_embeddings,
        pre_process=pre_process, post_process=post_process,
        hybrid_pattern=pattern, attention_spec=attn_spec,
        share_embeddings_and_output_weights=not
        args.untie_embeddings_and_output_weights)


def forward_step(data_iterator, model):
    from functools import partial
    tokens, labels, loss_mask, attention_mask, position_ids = get_batch(
        data_iterator)
    output_tensor = model(tokens, position_ids, attention_mask=attention_mask,
                          labels=labels)
    return output_tensor, partial(loss_func, loss_mask)


def extra_args(parser):
    group = parser.add_argument_group("mamba")
    group.add_argument("--hybrid-pattern", default=None,
                       help="per-layer 'M'/'*' pattern (mamba/attention)")
    return parser


if __name__ == "__main__":
    pretrain(train_valid_test_datasets_provider, model_provider,
             ModelType.encoder_or_decoder, forward_step,
             extra_args_provider=extra_args)
