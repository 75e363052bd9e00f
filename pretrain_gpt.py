#!/usr/bin/env python3
"""GPT pretraining entry point (reference pretrain_gpt.py).

Defines model_provider / dataset provider / forward_step and calls
megatronapp_amd.training.pretrain().  Launch with torchrun, one process
per GPU, e.g.:

  torchrun --nproc-per-node 4 --master-addr 127.0.0.1 pretrain_gpt.py \
      --num-layers 24 --hidden-size 2048 --num-attention-heads 16 \
      --seq-length 2048 --micro-batch-size 2 --global-batch-size 16 \
      --pipeline-model-parallel-size 4 --bf16 --mock-data \
      --train-iters 20 --lr 1e-4 [--trace] [--use-dpp] [...]
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
    MockGPTDataset,
)
from megatronapp_amd.core.enums import ModelType
from megatronapp_amd.core.models.gpt import GPTModel
from megatronapp_amd.core.models.gpt.gpt_layer_specs import (
    get_gpt_decoder_block_spec,
    get_gpt_layer_local_spec,
)
from megatronapp_amd.core.tensor_parallel.data import broadcast_data
from megatronapp_amd.core.utils import get_batch_on_this_cp_rank
from megatronapp_amd.training.arguments import core_transformer_config_from_args
from megatronapp_amd.training.global_vars import get_args
from megatronapp_amd.training.training import pretrain


def model_provider(pre_process=True, post_process=True, vp_stage=None):
    args = get_args()
    config = core_transformer_config_from_args(args)
    use_flash = args.attention_backend in ("auto", "flash")
    if args.num_experts is not None:
        spec = get_gpt_decoder_block_spec(config)
    else:
        spec = get_gpt_layer_local_spec(
            normalization=args.normalization, qk_layernorm=args.qk_layernorm,
            use_flash=use_flash)
    model = GPTModel(
        config=config,
        transformer_layer_spec=spec,
        vocab_size=args.padded_vocab_size,
        max_sequence_length=args.max_position_embeddings,
        pre_process=pre_process, post_process=post_process,
        parallel_output=True,
        share_embeddings_and_output_weights=not args.untie_embeddings_and_output_weights,
        position_embedding_type=args.position_embedding_type,
        rotary_percent=args.rotary_percent, rotary_base=args.rotary_base,
        vp_stage=vp_stage)
    return model


def train_valid_test_datasets_provider(train_val_test_num_samples):
    args = get_args()
    config = GPTDatasetConfig(
        random_seed=args.seed, sequence_length=args.seq_length,
        vocab_size=args.padded_vocab_size,
        reset_position_ids=args.reset_position_ids,
        reset_attention_mask=args.reset_attention_mask,
        eod_mask_loss=args.eod_mask_loss,
        create_attention_mask=args.create_attention_mask_in_dataloader,
        blend=args.data_path, split=args.split,
        mock=args.mock_data or args.data_path is None)
    builder = BlendedMegatronDatasetBuilder(
        GPTDataset, train_val_test_num_samples, lambda: True, config)
    return builder.build()


def get_batch(data_iterator):
    """Pull + broadcast one microbatch (reference pretrain_gpt get_batch)."""
    args = get_args()
    if (not parallel_state.is_pipeline_first_stage()) and \
            (not parallel_state.is_pipeline_last_stage()):
        return None, None, None, None, None
    data = next(data_iterator) if data_iterator is not None else None
    keys = ["tokens", "labels", "loss_mask", "position_ids"]
    if args.create_attention_mask_in_dataloader:
        keys.append("attention_mask")
    if parallel_state.get_tensor_model_parallel_world_size() > 1:
        data_b = broadcast_data(keys[:2], data, torch.int64)
        data_f = broadcast_data(keys[2:4], data, torch.float32)
        tokens, labels = data_b["tokens"], data_b["labels"]
        loss_mask = data_f["loss_mask"]
        position_ids = data_f["position_ids"].long()
        attention_mask = None
    else:
        device = "cuda" if torch.cuda.is_available() else "cpu"
        tokens = data["tokens"].to(device, non_blocking=True)
        labels = data["labels"].to(device, non_blocking=True)
        loss_mask = data["loss_mask"].to(device, non_blocking=True)
        position_ids = data["position_ids"].to(device, non_blocking=True)
        attention_mask = data.get("attention_mask")
        if attention_mask is not None:
            attention_mask = attention_mask.to(device, non_blocking=True)
    batch = {"tokens": tokens, "labels": labels, "loss_mask": loss_mask,
             "position_ids": position_ids, "attention_mask": attention_mask}
    batch = get_batch_on_this_cp_rank(batch)
    return (batch["tokens"], batch["labels"], batch["loss_mask"],
            batch["attention_mask"], batch["position_ids"])


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
    tokens, labels, loss_mask, attention_mask, position_ids = get_batch(
        data_iterator)
    output_tensor = model(tokens, position_ids, attention_mask=attention_mask,
                          labels=labels)
    import functools
    return output_tensor, functools.partial(loss_func, loss_mask)


if __name__ == "__main__":
    pretrain(train_valid_test_datasets_provider, model_provider,
             ModelType.encoder_or_decoder, forward_step,
             args_defaults={"tokenizer_type": "NullTokenizer"})
