#!/usr/bin/env python3
"""BERT pretraining entry (reference pretrain_bert.py): masked LM (+ NSP
binary head) over the bidirectional encoder."""

import functools
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

import torch

from megatronapp_amd.core import parallel_state
from megatronapp_amd.core.datasets.gpt_dataset import GPTDatasetConfig
from megatronapp_amd.core.datasets.masked_dataset import MockBertDataset
from megatronapp_amd.core.enums import ModelType
from megatronapp_amd.core.models.bert import BertModel
from megatronapp_amd.core.models.bert.bert_layer_specs import (
    get_bert_layer_local_spec,
)
from megatronapp_amd.training.arguments import core_transformer_config_from_args
from megatronapp_amd.training.global_vars import get_args
from megatronapp_amd.training.training import pretrain


def model_provider(pre_process=True, post_process=True, vp_stage=None):
    args = get_args()
    config = core_transformer_config_from_args(args)
    return BertModel(
        config=config,
        transformer_layer_spec=get_bert_layer_local_spec(
            normalization=args.normalization),
        vocab_size=args.padded_vocab_size,
        max_sequence_length=args.max_position_embeddings,
        pre_process=pre_process, post_process=post_process,
        add_binary_head=True,
        share_embeddings_and_output_weights=not args.untie_embeddings_and_output_weights)


def train_valid_test_datasets_provider(train_val_test_num_samples):
    args = get_args()
    config = GPTDatasetConfig(random_seed=args.seed,
                              sequence_length=args.seq_length,
                              vocab_size=args.padded_vocab_size)
    return [MockBertDataset(config, n or 1) if n else None
            for n in train_val_test_num_samples]


def get_batch(data_iterator):
    data = next(data_iterator)
    device = "cuda" if torch.cuda.is_available() else "cpu"
    return {k: v.to(device, non_blocking=True) for k, v in data.items()}


def loss_func(loss_mask, sentence_order, output_tensor):
    lm_loss_, binary_logits = output_tensor
    lm_loss_ = lm_loss_.float()
    loss_mask = loss_mask.float()
    lm_loss = torch.sum(lm_loss_.reshape(-1) * loss_mask.reshape(-1)) / \
        loss_mask.sum().clamp(min=1)
    loss = lm_loss
    averaged = {"lm loss": lm_loss.detach()}
    if binary_logits is not None:
        sop_loss = torch.nn.functional.cross_entropy(
            binary_logits.float(), sentence_order)
        loss = loss + sop_loss
        averaged["sop loss"] = sop_loss.detach()
    return loss, averaged


def forward_step(data_iterator, model):
    batch = get_batch(data_iterator)
    labels = batch["labels"].clamp(min=0)  # -1 -> 0 (masked out by loss)
    output = model(batch["text"], batch["padding_mask"],
                   tokentype_ids=batch["types"], lm_labels=labels)
    return output, functools.partial(loss_func, batch["loss_mask"],
                                     batch["is_random"])


if __name__ == "__main__":
    pretrain(train_valid_test_datasets_provider, model_provider,
             ModelType.encoder_or_decoder, forward_step,
             args_defaults={"tokenizer_type": "NullTokenizer"})
