#!/usr/bin/env python3
"""LLaVA vision-language pretraining entry point (reference
pretrain_vlm.py).

Builds a CLIP ViT tower + projector + GPT decoder and trains on
image+text batches where each ``<image>`` token expands into the tile's
patch embeddings.  This environment has no image corpus, so the dataset
is synthetic (random tiles + token streams with one image token per
sample); swap ``MockVLMDataset`` for a real pipeline to train on data.

  torchrun --nproc-per-node 1 --master-addr 127.0.0.1 pretrain_vlm.py \
      --num-layers 4 --hidden-size 256 --num-attention-heads 8 \
      --seq-length 128 --micro-batch-size 2 --global-batch-size 4 \
      --img-h 64 --img-w 64 --patch-dim 16 --train-iters 10 --lr 1e-4
"""

import functools
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

import torch

from megatronapp_amd.core import parallel_state
from megatronapp_amd.core.enums import ModelType
from megatronapp_amd.core.models.multimodal import (
    DEFAULT_IMAGE_TOKEN_INDEX,
    LLaVAModel,
)
from megatronapp_amd.core.models.gpt.gpt_layer_specs import (
    get_gpt_layer_local_spec,
)
from megatronapp_amd.core.models.vision import get_vit_layer_local_spec
from megatronapp_amd.core.transformer_config import TransformerConfig
from megatronapp_amd.training.arguments import (
    core_transformer_config_from_args,
)
from megatronapp_amd.training.global_vars import get_args
from megatronapp_amd.training.training import pretrain


def add_vlm_extra_args(parser):
    g = parser.add_argument_group("multimodal")
    g.add_argument("--img-h", type=int, default=336)
    g.add_argument("--img-w", type=int, default=336)
    g.add_argument("--patch-dim", type=int, default=14)
    g.add_argument("--vision-num-layers", type=int, default=2)
    g.add_argument("--vision-hidden-size", type=int, default=256)
    g.add_argument("--vision-num-attention-heads", type=int, default=8)
    g.add_argument("--disable-vision-class-token", action="store_true")
    g.add_argument("--freeze-LM", action="store_true")
    g.add_argument("--freeze-ViT", action="store_true")
    return parser


def model_provider(pre_process=True, post_process=True, vp_stage=None):
    args = get_args()
    language_config = core_transformer_config_from_args(args)
    vision_config = TransformerConfig(
        num_layers=args.vision_num_layers,
        hidden_size=args.vision_hidden_size,
        num_attention_heads=args.vision_num_attention_heads,
        ffn_hidden_size=4 * args.vision_hidden_size,
        hidden_dropout=0.0, attention_dropout=0.0,
        bf16=language_config.bf16, params_dtype=language_config.params_dtype,
        masked_softmax_fusion=False)
    use_flash = args.attention_backend in ("auto", "flash")
    model = LLaVAModel(
        language_transformer_config=language_config,
        language_transformer_layer_spec=get_gpt_layer_local_spec(
            normalization=args.normalization, use_flash=use_flash),
        language_vocab_size=args.padded_vocab_size,
        language_max_sequence_length=args.max_position_embeddings,
        vision_transformer_config=vision_config,
        vision_transformer_layer_spec=get_vit_layer_local_spec(),
        drop_vision_class_token=not args.disable_vision_class_token,
        img_h=args.img_h, img_w=args.img_w, patch_dim=args.patch_dim,
        position_embedding_type=args.position_embedding_type,
        share_embeddings_and_output_weights=not args.untie_embeddings_and_output_weights,
    )
    model.freeze(freeze_language_model=args.freeze_LM,
                 freeze_vision_model=args.freeze_ViT,
                 freeze_vision_projection=False)
    return model


class MockVLMDataset(torch.utils.data.Dataset):
    """Random image tiles + token streams with one image token."""

    def __init__(self, n, seq_length, vocab_size, img_h, img_w, seed=1234):
        self.n = n
        self.seq_length = seq_length
        self.vocab_size = vocab_size
        self.img_h = img_h
        self.img_w = img_w
        self.seed = seed

    def __len__(self):
        return self.n

    def __getitem__(self, idx):
        g = torch.Generator().manual_seed(self.seed + idx)
        s = self.seq_length
        tokens = torch.randint(0, self.vocab_size, (s,), generator=g)
        pos = int(torch.randint(1, max(2, s // 4), (1,), generator=g))
        tokens[pos] = DEFAULT_IMAGE_TOKEN_INDEX
        labels = torch.roll(tokens, -1)
        labels[-1] = 0
        loss_mask = torch.ones(s)
        loss_mask[pos] = 0.0
        return {
            "tokens": tokens,
            "labels": labels,
            "loss_mask": loss_mask,
            "position_ids": torch.arange(s),
            "images": torch.randn(3, self.img_h, self.img_w, generator=g),
        }


def train_valid_test_datasets_provider(train_val_test_num_samples):
    args = get_args()
    mk = lambda n, seed: MockVLMDataset(
        max(n or 0, 1), args.seq_length, args.padded_vocab_size,
        args.img_h, args.img_w, seed)
    return (mk(train_val_test_num_samples[0], 1234),
            mk(train_val_test_num_samples[1], 4321),
            mk(train_val_test_num_samples[2], 5678))


def get_batch(data_iterator):
    data = next(data_iterator)
    device = "cuda" if torch.cuda.is_available() else "cpu"
    return {k: v.to(device, non_blocking=True) for k, v in data.items()}


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
    b = get_batch(data_iterator)
    output_tensor, new_loss_mask = model(
        b["images"], b["tokens"], b["position_ids"],
        labels=b["labels"], loss_mask=b["loss_mask"])
    return output_tensor, functools.partial(loss_func, new_loss_mask)


if __name__ == "__main__":
    pretrain(train_valid_test_datasets_provider, model_provider,
             ModelType.encoder_or_decoder, forward_step,
             extra_args_provider=add_vlm_extra_args,
             args_defaults={"tokenizer_type": "NullTokenizer"})
