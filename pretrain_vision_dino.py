#!/usr/bin/env python3
"""DINO self-distillation pretraining for ViT (reference
pretrain_vision_dino.py, compact): a student tower learns to match a
momentum (EMA) teacher's sharpened, centered output distribution over
two augmented crops of each image.  The teacher EMA update runs after
every step via the loss closure."""

import functools
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

import torch

from megatronapp_amd.core import parallel_state
from megatronapp_amd.core.enums import ModelType
from megatronapp_amd.core.models.vision import (
    DinoPretrainModel,
    get_vit_layer_local_spec,
)
from megatronapp_amd.training.arguments import (
    core_transformer_config_from_args,
)
from megatronapp_amd.training.global_vars import get_args
from megatronapp_amd.training.training import pretrain


def add_vision_extra_args(parser):
    g = parser.add_argument_group("vision")
    g.add_argument("--img-h", type=int, default=224)
    g.add_argument("--img-w", type=int, default=224)
    g.add_argument("--patch-dim", type=int, default=16)
    g.add_argument("--dino-out-dim", type=int, default=4096)
    g.add_argument("--dino-momentum", type=float, default=0.996)
    g.add_argument("--dino-teacher-temp", type=float, default=0.04)
    return parser


def model_provider(pre_process=True, post_process=True, vp_stage=None):
    args = get_args()
    config = core_transformer_config_from_args(args)
    return DinoPretrainModel(
        config=config, transformer_layer_spec=get_vit_layer_local_spec(),
        out_dim=args.dino_out_dim, patch_dim=args.patch_dim,
        img_h=args.img_h, img_w=args.img_w,
        momentum=args.dino_momentum,
        teacher_temp=args.dino_teacher_temp)


class MockCropsDataset(torch.utils.data.Dataset):
    """Two noisy crops of the same underlying random image."""

    def __init__(self, n, img_h, img_w, seed=1234):
        self.n = n
        self.img_h = img_h
        self.img_w = img_w
        self.seed = seed

    def __len__(self):
        return self.n

    def __getitem__(self, idx):
        g = torch.Generator().manual_seed(self.seed + idx)
        base = torch.randn(3, self.img_h, self.img_w, generator=g)
        a = base + 0.1 * torch.randn_like(base)
        b = base + 0.1 * torch.randn_like(base)
        return {"crop_a": a, "crop_b": b}


def train_valid_test_datasets_provider(train_val_test_num_samples):
    args = get_args()
    mk = lambda n, seed: MockCropsDataset(
        max(n or 0, 1), args.img_h, args.img_w, seed)
    return (mk(train_val_test_num_samples[0], 1234),
            mk(train_val_test_num_samples[1], 4321),
            mk(train_val_test_num_samples[2], 5678))


def loss_func(model, loss):
    if model.training:
        model.momentum_update()
    averaged = loss.detach().clone()
    if parallel_state.get_data_parallel_world_size() > 1:
        torch.distributed.all_reduce(
            averaged, group=parallel_state.get_data_parallel_group())
        averaged /= parallel_state.get_data_parallel_world_size()
    return loss, {"lm loss": averaged}


def forward_step(data_iterator, model):
    data = next(data_iterator)
    device = "cuda" if torch.cuda.is_available() else "cpu"
    a = data["crop_a"].to(device, non_blocking=True)
    b = data["crop_b"].to(device, non_blocking=True)
    unwrapped = model
    while hasattr(unwrapped, "module"):
        unwrapped = unwrapped.module
    loss = model(a, b)
    return loss, functools.partial(loss_func, unwrapped)


if __name__ == "__main__":
    pretrain(train_valid_test_datasets_provider, model_provider,
             ModelType.encoder_or_decoder, forward_step,
             extra_args_provider=add_vision_extra_args,
             args_defaults={"tokenizer_type": "NullTokenizer",
                            "vocab_size": 1})
