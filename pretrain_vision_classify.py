#!/usr/bin/env python3
"""ViT image classification pretraining (reference
pretrain_vision_classify.py).

No image corpus exists in this environment, so the dataset is synthetic
with class-dependent statistics (so accuracy is learnable); swap
``MockImageDataset`` for a real loader to train on data.

  torchrun --nproc-per-node 1 --master-addr 127.0.0.1 \
      pretrain_vision_classify.py --num-layers 4 --hidden-size 256 \
      --num-attention-heads 8 --img-h 64 --img-w 64 --patch-dim 16 \
      --num-classes 10 --micro-batch-size 8 --global-batch-size 8 \
      --train-iters 100 --lr 1e-4
"""

import functools
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

import torch
import torch.nn.functional as F

from megatronapp_amd.core import parallel_state
from megatronapp_amd.core.enums import ModelType
from megatronapp_amd.core.models.vision import (
    VitClassificationModel,
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
    g.add_argument("--num-classes", type=int, default=1000)
    g.add_argument("--finetune-head", action="store_true")
    return parser


def model_provider(pre_process=True, post_process=True, vp_stage=None):
    args = get_args()
    config = core_transformer_config_from_args(args)
    return VitClassificationModel(
        config=config, transformer_layer_spec=get_vit_layer_local_spec(),
        num_classes=args.num_classes, patch_dim=args.patch_dim,
        img_h=args.img_h, img_w=args.img_w, finetune=args.finetune_head)


class MockImageDataset(torch.utils.data.Dataset):
    """Random images whose per-class channel bias makes labels
    learnable."""

    def __init__(self, n, img_h, img_w, num_classes, seed=1234):
        self.n = n
        self.img_h = img_h
        self.img_w = img_w
        self.num_classes = num_classes
        self.seed = seed

    def __len__(self):
        return self.n

    def __getitem__(self, idx):
        g = torch.Generator().manual_seed(self.seed + idx)
        label = int(torch.randint(0, self.num_classes, (1,), generator=g))
        img = torch.randn(3, self.img_h, self.img_w, generator=g)
        img += 0.5 * (label - self.num_classes / 2) / self.num_classes
        return {"images": img, "labels": torch.tensor(label)}


def train_valid_test_datasets_provider(train_val_test_num_samples):
    args = get_args()
    mk = lambda n, seed: MockImageDataset(
        max(n or 0, 1), args.img_h, args.img_w, args.num_classes, seed)
    return (mk(train_val_test_num_samples[0], 1234),
            mk(train_val_test_num_samples[1], 4321),
            mk(train_val_test_num_samples[2], 5678))


def loss_func(labels, output_tensor):
    logits = output_tensor.contiguous().float()
    loss = F.cross_entropy(logits, labels)
    acc = (logits.argmax(-1) == labels).float().mean()
    averaged = loss.detach().clone()
    if parallel_state.get_data_parallel_world_size() > 1:
        torch.distributed.all_reduce(
            averaged, group=parallel_state.get_data_parallel_group())
        averaged /= parallel_state.get_data_parallel_world_size()
    return loss, {"lm loss": averaged, "accuracy": acc * 100}


def forward_step(data_iterator, model):
    data = next(data_iterator)
    device = "cuda" if torch.cuda.is_available() else "cpu"
    images = data["images"].to(device, non_blocking=True)
    labels = data["labels"].to(device, non_blocking=True)
    output_tensor = model(images)
    return output_tensor, functools.partial(loss_func, labels)


if __name__ == "__main__":
    pretrain(train_valid_test_datasets_provider, model_provider,
             ModelType.encoder_or_decoder, forward_step,
             extra_args_provider=add_vision_extra_args,
             args_defaults={"tokenizer_type": "NullTokenizer",
                            "vocab_size": 1})
