#!/usr/bin/env python3
"""ViT masked-patch inpainting pretraining (reference
pretrain_vision_inpaint.py): random patches are zero-masked, the model
reconstructs them, and the loss is MSE restricted to the masked
regions.  Images are synthetic (smooth random fields) in this offline
environment."""

import functools
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

import torch
import torch.nn.functional as F

from megatronapp_amd.core import parallel_state
from megatronapp_amd.core.enums import ModelType
from megatronapp_amd.core.models.vision import (
    VitInpaintingModel,
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
    g.add_argument("--mask-factor", type=float, default=0.25,
                   help="fraction of patches masked out")
    return parser


def model_provider(pre_process=True, post_process=True, vp_stage=None):
    args = get_args()
    config = core_transformer_config_from_args(args)
    return VitInpaintingModel(
        config=config, transformer_layer_spec=get_vit_layer_local_spec(),
        patch_dim=args.patch_dim, img_h=args.img_h, img_w=args.img_w)


class MockInpaintDataset(torch.utils.data.Dataset):
    """Smooth random images + random patch masks."""

    def __init__(self, n, img_h, img_w, patch, mask_factor, seed=1234):
        self.n = n
        self.img_h = img_h
        self.img_w = img_w
        self.patch = patch
        self.mask_factor = mask_factor
        self.seed = seed

    def __len__(self):
        return self.n

    def __getitem__(self, idx):
        g = torch.Generator().manual_seed(self.seed + idx)
        low = torch.randn(3, self.img_h // 8, self.img_w // 8, generator=g)
        img = F.interpolate(low.unsqueeze(0), (self.img_h, self.img_w),
                            mode="bilinear", align_corners=False)[0]
        gh, gw = self.img_h // self.patch, self.img_w // self.patch
        patch_mask = (torch.rand(gh, gw, generator=g) <
                      self.mask_factor).float()
        mask = patch_mask.repeat_interleave(
            self.patch, 0).repeat_interleave(self.patch, 1)
        mask = mask.unsqueeze(0).expand(3, -1, -1)   # 1 = masked
        return {"images": img, "masks": mask}


def train_valid_test_datasets_provider(train_val_test_num_samples):
    args = get_args()
    mk = lambda n, seed: MockInpaintDataset(
        max(n or 0, 1), args.img_h, args.img_w, args.patch_dim,
        args.mask_factor, seed)
    return (mk(train_val_test_num_samples[0], 1234),
            mk(train_val_test_num_samples[1], 4321),
            mk(train_val_test_num_samples[2], 5678))


def loss_func(images, masks, output_tensor):
    """MSE over the masked pixels only
    (reference pretrain_vision_inpaint.py loss_func)."""
    recon = output_tensor.float() * masks
    target = images.float() * masks
    loss = F.mse_loss(recon, target)
    averaged = loss.detach().clone()
    if parallel_state.get_data_parallel_world_size() > 1:
        torch.distributed.all_reduce(
            averaged, group=parallel_state.get_data_parallel_group())
        averaged /= parallel_state.get_data_parallel_world_size()
    return loss, {"lm loss": averaged}


def forward_step(data_iterator, model):
    data = next(data_iterator)
    device = "cuda" if torch.cuda.is_available() else "cpu"
    images = data["images"].to(device, non_blocking=True)
    masks = data["masks"].to(device, non_blocking=True)
    masked = images * (1 - masks)
    output_tensor = model(masked)
    return output_tensor, functools.partial(loss_func, images, masks)


if __name__ == "__main__":
    pretrain(train_valid_test_datasets_provider, model_provider,
             ModelType.encoder_or_decoder, forward_step,
             extra_args_provider=add_vision_extra_args,
             args_defaults={"tokenizer_type": "NullTokenizer",
                            "vocab_size": 1})
