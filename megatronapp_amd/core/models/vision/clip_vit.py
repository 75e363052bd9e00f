"""CLIP/SigLIP ViT vision tower (reference:
core/models/vision/clip_vit_model.py:25-186).

Behavior parity:

* conv patch embed (stride = kernel = patch_dim; no bias for 'clip',
  bias for 'siglip'), optional class token prepended, learned position
  embeddings over ``num_patches + class_token_len``
* 'clip' applies a pre-norm before the transformer; 'siglip' applies a
  post-norm after it and forbids class tokens
* the transformer stack is our TransformerBlock without a final norm
  (reference passes post_process=False), bidirectional attention
* forward: [b, 3, H, W] -> [b, s, h]

MI355X notes: the block runs the fused HIP norm / bias-gelu kernels; the
conv patch embed stays on MIOpen (one conv per step, never hot).
"""

from __future__ import annotations

from typing import Optional

import torch

from ...transformer.module import MegatronModule
from ...transformer.spec_utils import ModuleSpec
from ...transformer.transformer_block import TransformerBlock
from ...transformer_config import TransformerConfig
from ...enums import ModelType


class CLIPViTModel(MegatronModule):
    """ViT image encoder producing one embedding per patch (+class)."""

    def __init__(
        self,
        transformer_config: TransformerConfig,
        transformer_layer_spec: ModuleSpec,
        add_class_token: bool = True,
        class_token_len: int = 1,
        patch_dim: int = 14,
        img_h: int = 336,
        img_w: int = 336,
        model_subtype: str = "clip",
    ) -> None:
        assert model_subtype in ("clip", "siglip"), model_subtype
        if model_subtype == "siglip":
            assert not add_class_token and class_token_len == 0, \
                "SigLIP does not use class tokens"
        super().__init__(config=transformer_config)

        h = transformer_config.hidden_size
        self.visual_hidden_size = h
        self.patch_dim = patch_dim
        self.img_h = img_h
        self.img_w = img_w
        assert img_h % patch_dim == 0 and img_w % patch_dim == 0
        self.num_patches = (img_h // patch_dim) * (img_w // patch_dim)
        self.add_class_token = add_class_token
        self.class_token_len = class_token_len
        self.seq_length = self.num_patches + \
            (class_token_len if add_class_token else 0)

        eps = transformer_config.layernorm_epsilon
        pdt = transformer_config.params_dtype
        self.ln_pre = self.ln_post = None
        if model_subtype == "clip":
            self.ln_pre = torch.nn.LayerNorm(h, eps=eps, dtype=pdt)
            conv_bias = False
        else:  # siglip
            self.ln_post = torch.nn.LayerNorm(h, eps=eps, dtype=pdt)
            conv_bias = True

        self.conv1 = torch.nn.Conv2d(
            in_channels=3, out_channels=h, kernel_size=patch_dim,
            stride=patch_dim, bias=conv_bias, dtype=pdt)
        self.position_embeddings = torch.nn.Embedding(self.seq_length, h,
                                                      dtype=pdt)
        self.register_buffer(
            "position_ids",
            torch.arange(self.seq_length).unsqueeze(0), persistent=False)
        if add_class_token:
            self.class_token = torch.nn.Parameter(
                torch.randn(1, class_token_len, h, dtype=pdt))

        self.model_type = ModelType.encoder_or_decoder
        self.decoder = TransformerBlock(
            config=transformer_config, spec=transformer_layer_spec,
            pre_process=True, post_process=False)

    def set_input_tensor(self, input_tensor: torch.Tensor) -> None:
        self.decoder.set_input_tensor(input_tensor)

    def forward(self, x: torch.Tensor,
                attention_mask: Optional[torch.Tensor] = None) -> torch.Tensor:
        x = self.conv1(x.to(self.conv1.weight.dtype))  # [b, h, gh, gw]
        x = x.flatten(2).permute(0, 2, 1)        # [b, patches, h]
        if self.add_class_token:
            cls = self.class_token.expand(x.shape[0], -1, -1)
            x = torch.cat([cls, x], dim=1)
        assert x.shape[1] == self.seq_length, (x.shape[1], self.seq_length)
        x = x + self.position_embeddings(self.position_ids)
        if self.ln_pre is not None:
            x = self.ln_pre(x)
        x = x.permute(1, 0, 2).contiguous()      # [s, b, h]
        x = self.decoder(x, attention_mask)
        x = x.permute(1, 0, 2).contiguous()      # [b, s, h]
        if self.ln_post is not None:
            x = self.ln_post(x)
        return x


def get_num_image_embeddings(img_h: int, img_w: int, patch_dim: int,
                             vision_model_type: str = "clip",
                             disable_vision_class_token: bool = False,
                             class_token_len: int = 1) -> int:
    """Embeddings per image tile after optional class-token drop
    (reference clip_vit_model.py:189-242, without tile tags)."""
    if vision_model_type == "siglip":
        keep_class = False
    else:
        keep_class = not disable_vision_class_token
    num_patches = (img_h // patch_dim) * (img_w // patch_dim)
    return num_patches + (class_token_len if keep_class else 0)
