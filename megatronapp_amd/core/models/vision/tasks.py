"""Vision task heads over the ViT backbone (reference
legacy/model/vision/{classification,inpainting}.py and
pretrain_vision_{classify,inpaint,dino}.py).

* ``VitClassificationModel`` — CLS-token MLP head (dense→tanh→dense,
  out-bias init −10 per the reference's VitMlpHead)
* ``VitInpaintingModel`` — class-token-free backbone + linear decoder
  from each patch embedding back to its pixel patch
* ``DinoPretrainModel`` — student/teacher self-distillation: teacher is
  an EMA copy, loss is cross-entropy of sharpened teacher targets
  (centered) against the student over crop pairs
"""

from __future__ import annotations

import torch
import torch.nn.functional as F

from ...transformer.module import MegatronModule
from .clip_vit import CLIPViTModel


class VitMlpHead(MegatronModule):
    def __init__(self, config, hidden_size: int, num_classes: int):
        super().__init__(config=config)
        pdt = config.params_dtype
        self.dense_in = torch.nn.Linear(hidden_size, hidden_size, dtype=pdt)
        self.dense_out = torch.nn.Linear(hidden_size, num_classes, dtype=pdt)
        torch.nn.init.constant_(self.dense_out.bias, -10)

    def forward(self, hidden_states):
        return self.dense_out(torch.tanh(self.dense_in(hidden_states)))


class VitClassificationModel(MegatronModule):
    """ViT + CLS classification head
    (reference classification.py:13-54)."""

    def __init__(self, config, transformer_layer_spec, num_classes: int,
                 patch_dim: int = 16, img_h: int = 224, img_w: int = 224,
                 finetune: bool = False):
        super().__init__(config=config)
        self.num_classes = num_classes
        self.backbone = CLIPViTModel(
            config, transformer_layer_spec,
            patch_dim=patch_dim, img_h=img_h, img_w=img_w)
        if finetune:
            self.head = torch.nn.Linear(config.hidden_size, num_classes,
                                        dtype=config.params_dtype)
            torch.nn.init.zeros_(self.head.weight)
        else:
            self.head = VitMlpHead(config, config.hidden_size, num_classes)

    def set_input_tensor(self, input_tensor):
        self.backbone.set_input_tensor(input_tensor)

    def forward(self, images):
        hidden = self.backbone(images)        # [b, s, h]
        return self.head(hidden[:, 0])        # CLS token -> [b, classes]


class VitInpaintingModel(MegatronModule):
    """ViT + per-patch pixel decoder (reference inpainting.py:19-66)."""

    def __init__(self, config, transformer_layer_spec,
                 patch_dim: int = 16, img_h: int = 224, img_w: int = 224):
        super().__init__(config=config)
        self.patch_dim = patch_dim
        self.img_h = img_h
        self.img_w = img_w
        self.backbone = CLIPViTModel(
            config, transformer_layer_spec, add_class_token=False,
            class_token_len=0, patch_dim=patch_dim, img_h=img_h,
            img_w=img_w, model_subtype="siglip")
        self.flatten_dim = 3 * patch_dim * patch_dim
        self.linear_decoder = torch.nn.Linear(
            config.hidden_size, self.flatten_dim,
            dtype=config.params_dtype)
        torch.nn.init.zeros_(self.linear_decoder.weight)

    def set_input_tensor(self, input_tensor):
        self.backbone.set_input_tensor(input_tensor)

    def forward(self, images):
        hidden = self.backbone(images)                 # [b, patches, h]
        decoded = self.linear_decoder(hidden)          # [b, patches, 3pp]
        b = decoded.shape[0]
        gh = self.img_h // self.patch_dim
        gw = self.img_w // self.patch_dim
        p = self.patch_dim
        x = decoded.view(b, gh, gw, 3, p, p)
        x = x.permute(0, 3, 1, 4, 2, 5).reshape(b, 3, self.img_h,
                                                self.img_w)
        return x


class DinoPretrainModel(MegatronModule):
    """Self-distillation (DINO) over two crops
    (reference pretrain_vision_dino.py, compact form)."""

    def __init__(self, config, transformer_layer_spec, out_dim: int = 4096,
                 patch_dim: int = 16, img_h: int = 224, img_w: int = 224,
                 momentum: float = 0.996, teacher_temp: float = 0.04,
                 student_temp: float = 0.1, center_momentum: float = 0.9):
        super().__init__(config=config)
        self.momentum = momentum
        self.teacher_temp = teacher_temp
        self.student_temp = student_temp
        self.center_momentum = center_momentum

        def tower():
            backbone = CLIPViTModel(
                config, transformer_layer_spec,
                patch_dim=patch_dim, img_h=img_h, img_w=img_w)
            head = VitMlpHead(config, config.hidden_size, out_dim)
            return torch.nn.ModuleDict(
                {"backbone": backbone, "head": head})

        self.student = tower()
        self.teacher = tower()
        self.teacher.load_state_dict(self.student.state_dict())
        for p in self.teacher.parameters():
            p.requires_grad = False
        self.register_buffer("center", torch.zeros(1, out_dim))

    def set_input_tensor(self, input_tensor):
        self.student["backbone"].set_input_tensor(input_tensor)

    def _embed(self, tower, images):
        return tower["head"](tower["backbone"](images)[:, 0])

    @torch.no_grad()
    def momentum_update(self):
        for ps, pt in zip(self.student.parameters(),
                          self.teacher.parameters()):
            pt.mul_(self.momentum).add_(ps, alpha=1 - self.momentum)

    def forward(self, crop_a, crop_b):
        """Returns the symmetric DINO loss over the two crops."""
        s_a = self._embed(self.student, crop_a)
        s_b = self._embed(self.student, crop_b)
        with torch.no_grad():
            t_a = self._embed(self.teacher, crop_a)
            t_b = self._embed(self.teacher, crop_b)
            targets_a = F.softmax(
                (t_a - self.center) / self.teacher_temp, dim=-1)
            targets_b = F.softmax(
                (t_b - self.center) / self.teacher_temp, dim=-1)
            batch_center = torch.cat([t_a, t_b]).mean(0, keepdim=True)
            self.center.mul_(self.center_momentum).add_(
                batch_center, alpha=1 - self.center_momentum)
        loss = (-(targets_b * F.log_softmax(
                    s_a / self.student_temp, dim=-1)).sum(-1).mean()
                - (targets_a * F.log_softmax(
                    s_b / self.student_temp, dim=-1)).sum(-1).mean()) / 2
        return loss
