"""LLaVA multimodal model (reference:
core/models/multimodal/llava_model.py:50-1005).

Architecture parity: a CLIP ViT vision tower encodes image tiles, a
projector maps them to the language hidden size, and each image token
(``image_token_index``, default -200) in ``input_ids`` is *expanded
in place* into the tile's image embeddings before the GPT decoder runs
on the combined sequence (reference ``_preprocess_data``:406-578).
Labels over image spans become IGNORE_INDEX and the loss mask is zeroed
there, exactly as the reference.  ``freeze()`` matches the reference's
selective-freeze API.

The combined sequence enters the language model through its
``decoder_input`` hook — RoPE (when enabled) is applied there, one
position per embedding, like the reference.
"""

from __future__ import annotations

from typing import List, Optional

import torch

from ...transformer.module import MegatronModule
from ...transformer.spec_utils import ModuleSpec
from ...transformer_config import TransformerConfig
from ...transformer.mlp import MLPSubmodules
from ...tensor_parallel.layers import ColumnParallelLinear, RowParallelLinear
from ..gpt import GPTModel
from ..vision import CLIPViTModel, MultimodalProjector

IGNORE_INDEX = -100
DEFAULT_IMAGE_TOKEN_INDEX = -200
IMAGE_TOKEN = "<image>"


class LLaVAModel(MegatronModule):
    """Vision tower + projector + GPT language model."""

    def __init__(
        self,
        language_transformer_config: TransformerConfig,
        language_transformer_layer_spec: ModuleSpec,
        language_vocab_size: int,
        language_max_sequence_length: int,
        vision_transformer_config: TransformerConfig,
        vision_transformer_layer_spec: ModuleSpec,
        drop_vision_class_token: bool = False,
        vision_projection_config: Optional[TransformerConfig] = None,
        vision_projection_type: str = "mlp",
        img_h: int = 336,
        img_w: int = 336,
        patch_dim: int = 14,
        image_token_index: int = DEFAULT_IMAGE_TOKEN_INDEX,
        position_embedding_type: str = "learned_absolute",
        rotary_percent: float = 1.0,
        parallel_output: bool = True,
        share_embeddings_and_output_weights: bool = False,
    ) -> None:
        super().__init__(config=language_transformer_config)
        self.image_token_index = image_token_index
        self._drop_vision_class_token = drop_vision_class_token

        self.language_model = GPTModel(
            config=language_transformer_config,
            transformer_layer_spec=language_transformer_layer_spec,
            vocab_size=language_vocab_size,
            max_sequence_length=language_max_sequence_length,
            parallel_output=parallel_output,
            position_embedding_type=position_embedding_type,
            rotary_percent=rotary_percent,
            share_embeddings_and_output_weights=share_embeddings_and_output_weights,
        )
        self.vision_model = CLIPViTModel(
            vision_transformer_config, vision_transformer_layer_spec,
            patch_dim=patch_dim, img_h=img_h, img_w=img_w)
        proj_cfg = vision_projection_config or language_transformer_config
        self.vision_projection = MultimodalProjector(
            proj_cfg,
            MLPSubmodules(linear_fc1=ColumnParallelLinear,
                          linear_fc2=RowParallelLinear),
            vision_projection_type,
            input_size=vision_transformer_config.hidden_size)

    def set_input_tensor(self, input_tensor) -> None:
        self.language_model.set_input_tensor(input_tensor)

    def shared_embedding_or_output_weight(self):
        return self.language_model.shared_embedding_or_output_weight()

    def freeze(self, freeze_language_model: bool, freeze_vision_model: bool,
               freeze_vision_projection: bool) -> None:
        """Selective fine-tune freezing (reference llava_model.py freeze)."""
        modules = []
        if freeze_language_model:
            modules.append(self.language_model)
        if freeze_vision_model:
            modules.append(self.vision_model)
        if freeze_vision_projection:
            modules.append(self.vision_projection)
        for m in modules:
            for p in m.parameters():
                p.requires_grad = False

    def _merge(self, image_embeddings, language_embeddings, input_ids,
               labels, loss_mask):
        """Expand each image token into its tile's embeddings
        (reference _preprocess_data:406-578, single-tile-per-token form).

        image_embeddings: [img_seq_len, num_tiles, h]; tiles are consumed
        in order of image-token appearance (row-major over the batch).
        Returns combined [s, b, h] embeddings plus per-sample labels /
        loss mask padded to the longest combined length.
        """
        b = input_ids.shape[0]
        h = language_embeddings.shape[-1]
        img_seq_len = image_embeddings.shape[0] if \
            image_embeddings.numel() else 0
        device = language_embeddings.device
        tile = 0
        seqs, labs, masks = [], [], []
        for i in range(b):
            pieces, lpieces, mpieces = [], [], []
            prev = 0
            positions = (input_ids[i] == self.image_token_index) \
                .nonzero(as_tuple=True)[0].tolist()
            for pos in positions:
                if pos > prev:
                    pieces.append(language_embeddings[i, prev:pos])
                    if labels is not None:
                        lpieces.append(labels[i, prev:pos])
                        mpieces.append(loss_mask[i, prev:pos])
                pieces.append(image_embeddings[:, tile])
                if labels is not None:
                    lpieces.append(torch.full((img_seq_len,), IGNORE_INDEX,
                                              dtype=labels.dtype,
                                              device=device))
                    mpieces.append(torch.zeros(img_seq_len,
                                               dtype=loss_mask.dtype,
                                               device=device))
                tile += 1
                prev = pos + 1
            pieces.append(language_embeddings[i, prev:])
            if labels is not None:
                lpieces.append(labels[i, prev:])
                mpieces.append(loss_mask[i, prev:])
            seqs.append(torch.cat(pieces, dim=0))
            if labels is not None:
                labs.append(torch.cat(lpieces, dim=0))
                masks.append(torch.cat(mpieces, dim=0))

        max_len = max(s.shape[0] for s in seqs)
        combined = torch.zeros(max_len, b, h, dtype=seqs[0].dtype,
                               device=device)
        new_labels = new_mask = None
        if labels is not None:
            new_labels = torch.full((b, max_len), IGNORE_INDEX,
                                    dtype=labels.dtype, device=device)
            new_mask = torch.zeros(b, max_len, dtype=loss_mask.dtype,
                                   device=device)
        for i in range(b):
            n = seqs[i].shape[0]
            combined[:n, i] = seqs[i]
            if labels is not None:
                new_labels[i, :n] = labs[i]
                new_mask[i, :n] = masks[i]
        return combined, new_labels, new_mask

    def forward(
        self,
        images: torch.Tensor,
        input_ids: torch.Tensor,
        position_ids: torch.Tensor,
        attention_mask: Optional[torch.Tensor] = None,
        labels: Optional[torch.Tensor] = None,
        loss_mask: Optional[torch.Tensor] = None,
        inference_context=None,
        num_image_tiles: Optional[List[int]] = None,
        runtime_gather_output: Optional[bool] = None,
    ):
        """Returns (loss [b, s] if labels given else logits, loss_mask)."""
        has_images = images is not None and images.shape[0] > 0
        if has_images:
            image_embeddings = self.vision_model(images)  # [tiles, s_img, h]
            if self._drop_vision_class_token:
                image_embeddings = \
                    image_embeddings[:, self.vision_model.class_token_len:, :]
            image_embeddings = image_embeddings.permute(1, 0, 2).contiguous()
            image_embeddings = self.vision_projection(image_embeddings)
        else:
            image_embeddings = torch.empty(
                0, 0, self.config.hidden_size,
                dtype=self.language_model.embedding.word_embeddings.weight.dtype,
                device=input_ids.device)

        input_ids_text = input_ids.clone()
        input_ids_text[input_ids_text == self.image_token_index] = 0
        language_embeddings = self.language_model.embedding(
            input_ids_text, position_ids)          # [s, b, h]
        language_embeddings = \
            language_embeddings.transpose(0, 1).contiguous()  # [b, s, h]

        combined, new_labels, new_loss_mask = self._merge(
            image_embeddings, language_embeddings, input_ids, labels,
            loss_mask)

        output = self.language_model(
            input_ids=None, position_ids=None,
            attention_mask=attention_mask,
            decoder_input=combined, labels=new_labels,
            inference_context=inference_context,
            runtime_gather_output=runtime_gather_output)
        return output, new_loss_mask
