"""CLIP ViT vision tower + LLaVA multimodal model (reference
core/models/vision/clip_vit_model.py, multimodal/llava_model.py)."""
import pytest
import torch

from tests.utils import initialize_model_parallel, destroy
from megatronapp_amd.core.transformer_config import TransformerConfig


def _vision_cfg():
    return TransformerConfig(
        num_layers=2, hidden_size=64, num_attention_heads=4,
        ffn_hidden_size=128, hidden_dropout=0.0, attention_dropout=0.0,
        masked_softmax_fusion=False)


def _lang_cfg():
    return TransformerConfig(
        num_layers=2, hidden_size=96, num_attention_heads=4,
        ffn_hidden_size=192, hidden_dropout=0.0, attention_dropout=0.0,
        masked_softmax_fusion=False)


def test_clip_vit_shapes_and_grad():
    initialize_model_parallel()
    try:
        from megatronapp_amd.core.models.vision import (
            CLIPViTModel, get_vit_layer_local_spec)
        torch.manual_seed(0)
        m = CLIPViTModel(_vision_cfg(), get_vit_layer_local_spec(),
                         patch_dim=8, img_h=32, img_w=32)
        assert m.seq_length == 16 + 1        # 4x4 patches + class token
        x = torch.randn(3, 3, 32, 32)
        out = m(x)
        assert out.shape == (3, 17, 64)
        out.sum().backward()
        assert m.conv1.weight.grad is not None
        assert m.class_token.grad is not None
    finally:
        destroy()


def test_siglip_subtype_no_class_token():
    initialize_model_parallel()
    try:
        from megatronapp_amd.core.models.vision import (
            CLIPViTModel, get_vit_layer_local_spec)
        m = CLIPViTModel(_vision_cfg(), get_vit_layer_local_spec(),
                         add_class_token=False, class_token_len=0,
                         patch_dim=8, img_h=32, img_w=32,
                         model_subtype="siglip")
        out = m(torch.randn(2, 3, 32, 32))
        assert out.shape == (2, 16, 64)
        assert m.ln_post is not None and m.ln_pre is None
    finally:
        destroy()


def test_get_num_image_embeddings():
    from megatronapp_amd.core.models.vision import get_num_image_embeddings
    assert get_num_image_embeddings(336, 336, 14, "clip", False, 1) == 577
    assert get_num_image_embeddings(336, 336, 14, "clip", True, 1) == 576
    assert get_num_image_embeddings(336, 336, 14, "siglip", False, 0) == 576


def _build_llava(drop_class=True):
    from megatronapp_amd.core.models.multimodal import LLaVAModel
    from megatronapp_amd.core.models.gpt.gpt_layer_specs import (
        get_gpt_layer_local_spec)
    from megatronapp_amd.core.models.vision import get_vit_layer_local_spec
    return LLaVAModel(
        language_transformer_config=_lang_cfg(),
        language_transformer_layer_spec=get_gpt_layer_local_spec(
            use_flash=False),
        language_vocab_size=128,
        language_max_sequence_length=64,
        vision_transformer_config=_vision_cfg(),
        vision_transformer_layer_spec=get_vit_layer_local_spec(),
        drop_vision_class_token=drop_class,
        img_h=32, img_w=32, patch_dim=8)


def test_llava_forward_with_images():
    initialize_model_parallel()
    try:
        from megatronapp_amd.core.models.multimodal import IGNORE_INDEX
        torch.manual_seed(1)
        m = _build_llava()
        b, s = 2, 10
        img_tok = m.image_token_index
        input_ids = torch.randint(0, 128, (b, s))
        input_ids[0, 2] = img_tok
        input_ids[1, 5] = img_tok
        position_ids = torch.arange(s).expand(b, -1)
        labels = torch.randint(0, 128, (b, s))
        loss_mask = torch.ones(b, s)
        images = torch.randn(2, 3, 32, 32)     # one tile per image token
        loss, new_mask = m(images, input_ids, position_ids, labels=labels,
                           loss_mask=loss_mask)
        # class token dropped: 16 image embeddings replace 1 token
        combined = s - 1 + 16
        assert loss.shape == (b, combined)
        assert new_mask.shape == (b, combined)
        # image spans excluded from the loss
        assert new_mask[0, 2:18].sum() == 0 and new_mask[0, :2].sum() == 2
        (loss * new_mask).sum().backward()
        assert m.vision_model.conv1.weight.grad is not None
        assert m.language_model.embedding.word_embeddings.weight.grad \
            is not None
    finally:
        destroy()


def test_llava_no_images_matches_gpt():
    """Without images the model reduces to its language model."""
    initialize_model_parallel()
    try:
        torch.manual_seed(2)
        m = _build_llava()
        b, s = 2, 12
        input_ids = torch.randint(0, 128, (b, s))
        position_ids = torch.arange(s).expand(b, -1)
        logits, _ = m(None, input_ids, position_ids)
        ref = m.language_model(input_ids, position_ids)
        assert torch.allclose(logits, ref, atol=1e-5)
    finally:
        destroy()


def test_llava_freeze():
    initialize_model_parallel()
    try:
        m = _build_llava()
        m.freeze(freeze_language_model=True, freeze_vision_model=True,
                 freeze_vision_projection=False)
        assert not any(p.requires_grad
                       for p in m.language_model.parameters())
        assert not any(p.requires_grad for p in m.vision_model.parameters())
        assert all(p.requires_grad
                   for p in m.vision_projection.parameters())
    finally:
        destroy()


def test_pretrain_vlm_entry_runs(tmp_path):
    """pretrain_vlm.py trains end to end on the synthetic VLM data."""
    import os
    import subprocess
    import sys

    REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    env = dict(os.environ, MASTER_ADDR="127.0.0.1", MASTER_PORT="29677",
               RANK="0", WORLD_SIZE="1", LOCAL_RANK="0")
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, "pretrain_vlm.py"),
         "--num-layers", "2", "--hidden-size", "64",
         "--num-attention-heads", "4", "--seq-length", "32",
         "--max-position-embeddings", "128", "--micro-batch-size", "2",
         "--global-batch-size", "2", "--vocab-size", "128",
         "--img-h", "32", "--img-w", "32", "--patch-dim", "8",
         "--vision-num-layers", "2", "--vision-hidden-size", "64",
         "--vision-num-attention-heads", "4",
         "--train-iters", "2", "--lr", "1e-4", "--eval-iters", "1",
         "--hidden-dropout", "0", "--attention-dropout", "0"],
        capture_output=True, text=True, cwd=REPO, env=env, timeout=420)
    assert out.returncode == 0, out.stderr[-2000:]
    assert "lm loss" in out.stdout
