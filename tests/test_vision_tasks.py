"""Vision task models: classification / inpainting / DINO (reference
legacy/model/vision/* + pretrain_vision_*.py)."""
import os
import subprocess
import sys

import pytest
import torch

from tests.utils import initialize_model_parallel, destroy
from megatronapp_amd.core.transformer_config import TransformerConfig

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _cfg():
    return TransformerConfig(
        num_layers=2, hidden_size=64, num_attention_heads=4,
        ffn_hidden_size=128, hidden_dropout=0.0, attention_dropout=0.0,
        masked_softmax_fusion=False)


def test_vit_classification_forward_grad():
    initialize_model_parallel()
    try:
        from megatronapp_amd.core.models.vision import (
            VitClassificationModel, get_vit_layer_local_spec)
        torch.manual_seed(0)
        m = VitClassificationModel(_cfg(), get_vit_layer_local_spec(),
                                   num_classes=10, patch_dim=8,
                                   img_h=32, img_w=32)
        logits = m(torch.randn(3, 3, 32, 32))
        assert logits.shape == (3, 10)
        torch.nn.functional.cross_entropy(
            logits, torch.tensor([1, 2, 3])).backward()
        assert m.backbone.conv1.weight.grad is not None
    finally:
        destroy()


def test_vit_inpainting_round_trip_shape():
    initialize_model_parallel()
    try:
        from megatronapp_amd.core.models.vision import (
            VitInpaintingModel, get_vit_layer_local_spec)
        m = VitInpaintingModel(_cfg(), get_vit_layer_local_spec(),
                               patch_dim=8, img_h=32, img_w=32)
        x = torch.randn(2, 3, 32, 32)
        out = m(x)
        assert out.shape == x.shape
        # patch layout inverse: decoder output patch (i,j) comes from
        # patch embedding (i,j) — check via a delta on one embedding
        torch.nn.functional.mse_loss(out, x).backward()
        assert m.linear_decoder.weight.grad is not None
    finally:
        destroy()


def test_dino_teacher_ema_and_loss():
    initialize_model_parallel()
    try:
        from megatronapp_amd.core.models.vision import (
            DinoPretrainModel, get_vit_layer_local_spec)
        torch.manual_seed(0)
        m = DinoPretrainModel(_cfg(), get_vit_layer_local_spec(),
                              out_dim=32, patch_dim=8, img_h=32, img_w=32,
                              momentum=0.5)
        a = torch.randn(2, 3, 32, 32)
        b = torch.randn(2, 3, 32, 32)
        loss = m(a, b)
        assert loss.dim() == 0 and torch.isfinite(loss)
        loss.backward()
        # teacher got no gradient; student did
        assert all(p.grad is None for p in m.teacher.parameters())
        assert any(p.grad is not None and p.grad.abs().sum() > 0
                   for p in m.student.parameters())
        # EMA moves teacher toward student
        with torch.no_grad():
            for p in m.student.parameters():
                p.add_(1.0)
        s0 = next(iter(m.student.parameters())).clone()
        t_before = next(iter(m.teacher.parameters())).clone()
        m.momentum_update()
        t_after = next(iter(m.teacher.parameters()))
        assert torch.allclose(t_after, 0.5 * t_before + 0.5 * s0, atol=1e-6)
    finally:
        destroy()


@pytest.mark.parametrize("script,extra", [
    ("pretrain_vision_classify.py", ["--num-classes", "4"]),
    ("pretrain_vision_inpaint.py", []),
    ("pretrain_vision_dino.py", ["--dino-out-dim", "32"]),
])
def test_vision_entry_runs(tmp_path, script, extra):
    env = dict(os.environ, MASTER_ADDR="127.0.0.1",
               MASTER_PORT=str(29690 + hash(script) % 7), RANK="0",
               WORLD_SIZE="1", LOCAL_RANK="0")
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, script),
         "--num-layers", "2", "--hidden-size", "64",
         "--num-attention-heads", "4", "--seq-length", "16",
         "--max-position-embeddings", "64",
         "--img-h", "32", "--img-w", "32", "--patch-dim", "8",
         "--micro-batch-size", "2", "--global-batch-size", "2",
         "--vocab-size", "8",
         "--train-iters", "2", "--lr", "1e-4", "--eval-iters", "1",
         "--hidden-dropout", "0", "--attention-dropout", "0"] + extra,
        capture_output=True, text=True, cwd=REPO, env=env, timeout=420)
    assert out.returncode == 0, out.stderr[-2000:]
    assert "lm loss" in out.stdout
