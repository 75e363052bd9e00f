"""On-GPU train-step smokes for the newer model families and wrappers:
MoE, multi-latent attention, multi-token prediction, ZeRO-3 FSDP."""
import pytest
import torch

from tests.utils import initialize_model_parallel, destroy

pytestmark = pytest.mark.gpu


def _gpt(cfg_kwargs, spec_kwargs):
    from megatronapp_amd.core.models.gpt import GPTModel
    from megatronapp_amd.core.models.gpt.gpt_layer_specs import (
        get_gpt_layer_local_spec)
    from megatronapp_amd.core.transformer_config import (
        MLATransformerConfig, TransformerConfig)
    from megatronapp_amd.core.tensor_parallel.random import (
        model_parallel_cuda_manual_seed)
    model_parallel_cuda_manual_seed(21)
    torch.manual_seed(21)
    cls = cfg_kwargs.pop("_cls", TransformerConfig)
    cfg = cls(num_layers=2, hidden_size=256, num_attention_heads=4,
              ffn_hidden_size=512, hidden_dropout=0.0, attention_dropout=0.0,
              bf16=True, params_dtype=torch.bfloat16, **cfg_kwargs)
    with torch.device("cuda"):
        return GPTModel(config=cfg,
                        transformer_layer_spec=get_gpt_layer_local_spec(
                            **spec_kwargs),
                        vocab_size=512, max_sequence_length=256,
                        pre_process=True, post_process=True)


def _step(m):
    tok = torch.randint(0, 512, (2, 128), device="cuda")
    pos = torch.arange(128, device="cuda").unsqueeze(0).expand(2, -1)
    loss = m(tok, pos, None, labels=tok).float().mean()
    loss.backward()
    assert torch.isfinite(loss)
    for n, p in m.named_parameters():
        if p.grad is not None:
            assert torch.isfinite(p.grad.float()).all(), n
    return float(loss)


def test_moe_gpu():
    initialize_model_parallel()
    m = _gpt(dict(num_moe_experts=4, moe_router_topk=2,
                  moe_aux_loss_coeff=0.01,
                  moe_router_load_balancing_type="aux_loss"),
             dict(num_experts=4, use_flash=False))
    _step(m)
    destroy()


def test_mla_gpu():
    from megatronapp_amd.core.transformer_config import MLATransformerConfig
    initialize_model_parallel()
    m = _gpt(dict(_cls=MLATransformerConfig, q_lora_rank=64, kv_lora_rank=64,
                  qk_head_dim=48, qk_pos_emb_head_dim=16, v_head_dim=64),
             dict(multi_latent_attention=True, use_flash=False))
    _step(m)
    destroy()


def test_mtp_gpu():
    initialize_model_parallel()
    m = _gpt(dict(mtp_num_layers=1), dict(use_flash=False))
    _step(m)
    from megatronapp_amd.core.transformer.multi_token_prediction import (
        MTPLossLoggingHelper)
    vals = MTPLossLoggingHelper.get_and_clear()
    assert vals is not None and (vals > 0).all()
    destroy()


def test_fsdp_gpu():
    from megatronapp_amd.core.distributed.fsdp import (
        FullyShardedDataParallel)
    initialize_model_parallel()
    m = _gpt({}, dict(use_flash=False))
    fsdp = FullyShardedDataParallel(m, lr=1e-3, clip_grad=1.0)
    for _ in range(2):
        tok = torch.randint(0, 512, (2, 128), device="cuda")
        pos = torch.arange(128, device="cuda").unsqueeze(0).expand(2, -1)
        loss = fsdp(tok, pos, None, labels=tok).float().mean()
        loss.backward()
        ok, norm = fsdp.optimizer_step()
        assert ok and torch.isfinite(loss)
    # params are released between steps
    assert all(u.full is None for u in fsdp.units)
    destroy()


@pytest.mark.gpu
def test_bf16_training_learns_on_gpu():
    """End-to-end learning guard: bf16 training on MI355X must converge on
    a structured corpus like fp32/CPU does (caught a dropped residual-grad
    bug in the fused norm backward)."""
    import subprocess
    import sys
    r = subprocess.run([sys.executable, "scripts/train_sanity.py",
                        "--iters", "120"], capture_output=True, text=True,
                       timeout=400)
    assert "LEARNING SANITY OK" in r.stdout, r.stdout[-1500:] + r.stderr[-800:]


@pytest.mark.gpu
def test_fp16_training_step_gpu():
    """fp16 with dynamic loss scaling: one step on MI355X (fp16 params,
    scaled loss, unscaled master update)."""
    from megatronapp_amd.core.distributed import (
        DistributedDataParallel, DistributedDataParallelConfig)
    from megatronapp_amd.core.optimizer import (
        OptimizerConfig, get_megatron_optimizer)
    initialize_model_parallel()
    from megatronapp_amd.core.models.gpt import GPTModel
    from megatronapp_amd.core.models.gpt.gpt_layer_specs import (
        get_gpt_layer_local_spec)
    from megatronapp_amd.core.transformer_config import TransformerConfig
    from megatronapp_amd.core.tensor_parallel.random import (
        model_parallel_cuda_manual_seed)
    model_parallel_cuda_manual_seed(8)
    cfg = TransformerConfig(
        num_layers=2, hidden_size=128, num_attention_heads=4,
        ffn_hidden_size=256, hidden_dropout=0.0, attention_dropout=0.0,
        fp16=True, params_dtype=torch.float16)
    with torch.device("cuda"):
        m = GPTModel(config=cfg,
                     transformer_layer_spec=get_gpt_layer_local_spec(
                         use_flash=False),
                     vocab_size=256, max_sequence_length=64,
                     pre_process=True, post_process=True)
    ddp = DistributedDataParallel(
        cfg, DistributedDataParallelConfig(overlap_grad_reduce=False), m)
    opt = get_megatron_optimizer(
        OptimizerConfig(optimizer="adam", lr=1e-4, min_lr=0.0, fp16=True,
                        weight_decay=0.0, clip_grad=1.0,
                        initial_loss_scale=2 ** 16), [ddp])
    tok = torch.randint(0, 256, (2, 64), device="cuda")
    pos = torch.arange(64, device="cuda").unsqueeze(0).expand(2, -1)
    ddp.zero_grad_buffer()
    loss = ddp(input_ids=tok, position_ids=pos, attention_mask=None,
               labels=tok).float().mean()
    opt.scale_loss(loss).backward()
    ok, norm, _ = opt.step()
    assert ok and norm is not None and torch.isfinite(loss)
    destroy()


@pytest.mark.gpu
def test_fp8_training_learns_on_gpu():
    """fp8 (e4m3 forward+dgrad, bf16 wgrad) end-to-end convergence on the
    structured corpus — the fp8 analogue of the bf16 learning guard."""
    import os
    import subprocess
    import sys
    r = subprocess.run([sys.executable, "scripts/train_sanity.py",
                        "--iters", "120"], capture_output=True, text=True,
                       timeout=400, env={**os.environ, "SANITY_FP8": "1"})
    assert "LEARNING SANITY OK" in r.stdout, r.stdout[-1500:] + r.stderr[-800:]
