"""Multi-latent attention (reference multi_latent_attention.py)."""
import pytest
import torch

from tests.utils import initialize_model_parallel, destroy


def _mla_config(q_lora):
    from megatronapp_amd.core.transformer_config import MLATransformerConfig
    return MLATransformerConfig(
        num_layers=2, hidden_size=128, num_attention_heads=4,
        ffn_hidden_size=256, hidden_dropout=0.0, attention_dropout=0.0,
        q_lora_rank=q_lora, kv_lora_rank=32, qk_head_dim=24,
        qk_pos_emb_head_dim=8, v_head_dim=16, add_bias_linear=False,
        masked_softmax_fusion=False)


@pytest.mark.parametrize("q_lora", [None, 48])
def test_mla_forward_backward(q_lora):
    from megatronapp_amd.core.models.gpt import GPTModel
    from megatronapp_amd.core.models.gpt.gpt_layer_specs import (
        get_gpt_layer_local_spec)
    from megatronapp_amd.core.tensor_parallel.random import (
        model_parallel_cuda_manual_seed)
    initialize_model_parallel()
    model_parallel_cuda_manual_seed(7)
    torch.manual_seed(7)
    cfg = _mla_config(q_lora)
    m = GPTModel(config=cfg,
                 transformer_layer_spec=get_gpt_layer_local_spec(
                     multi_latent_attention=True, use_flash=False),
                 vocab_size=512, max_sequence_length=64,
                 pre_process=True, post_process=True)
    tok = torch.randint(0, 512, (2, 32))
    pos = torch.arange(32).unsqueeze(0).expand(2, -1)
    loss = m(tok, pos, None, labels=tok).float().mean()
    loss.backward()
    grads = [p.grad for p in m.parameters() if p.requires_grad]
    assert all(g is not None for g in grads)
    assert all(torch.isfinite(g).all() for g in grads)
    # latent compressions exist and are the declared ranks
    attn = m.decoder.layers[0].self_attention
    assert attn.linear_kv_down_proj.out_features == 32 + 8
    destroy()


def test_mla_kv_cache_decode_matches_full():
    """Prefill+decode through the latent KV cache == one full forward."""
    from megatronapp_amd.core.models.gpt import GPTModel
    from megatronapp_amd.core.models.gpt.gpt_layer_specs import (
        get_gpt_layer_local_spec)
    from megatronapp_amd.core.inference_params import InferenceParams
    from megatronapp_amd.core.tensor_parallel.random import (
        model_parallel_cuda_manual_seed)
    initialize_model_parallel()
    model_parallel_cuda_manual_seed(11)
    torch.manual_seed(11)
    cfg = _mla_config(None)
    m = GPTModel(config=cfg,
                 transformer_layer_spec=get_gpt_layer_local_spec(
                     multi_latent_attention=True, use_flash=False),
                 vocab_size=512, max_sequence_length=64,
                 pre_process=True, post_process=True).eval()
    tok = torch.randint(0, 512, (1, 16))
    pos = torch.arange(16).unsqueeze(0)
    with torch.no_grad():
        full = m(tok, pos, None)
        inf = InferenceParams(max_batch_size=1, max_sequence_length=64)
        pre = m(tok[:, :15], pos[:, :15], None, inference_context=inf)
        inf.sequence_len_offset = 15
        step = m(tok[:, 15:16], pos[:, 15:16], None, inference_context=inf)
    assert torch.allclose(full[:, :15], pre, atol=1e-4)
    assert torch.allclose(full[:, 15:16], step, atol=1e-4)
    destroy()
