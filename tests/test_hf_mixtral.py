"""HF Mixtral converter logit parity (reference tools/checkpoint
loader_mixtral_hf.py)."""
import pytest
import torch

from tests.utils import initialize_model_parallel, destroy


def test_hf_mixtral_export_logit_parity(tmp_path):
    transformers = pytest.importorskip("transformers")
    initialize_model_parallel()
    try:
        from megatronapp_amd.core.models.gpt import GPTModel
        from megatronapp_amd.core.models.gpt.gpt_layer_specs import (
            get_gpt_decoder_block_spec)
        from megatronapp_amd.core.transformer_config import (
            TransformerConfig)
        from megatronapp_amd.core.tensor_parallel.random import (
            model_parallel_cuda_manual_seed)
        model_parallel_cuda_manual_seed(8)
        torch.manual_seed(8)
        cfg = TransformerConfig(
            num_layers=2, hidden_size=64, num_attention_heads=4,
            num_query_groups=2, ffn_hidden_size=128,
            hidden_dropout=0.0, attention_dropout=0.0,
            normalization="RMSNorm", gated_linear_unit=True,
            activation_func="silu", add_bias_linear=False,
            num_moe_experts=4, moe_router_topk=2,
            moe_router_renormalize=True, moe_grouped_gemm=True,
            moe_aux_loss_coeff=0.0, masked_softmax_fusion=False)
        spec = get_gpt_decoder_block_spec(cfg, normalization="RMSNorm")
        m = GPTModel(config=cfg, transformer_layer_spec=spec,
                     vocab_size=96, max_sequence_length=32,
                     position_embedding_type="rope",
                     share_embeddings_and_output_weights=False).eval()

        full = {"model." + k: v for k, v in m.state_dict().items()
                if torch.is_tensor(v)}
        import sys as _s
        _s.path.insert(0, "tools/checkpoint")
        from saver_hf_mixtral import load_hf_mixtral, save_hf_mixtral
        save_hf_mixtral(full, {"args": {"num_attention_heads": 4,
                                        "num_query_groups": 2,
                                        "moe_router_topk": 2,
                                        "max_position_embeddings": 32}},
                        str(tmp_path / "hf"))

        hf_cfg = transformers.MixtralConfig(
            vocab_size=96, hidden_size=64, intermediate_size=128,
            num_hidden_layers=2, num_attention_heads=4,
            num_key_value_heads=2, num_local_experts=4,
            num_experts_per_tok=2, max_position_embeddings=32,
            rope_theta=10000.0,
            rms_norm_eps=1e-5, tie_word_embeddings=False,
            router_aux_loss_coef=0.0)
        hf = transformers.MixtralForCausalLM(hf_cfg).eval()
        sd = torch.load(tmp_path / "hf" / "pytorch_model.bin",
                        weights_only=False)
        missing, unexpected = hf.load_state_dict(sd, strict=False)
        assert not unexpected, unexpected
        assert not missing, missing

        tok = torch.randint(0, 96, (2, 24))
        pos = torch.arange(24).unsqueeze(0).expand(2, -1)
        with torch.no_grad():
            ours = m(tok, pos, None)
            theirs = hf(tok).logits
        err = (ours - theirs).abs().max()
        assert err < 2e-4, float(err)

        full2, _ = load_hf_mixtral(str(tmp_path / "hf"))
        for k, v in full.items():
            assert torch.allclose(full2[k].float(), v.float(),
                                  atol=1e-6), k
    finally:
        destroy()
