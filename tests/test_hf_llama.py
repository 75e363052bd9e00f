"""HF Llama converter logit parity (reference tools/checkpoint
loader_llama_mistral.py)."""
import pytest
import torch

from tests.utils import initialize_model_parallel, destroy


def _build_llama(ng, vocab=96, nl=2, h=64, nh=4, ffn=128, seq=32):
    from megatronapp_amd.core.models.gpt import GPTModel
    from megatronapp_amd.core.models.gpt.gpt_layer_specs import (
        get_gpt_layer_local_spec)
    from megatronapp_amd.core.transformer_config import TransformerConfig
    from megatronapp_amd.core.tensor_parallel.random import (
        model_parallel_cuda_manual_seed)
    model_parallel_cuda_manual_seed(7)
    torch.manual_seed(7)
    cfg = TransformerConfig(
        num_layers=nl, hidden_size=h, num_attention_heads=nh,
        num_query_groups=ng, ffn_hidden_size=ffn,
        hidden_dropout=0.0, attention_dropout=0.0,
        normalization="RMSNorm", gated_linear_unit=True,
        activation_func="silu", add_bias_linear=False,
        masked_softmax_fusion=False)
    return GPTModel(
        config=cfg,
        transformer_layer_spec=get_gpt_layer_local_spec(
            normalization="RMSNorm", use_flash=False),
        vocab_size=vocab, max_sequence_length=seq,
        position_embedding_type="rope",
        share_embeddings_and_output_weights=False).eval()


@pytest.mark.parametrize("ng", [4, 2])
def test_hf_llama_export_logit_parity(tmp_path, ng):
    transformers = pytest.importorskip("transformers")
    initialize_model_parallel()
    try:
        m = _build_llama(ng)
        full = {"model." + k: v for k, v in m.state_dict().items()
                if torch.is_tensor(v)}
        import sys as _s
        _s.path.insert(0, "tools/checkpoint")
        from saver_hf_llama import load_hf_llama, save_hf_llama
        save_hf_llama(full, {"args": {"num_attention_heads": 4,
                                      "num_query_groups": ng,
                                      "max_position_embeddings": 32}},
                      str(tmp_path / "hf"))

        hf_cfg = transformers.LlamaConfig(
            vocab_size=96, hidden_size=64, intermediate_size=128,
            num_hidden_layers=2, num_attention_heads=4,
            num_key_value_heads=ng, max_position_embeddings=32,
            rms_norm_eps=1e-5, attention_bias=False, mlp_bias=False,
            tie_word_embeddings=False)
        hf = transformers.LlamaForCausalLM(hf_cfg).eval()
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

        # round trip back into our naming
        full2, common2 = load_hf_llama(str(tmp_path / "hf"))
        assert common2["hf_config"]["num_key_value_heads"] == ng
        for k, v in full.items():
            assert torch.allclose(full2[k].float(), v.float(),
                                  atol=1e-6), k
    finally:
        destroy()


def test_gguf_export_round_trip(tmp_path):
    """GGUF writer: llama-family checkpoint -> .gguf, verified by an
    independent minimal reader (header, metadata, aligned tensor data)."""
    import os
    import sys
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    sys.path.insert(0, os.path.join(repo, "tools", "checkpoint"))
    try:
        import importlib
        import saver_gguf
        importlib.reload(saver_gguf)
        h, nh, ng, ffn, vocab, L = 64, 4, 2, 128, 256, 2
        torch.manual_seed(3)
        full = {"model.embedding.word_embeddings.weight":
                torch.randn(vocab, h)}
        full["model.decoder.final_layernorm.weight"] = torch.randn(h)
        full["model.output_layer.weight"] = torch.randn(vocab, h)
        for i in range(L):
            p = f"model.decoder.layers.{i}."
            full[p + "self_attention.linear_qkv.weight"] = torch.randn(
                (nh + 2 * ng) * (h // nh), h)
            full[p + "self_attention.linear_proj.weight"] = torch.randn(h, h)
            full[p + "input_layernorm.weight"] = torch.randn(h)
            full[p + "pre_mlp_layernorm.weight"] = torch.randn(h)
            full[p + "mlp.linear_fc1.weight"] = torch.randn(2 * ffn, h)
            full[p + "mlp.linear_fc2.weight"] = torch.randn(h, ffn)
        common = {"args": {"num_attention_heads": nh,
                           "num_query_groups": ng}}
        out = str(tmp_path / "model.gguf")
        saver_gguf.save_gguf(full, common, out, dtype="f32")
        meta, tensors = saver_gguf.read_gguf(out)
        assert meta["general.architecture"] == "llama"
        assert meta["llama.block_count"] == L
        assert meta["llama.attention.head_count_kv"] == ng
        assert tensors["token_embd.weight"].shape == (vocab, h)
        assert torch.allclose(
            tensors["token_embd.weight"],
            full["model.embedding.word_embeddings.weight"])
        assert tensors["blk.1.ffn_down.weight"].shape == (h, ffn)
        assert torch.allclose(tensors["blk.1.ffn_down.weight"],
                              full["model.decoder.layers.1.mlp."
                                   "linear_fc2.weight"])
        assert tensors["output.weight"].shape == (vocab, h)
    finally:
        sys.path.pop(0)
