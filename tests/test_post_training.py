"""Post-training int8 quantization (reference megatron/post_training)."""
import copy

import torch

from tests.utils import initialize_model_parallel, destroy


def test_weight_quant_roundtrip_error_bound():
    from megatronapp_amd.post_training import (
        dequantize_weight, quantize_weight_int8)
    torch.manual_seed(0)
    w = torch.randn(64, 32)
    q, scale = quantize_weight_int8(w)
    assert q.dtype == torch.int8 and scale.shape == (64,)
    deq = dequantize_weight(q, scale)
    # error bounded by half a quantization step per channel
    step = scale.unsqueeze(1)
    assert ((w - deq).abs() <= 0.5 * step + 1e-6).all()


def test_quantize_gpt_model_outputs_close():
    initialize_model_parallel()
    try:
        from megatronapp_amd.core.transformer_config import (
            TransformerConfig)
        from megatronapp_amd.core.models.gpt import GPTModel
        from megatronapp_amd.core.models.gpt.gpt_layer_specs import (
            get_gpt_layer_local_spec)
        from megatronapp_amd.post_training import (
            export_int8_state_dict, quantize_model)
        torch.manual_seed(0)
        cfg = TransformerConfig(
            num_layers=2, hidden_size=64, num_attention_heads=4,
            ffn_hidden_size=128, hidden_dropout=0.0,
            attention_dropout=0.0, masked_softmax_fusion=False)
        m = GPTModel(config=cfg,
                     transformer_layer_spec=get_gpt_layer_local_spec(
                         use_flash=False),
                     vocab_size=128, max_sequence_length=64).eval()
        ids = torch.randint(0, 128, (2, 16))
        pos = torch.arange(16).expand(2, -1)
        with torch.no_grad():
            ref = m(ids, pos)
        mq = copy.deepcopy(m)
        n = quantize_model(mq)
        assert n >= 8            # qkv/proj/fc1/fc2 per layer
        with torch.no_grad():
            out = mq(ids, pos)
        # int8 weight-only stays close in logit space
        rel = (out - ref).abs().max() / ref.abs().max()
        assert rel < 0.05, rel
        exported = export_int8_state_dict(mq)
        assert len(exported) == n
        q, s = next(iter(exported.values()))
        assert q.dtype == torch.int8 and s.dtype == torch.float32
        # embeddings / output layer untouched
        assert not any("embedding" in k or "output_layer" in k
                       for k in exported)
    finally:
        destroy()


def test_activation_calibration():
    from megatronapp_amd.post_training import calibrate_activation_scales
    lin = torch.nn.Linear(8, 8)
    model = torch.nn.Sequential(lin)
    batches = [((torch.randn(4, 8) * 3,),) for _ in range(4)]
    scales = calibrate_activation_scales(
        model, [b[0] for b in batches],
        forward=lambda m, x: m(*x))
    assert "0" in scales and scales["0"] > 0
