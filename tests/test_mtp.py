"""Multi-token prediction (reference multi_token_prediction.py)."""
import torch

from tests.utils import initialize_model_parallel, destroy


def _model(mtp_layers):
    from megatronapp_amd.core.models.gpt import GPTModel
    from megatronapp_amd.core.models.gpt.gpt_layer_specs import (
        get_gpt_layer_local_spec)
    from megatronapp_amd.core.transformer_config import TransformerConfig
    from megatronapp_amd.core.tensor_parallel.random import (
        model_parallel_cuda_manual_seed)
    model_parallel_cuda_manual_seed(5)
    torch.manual_seed(5)
    cfg = TransformerConfig(
        num_layers=2, hidden_size=64, num_attention_heads=4,
        ffn_hidden_size=128, hidden_dropout=0.0, attention_dropout=0.0,
        mtp_num_layers=mtp_layers, mtp_loss_scaling_factor=0.2)
    return GPTModel(config=cfg,
                    transformer_layer_spec=get_gpt_layer_local_spec(
                        use_flash=False),
                    vocab_size=128, max_sequence_length=64,
                    pre_process=True, post_process=True)


def test_mtp_trains_and_tracks_losses():
    from megatronapp_amd.core.transformer.multi_token_prediction import (
        MTPLossLoggingHelper)
    initialize_model_parallel()
    m = _model(2)
    assert m.mtp is not None and len(m.mtp.layers) == 2
    tok = torch.randint(0, 128, (2, 32))
    pos = torch.arange(32).unsqueeze(0).expand(2, -1)
    loss = m(tok, pos, None, labels=tok).mean()
    loss.backward()
    # every MTP depth contributed a tracked loss
    vals = MTPLossLoggingHelper.get_and_clear()
    assert vals is not None and vals.shape == (2,)
    assert (vals > 0).all()
    # MTP parameters received gradients through the attached losses
    for name, p in m.named_parameters():
        assert p.grad is not None, name
    eh = dict(m.named_parameters())["mtp.layers.0.eh_proj.weight"]
    assert eh.grad.abs().sum() > 0
    destroy()


def test_mtp_grad_reaches_main_stream():
    """The attached depth losses must backprop into the MAIN decoder
    (hidden_states feeds the MTP chain)."""
    initialize_model_parallel()
    m0 = _model(0 or None)  # no mtp
    torch.manual_seed(7)
    tok = torch.randint(0, 128, (2, 32))
    pos = torch.arange(32).unsqueeze(0).expand(2, -1)
    m0(tok, pos, None, labels=tok).mean().backward()
    g0 = dict(m0.named_parameters())[
        "decoder.layers.0.self_attention.linear_qkv.weight"].grad.clone()

    m1 = _model(1)
    m1(tok, pos, None, labels=tok).mean().backward()
    g1 = dict(m1.named_parameters())[
        "decoder.layers.0.self_attention.linear_qkv.weight"].grad
    assert not torch.allclose(g0, g1)  # MTP changed the main-stream grads
    destroy()
