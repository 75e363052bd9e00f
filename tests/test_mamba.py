"""Mamba / selective-scan (reference core/ssm + models/mamba)."""
import pytest
import torch

from tests.utils import initialize_model_parallel, destroy


def _scan_inputs(b=2, l=96, d=8, n=16, device="cpu", dtype=torch.float32):
    g = torch.Generator().manual_seed(4)
    x = torch.randn(b, l, d, generator=g).to(device, dtype)
    dt = (torch.rand(b, l, d, generator=g) * 0.1 + 0.01).to(device, dtype)
    A = (-torch.rand(d, n, generator=g) - 0.5).to(device, torch.float32)
    B = torch.randn(b, l, n, generator=g).to(device, dtype)
    C = torch.randn(b, l, n, generator=g).to(device, dtype)
    D = torch.randn(d, generator=g).to(device, torch.float32)
    return x, dt, A, B, C, D


def test_chunked_scan_matches_reference():
    from megatronapp_amd.core.ssm import (selective_scan_chunked,
                                          selective_scan_ref)
    x, dt, A, B, C, D = _scan_inputs()
    ref = selective_scan_ref(x, dt, A, B, C, D)
    got = selective_scan_chunked(x, dt, A, B, C, D, chunk=32)
    assert (ref - got).abs().max() < 1e-4


def test_chunked_scan_state_carry():
    """Scanning in two halves with the carried state == one scan."""
    from megatronapp_amd.core.ssm import selective_scan_chunked
    x, dt, A, B, C, D = _scan_inputs(l=64)
    full = selective_scan_chunked(x, dt, A, B, C, D)
    y1, h = selective_scan_chunked(x[:, :40], dt[:, :40], A, B[:, :40],
                                   C[:, :40], D, return_state=True)
    y2 = selective_scan_chunked(x[:, 40:], dt[:, 40:], A, B[:, 40:],
                                C[:, 40:], D, h0=h)
    assert (torch.cat([y1, y2], 1) - full).abs().max() < 1e-4


@pytest.mark.gpu
def test_scan_kernel_matches_chunked_gpu():
    from megatronapp_amd.core.ssm import (selective_scan,
                                          selective_scan_chunked)
    x, dt, A, B, C, D = _scan_inputs(device="cuda", dtype=torch.bfloat16)
    ref = selective_scan_chunked(x.float(), dt.float(), A, B.float(),
                                 C.float(), D)
    with torch.no_grad():
        got = selective_scan(x, dt, A, B, C, D)   # HIP kernel path
    assert (ref - got.float()).abs().max() < 0.05


def _model(device="cpu", dtype=torch.float32):
    from megatronapp_amd.core.models.mamba import MambaModel
    from megatronapp_amd.core.transformer_config import TransformerConfig
    from megatronapp_amd.core.tensor_parallel.random import (
        model_parallel_cuda_manual_seed)
    model_parallel_cuda_manual_seed(9)
    torch.manual_seed(9)
    cfg = TransformerConfig(
        num_layers=2, hidden_size=64, num_attention_heads=4,
        hidden_dropout=0.0, attention_dropout=0.0,
        params_dtype=dtype, bf16=(dtype == torch.bfloat16))
    with torch.device(device):
        return MambaModel(config=cfg, vocab_size=128, max_sequence_length=64)


def test_mamba_learns_memorizable_batch():
    initialize_model_parallel()
    m = _model()
    opt = torch.optim.Adam(m.parameters(), lr=3e-3)
    g = torch.Generator().manual_seed(1)
    tok = torch.randint(0, 128, (4, 32), generator=g)
    losses = []
    for _ in range(30):
        opt.zero_grad()
        loss = m(tok, labels=tok).mean()
        loss.backward()
        opt.step()
        losses.append(float(loss))
    assert losses[-1] < losses[0] * 0.6, losses[::6]
    destroy()


def test_mamba_decode_matches_full():
    """Stateful decode (conv window + ssm state) == full forward."""
    from megatronapp_amd.core.inference_params import InferenceParams
    initialize_model_parallel()
    m = _model().eval()
    tok = torch.randint(0, 128, (1, 24))
    with torch.no_grad():
        full = m(tok)
        ctx = InferenceParams(1, 64)
        pre = m(tok[:, :20], inference_context=ctx)
        steps = []
        for i in range(20, 24):
            steps.append(m(tok[:, i:i + 1], inference_context=ctx))
    assert torch.allclose(pre, full[:, :20], atol=1e-4)
    dec = torch.cat(steps, dim=1)
    assert torch.allclose(dec, full[:, 20:], atol=1e-4), \
        (dec - full[:, 20:]).abs().max()
    destroy()


@pytest.mark.gpu
def test_mamba_gpu_train_step():
    initialize_model_parallel()
    m = _model("cuda", torch.bfloat16)
    tok = torch.randint(0, 128, (2, 64), device="cuda")
    loss = m(tok, labels=tok).float().mean()
    loss.backward()
    assert torch.isfinite(loss)
    destroy()
