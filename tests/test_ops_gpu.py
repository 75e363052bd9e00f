"""Numerics tests for the CDNA4 HIP kernels vs plain fp32 torch references.

Run on an MI355X via gpurun; each kernel must match the fp32 reference to
bf16 tolerance.
"""

import pytest
import torch

pytestmark = pytest.mark.gpu

ATOL = 2e-2
RTOL = 2e-2


def _ops():
    from megatronapp_amd import ops
    return ops.get_ops()


def test_extension_loaded_from_tree():
    from megatronapp_amd import ops
    assert ops.have_ops(), f"HIP extension missing on GPU box: {ops._LOAD_ERROR}"
    import megatronapp_amd.ops as m
    import os
    so = os.path.join(os.path.dirname(m.__file__), "_C.so")
    assert os.path.exists(so)


@pytest.mark.parametrize("shape", [(128, 2048), (64, 128), (1024, 4096)])
def test_rmsnorm_fwd_bwd(shape):
    N, H = shape
    torch.manual_seed(0)
    x = torch.randn(N, H, device="cuda", dtype=torch.bfloat16)
    w = torch.randn(H, device="cuda", dtype=torch.bfloat16)
    eps = 1e-5

    y, invrms = _ops().rmsnorm_fwd(x, w, eps)
    xf = x.float()
    ref_inv = torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + eps)
    ref_y = xf * ref_inv * w.float()
    assert torch.allclose(y.float(), ref_y, atol=ATOL, rtol=RTOL)
    assert torch.allclose(invrms, ref_inv.squeeze(-1), atol=1e-5, rtol=1e-4)

    dy = torch.randn_like(x)
    dx, dw = _ops().rmsnorm_bwd(dy, x, w, invrms)
    xr = xf.clone().requires_grad_(True)
    wr = w.float().clone().requires_grad_(True)
    yr = xr * torch.rsqrt(xr.pow(2).mean(-1, keepdim=True) + eps) * wr
    yr.backward(dy.float())
    assert torch.allclose(dx.float(), xr.grad, atol=ATOL, rtol=RTOL), \
        (dx.float() - xr.grad).abs().max()
    assert torch.allclose(dw, wr.grad, atol=0.1, rtol=2e-2), \
        (dw - wr.grad).abs().max()


@pytest.mark.parametrize("shape", [(256, 1024), (128, 2048)])
def test_layernorm_fwd_bwd(shape):
    N, H = shape
    torch.manual_seed(1)
    x = torch.randn(N, H, device="cuda", dtype=torch.bfloat16)
    w = torch.randn(H, device="cuda", dtype=torch.bfloat16)
    b = torch.randn(H, device="cuda", dtype=torch.bfloat16)
    eps = 1e-5
    y, mean, invstd = _ops().layernorm_fwd(x, w, b, eps)
    ref = torch.nn.functional.layer_norm(x.float(), (H,), w.float(), b.float(), eps)
    assert torch.allclose(y.float(), ref, atol=ATOL, rtol=RTOL)

    dy = torch.randn_like(x)
    dx, dw, db = _ops().layernorm_bwd(dy, x, w, mean, invstd)
    xr = x.float().clone().requires_grad_(True)
    wr = w.float().clone().requires_grad_(True)
    br = b.float().clone().requires_grad_(True)
    torch.nn.functional.layer_norm(xr, (H,), wr, br, eps).backward(dy.float())
    assert torch.allclose(dx.float(), xr.grad, atol=ATOL, rtol=RTOL)
    assert torch.allclose(dw, wr.grad, atol=0.1, rtol=2e-2)
    assert torch.allclose(db, br.grad, atol=0.1, rtol=2e-2)


def test_bias_swiglu():
    torch.manual_seed(2)
    N, F = 512, 1024
    x = torch.randn(N, 2 * F, device="cuda", dtype=torch.bfloat16)
    bias = torch.randn(2 * F, device="cuda", dtype=torch.bfloat16)
    y = _ops().bias_swiglu_fwd(x, bias)
    xf = (x.float() + bias.float())
    x1, x2 = xf.chunk(2, dim=-1)
    ref = torch.nn.functional.silu(x1) * x2
    assert torch.allclose(y.float(), ref, atol=ATOL, rtol=RTOL)

    dy = torch.randn(N, F, device="cuda", dtype=torch.bfloat16)
    dx = _ops().bias_swiglu_bwd(dy, x, bias)
    xr = x.float().clone().requires_grad_(True)
    xf2 = xr + bias.float()
    a, c = xf2.chunk(2, dim=-1)
    (torch.nn.functional.silu(a) * c).backward(dy.float())
    assert torch.allclose(dx.float(), xr.grad, atol=ATOL, rtol=RTOL)


def test_bias_gelu():
    torch.manual_seed(3)
    N, F = 512, 2048
    x = torch.randn(N, F, device="cuda", dtype=torch.bfloat16)
    bias = torch.randn(F, device="cuda", dtype=torch.bfloat16)
    y = _ops().bias_gelu_fwd(x, bias)
    ref = torch.nn.functional.gelu((x.float() + bias.float()), approximate="tanh")
    assert torch.allclose(y.float(), ref, atol=ATOL, rtol=RTOL)

    dy = torch.randn_like(x)
    dx = _ops().bias_gelu_bwd(dy, x, bias)
    xr = x.float().clone().requires_grad_(True)
    torch.nn.functional.gelu(xr + bias.float(), approximate="tanh").backward(dy.float())
    assert torch.allclose(dx.float(), xr.grad, atol=ATOL, rtol=RTOL)


def test_rope_fwd_bwd():
    torch.manual_seed(4)
    s, b, nh, d = 128, 2, 4, 128
    t = torch.randn(s, b, nh, d, device="cuda", dtype=torch.bfloat16)
    inv_freq = 1.0 / (10000 ** (torch.arange(0, d, 2).float() / d))
    freqs = torch.outer(torch.arange(s).float(), inv_freq)
    emb = torch.cat([freqs, freqs], dim=-1)[:, None, None, :].cuda()
    cos, sin = torch.cos(emb), torch.sin(emb)

    y = _ops().rope_fwd(t, cos.contiguous(), sin.contiguous())

    def rotate_half(x):
        x1, x2 = x.chunk(2, dim=-1)
        return torch.cat((-x2, x1), dim=-1)

    ref = t.float() * cos + rotate_half(t.float()) * sin
    assert torch.allclose(y.float(), ref, atol=ATOL, rtol=RTOL)

    dy = torch.randn_like(t)
    dx = _ops().rope_bwd(dy, cos.contiguous(), sin.contiguous())
    tr = t.float().clone().requires_grad_(True)
    (tr * cos + rotate_half(tr) * sin).backward(dy.float())
    assert torch.allclose(dx.float(), tr.grad, atol=ATOL, rtol=RTOL)


@pytest.mark.parametrize("sq", [128, 512, 2048, 4096])
def test_softmax_causal(sq):
    torch.manual_seed(5)
    b_np = 8
    x = torch.randn(b_np, sq, sq, device="cuda", dtype=torch.bfloat16)
    scale = 0.125
    y = _ops().scaled_upper_triang_masked_softmax_fwd(x, scale)
    mask = torch.triu(torch.ones(sq, sq, dtype=torch.bool, device="cuda"), 1)
    ref = torch.softmax((x.float() * scale).masked_fill(mask, float("-inf")), -1)
    assert torch.allclose(y.float(), ref, atol=ATOL, rtol=RTOL)

    dy = torch.randn_like(x)
    xr = x.float().clone().requires_grad_(True)
    torch.softmax((xr * scale).masked_fill(mask, float("-inf")), -1).backward(dy.float())
    dx = _ops().scaled_softmax_bwd(dy, y, scale)
    assert torch.allclose(dx.float(), xr.grad, atol=ATOL, rtol=RTOL)
    # causal-aware bwd (register-cached prefix reads)
    dx_c = _ops().scaled_upper_triang_masked_softmax_bwd(dy, y, scale)
    assert torch.allclose(dx_c.float(), xr.grad, atol=ATOL, rtol=RTOL)


def test_softmax_masked():
    torch.manual_seed(6)
    b, np_, sq, sk = 2, 4, 64, 64
    x = torch.randn(b, np_, sq, sk, device="cuda", dtype=torch.bfloat16)
    mask = torch.rand(b, 1, sq, sk, device="cuda") > 0.7
    y = _ops().scaled_masked_softmax_fwd(x, mask.expand(b, 1, sq, sk).reshape(b, sq, sk).contiguous(), 0.5)
    ref_in = (x.float() * 0.5).masked_fill(mask, float("-inf"))
    ref = torch.softmax(ref_in, -1).nan_to_num(0.0)
    assert torch.allclose(y.float(), ref, atol=ATOL, rtol=RTOL)


def test_adamw_flat_matches_torch():
    torch.manual_seed(7)
    n = 10007
    p = torch.randn(n, device="cuda")
    g = torch.randn(n, device="cuda")
    m = torch.zeros(n, device="cuda")
    v = torch.zeros(n, device="cuda")
    p_ref = p.clone()

    lr, b1, b2, eps, wd = 1e-3, 0.9, 0.999, 1e-8, 0.01
    p16 = torch.zeros(n, device="cuda", dtype=torch.bfloat16)
    for step in range(1, 4):
        _ops().adamw_flat(p, g, m, v, lr, b1, b2, eps, wd, step, p16)
    # the fused bf16 param mirror must equal the cast of the master
    assert torch.equal(p16, p.bfloat16())

    mt = torch.zeros(n, device="cuda")
    vt = torch.zeros(n, device="cuda")
    for step in range(1, 4):
        p_ref.mul_(1 - lr * wd)
        mt.mul_(b1).add_(g, alpha=1 - b1)
        vt.mul_(b2).addcmul_(g, g, value=1 - b2)
        bc1 = 1 - b1 ** step
        bc2 = 1 - b2 ** step
        p_ref.addcdiv_(mt, (vt / bc2).sqrt().add(eps), value=-lr / bc1)
    assert torch.allclose(p, p_ref, atol=1e-6, rtol=1e-5), (p - p_ref).abs().max()


def test_model_layer_gpu_matches_cpu_fallback():
    """One fused transformer stack forward on GPU vs fp32 CPU reference."""
    from megatronapp_amd.core import parallel_state
    from tests.utils import initialize_model_parallel, destroy
    from megatronapp_amd.core.transformer_config import TransformerConfig
    from megatronapp_amd.core.models.gpt import GPTModel
    from megatronapp_amd.core.models.gpt.gpt_layer_specs import (
        get_gpt_layer_local_spec)
    from megatronapp_amd.core.tensor_parallel.random import (
        model_parallel_cuda_manual_seed)

    initialize_model_parallel()
    model_parallel_cuda_manual_seed(99)
    torch.manual_seed(99)
    config = TransformerConfig(
        num_layers=2, hidden_size=256, num_attention_heads=4,
        hidden_dropout=0.0, attention_dropout=0.0, bf16=True,
        position_embedding_type="rope", normalization="RMSNorm",
        activation_func="swiglu", add_bias_linear=False)
    with torch.device("cuda"):
        m = GPTModel(config=config,
                     transformer_layer_spec=get_gpt_layer_local_spec(
                         normalization="RMSNorm", use_flash=False),
                     vocab_size=1024, max_sequence_length=64,
                     position_embedding_type="rope")
    tokens = torch.randint(0, 1024, (2, 64), device="cuda")
    pos = torch.arange(64, device="cuda").unsqueeze(0).expand(2, -1)
    logits = m(tokens, pos)
    assert logits.shape == (2, 64, 1024)
    assert torch.isfinite(logits.float()).all()
    destroy()


def _sdpa_ref(q, k, v, scale, causal):
    """fp32 full-tensor reference; q [sq,b,nh,d], k/v [sk,b,ng,d]."""
    sq, b, nh, d = q.shape
    sk, _, ng, _ = k.shape
    qf = q.float().permute(1, 2, 0, 3)          # [b,nh,sq,d]
    kf = k.float().permute(1, 2, 0, 3)
    vf = v.float().permute(1, 2, 0, 3)
    if ng != nh:
        rep = nh // ng
        kf = kf.repeat_interleave(rep, dim=1)
        vf = vf.repeat_interleave(rep, dim=1)
    s = torch.matmul(qf, kf.transpose(-1, -2)) * scale
    if causal:
        mask = torch.triu(torch.ones(sq, sk, dtype=torch.bool,
                                     device=q.device), 1 + sk - sq)
        s = s.masked_fill(mask, float("-inf"))
    p = torch.softmax(s, dim=-1)
    o = torch.matmul(p, vf)                     # [b,nh,sq,d]
    lse = torch.logsumexp(s, dim=-1)            # [b,nh,sq]
    return o.permute(2, 0, 1, 3), lse


@pytest.mark.parametrize("d,gqa,sq", [(128, 1, 256), (128, 4, 256),
                                      (64, 1, 128), (128, 1, 2048)])
def test_flash_attn_fwd(d, gqa, sq):
    torch.manual_seed(10)
    b, nh = 2, 8
    ng = nh // gqa
    scale = d ** -0.5
    q = torch.randn(sq, b, nh, d, device="cuda", dtype=torch.bfloat16)
    k = torch.randn(sq, b, ng, d, device="cuda", dtype=torch.bfloat16)
    v = torch.randn(sq, b, ng, d, device="cuda", dtype=torch.bfloat16)
    o, lse = _ops().attn_fwd(q, k, v, scale, True)
    o_ref, lse_ref = _sdpa_ref(q, k, v, scale, True)
    err = (o.float() - o_ref).abs().max().item()
    assert err < 3e-2, f"fwd max err {err}"
    lse_err = (lse - lse_ref).abs().max().item()
    assert lse_err < 1e-2, f"lse max err {lse_err}"


@pytest.mark.parametrize("d,gqa", [(128, 1), (128, 4), (64, 1)])
def test_flash_attn_bwd(d, gqa):
    torch.manual_seed(11)
    sq, b, nh = 256, 2, 8
    ng = nh // gqa
    scale = d ** -0.5
    q = torch.randn(sq, b, nh, d, device="cuda", dtype=torch.bfloat16)
    k = torch.randn(sq, b, ng, d, device="cuda", dtype=torch.bfloat16)
    v = torch.randn(sq, b, ng, d, device="cuda", dtype=torch.bfloat16)
    do = torch.randn(sq, b, nh, d, device="cuda", dtype=torch.bfloat16)

    o, lse = _ops().attn_fwd(q, k, v, scale, True)
    dq, dk, dv = _ops().attn_bwd(do, q, k, v, o, lse, scale, True)

    qr = q.float().clone().requires_grad_(True)
    kr = k.float().clone().requires_grad_(True)
    vr = v.float().clone().requires_grad_(True)
    o_ref, _ = _sdpa_ref(qr, kr, vr, scale, True)
    o_ref.backward(do.float())
    for got, ref, name in ((dq, qr.grad, "dq"), (dk, kr.grad, "dk"),
                           (dv, vr.grad, "dv")):
        err = (got.float() - ref).abs().max().item()
        rel = err / (ref.abs().max().item() + 1e-6)
        assert rel < 5e-2, f"{name} max err {err} rel {rel}"


def test_flash_attn_noncausal():
    torch.manual_seed(12)
    sq, b, nh, d = 128, 2, 4, 128
    scale = d ** -0.5
    q = torch.randn(sq, b, nh, d, device="cuda", dtype=torch.bfloat16)
    k = torch.randn(sq, b, nh, d, device="cuda", dtype=torch.bfloat16)
    v = torch.randn(sq, b, nh, d, device="cuda", dtype=torch.bfloat16)
    o, lse = _ops().attn_fwd(q, k, v, scale, False)
    o_ref, _ = _sdpa_ref(q, k, v, scale, False)
    assert (o.float() - o_ref).abs().max().item() < 3e-2


@pytest.mark.gpu
def test_train_step_grads_fused_vs_eager():
    """Full fwd+bwd main_grad parity: HIP fused paths (wgrad_accum, colsum
    bias grads, fused norms/activations) vs the eager torch fallback,
    both on GPU.  Catches any fused grad landing in the wrong buffer."""
    import megatronapp_amd.ops as _ops_mod
    from megatronapp_amd.core import parallel_state
    from tests.utils import initialize_model_parallel, destroy
    from megatronapp_amd.core.transformer_config import TransformerConfig
    from megatronapp_amd.core.models.gpt import GPTModel
    from megatronapp_amd.core.models.gpt.gpt_layer_specs import (
        get_gpt_layer_local_spec)
    from megatronapp_amd.core.distributed import (
        DistributedDataParallel, DistributedDataParallelConfig)
    from megatronapp_amd.core.tensor_parallel.random import (
        model_parallel_cuda_manual_seed)

    initialize_model_parallel()

    def build():
        model_parallel_cuda_manual_seed(17)
        torch.manual_seed(17)
        config = TransformerConfig(
            num_layers=2, hidden_size=256, num_attention_heads=4,
            ffn_hidden_size=512, hidden_dropout=0.0, attention_dropout=0.0,
            bf16=True, params_dtype=torch.bfloat16, add_bias_linear=True,
            masked_softmax_fusion=True)
        with torch.device("cuda"):
            m = GPTModel(config=config,
                         transformer_layer_spec=get_gpt_layer_local_spec(
                             use_flash=False),
                         vocab_size=512, max_sequence_length=128,
                         pre_process=True, post_process=True)
        return DistributedDataParallel(
            config, DistributedDataParallelConfig(), m)

    torch.manual_seed(3)
    tokens = torch.randint(0, 512, (2, 128), device="cuda")
    pos = torch.arange(128, device="cuda").unsqueeze(0).expand(2, -1)

    def run(ddp):
        ddp.zero_grad_buffer()
        loss = ddp(input_ids=tokens, position_ids=pos, attention_mask=None,
                   labels=tokens).float().mean()
        loss.backward()
        return {n: p.main_grad.clone()
                for n, p in ddp.module.named_parameters()}

    fused = run(build())
    orig = _ops_mod.have_ops
    _ops_mod.have_ops = lambda: False
    try:
        eager = run(build())
    finally:
        _ops_mod.have_ops = orig

    for name in fused:
        f, e = fused[name], eager[name]
        denom = e.abs().max().clamp(min=1e-3)
        rel = (f - e).abs().max() / denom
        assert rel < 0.06, f"{name}: rel diff {rel:.4f}"
    destroy()


@pytest.mark.gpu
def test_adamw_flat_ranged_matches_per_range():
    """Single-launch ranged AdamW == stitched per-range adamw_flat."""
    import megatronapp_amd.ops as _ops_mod
    ops = _ops_mod.get_ops()
    torch.manual_seed(5)
    n = 40960
    ranges = [(4096, 8192), (12288, 12288 + 2048), (20480, 24576)]
    lr, b1, b2, eps, wd, step = 1e-3, 0.9, 0.95, 1e-8, 0.1, 7

    def init():
        p = torch.randn(n, device="cuda", dtype=torch.float32)
        g = torch.randn(n, device="cuda", dtype=torch.float32)
        m = torch.randn(n, device="cuda", dtype=torch.float32).abs()
        v = torch.randn(n, device="cuda", dtype=torch.float32).abs()
        return p, g, m, v

    torch.manual_seed(5)
    p1, g1, m1, v1 = init()
    torch.manual_seed(5)
    p2, g2, m2, v2 = init()

    nw_s = torch.tensor([r[0] for r in ranges], dtype=torch.int64,
                        device="cuda")
    nw_e = torch.tensor([r[1] for r in ranges], dtype=torch.int64,
                        device="cuda")
    ops.adamw_flat_ranged(p1, g1, m1, v1, nw_s, nw_e, lr, b1, b2, eps, wd,
                          step)

    cursor = 0
    for (s, e) in ranges + [(n, n)]:
        if cursor < s:
            ops.adamw_flat(p2[cursor:s], g2[cursor:s], m2[cursor:s],
                           v2[cursor:s], lr, b1, b2, eps, wd, step)
        if s < e:
            ops.adamw_flat(p2[s:e], g2[s:e], m2[s:e], v2[s:e], lr, b1, b2,
                           eps, 0.0, step)
        cursor = max(cursor, e)

    for a, b, name in ((p1, p2, "p"), (m1, m2, "m"), (v1, v2, "v")):
        assert torch.equal(a, b), f"{name} differs"


@pytest.mark.gpu
def test_bias_add_residual():
    import megatronapp_amd.ops as _ops_mod
    ops = _ops_mod.get_ops()
    torch.manual_seed(9)
    x = torch.randn(128, 4, 256, device="cuda", dtype=torch.bfloat16)
    r = torch.randn_like(x)
    b = torch.randn(256, device="cuda", dtype=torch.bfloat16)
    out = ops.bias_add_residual(x, b, r)
    ref = (x.float() + b.float() + r.float()).to(torch.bfloat16)
    assert (out.float() - ref.float()).abs().max() < 2e-2


@pytest.mark.gpu
def test_fused_cross_entropy_matches_eager():
    """Fused CE (bf16 single-pass kernels) vs the eager fp32 path: loss and
    dlogits."""
    import megatronapp_amd.ops as _ops_mod
    from tests.utils import initialize_model_parallel, destroy
    from megatronapp_amd.core.tensor_parallel.cross_entropy import (
        vocab_parallel_cross_entropy)
    initialize_model_parallel()
    torch.manual_seed(13)
    logits = torch.randn(64, 2, 1024, device="cuda",
                         dtype=torch.bfloat16) * 4
    target = torch.randint(0, 1024, (64, 2), device="cuda")

    l1 = logits.clone().requires_grad_(True)
    loss_fused = vocab_parallel_cross_entropy(l1, target)
    loss_fused.mean().backward()

    orig = _ops_mod.have_ops
    _ops_mod.have_ops = lambda: False
    try:
        l2 = logits.clone().requires_grad_(True)
        loss_eager = vocab_parallel_cross_entropy(l2, target)
        loss_eager.mean().backward()
    finally:
        _ops_mod.have_ops = orig

    # the eager path rounds (x - max) to bf16 before exp; the fused path
    # keeps it fp32 — so they differ by bf16 ulps on ~20-magnitude losses
    rel = (loss_fused - loss_eager).abs().max() / loss_eager.abs().max()
    assert rel < 3e-3, rel
    assert (l1.grad.float() - l2.grad.float()).abs().max() < 2e-2
    destroy()


@pytest.mark.gpu
def test_swept_gemms_match_torch():
    ops = _ops()
    torch.manual_seed(17)
    a = torch.randn(256, 512, device="cuda", dtype=torch.bfloat16)
    w = torch.randn(384, 512, device="cuda", dtype=torch.bfloat16)
    ref = torch.matmul(a.float(), w.t().float())
    out = ops.gemm_nt(a, w)
    assert (out.float() - ref).abs().max() / ref.abs().max() < 2e-2
    g = torch.randn(256, 384, device="cuda", dtype=torch.bfloat16)
    refnn = torch.matmul(g.float(), w.float())
    outnn = ops.gemm_nn(g, w)
    assert (outnn.float() - refnn).abs().max() / refnn.abs().max() < 2e-2


@pytest.mark.gpu
@pytest.mark.parametrize("rows,H", [(2048, 256), (512, 2048), (333, 512)])
def test_norm_residual_backward_matches_eager(rows, H):
    """The two-output norm (residual-grad fused into dx) vs fp32 autograd,
    including the main_grad-accumulate branch."""
    from megatronapp_amd.core.fusions.fused_layer_norm import (
        _LayerNormResidualFn, _RMSNormResidualFn)
    torch.manual_seed(3)
    for fn, has_bias in ((_LayerNormResidualFn, True),
                        (_RMSNormResidualFn, False)):
        x = torch.randn(rows, H, device="cuda",
                        dtype=torch.bfloat16).requires_grad_(True)
        w = torch.nn.Parameter(torch.rand(H, device="cuda",
                                          dtype=torch.bfloat16) + 0.5)
        b = torch.nn.Parameter(torch.randn(H, device="cuda",
                                           dtype=torch.bfloat16))
        dy = torch.randn(rows, H, device="cuda", dtype=torch.bfloat16)
        dres = torch.randn(rows, H, device="cuda", dtype=torch.bfloat16)

        # fp32 reference
        xf = x.detach().float().requires_grad_(True)
        wf = w.detach().float().requires_grad_(True)
        bf = b.detach().float().requires_grad_(True)
        if has_bias:
            yr = torch.nn.functional.layer_norm(xf, (H,), wf, bf, 1e-5)
        else:
            r = torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + 1e-5)
            yr = xf * r * wf
        (yr * dy.float()).sum().backward()
        ref_dx = xf.grad + dres.float()

        # fused two-output path, main_grad branch
        w.main_grad = torch.zeros(H, device="cuda", dtype=torch.float32)
        w.grad_added_to_main_grad = False
        if has_bias:
            b.main_grad = torch.zeros(H, device="cuda", dtype=torch.float32)
            b.grad_added_to_main_grad = False
            y, passthrough = fn.apply(x, w, b, 1e-5)
        else:
            y, passthrough = fn.apply(x, w, 1e-5)
        (y.float() * dy.float()).sum().backward(retain_graph=True)
        # feed the residual grad through the passthrough output
        x.grad = None
        w.main_grad.zero_()              # first backward accumulated once
        w.grad_added_to_main_grad = False
        if has_bias:
            b.main_grad.zero_()
            b.grad_added_to_main_grad = False
        if has_bias:
            y2, p2 = fn.apply(x, w, b, 1e-5)
        else:
            y2, p2 = fn.apply(x, w, 1e-5)
        ((y2.float() * dy.float()).sum()
         + (p2.float() * dres.float()).sum()).backward()
        got_dx = x.grad.float()
        err = (got_dx - ref_dx).abs().max()
        assert err < 0.1, (fn.__name__, rows, H, float(err))
        werr = (w.main_grad - wf.grad).abs().max() / wf.grad.abs().max()
        assert werr < 0.05, (fn.__name__, float(werr))


@pytest.mark.gpu
@pytest.mark.parametrize("sq,gqa", [(256, 1), (256, 4), (2048, 1)])
def test_attn_fwd_transposed_matches_ref(sq, gqa):
    ops = _ops()
    torch.manual_seed(23)
    b, nh, d = 2, 8, 128
    ng = nh // gqa
    q = torch.randn(sq, b, nh, d, device="cuda", dtype=torch.bfloat16)
    k = torch.randn(sq, b, ng, d, device="cuda", dtype=torch.bfloat16)
    v = torch.randn_like(k)
    scale = d ** -0.5
    o_ref, lse_ref = _sdpa_ref(q, k, v, scale, True)
    o, lse = ops.attn_fwd_t(q, k, v, scale, True)
    assert torch.allclose(o.float(), o_ref.float(), atol=2e-2, rtol=2e-2), \
        (o.float() - o_ref.float()).abs().max()
    lse_got = lse.view(b, nh, sq)
    assert torch.allclose(lse_got, lse_ref.float(), atol=1e-3), \
        (lse_got - lse_ref.float()).abs().max()


@pytest.mark.gpu
@pytest.mark.parametrize("sq,gqa,causal,d", [(256, 1, True, 128),
                                             (256, 4, True, 128),
                                             (2048, 1, True, 128),
                                             (512, 1, False, 128),
                                             (2048, 4, True, 128),
                                             (256, 1, True, 64),
                                             (2048, 2, True, 64)])
def test_attn_fwd2_matches_ref(sq, gqa, causal, d):
    """Round-2 32x32-MFMA forward (swapped QK^T, tr16 V reads, in-register
    P^T) vs the fp32 reference."""
    ops = _ops()
    torch.manual_seed(29)
    b, nh = 2, 8
    ng = nh // gqa
    q = torch.randn(sq, b, nh, d, device="cuda", dtype=torch.bfloat16)
    k = torch.randn(sq, b, ng, d, device="cuda", dtype=torch.bfloat16)
    v = torch.randn_like(k)
    scale = d ** -0.5
    o_ref, lse_ref = _sdpa_ref(q, k, v, scale, causal)
    o, lse = ops.attn_fwd2(q, k, v, scale, causal)
    assert torch.allclose(o.float(), o_ref.float(), atol=2e-2, rtol=2e-2), \
        (o.float() - o_ref.float()).abs().max()
    lse_got = lse.view(b, nh, sq)
    assert torch.allclose(lse_got, lse_ref.float(), atol=1e-3), \
        (lse_got - lse_ref.float()).abs().max()


@pytest.mark.gpu
@pytest.mark.parametrize("sq,gqa,causal,d", [(256, 1, True, 128),
                                             (256, 4, True, 128),
                                             (512, 1, False, 128),
                                             (2048, 1, True, 128),
                                             (2048, 4, True, 128),
                                             (256, 1, True, 64),
                                             (2048, 2, True, 64)])
def test_attn_bwd2_matches_ref(sq, gqa, causal, d):
    """Round-2 backward (dq2 + dv2/dk2 kernels; dispatched by attn_bwd
    for sq,sk multiples of 256 at d in {64,128}) vs fp32 autograd."""
    ops = _ops()
    torch.manual_seed(31)
    b, nh = 2, 8
    ng = nh // gqa
    scale = d ** -0.5
    q = torch.randn(sq, b, nh, d, device="cuda", dtype=torch.bfloat16)
    k = torch.randn(sq, b, ng, d, device="cuda", dtype=torch.bfloat16)
    v = torch.randn_like(k)
    do = torch.randn_like(q)
    o, lse = ops.attn_fwd2(q, k, v, scale, causal)
    dq, dk, dv = ops.attn_bwd(do, q, k, v, o, lse, scale, causal)
    qr = q.float().clone().requires_grad_(True)
    kr = k.float().clone().requires_grad_(True)
    vr = v.float().clone().requires_grad_(True)
    o_ref, _ = _sdpa_ref(qr, kr, vr, scale, causal)
    o_ref.backward(do.float())
    for got, ref, name in ((dq, qr.grad, "dq"), (dk, kr.grad, "dk"),
                           (dv, vr.grad, "dv")):
        err = (got.float() - ref).abs().max().item()
        rel = err / (ref.abs().max().item() + 1e-6)
        assert rel < 5e-2, f"{name} max err {err} rel {rel}"


@pytest.mark.gpu
def test_grouped_mlp_grouped_mm_matches_loop():
    """GroupedMLP's single-launch torch._grouped_mm path (fwd + custom
    autograd bwd) vs the segmented per-expert loop."""
    from megatronapp_amd.core.transformer.moe.experts import (
        GroupedMLP, _GroupedMMFn)
    torch.manual_seed(41)
    E, h, f, T = 4, 256, 512, 1024
    x = torch.randn(T, h, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    w = (torch.randn(E, h, f, device="cuda", dtype=torch.bfloat16) *
         0.05).requires_grad_(True)
    sizes = torch.tensor([300, 200, 324, 200], device="cuda")
    offs = torch.cumsum(sizes, 0).to(torch.int32)
    y = _GroupedMMFn.apply(x, w, offs)
    dy = torch.randn_like(y)
    y.backward(dy)

    xr = x.detach().float().requires_grad_(True)
    wr = w.detach().float().requires_grad_(True)
    outs, start = [], 0
    for e in range(E):
        n = int(sizes[e])
        outs.append(xr[start:start + n] @ wr[e])
        start += n
    yr = torch.cat(outs)
    yr.backward(dy.float())
    assert torch.allclose(y.float(), yr, atol=1e-1, rtol=5e-2)
    for got, ref, name in ((x.grad, xr.grad, "dx"), (w.grad, wr.grad, "dw")):
        rel = (got.float() - ref).abs().max() / (ref.abs().max() + 1e-6)
        assert rel < 5e-2, (name, float(rel))


@pytest.mark.gpu
def test_fp8_linear_matches_bf16():
    """fp8 serving fast path (e4m3 _scaled_mm, per-row weight scales +
    per-token act scales) vs the bf16 linear."""
    from megatronapp_amd.inference.fp8 import (_quantize_weight, fp8_linear)
    torch.manual_seed(51)
    M, K, N = 512, 2048, 4096
    x = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
    w = torch.nn.Parameter(
        torch.randn(N, K, device="cuda", dtype=torch.bfloat16) * 0.02)
    ref = x @ w.t()
    w.fp8_data, w.fp8_scale = _quantize_weight(w.data)
    y = fp8_linear(x, w)
    rel = (y.float() - ref.float()).norm() / ref.float().norm()
    # w8a8 e4m3 with dynamic scales lands ~3-4% relative norm error
    assert rel < 0.06, float(rel)


@pytest.mark.gpu
def test_fp8_model_decode_close_to_bf16():
    """Whole-model fp8 serving: logits stay close and greedy tokens
    mostly agree with bf16 on a small GPT."""
    from tests.utils import initialize_model_parallel
    initialize_model_parallel()
    from megatronapp_amd.core.models.gpt import GPTModel
    from megatronapp_amd.core.models.gpt.gpt_layer_specs import (
        get_gpt_layer_local_spec)
    from megatronapp_amd.core.transformer_config import TransformerConfig
    from megatronapp_amd.core.tensor_parallel.random import (
        model_parallel_cuda_manual_seed)
    from megatronapp_amd.inference.fp8 import (dequantize_model_fp8,
                                               quantize_model_fp8)
    model_parallel_cuda_manual_seed(7)
    cfg = TransformerConfig(
        num_layers=4, hidden_size=256, num_attention_heads=4,
        ffn_hidden_size=512, hidden_dropout=0.0, attention_dropout=0.0,
        bf16=True, params_dtype=torch.bfloat16)
    with torch.device("cuda"):
        m = GPTModel(config=cfg,
                     transformer_layer_spec=get_gpt_layer_local_spec(),
                     vocab_size=512, max_sequence_length=256,
                     pre_process=True, post_process=True).eval()
    toks = torch.randint(0, 512, (2, 256), device="cuda")
    pos = torch.arange(256, device="cuda").expand(2, 256)
    mask = None
    with torch.no_grad():
        ref = m(toks, pos, mask).float()
        n = quantize_model_fp8(m)
        assert n > 0
        got = m(toks, pos, mask).float()
        dequantize_model_fp8(m)
    agree = (ref.argmax(-1) == got.argmax(-1)).float().mean()
    # random-init logits are near-uniform, so argmax flips easily; the
    # bar is rough agreement plus bounded logit error
    rel = (got - ref).norm() / ref.norm()
    assert agree > 0.6 and rel < 0.1, (float(agree), float(rel))


@pytest.mark.gpu
def test_quantize_rows_e4m3_matches_eager():
    ops = _ops()
    torch.manual_seed(61)
    for M, K in ((128, 2048), (1000, 8192)):
        x = torch.randn(M, K, device="cuda", dtype=torch.bfloat16) * 3
        q, s = ops.quantize_rows_e4m3(x)
        s_ref = torch.clamp(x.abs().amax(dim=1, keepdim=True).float() / 448.0,
                            min=1e-12)
        q_ref = (x.float() / s_ref).clamp(-448, 448).to(torch.float8_e4m3fn)
        assert torch.allclose(s, s_ref, rtol=1e-3)
        # reconstruction error bounded by the e4m3 quantization step
        # (x*inv vs x/s can differ by one ULP at the top bin, so compare
        # against the TRUE values, not bit-exactly against eager)
        deq = q.float() * s
        ulp_bound = 0.07 * x.abs().amax().float() + 1e-3
        err = (deq - x.float()).abs().max()
        assert err < ulp_bound, (float(err), float(ulp_bound))
        deq_ref = q_ref.float() * s_ref
        err_ref = (deq_ref - x.float()).abs().max()
        assert err < err_ref * 1.5 + 1e-3


@pytest.mark.gpu
def test_moe_fused_combine_matches_eager():
    """Fused unpermute+combine (moe.hip) vs the eager 3-pass version,
    forward and gradients."""
    from megatronapp_amd.core.transformer.moe.token_dispatcher import (
        _FusedCombineFn, permute)
    torch.manual_seed(71)
    n, topk, h, E = 512, 2, 256, 8
    tokens = torch.randn(n, h, device="cuda", dtype=torch.bfloat16)
    indices = torch.randint(0, E, (n, topk), device="cuda")
    probs = torch.softmax(torch.randn(n, topk, device="cuda"), dim=-1)
    permuted, sort_idx = permute(tokens, indices)

    pg = permuted.detach().requires_grad_(True)
    prg = probs.detach().requires_grad_(True)
    out = _FusedCombineFn.apply(pg, sort_idx, prg, n)
    dout = torch.randn_like(out)
    out.backward(dout)

    pe = permuted.detach().float().requires_grad_(True)
    pre = probs.detach().float().requires_grad_(True)
    unsorted = torch.zeros_like(pe).index_copy(0, sort_idx, pe)
    ref = (unsorted.reshape(n, topk, h) * pre.unsqueeze(-1)).sum(1)
    ref.backward(dout.float())

    assert torch.allclose(out.float(), ref, atol=3e-2, rtol=3e-2)
    rel = (pg.grad.float() - pe.grad).abs().max() / pe.grad.abs().max()
    assert rel < 3e-2, float(rel)
    relp = (prg.grad.float() - pre.grad).abs().max() / pre.grad.abs().max()
    assert relp < 3e-2, float(relp)


@pytest.mark.gpu
def test_bias_geglu_matches_eager():
    """GeGLU fused kernels vs the fp32 eager formula, fwd + bwd."""
    from megatronapp_amd.core.fusions.fused_bias_act import bias_geglu_impl
    torch.manual_seed(81)
    N, F = 512, 256
    x = torch.randn(N, 2 * F, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    b = torch.randn(2 * F, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    y = bias_geglu_impl(x, b)
    dy = torch.randn_like(y)
    y.backward(dy)

    xr = x.detach().float().requires_grad_(True)
    br = b.detach().float().requires_grad_(True)
    x1, x2 = (xr + br).chunk(2, dim=-1)
    ref = torch.nn.functional.gelu(x1, approximate="tanh") * x2
    ref.backward(dy.float())
    assert torch.allclose(y.float(), ref, atol=5e-2, rtol=5e-2)
    for got, want in ((x.grad, xr.grad), (b.grad, br.grad)):
        rel = (got.float() - want).abs().max() / (want.abs().max() + 1e-6)
        assert rel < 5e-2, float(rel)


@pytest.mark.gpu
def test_scaled_softmax_no_mask_fwd():
    ops = _ops()
    x = torch.randn(4, 64, 72, device="cuda", dtype=torch.bfloat16)
    y = ops.scaled_softmax_fwd(x, 0.5)
    ref = torch.softmax(x.float() * 0.5, dim=-1)
    assert torch.allclose(y.float(), ref, atol=2e-2)


@pytest.mark.gpu
def test_fused_bias_dropout_add_statistics_and_grads():
    """Philox fused bias-dropout-add: keep-rate within tolerance, kept
    elements exactly (x+bias)/(1-p)+residual, backward masks dy
    identically."""
    from megatronapp_amd.core.fusions.fused_bias_dropout import (
        _FusedBiasDropoutAddFn)
    torch.manual_seed(91)
    n, F, p = 4096, 256, 0.3
    x = torch.randn(n, F, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    bias = torch.randn(F, device="cuda", dtype=torch.bfloat16,
                       requires_grad=True)
    res = torch.randn(n, F, device="cuda", dtype=torch.bfloat16)
    out = _FusedBiasDropoutAddFn.apply(x, bias, res, p)
    pre = (x + bias).detach()
    dy = torch.randn_like(out)
    dy[dy.abs() < 1e-2] = 0.1   # make dy nonzero so grads reveal the mask
    out.backward(dy)
    # ground-truth keep mask comes from the backward (dx != 0 iff kept)
    kept = x.grad != 0
    rate = kept.float().mean().item()
    assert abs(rate - (1 - p)) < 0.02, rate
    scaled = (pre / (1 - p) + res).to(torch.bfloat16)
    assert torch.allclose(out[kept].float(), scaled[kept].float(),
                          atol=3e-2, rtol=3e-2)
    assert torch.allclose(out[~kept].float(), res[~kept].float(),
                          atol=1e-6)
    assert torch.allclose(x.grad[kept].float(),
                          (dy[kept].float() / (1 - p)), atol=3e-2,
                          rtol=3e-2)


@pytest.mark.gpu
def test_wgrad_accum_bgrad_matches_eager():
    """BGRADB-epilogue wgrad: main_grad += g^T @ x AND dbias = colsum(g)
    in one hipblasLt call (tensor_parallel/layers.py backward)."""
    torch.manual_seed(5)
    ops = _ops()
    supported = 0
    for rows, out, in_ in [(4096, 1536, 512), (512, 768, 256),
                           (333, 128, 96)]:
        g = torch.randn(rows, out, device="cuda", dtype=torch.bfloat16)
        x = torch.randn(rows, in_, device="cuda", dtype=torch.bfloat16)
        mg = torch.randn(out, in_, device="cuda", dtype=torch.float32)
        mg0 = mg.clone()
        dbias = torch.full((out,), 7.0, device="cuda", dtype=torch.float32)
        if not ops.wgrad_accum_bgrad(g, x, mg, dbias):
            # no epilogue algo for this shape: main_grad must be untouched
            assert torch.equal(mg, mg0), "failed call corrupted main_grad"
            continue
        supported += 1
        ref_w = mg0 + g.float().t() @ x.float()
        ref_b = g.float().sum(0)
        werr = (mg - ref_w).abs().max().item()
        wrel = werr / (ref_w.abs().max().item() + 1e-6)
        berr = (dbias - ref_b).abs().max().item()
        brel = berr / (ref_b.abs().max().item() + 1e-6)
        assert wrel < 2e-2, f"wgrad rel err {wrel} at {rows}x{out}x{in_}"
        assert brel < 2e-2, f"dbias rel err {brel} at {rows}x{out}x{in_}"
    if supported == 0:
        # this hipblaslt only offers BGRADB with bf16 D (tools/probe_bgrad)
        # which cannot carry the fp32 beta=1 main_grad accumulation; the
        # path auto-activates if a future build adds fp32-D support
        pytest.skip("no fp32-D BGRADB algos in this hipblaslt; "
                    "fallback verified")


@pytest.mark.gpu
def test_colsum_accum_matches_and_deterministic():
    """Two-stage colsum: numerics vs fp32 sum and bitwise run-to-run
    determinism (the atomicAdd version it replaced was neither fast nor
    deterministic)."""
    ops = _ops()
    torch.manual_seed(9)
    for R, F in [(32768, 2048), (4096, 8192), (1000, 120), (777, 333)]:
        dy = torch.randn(R, F, device="cuda", dtype=torch.bfloat16)
        out1 = torch.randn(F, device="cuda", dtype=torch.float32)
        base = out1.clone()
        ops.colsum_accum(dy, out1)
        ref = base + dy.float().sum(0)
        rel = ((out1 - ref).abs().max() /
               (ref.abs().max() + 1e-6)).item()
        assert rel < 2e-2, f"colsum rel err {rel} at {R}x{F}"
        if F % 8 == 0:  # vector path is deterministic by construction
            out2 = base.clone()
            ops.colsum_accum(dy, out2)
            assert torch.equal(out1, out2)


@pytest.mark.gpu
def test_grouped_mm_fused_main_grad_wgrad():
    """_GroupedMMFn with host_counts + a DDP-style fp32 main_grad: the
    per-expert hipBLASLt wgrad must accumulate the same dw the grouped
    bf16 path produces (and mark grad_added_to_main_grad)."""
    from megatronapp_amd.core.transformer.moe.experts import _GroupedMMFn
    torch.manual_seed(43)
    E, h, f, T = 4, 256, 512, 1024
    x = torch.randn(T, h, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    w = (torch.randn(E, h, f, device="cuda", dtype=torch.bfloat16) *
         0.05).requires_grad_(True)
    counts = [300, 200, 324, 200]
    offs = torch.cumsum(torch.tensor(counts, device="cuda"), 0).to(torch.int32)
    w.main_grad = torch.randn(E, h, f, device="cuda", dtype=torch.float32)
    w.grad_added_to_main_grad = False
    base = w.main_grad.clone()

    y = _GroupedMMFn.apply(x, w, offs, counts)
    dy = torch.randn_like(y)
    y.backward(dy)
    assert w.grad_added_to_main_grad

    ref = base.clone()
    start = 0
    for e, n in enumerate(counts):
        ref[e] += (x[start:start + n].float().t()
                   @ dy[start:start + n].float())
        start += n
    rel = ((w.main_grad - ref).abs().max() / (ref.abs().max() + 1e-6)).item()
    assert rel < 5e-2, rel
    # dx unchanged vs the no-fusion path
    x2 = x.detach().clone().requires_grad_(True)
    y2 = _GroupedMMFn.apply(x2, w.detach().clone().requires_grad_(True), offs)
    y2.backward(dy)
    assert torch.equal(x.grad, x2.grad)


@pytest.mark.gpu
def test_adamw_flat_bf16_states_matches_fp32_math():
    """Precision-aware optimizer: bf16 exp_avg/exp_avg_sq storage with
    fp32 in-kernel math tracks the fp32-state reference to bf16 rounding
    over several steps."""
    torch.manual_seed(8)
    n = 10007
    p = torch.randn(n, device="cuda")
    g = torch.randn(n, device="cuda")
    m = torch.zeros(n, device="cuda", dtype=torch.bfloat16)
    v = torch.zeros(n, device="cuda", dtype=torch.bfloat16)
    p_ref = p.clone()
    mt = torch.zeros(n, device="cuda")
    vt = torch.zeros(n, device="cuda")
    lr, b1, b2, eps, wd = 1e-3, 0.9, 0.999, 1e-8, 0.01
    for step in range(1, 5):
        _ops().adamw_flat(p, g, m, v, lr, b1, b2, eps, wd, step)
        # reference with bf16 state rounding applied the same way
        p_ref.mul_(1 - lr * wd)
        mf = mt.bfloat16().float().mul_(b1).add_(g, alpha=1 - b1)
        vf = vt.bfloat16().float().mul_(b2).addcmul_(g, g, value=1 - b2)
        bc1, bc2 = 1 - b1 ** step, 1 - b2 ** step
        p_ref.addcdiv_(mf, (vf / bc2).sqrt().add(eps), value=-lr / bc1)
        mt, vt = mf, vf
    # fp32 FMA contraction in the kernel can differ from torch's
    # separate mul/addcmul by 1 ULP, flipping bf16 rounding near a
    # boundary — compare to bf16 precision, not exactly
    assert torch.allclose(m.float(), mt.bfloat16().float(),
                          rtol=2e-2, atol=1e-10), "m mismatch"
    assert torch.allclose(v.float(), vt.bfloat16().float(),
                          rtol=2e-2, atol=1e-10), "v mismatch"
    err = (p - p_ref).abs().max().item()
    assert err < 1e-4, err


@pytest.mark.gpu
def test_quantize_transpose_e4m3():
    """[R, C] bf16 -> [C, R] e4m3 with a per-tensor scale: dequantized
    transpose must reconstruct the input to e4m3 precision, including
    non-multiple-of-64 edges."""
    ops = _ops()
    torch.manual_seed(11)
    for R, C in [(256, 192), (1000, 120), (130, 70)]:
        x = torch.randn(R, C, device="cuda", dtype=torch.bfloat16) * 3
        q, s = ops.quantize_transpose_e4m3(x)
        assert q.shape == (C, R) and q.dtype == torch.float8_e4m3fn
        rec = q.float().t() * s
        err = (rec - x.float()).abs().max().item()
        amax = x.float().abs().max().item()
        # e4m3 mantissa: 3 bits -> rel step 1/16 of the top bin
        assert err <= amax / 448 * 32, (R, C, err, amax)


@pytest.mark.gpu
def test_fp8_wgrad_accumulates_close_to_bf16():
    """fp8_wgrad (transpose-quantized e4m3 GEMM, fp32 accumulate) tracks
    the exact wgrad within fp8 tolerance."""
    from megatronapp_amd.core.fp8 import fp8_wgrad
    torch.manual_seed(12)
    rows, out, in_ = 4096, 1536, 512
    g = torch.randn(rows, out, device="cuda", dtype=torch.bfloat16)
    x = torch.randn(rows, in_, device="cuda", dtype=torch.bfloat16)
    mg = torch.randn(out, in_, device="cuda", dtype=torch.float32)
    base = mg.clone()
    assert fp8_wgrad(g, x, mg)
    ref = base + g.float().t() @ x.float()
    delta = ref - base          # the wgrad contribution itself
    err = (mg - ref).abs().max().item()
    rel = err / (delta.abs().max().item() + 1e-6)
    assert rel < 8e-2, rel


@pytest.mark.gpu
def test_attn_bwd_into_matches_attn_bwd():
    """Strided-output backward (writes into views of a fused QKV-grad
    buffer) must produce bitwise the same dQ/dK/dV as the contiguous
    attn_bwd path."""
    ops = _ops()
    torch.manual_seed(13)
    sq, b, nh, d = 512, 2, 4, 128
    q = torch.randn(sq, b, nh, d, device="cuda", dtype=torch.bfloat16)
    k = torch.randn(sq, b, nh, d, device="cuda", dtype=torch.bfloat16)
    v = torch.randn(sq, b, nh, d, device="cuda", dtype=torch.bfloat16)
    scale = d ** -0.5
    o, lse = ops.attn_fwd(q, k, v, scale, True)
    do = torch.randn_like(o)
    dq, dk, dv = ops.attn_bwd(do, q, k, v, o, lse, scale, True)

    dm = torch.full((sq, b, nh, 3 * d), 7.0, device="cuda",
                    dtype=torch.bfloat16)
    dq_v, dk_v, dv_v = dm[..., :d], dm[..., d:2 * d], dm[..., 2 * d:]
    assert ops.attn_bwd_into(do, q, k, v, o, lse, scale, True,
                             dq_v, dk_v, dv_v)
    assert torch.equal(dq_v, dq)
    assert torch.equal(dk_v, dk)
    assert torch.equal(dv_v, dv)
    # odd seq falls back gracefully
    q2 = torch.randn(128, 2, 4, 128, device="cuda", dtype=torch.bfloat16)
    o2, lse2 = ops.attn_fwd(q2, q2, q2, scale, True)
    assert not ops.attn_bwd_into(
        torch.randn_like(o2), q2, q2, q2, o2, lse2, scale, True,
        torch.empty_like(q2), torch.empty_like(q2), torch.empty_like(q2))


@pytest.mark.gpu
def test_split_qkv_flash_zero_copy_grads():
    """End-to-end: _SplitQKV views -> flash attention -> backward; the
    fused-buffer zero-copy path must give the same mixed-QKV grad as the
    copy fallback."""
    from megatronapp_amd.core.transformer.attention import _SplitQKV
    from megatronapp_amd.core.transformer.dot_product_attention import (
        _FlashAttnFn)
    torch.manual_seed(14)
    sq, b, ng, hn = 256, 2, 4, 128
    mixed = torch.randn(sq, b, ng * 3 * hn, device="cuda",
                        dtype=torch.bfloat16, requires_grad=True)
    q, k, v = _SplitQKV.apply(mixed, ng, 1, hn)
    o = _FlashAttnFn.apply(q, k, v, hn ** -0.5, True, 0.0)
    do = torch.randn_like(o)
    o.backward(do)
    got = mixed.grad.clone()

    mixed2 = mixed.detach().clone().requires_grad_(True)
    q2, k2, v2 = _SplitQKV.apply(mixed2, ng, 1, hn)
    dq, dk, dv = _ops().attn_bwd(
        do, q2, k2, v2, *_ops().attn_fwd(q2, k2, v2, hn ** -0.5, True),
        hn ** -0.5, True)
    ref = torch.cat([dq, dk, dv], dim=-1).reshape(sq, b, -1)
    assert torch.equal(got, ref)
