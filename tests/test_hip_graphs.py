"""hipGraph decode capture (reference cuda_graphs.py equivalent)."""
import pytest
import torch

from tests.utils import initialize_model_parallel, destroy


def _tiny(dtype=torch.float32, max_seq=64):
    from megatronapp_amd.core.models.gpt import GPTModel
    from megatronapp_amd.core.models.gpt.gpt_layer_specs import (
        get_gpt_layer_local_spec)
    from megatronapp_amd.core.transformer_config import TransformerConfig
    from megatronapp_amd.core.tensor_parallel.random import (
        model_parallel_cuda_manual_seed)
    model_parallel_cuda_manual_seed(3)
    torch.manual_seed(3)
    cfg = TransformerConfig(
        num_layers=2, hidden_size=64, num_attention_heads=4,
        ffn_hidden_size=128, hidden_dropout=0.0, attention_dropout=0.0,
        params_dtype=dtype, bf16=(dtype == torch.bfloat16),
        masked_softmax_fusion=True, add_bias_linear=True)
    return GPTModel(config=cfg,
                    transformer_layer_spec=get_gpt_layer_local_spec(
                        use_flash=False),
                    vocab_size=256, max_sequence_length=max_seq,
                    pre_process=True, post_process=True)


def test_graph_context_masked_decode_matches_plain_cpu():
    """GraphDecodeContext semantics (fixed window + padding mask) must
    reproduce the plain growing-window KV cache path — checked on CPU
    where no graph is captured, exercising the same code."""
    from megatronapp_amd.core.hip_graphs import GraphDecodeContext
    from megatronapp_amd.core.inference_params import InferenceParams
    initialize_model_parallel()
    m = _tiny().eval()
    tok = torch.randint(0, 256, (1, 12))
    pos = torch.arange(12).unsqueeze(0)
    with torch.no_grad():
        plain = InferenceParams(1, 32)
        ref_pre = m(tok[:, :8], pos[:, :8], None, inference_context=plain)
        plain.sequence_len_offset = 8
        ref_steps = []
        for i in range(8, 12):
            ref_steps.append(m(tok[:, i:i + 1], pos[:, i:i + 1], None,
                               inference_context=plain))
            plain.sequence_len_offset += 1

        g = GraphDecodeContext.__new__(GraphDecodeContext)
        InferenceParams.__init__(g, 1, 32)
        g.device = torch.device("cpu")
        g.cur_len = torch.zeros(1, dtype=torch.long)
        g._arange = torch.arange(32)
        g.graph_mode = False
        got_pre = m(tok[:, :8], pos[:, :8], None, inference_context=g)
        g.sequence_len_offset = 8
        got_steps = []
        for i in range(8, 12):
            got_steps.append(m(tok[:, i:i + 1], pos[:, i:i + 1], None,
                               inference_context=g))
            g.sequence_len_offset += 1
    assert torch.allclose(got_pre, ref_pre, atol=1e-5)
    for a, b in zip(got_steps, ref_steps):
        assert torch.allclose(a, b, atol=1e-5)
    destroy()


@pytest.mark.gpu
def test_graphed_decode_matches_eager_gpu():
    """Captured hipGraph decode == eager decode, greedy tokens."""
    from megatronapp_amd.core.hip_graphs import (GraphDecodeContext,
                                                 GraphedDecodeStep)
    from megatronapp_amd.core.inference_params import InferenceParams
    initialize_model_parallel()
    with torch.device("cuda"):
        m = _tiny(torch.bfloat16).eval()
    tok = torch.randint(0, 256, (2, 8), device="cuda")
    pos = torch.arange(8, device="cuda").unsqueeze(0).expand(2, -1)
    n_new = 8
    with torch.no_grad():
        # eager reference
        plain = InferenceParams(2, 32)
        logits = m(tok, pos, None, inference_context=plain)
        plain.sequence_len_offset = 8
        cur = logits[:, -1].argmax(-1, keepdim=True)
        ref = [cur.clone()]
        for i in range(n_new - 1):
            p = torch.full((2, 1), 8 + i, device="cuda", dtype=torch.long)
            logits = m(cur, p, None, inference_context=plain)
            plain.sequence_len_offset += 1
            cur = logits[:, -1].argmax(-1, keepdim=True)
            ref.append(cur.clone())

        # graphed decode
        g = GraphDecodeContext(2, 32)
        logits = m(tok, pos, None, inference_context=g)
        g.sequence_len_offset = 8
        cur = logits[:, -1].argmax(-1, keepdim=True)
        got = [cur.clone()]
        step = GraphedDecodeStep(m, g, batch_size=2)
        for i in range(n_new - 1):
            p = torch.full((2, 1), 8 + i, device="cuda", dtype=torch.long)
            logits = step(cur, p)
            cur = logits[:, -1].argmax(-1, keepdim=True)
            got.append(cur.clone())
    for a, b in zip(got, ref):
        assert torch.equal(a, b), (a, b)
    destroy()


@pytest.mark.gpu
def test_graphed_decode_rope_model_gpu():
    """hipGraph decode with rotary embeddings: the rope slice is fetched
    by device-tensor index_select, so replays see the right positions."""
    from megatronapp_amd.core.hip_graphs import (GraphDecodeContext,
                                                 GraphedDecodeStep)
    from megatronapp_amd.core.inference_params import InferenceParams
    from megatronapp_amd.core.models.gpt import GPTModel
    from megatronapp_amd.core.models.gpt.gpt_layer_specs import (
        get_gpt_layer_local_spec)
    from megatronapp_amd.core.transformer_config import TransformerConfig
    from megatronapp_amd.core.tensor_parallel.random import (
        model_parallel_cuda_manual_seed)
    initialize_model_parallel()
    model_parallel_cuda_manual_seed(4)
    torch.manual_seed(4)
    cfg = TransformerConfig(
        num_layers=2, hidden_size=64, num_attention_heads=4,
        ffn_hidden_size=128, hidden_dropout=0.0, attention_dropout=0.0,
        params_dtype=torch.bfloat16, bf16=True, add_bias_linear=False,
        position_embedding_type="rope", normalization="RMSNorm",
        gated_linear_unit=True, activation_func="silu")
    with torch.device("cuda"):
        m = GPTModel(config=cfg,
                     transformer_layer_spec=get_gpt_layer_local_spec(
                         normalization="RMSNorm", use_flash=False),
                     vocab_size=256, max_sequence_length=64,
                     position_embedding_type="rope",
                     pre_process=True, post_process=True).eval()
    tok = torch.randint(0, 256, (2, 8), device="cuda")
    pos = torch.arange(8, device="cuda").unsqueeze(0).expand(2, -1)
    with torch.no_grad():
        plain = InferenceParams(2, 32)
        logits = m(tok, pos, None, inference_context=plain)
        plain.sequence_len_offset = 8
        cur = logits[:, -1].argmax(-1, keepdim=True)
        ref = [cur.clone()]
        for i in range(5):
            p = torch.full((2, 1), 8 + i, device="cuda", dtype=torch.long)
            logits = m(cur, p, None, inference_context=plain)
            plain.sequence_len_offset += 1
            cur = logits[:, -1].argmax(-1, keepdim=True)
            ref.append(cur.clone())

        g = GraphDecodeContext(2, 32)
        logits = m(tok, pos, None, inference_context=g)
        g.sequence_len_offset = 8
        cur = logits[:, -1].argmax(-1, keepdim=True)
        got = [cur.clone()]
        step = GraphedDecodeStep(m, g, batch_size=2)
        for i in range(5):
            p = torch.full((2, 1), 8 + i, device="cuda", dtype=torch.long)
            logits = step(cur, p)
            cur = logits[:, -1].argmax(-1, keepdim=True)
            got.append(cur.clone())
    for a, b in zip(got, ref):
        assert torch.equal(a, b), (got, ref)
    destroy()


@pytest.mark.gpu
def test_bucketed_graph_decode_crosses_buckets():
    """Lengths-bucketed graphs: decoding across a bucket boundary (256 ->
    512 window) matches eager decode token for token."""
    from megatronapp_amd.core.hip_graphs import (BucketedGraphedDecodeStep,
                                                 GraphDecodeContext)
    from megatronapp_amd.core.inference_params import InferenceParams
    initialize_model_parallel()
    with torch.device("cuda"):
        m = _tiny(torch.bfloat16, max_seq=1024).eval()
    b, plen, n_new = 2, 250, 16    # crosses 256 at step 6
    tok = torch.randint(0, 256, (b, plen), device="cuda")
    pos = torch.arange(plen, device="cuda").unsqueeze(0).expand(b, -1)

    def run(graphed):
        torch.manual_seed(0)
        ctx = GraphDecodeContext(b, 1024) if graphed else \
            InferenceParams(b, 1024)
        with torch.no_grad():
            logits = m(tok, pos, None, inference_context=ctx)
            ctx.increment_sequence_len_offset(plen)
            out = [logits[:, -1].argmax(-1)]
            step = BucketedGraphedDecodeStep(m, ctx, b) if graphed else None
            for i in range(n_new - 1):
                t = out[-1].unsqueeze(1)
                p = torch.full((b, 1), plen + i, device="cuda",
                               dtype=torch.long)
                if graphed:
                    lg = step(t, p)
                else:
                    lg = m(t, p, None, inference_context=ctx)
                    ctx.increment_sequence_len_offset(1)
                out.append(lg[:, -1].argmax(-1))
        return torch.stack(out, 1)

    eager = run(False)
    graphed = run(True)
    assert torch.equal(eager, graphed), (eager, graphed)
