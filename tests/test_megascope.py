"""MegaScope tests: tensor tracer taps, disturbance, WS protocol, inference
engine (all CPU)."""

import json
import threading
import time

import pytest
import torch

from .utils import destroy, initialize_model_parallel


def _tiny_model(vocab=64, seq=16):
    from megatronapp_amd.core.models.gpt import GPTModel
    from megatronapp_amd.core.models.gpt.gpt_layer_specs import (
        get_gpt_layer_local_spec)
    from megatronapp_amd.core.transformer_config import TransformerConfig
    config = TransformerConfig(
        num_layers=2, hidden_size=32, num_attention_heads=4,
        hidden_dropout=0.0, attention_dropout=0.0,
        position_embedding_type="rope", normalization="RMSNorm",
        activation_func="swiglu", add_bias_linear=False)
    return GPTModel(config=config,
                    transformer_layer_spec=get_gpt_layer_local_spec(
                        normalization="RMSNorm", use_flash=False),
                    vocab_size=vocab, max_sequence_length=seq + 64,
                    position_embedding_type="rope")


def test_tensor_tracer_taps_fire():
    initialize_model_parallel()
    from megatronapp_amd.core.tensor_tracer import (
        FlagType, enable_tensor_tracers)
    tt = enable_tensor_tracers()
    tt.set_num_layers(2)
    reports = []
    tt.set_report(reports.append)
    tt.tt_flags.set_by_configs({"QKV_mat_mul": "True", "MLP_1": "True",
                                "MLP_2": "True",
                                "Raw_attention_score": "True"})
    tt.set_compressor_configs({"QKV": {"pixels": 8, "method": "mean"},
                               "MLP": {"pixels": 4, "method": "max"}})
    model = _tiny_model()
    tokens = torch.randint(0, 64, (2, 16))
    pos = torch.arange(16).unsqueeze(0).expand(2, -1)
    model(tokens, pos)
    types = {r["update_type"] for r in reports}
    assert FlagType.QKV.value in types
    assert FlagType.MLP1.value in types
    assert FlagType.MLP2.value in types
    assert FlagType.RawAttentionScore.value in types
    for r in reports:
        assert r["type"] == "update"
        assert "layer_id" in r and "result" in r and "args" in r
    tt.set_report(None)
    destroy()


def test_disturbance_noise():
    from megatronapp_amd.core.tensor_disturbance import get_disturbance
    d = get_disturbance()
    d.set_by_configs({"system_perturbation": "True",
                      "system_perturbation_fn": "noise2",
                      "system_perturbation_coef": 0.1})
    x = torch.ones(1000)
    y = d.perturb_system(x)
    assert not torch.equal(x, y)
    assert (y - x).abs().max() <= 0.1 + 1e-6
    d.set_by_configs({})  # reset
    assert not d.any_enabled


def test_ws_server_roundtrip():
    from megatronapp_amd.utils.ws import WebSocketServer, ws_connect

    received = []

    def handler(conn):
        conn.send({"type": "start", "hello": 1})
        while conn.open:
            msg = conn.recv_message()
            if msg is None:
                break
            received.append(json.loads(msg))
            conn.send({"type": "echo", "payload": json.loads(msg)})

    server = WebSocketServer(host="127.0.0.1", port=0)
    # pick a free port
    import socket as s
    sock = s.socket()
    sock.bind(("127.0.0.1", 0))
    port = sock.getsockname()[1]
    sock.close()
    server.port = port
    server.start_in_thread(handler)
    time.sleep(0.2)

    conn = ws_connect("127.0.0.1", port)
    first = json.loads(conn.recv_message())
    assert first == {"type": "start", "hello": 1}
    conn.send({"type": "ping", "n": 42})
    echo = json.loads(conn.recv_message())
    assert echo["payload"]["n"] == 42
    conn.close()
    server.stop()


def test_inference_engine_generates():
    initialize_model_parallel()
    from megatronapp_amd.core.inference.static_engine import (
        get_inference_engine, run_mcore_engine)
    from megatronapp_amd.training.tokenizer import NullTokenizer

    torch.manual_seed(0)
    model = _tiny_model()
    model.eval()
    tok = NullTokenizer(64)
    engine = get_inference_engine(model, tok, max_batch_size=4)
    out = run_mcore_engine(engine, ["1 2 3", "4 5"], tokens_to_generate=8,
                           top_k=1, logprobs=True)
    assert len(out["text"]) == 2
    # greedy decoding is deterministic
    out2 = run_mcore_engine(engine, ["1 2 3", "4 5"], tokens_to_generate=8,
                            top_k=1)
    assert out["text"] == out2["text"]
    assert len(out["logprobs"][0]) > 0
    # generated tokens are in-vocab ints
    for seg in out["segments"][0]:
        assert 0 <= int(seg) < 64
    destroy()


def test_kv_cache_matches_full_forward():
    """Decode with KV cache must equal full-context forward logits."""
    initialize_model_parallel()
    from megatronapp_amd.core.inference_params import InferenceParams
    torch.manual_seed(1)
    model = _tiny_model()
    model.eval()
    tokens = torch.randint(0, 64, (1, 8))
    pos = torch.arange(8).unsqueeze(0)
    with torch.no_grad():
        full = model(tokens, pos)

        params = InferenceParams(1, 32)
        out1 = model(tokens[:, :4], pos[:, :4], inference_context=params)
        params.increment_sequence_len_offset(4)
        outs = [out1]
        for i in range(4, 8):
            o = model(tokens[:, i:i + 1], pos[:, i:i + 1],
                      inference_context=params)
            params.increment_sequence_len_offset(1)
            outs.append(o)
    cached = torch.cat(outs, dim=1)
    assert torch.allclose(full, cached, atol=1e-4), (full - cached).abs().max()
    destroy()


@pytest.mark.gpu
def test_inference_engine_generates_gpu():
    """Serving path on MI355X: greedy decode through the KV cache on
    cuda:0, deterministic across runs."""
    initialize_model_parallel()
    from megatronapp_amd.core.inference.static_engine import (
        get_inference_engine, run_mcore_engine)
    from megatronapp_amd.training.tokenizer import NullTokenizer

    torch.manual_seed(0)
    model = _tiny_model().cuda().eval()
    tok = NullTokenizer(64)
    engine = get_inference_engine(model, tok, max_batch_size=4)
    out = run_mcore_engine(engine, ["1 2 3", "4 5"], tokens_to_generate=16,
                           top_k=1, logprobs=True)
    out2 = run_mcore_engine(engine, ["1 2 3", "4 5"], tokens_to_generate=16,
                            top_k=1)
    assert out["text"] == out2["text"]
    assert len(out["logprobs"][0]) > 0
    destroy()


def test_server_token_budget_guard():
    """--max-tokens-to-oom rejects oversized generation requests."""
    initialize_model_parallel()
    from megatronapp_amd.core.inference.static_engine import (
        get_inference_engine)
    from megatronapp_amd.inference.text_generation_server import (
        InferenceGenerate)
    from megatronapp_amd.training.tokenizer import NullTokenizer
    from megatronapp_amd.training import global_vars
    import argparse
    args = argparse.Namespace(max_tokens_to_oom=32,
                              inference_max_seq_length=None, rank=0)
    global_vars._GLOBAL_ARGS = args
    torch.manual_seed(0)
    engine = get_inference_engine(_tiny_model().eval(), NullTokenizer(64), 4)
    srv = InferenceGenerate(engine, num_layers=2)
    out = srv.run({"prompts": ["1 2 3"], "tokens_to_generate": 64}, None)
    assert out and out.get("type") == "error"
    global_vars._GLOBAL_ARGS = None
    destroy()
