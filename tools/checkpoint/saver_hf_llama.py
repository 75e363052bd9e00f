"""HF Llama/Mistral import + export (reference tools/checkpoint
loader_llama_mistral.py / HF export paths).

Maps a consolidated llama-family checkpoint (RMSNorm, RoPE, gated-SiLU
MLP, untied embeddings, no linear biases, optional GQA) onto
``transformers.LlamaForCausalLM`` naming:

* fused QKV is stored per query-group as [q_0..q_{rep-1} | k | v] rows;
  HF keeps separate q/k/v projections with query heads in global order,
  which equals group-major order, so the mapping is a pure reshape —
  both sides use the non-interleaved rotate-half RoPE, so no head-dim
  permutation is needed (verified by logit-parity tests).
* fc1 is [gate | up] halves -> HF gate_proj / up_proj.

Verified by logit equality against a transformers forward (tests).
"""

from __future__ import annotations

import json
import os

import torch


def _qkv_to_hf(w, ng, rep, hn):
    """[(rep+2)*ng*hn, h] fused rows -> (q [np*hn,h], k, v [ng*hn,h])."""
    h = w.shape[1]
    w = w.view(ng, (rep + 2) * hn, h)
    q = w[:, :rep * hn].reshape(ng * rep * hn, h)
    k = w[:, rep * hn:(rep + 1) * hn].reshape(ng * hn, h)
    v = w[:, (rep + 1) * hn:].reshape(ng * hn, h)
    return q.contiguous(), k.contiguous(), v.contiguous()


def _qkv_from_hf(q, k, v, ng, rep, hn):
    h = q.shape[1]
    q = q.view(ng, rep * hn, h)
    k = k.view(ng, hn, h)
    v = v.view(ng, hn, h)
    return torch.cat([q, k, v], dim=1).reshape(-1, h).contiguous()


def save_hf_llama(full: dict, common: dict, path: str) -> None:
    os.makedirs(path, exist_ok=True)

    def take(name):
        for prefix in ("model.", "model0."):
            if prefix + name in full:
                return full[prefix + name].float()
        raise KeyError(name)

    wte = take("embedding.word_embeddings.weight")
    vocab, h = wte.shape
    layers = sorted({int(k.split("decoder.layers.")[1].split(".")[0])
                     for k in full if "decoder.layers." in k})
    args = common.get("args", {}) if isinstance(common, dict) else {}
    nh = args.get("num_attention_heads") or h // 128
    ng = args.get("num_query_groups") or nh
    hn = h // nh
    rep = nh // ng

    sd = {"model.embed_tokens.weight": wte,
          "model.norm.weight": take("decoder.final_layernorm.weight")}
    def take_opt(name):
        for prefix in ("model.", "model0."):
            if prefix + name in full:
                return full[prefix + name].float()
        return None

    out_w = take_opt("output_layer.weight")
    tied = out_w is None
    sd["lm_head.weight"] = wte if tied else out_w
    ffn = None
    for i in layers:
        p = f"decoder.layers.{i}."
        o = f"model.layers.{i}."
        q, k, v = _qkv_to_hf(
            take(p + "self_attention.linear_qkv.weight"), ng, rep, hn)
        sd[o + "self_attn.q_proj.weight"] = q
        sd[o + "self_attn.k_proj.weight"] = k
        sd[o + "self_attn.v_proj.weight"] = v
        sd[o + "self_attn.o_proj.weight"] = take(
            p + "self_attention.linear_proj.weight")
        sd[o + "input_layernorm.weight"] = take(
            p + "input_layernorm.weight")
        sd[o + "post_attention_layernorm.weight"] = take(
            p + "pre_mlp_layernorm.weight")
        fc1 = take(p + "mlp.linear_fc1.weight")
        ffn = fc1.shape[0] // 2
        sd[o + "mlp.gate_proj.weight"] = fc1[:ffn].contiguous()
        sd[o + "mlp.up_proj.weight"] = fc1[ffn:].contiguous()
        sd[o + "mlp.down_proj.weight"] = take(
            p + "mlp.linear_fc2.weight")

    torch.save(sd, os.path.join(path, "pytorch_model.bin"))
    cfg = {"architectures": ["LlamaForCausalLM"], "model_type": "llama",
           "vocab_size": vocab, "hidden_size": h,
           "intermediate_size": ffn, "num_hidden_layers": len(layers),
           "num_attention_heads": nh, "num_key_value_heads": ng,
           "max_position_embeddings": args.get(
               "max_position_embeddings", 4096),
           "rope_theta": args.get("rotary_base", 10000.0),
           "rms_norm_eps": args.get("norm_epsilon", 1e-5),
           "hidden_act": "silu",
           "tie_word_embeddings": tied,
           "attention_bias": False, "mlp_bias": False}
    with open(os.path.join(path, "config.json"), "w") as f:
        json.dump(cfg, f, indent=2)
    print(f"wrote HF llama export: {len(layers)} layers, {nh} heads "
          f"({ng} kv groups), vocab {vocab} -> {path}")


def load_hf_llama(path):
    """transformers Llama directory -> full tensors in this framework's
    naming (inverse of save_hf_llama)."""
    cfg = json.load(open(os.path.join(path, "config.json")))
    nh = cfg["num_attention_heads"]
    ng = cfg.get("num_key_value_heads", nh)
    hn = cfg["hidden_size"] // nh
    rep = nh // ng
    bin_path = os.path.join(path, "pytorch_model.bin")
    if os.path.exists(bin_path):
        sd = torch.load(bin_path, map_location="cpu", weights_only=True)
    else:
        from safetensors.torch import load_file
        sd = {}
        for fn in sorted(os.listdir(path)):
            if fn.endswith(".safetensors"):
                sd.update(load_file(os.path.join(path, fn)))
    sd = {k.replace("model.", "", 1) if k.startswith("model.") else k: v
          for k, v in sd.items()}

    full = {"model.embedding.word_embeddings.weight":
            sd["embed_tokens.weight"],
            "model.decoder.final_layernorm.weight": sd["norm.weight"]}
    if not cfg.get("tie_word_embeddings", False) and \
            "lm_head.weight" in sd:
        full["model.output_layer.weight"] = sd["lm_head.weight"]
    for i in range(cfg["num_hidden_layers"]):
        s_ = f"layers.{i}."
        d = f"model.decoder.layers.{i}."
        full[d + "self_attention.linear_qkv.weight"] = _qkv_from_hf(
            sd[s_ + "self_attn.q_proj.weight"],
            sd[s_ + "self_attn.k_proj.weight"],
            sd[s_ + "self_attn.v_proj.weight"], ng, rep, hn)
        full[d + "self_attention.linear_proj.weight"] = \
            sd[s_ + "self_attn.o_proj.weight"]
        full[d + "input_layernorm.weight"] = \
            sd[s_ + "input_layernorm.weight"]
        full[d + "pre_mlp_layernorm.weight"] = \
            sd[s_ + "post_attention_layernorm.weight"]
        full[d + "mlp.linear_fc1.weight"] = torch.cat(
            [sd[s_ + "mlp.gate_proj.weight"],
             sd[s_ + "mlp.up_proj.weight"]], dim=0).contiguous()
        full[d + "mlp.linear_fc2.weight"] = \
            sd[s_ + "mlp.down_proj.weight"]
    return full, {"hf_config": cfg}
