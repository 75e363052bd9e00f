"""HF GPT-2 export (reference tools/checkpoint saver plugins export to HF
formats).  Maps a consolidated GPT checkpoint (learned positions, GeLU,
LayerNorm, tied embeddings — the gpt3-* configs here) onto
``transformers.GPT2LMHeadModel`` naming:

* fused QKV rows are reordered from per-head [q_h | k_h | v_h] blocks to
  HF's [Q_all | K_all | V_all] and transposed (HF Conv1D keeps [in, out])
* linear weights transpose for Conv1D; norms map 1:1

Verified by logit equality against a transformers forward (tests).
"""

from __future__ import annotations

import json
import os

import torch


def _qkv_to_hf(w, nh, hn, h):
    """[ (3*hn)*nh rows grouped per head, h ] -> [h, 3h] Conv1D weight."""
    w = w.view(nh, 3, hn, -1)                      # per head: q,k,v blocks
    q = w[:, 0].reshape(nh * hn, -1)
    k = w[:, 1].reshape(nh * hn, -1)
    v = w[:, 2].reshape(nh * hn, -1)
    return torch.cat([q, k, v], dim=0).t().contiguous()


def _qkv_bias_to_hf(b, nh, hn):
    b = b.view(nh, 3, hn)
    return torch.cat([b[:, 0].reshape(-1), b[:, 1].reshape(-1),
                      b[:, 2].reshape(-1)], dim=0).contiguous()


def save_hf_gpt2(full: dict, common: dict, path: str) -> None:
    os.makedirs(path, exist_ok=True)

    def take(name):
        for prefix in ("model.", "model0."):
            if prefix + name in full:
                return full[prefix + name].float()
        raise KeyError(name)

    wte = take("embedding.word_embeddings.weight")
    wpe = take("embedding.position_embeddings.weight")
    vocab, h = wte.shape
    n_positions = wpe.shape[0]
    layers = sorted({int(k.split("decoder.layers.")[1].split(".")[0])
                     for k in full if "decoder.layers." in k})
    n_layer = len(layers)
    qkv0 = take("decoder.layers.0.self_attention.linear_qkv.weight")
    nh3hn = qkv0.shape[0]
    # infer heads from args if present, else assume hn = 64
    args = common.get("args", {}) if isinstance(common, dict) else {}
    nh = args.get("num_attention_heads") or h // 64
    hn = nh3hn // (3 * nh)

    sd = {"transformer.wte.weight": wte,
          "transformer.wpe.weight": wpe,
          "transformer.ln_f.weight": take("decoder.final_layernorm.weight"),
          "transformer.ln_f.bias": take("decoder.final_layernorm.bias"),
          "lm_head.weight": wte}
    for i in layers:
        p = f"decoder.layers.{i}."
        o = f"transformer.h.{i}."
        sd[o + "ln_1.weight"] = take(p + "input_layernorm.weight")
        sd[o + "ln_1.bias"] = take(p + "input_layernorm.bias")
        sd[o + "attn.c_attn.weight"] = _qkv_to_hf(
            take(p + "self_attention.linear_qkv.weight"), nh, hn, h)
        sd[o + "attn.c_attn.bias"] = _qkv_bias_to_hf(
            take(p + "self_attention.linear_qkv.bias"), nh, hn)
        sd[o + "attn.c_proj.weight"] = take(
            p + "self_attention.linear_proj.weight").t().contiguous()
        sd[o + "attn.c_proj.bias"] = take(
            p + "self_attention.linear_proj.bias")
        sd[o + "ln_2.weight"] = take(p + "pre_mlp_layernorm.weight")
        sd[o + "ln_2.bias"] = take(p + "pre_mlp_layernorm.bias")
        sd[o + "mlp.c_fc.weight"] = take(
            p + "mlp.linear_fc1.weight").t().contiguous()
        sd[o + "mlp.c_fc.bias"] = take(p + "mlp.linear_fc1.bias")
        sd[o + "mlp.c_proj.weight"] = take(
            p + "mlp.linear_fc2.weight").t().contiguous()
        sd[o + "mlp.c_proj.bias"] = take(p + "mlp.linear_fc2.bias")

    torch.save(sd, os.path.join(path, "pytorch_model.bin"))
    cfg = {"architectures": ["GPT2LMHeadModel"], "model_type": "gpt2",
           "vocab_size": vocab, "n_positions": n_positions, "n_embd": h,
           "n_layer": n_layer, "n_head": nh,
           "n_inner": take("decoder.layers.0.mlp.linear_fc1.weight").shape[0],
           "activation_function": "gelu_new",
           "layer_norm_epsilon": 1e-5,
           "resid_pdrop": 0.0, "embd_pdrop": 0.0, "attn_pdrop": 0.0}
    with open(os.path.join(path, "config.json"), "w") as f:
        json.dump(cfg, f, indent=2)
    print(f"wrote HF GPT-2 export: {n_layer} layers, {nh} heads, "
          f"vocab {vocab} -> {path}")


def _qkv_from_hf(w, nh, hn):
    """HF c_attn [in, 3h] -> our fused rows [(3hn)*nh, in] grouped/head."""
    w = w.t().contiguous()                         # [3h, in]
    h3 = w.shape[0]
    q, k, v = w.split(h3 // 3, dim=0)
    q = q.view(nh, hn, -1)
    k = k.view(nh, hn, -1)
    v = v.view(nh, hn, -1)
    return torch.stack([q, k, v], dim=1).reshape(3 * nh * hn, -1).contiguous()


def _qkv_bias_from_hf(b, nh, hn):
    q, k, v = b.split(b.shape[0] // 3, dim=0)
    q = q.view(nh, hn)
    k = k.view(nh, hn)
    v = v.view(nh, hn)
    return torch.stack([q, k, v], dim=1).reshape(-1).contiguous()


def load_hf_gpt2(path):
    """transformers GPT-2 directory -> full tensors in this framework's
    naming (inverse of save_hf_gpt2)."""
    cfg = json.load(open(os.path.join(path, "config.json")))
    nh = cfg["n_head"]
    hn = cfg["n_embd"] // nh
    bin_path = os.path.join(path, "pytorch_model.bin")
    if os.path.exists(bin_path):
        sd = torch.load(bin_path, map_location="cpu", weights_only=True)
    else:
        from safetensors.torch import load_file
        sd = load_file(os.path.join(path, "model.safetensors"))
    sd = {k.replace("transformer.", ""): v for k, v in sd.items()}

    full = {"model.embedding.word_embeddings.weight": sd["wte.weight"],
            "model.embedding.position_embeddings.weight": sd["wpe.weight"],
            "model.decoder.final_layernorm.weight": sd["ln_f.weight"],
            "model.decoder.final_layernorm.bias": sd["ln_f.bias"]}
    for i in range(cfg["n_layer"]):
        s_ = f"h.{i}."
        d = f"model.decoder.layers.{i}."
        full[d + "input_layernorm.weight"] = sd[s_ + "ln_1.weight"]
        full[d + "input_layernorm.bias"] = sd[s_ + "ln_1.bias"]
        full[d + "self_attention.linear_qkv.weight"] = _qkv_from_hf(
            sd[s_ + "attn.c_attn.weight"], nh, hn)
        full[d + "self_attention.linear_qkv.bias"] = _qkv_bias_from_hf(
            sd[s_ + "attn.c_attn.bias"], nh, hn)
        full[d + "self_attention.linear_proj.weight"] =             sd[s_ + "attn.c_proj.weight"].t().contiguous()
        full[d + "self_attention.linear_proj.bias"] =             sd[s_ + "attn.c_proj.bias"]
        full[d + "pre_mlp_layernorm.weight"] = sd[s_ + "ln_2.weight"]
        full[d + "pre_mlp_layernorm.bias"] = sd[s_ + "ln_2.bias"]
        full[d + "mlp.linear_fc1.weight"] =             sd[s_ + "mlp.c_fc.weight"].t().contiguous()
        full[d + "mlp.linear_fc1.bias"] = sd[s_ + "mlp.c_fc.bias"]
        full[d + "mlp.linear_fc2.weight"] =             sd[s_ + "mlp.c_proj.weight"].t().contiguous()
        full[d + "mlp.linear_fc2.bias"] = sd[s_ + "mlp.c_proj.bias"]
    common = {"hf_config": cfg}
    return full, common
