"""HF Mixtral import + export (reference tools/checkpoint
loader_mixtral_hf.py).

Maps a consolidated Mixtral-style MoE checkpoint (RMSNorm, RoPE, GQA,
top-k softmax router with renormalized probs, gated-SiLU experts) onto
``transformers.MixtralForCausalLM`` naming:

* router: ``mlp.router.weight`` <-> ``mlp.gate.weight`` (transformers
  >= 5 fused naming; the loader also accepts the 4.x per-expert
  ``block_sparse_moe.experts.N.w1/w2/w3`` layout of released
  checkpoints)
* experts (GroupedMLP stacked weights): ``weight1[e]`` is [h, 2f] with
  [gate | up] column halves <-> HF fused ``experts.gate_up_proj``
  ([E, 2f, h]); ``weight2[e]`` [f, h] <-> ``experts.down_proj``
  ([E, h, f])
* attention/QKV mapping is shared with the llama converter

Verified by logit parity against a transformers forward with
``moe_router_renormalize=True`` (Mixtral normalizes its top-k probs).
"""

from __future__ import annotations

import json
import os

import torch

from saver_hf_llama import _qkv_from_hf, _qkv_to_hf


def save_hf_mixtral(full: dict, common: dict, path: str) -> None:
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
          "model.norm.weight": take("decoder.final_layernorm.weight"),
          "lm_head.weight": next(
              (full[pref + "output_layer.weight"].float()
               for pref in ("model.", "model0.")
               if pref + "output_layer.weight" in full), wte)}
    n_exp = ffn = None
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
        sd[o + "mlp.gate.weight"] = take(p + "mlp.router.weight")
        w1 = take(p + "mlp.experts.weight1")   # [E, h, 2f]
        w2 = take(p + "mlp.experts.weight2")   # [E, f, h]
        n_exp, _, f2 = w1.shape
        ffn = f2 // 2
        sd[o + "mlp.experts.gate_up_proj"] = \
            w1.transpose(1, 2).contiguous()     # [E, 2f, h]
        sd[o + "mlp.experts.down_proj"] = \
            w2.transpose(1, 2).contiguous()     # [E, h, f]

    torch.save(sd, os.path.join(path, "pytorch_model.bin"))
    cfg = {"architectures": ["MixtralForCausalLM"],
           "model_type": "mixtral", "vocab_size": vocab,
           "hidden_size": h, "intermediate_size": ffn,
           "num_hidden_layers": len(layers),
           "num_attention_heads": nh, "num_key_value_heads": ng,
           "num_local_experts": n_exp,
           "num_experts_per_tok": args.get("moe_router_topk", 2),
           "max_position_embeddings": args.get(
               "max_position_embeddings", 4096),
           "rope_theta": args.get("rotary_base", 10000.0),
           "rms_norm_eps": args.get("norm_epsilon", 1e-5),
           "tie_word_embeddings": False}
    with open(os.path.join(path, "config.json"), "w") as f:
        json.dump(cfg, f, indent=2)
    print(f"wrote HF mixtral export: {len(layers)} layers, "
          f"{n_exp} experts -> {path}")


def load_hf_mixtral(path):
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
            "model.decoder.final_layernorm.weight": sd["norm.weight"],
            "model.output_layer.weight": sd["lm_head.weight"]}
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
        gate_key = (s_ + "mlp.gate.weight"
                    if s_ + "mlp.gate.weight" in sd
                    else s_ + "block_sparse_moe.gate.weight")
        full[d + "mlp.router.weight"] = sd[gate_key]
        if s_ + "mlp.experts.gate_up_proj" in sd:       # transformers 5.x
            full[d + "mlp.experts.weight1"] = \
                sd[s_ + "mlp.experts.gate_up_proj"] \
                .transpose(1, 2).contiguous()
            full[d + "mlp.experts.weight2"] = \
                sd[s_ + "mlp.experts.down_proj"] \
                .transpose(1, 2).contiguous()
        else:                                           # 4.x per-expert
            w1s, w2s = [], []
            e = 0
            while s_ + f"block_sparse_moe.experts.{e}.w1.weight" in sd:
                eo = s_ + f"block_sparse_moe.experts.{e}."
                gate = sd[eo + "w1.weight"].t()      # [h, f]
                up = sd[eo + "w3.weight"].t()        # [h, f]
                w1s.append(torch.cat([gate, up], dim=1))
                w2s.append(sd[eo + "w2.weight"].t())  # [f, h]
                e += 1
            full[d + "mlp.experts.weight1"] = \
                torch.stack(w1s).contiguous()
            full[d + "mlp.experts.weight2"] = \
                torch.stack(w2s).contiguous()
    return full, {"hf_config": cfg}
