"""GGUF export for llama-family checkpoints.

The reference ships a TensorRT-LLM export path (core/export/trtllm) that
is NVIDIA-only; the MI355X-native serving story is this GGUF writer (for
llama.cpp-compatible runtimes) next to the HF exporters.  Implements
GGUF v3 from the spec: header, aligned KV metadata, tensor infos, then
32-byte-aligned tensor data.  Tensors are written F32 or F16.

Usage (via convert.py):
  python tools/checkpoint/convert.py --load ckpt/iter_N \
      --loader torch_dist --save model.gguf --saver gguf
"""

from __future__ import annotations

import struct

import torch

GGUF_MAGIC = 0x46554747      # "GGUF" little-endian
GGUF_VERSION = 3
ALIGNMENT = 32

# metadata value types
_U32, _F32, _BOOL, _STRING, _ARRAY, _U64 = 4, 6, 7, 8, 9, 10
# tensor dtypes
_GGML_F32, _GGML_F16 = 0, 1


def _ws(f, s: str):
    b = s.encode("utf-8")
    f.write(struct.pack("<Q", len(b)))
    f.write(b)


def _kv_str(f, key, val):
    _ws(f, key)
    f.write(struct.pack("<I", _STRING))
    _ws(f, val)


def _kv_u32(f, key, val):
    _ws(f, key)
    f.write(struct.pack("<I", _U32))
    f.write(struct.pack("<I", val))


def _kv_f32(f, key, val):
    _ws(f, key)
    f.write(struct.pack("<I", _F32))
    f.write(struct.pack("<f", val))


def _hf_name_to_gguf(name: str) -> str:
    """HF llama names -> GGUF tensor names (llama.cpp convention)."""
    rep = [("model.embed_tokens.weight", "token_embd.weight"),
           ("model.norm.weight", "output_norm.weight"),
           ("lm_head.weight", "output.weight")]
    for a, b in rep:
        if name == a:
            return b
    if name.startswith("model.layers."):
        rest = name[len("model.layers."):]
        li, sub = rest.split(".", 1)
        sub = {"self_attn.q_proj.weight": "attn_q.weight",
               "self_attn.k_proj.weight": "attn_k.weight",
               "self_attn.v_proj.weight": "attn_v.weight",
               "self_attn.o_proj.weight": "attn_output.weight",
               "mlp.gate_proj.weight": "ffn_gate.weight",
               "mlp.up_proj.weight": "ffn_up.weight",
               "mlp.down_proj.weight": "ffn_down.weight",
               "input_layernorm.weight": "attn_norm.weight",
               "post_attention_layernorm.weight": "ffn_norm.weight",
               }.get(sub, sub)
        return f"blk.{li}.{sub}"
    return name


def save_gguf(full: dict, common: dict, path: str,
              dtype: str = "f16") -> None:
    """full: megatron-name -> tensor dict (as produced by the loaders);
    converts through the HF llama mapping then writes GGUF."""
    from saver_hf_llama import _qkv_to_hf  # reuse the proven mapping

    # ---- megatron -> HF names (mirror of saver_hf_llama.save_hf_llama)
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
    rep_ = nh // ng

    sd = {"model.embed_tokens.weight": wte,
          "model.norm.weight": take("decoder.final_layernorm.weight")}
    out_w = None
    for prefix in ("model.", "model0."):
        if prefix + "output_layer.weight" in full:
            out_w = full[prefix + "output_layer.weight"].float()
    sd["lm_head.weight"] = wte if out_w is None else out_w
    ffn = None
    for i in layers:
        p = f"decoder.layers.{i}."
        o = f"model.layers.{i}."
        q, k, v = _qkv_to_hf(
            take(p + "self_attention.linear_qkv.weight"), ng, rep_, hn)
        sd[o + "self_attn.q_proj.weight"] = q
        sd[o + "self_attn.k_proj.weight"] = k
        sd[o + "self_attn.v_proj.weight"] = v
        sd[o + "self_attn.o_proj.weight"] = take(
            p + "self_attention.linear_proj.weight")
        sd[o + "input_layernorm.weight"] = take(p + "input_layernorm.weight")
        sd[o + "post_attention_layernorm.weight"] = take(
            p + "pre_mlp_layernorm.weight")
        fc1 = take(p + "mlp.linear_fc1.weight")
        ffn = fc1.shape[0] // 2
        sd[o + "mlp.gate_proj.weight"] = fc1[:ffn].contiguous()
        sd[o + "mlp.up_proj.weight"] = fc1[ffn:].contiguous()
        sd[o + "mlp.down_proj.weight"] = take(p + "mlp.linear_fc2.weight")

    tensors = {_hf_name_to_gguf(k): t for k, t in sd.items()}
    ggml_t = _GGML_F16 if dtype == "f16" else _GGML_F32
    torch_t = torch.float16 if dtype == "f16" else torch.float32
    esize = 2 if dtype == "f16" else 4

    kvs = [
        ("general.architecture", "str", "llama"),
        ("general.name", "str", "megatronapp-amd export"),
        ("llama.context_length", "u32",
         int(args.get("max_position_embeddings", 4096))),
        ("llama.embedding_length", "u32", int(h)),
        ("llama.block_count", "u32", len(layers)),
        ("llama.feed_forward_length", "u32", int(ffn or 4 * h)),
        ("llama.attention.head_count", "u32", int(nh)),
        ("llama.attention.head_count_kv", "u32", int(ng)),
        ("llama.attention.layer_norm_rms_epsilon", "f32",
         float(args.get("norm_epsilon", 1e-5))),
        ("llama.rope.freq_base", "f32",
         float(args.get("rotary_base", 10000.0))),
        ("llama.vocab_size", "u32", int(vocab)),
    ]

    with open(path, "wb") as f:
        f.write(struct.pack("<IIQQ", GGUF_MAGIC, GGUF_VERSION,
                            len(tensors), len(kvs)))
        for key, kind, val in kvs:
            if kind == "str":
                _kv_str(f, key, val)
            elif kind == "u32":
                _kv_u32(f, key, val)
            else:
                _kv_f32(f, key, val)

        # tensor infos; offsets are relative to the aligned data start
        offset = 0
        order = list(tensors.items())
        for name, t in order:
            _ws(f, name)
            dims = list(reversed(t.shape))   # GGUF stores ne[0]=innermost
            f.write(struct.pack("<I", len(dims)))
            for d in dims:
                f.write(struct.pack("<Q", d))
            f.write(struct.pack("<I", ggml_t))
            f.write(struct.pack("<Q", offset))
            nbytes = t.numel() * esize
            offset += (nbytes + ALIGNMENT - 1) // ALIGNMENT * ALIGNMENT
        # pad to data-start alignment
        pos = f.tell()
        pad = (-pos) % ALIGNMENT
        f.write(b"\x00" * pad)
        for name, t in order:
            data = t.to(torch_t).contiguous().numpy().tobytes()
            f.write(data)
            f.write(b"\x00" * ((-len(data)) % ALIGNMENT))
    print(f"wrote GGUF ({dtype}): {len(tensors)} tensors, "
          f"{len(layers)} layers -> {path}")


def read_gguf(path: str):
    """Minimal GGUF reader (round-trip verification)."""
    import numpy as np
    with open(path, "rb") as f:
        magic, version, n_tensors, n_kv = struct.unpack(
            "<IIQQ", f.read(24))
        assert magic == GGUF_MAGIC and version == GGUF_VERSION

        def rs():
            (n,) = struct.unpack("<Q", f.read(8))
            return f.read(n).decode()

        meta = {}
        for _ in range(n_kv):
            key = rs()
            (vt,) = struct.unpack("<I", f.read(4))
            if vt == _STRING:
                meta[key] = rs()
            elif vt == _U32:
                meta[key] = struct.unpack("<I", f.read(4))[0]
            elif vt == _F32:
                meta[key] = struct.unpack("<f", f.read(4))[0]
            else:
                raise ValueError(f"unhandled kv type {vt}")
        infos = []
        for _ in range(n_tensors):
            name = rs()
            (nd,) = struct.unpack("<I", f.read(4))
            dims = struct.unpack(f"<{nd}Q", f.read(8 * nd))
            gt, off = struct.unpack("<IQ", f.read(12))
            infos.append((name, dims, gt, off))
        pos = f.tell()
        data_start = pos + ((-pos) % ALIGNMENT)
        tensors = {}
        for name, dims, gt, off in infos:
            shape = tuple(reversed(dims))
            n = 1
            for d in shape:
                n *= d
            np_t = np.float16 if gt == _GGML_F16 else np.float32
            f.seek(data_start + off)
            arr = np.frombuffer(f.read(n * np_t().itemsize),
                                dtype=np_t).reshape(shape)
            tensors[name] = torch.from_numpy(arr.copy())
    return meta, tensors
