#!/usr/bin/env python3
"""Checkpoint format / topology converter.

Reference: tools/checkpoint/convert.py (loader/saver plugin protocol that
streams FULL, unsplit weights between a loader and a saver).  Same design
here, collapsed to the formats this framework writes:

  loaders:  torch_dist   sharded dir (index.json) at any TP/PP -> full tensors
            legacy       mp_rank_* layout (TP=PP=1 only; for sharded legacy
                         checkpoints re-save with --ckpt-format torch_dist)
            consolidated a single .pt of {name: tensor}
            hf_gpt2      a transformers GPT-2 directory (weights remapped
                         into this framework's naming)
  savers:   torch_dist   single-shard-per-key sharded dir; loadable at ANY
                         TP/PP by the overlap-window loader
            consolidated single .pt of {name: tensor}
            hf_gpt2      transformers-loadable GPT2LMHeadModel directory
                         (verified logit-exact against transformers)

Typical uses:
  # make a topology-free checkpoint from a TP=2,PP=2 run:
  python tools/checkpoint/convert.py \
      --load ckpt/iter_0001000 --loader torch_dist \
      --save ckpt_univ/iter_0001000 --saver torch_dist
  # export full weights for inspection / external tooling:
  python tools/checkpoint/convert.py \
      --load ckpt/iter_0001000 --loader torch_dist \
      --save model_full.pt --saver consolidated
"""

import argparse
import json
import os
import re
import sys

import torch

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", ".."))
sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))


def load_torch_dist(path):
    """Merge a sharded checkpoint's pieces into full tensors.  Reads the
    torch-DCP layout (.metadata + __N_M.distcp — ours or
    reference/upstream-produced) and the round-1 legacy index.json."""
    if os.path.exists(os.path.join(path, ".metadata")):
        from megatronapp_amd.core.dist_checkpointing.torch_dcp import (
            load_dcp_consolidated)
        full = load_dcp_consolidated(path)
        common = {}
        cpath = os.path.join(path, "common.pt")
        if os.path.exists(cpath):
            common = torch.load(cpath, map_location="cpu",
                                weights_only=False)
        return full, common
    with open(os.path.join(path, "index.json")) as f:
        index = json.load(f)
    cache = {}

    def piece(fname):
        if fname not in cache:
            cache[fname] = torch.load(os.path.join(path, fname),
                                      map_location="cpu", weights_only=False)
        return cache[fname]

    full = {}
    for key, metas in index.items():
        gshape = metas[0]["global_shape"]
        out = torch.empty(gshape, dtype=piece(metas[0]["file"])[key]["tensor"].dtype)
        covered = 0
        for meta in metas:
            t = piece(meta["file"])[key]["tensor"]
            slices = tuple(slice(o, o + s)
                           for o, s in zip(meta["offset"], meta["shape"]))
            out[slices] = t
            covered += t.numel()
        if covered < out.numel():
            raise RuntimeError(f"{key}: shards cover {covered}/{out.numel()} "
                               f"elements — checkpoint incomplete")
        full[key] = out
    common = {}
    cpath = os.path.join(path, "common.pt")
    if os.path.exists(cpath):
        common = torch.load(cpath, map_location="cpu", weights_only=False)
    return full, common


# TP shard axis by parameter name (Megatron-core GPT family), for
# merging reference-produced legacy checkpoints saved at TP>1:
#   dim 0 (column-parallel outputs / vocab rows): qkv, fc1, embeddings,
#          output layer; dim 1 (row-parallel inputs): proj, fc2.
#   replicated: norms and all biases of row-parallel layers.
def _legacy_tp_axis(name):
    if name.endswith("bias"):
        if any(t in name for t in ("linear_qkv", "linear_fc1", "linear_q",
                                   "linear_kv")):
            return 0
        return None
    if any(t in name for t in ("word_embeddings.weight",
                               "output_layer.weight",
                               "linear_qkv.weight", "linear_fc1.weight",
                               "linear_q_proj", "linear_q_down",
                               "linear_kv_down")):
        return 0
    if any(t in name for t in ("linear_proj.weight", "linear_fc2.weight")):
        return 1
    return None  # norms, rotary inv_freq, router, etc. are replicated


def _legacy_rank_dirs(path):
    """{(tp, pp): dir} from mp_rank_XX[_YYY] directory names."""
    out = {}
    for d in os.listdir(path):
        if not d.startswith("mp_rank_"):
            continue
        parts = d[len("mp_rank_"):].split("_")
        tp = int(parts[0])
        pp = int(parts[1]) if len(parts) > 1 else 0
        out[(tp, pp)] = d
    return out


def load_legacy(path):
    """mp_rank_XX[_YYY]/model_optim_rng.pt — reference legacy layout at
    any TP (name-based shard-axis merge) and any PP (disjoint layer
    namespaces merged with globalized layer indices)."""
    ranks = _legacy_rank_dirs(path)
    if not ranks:
        raise SystemExit(f"no mp_rank_* dirs under {path}")
    tps = sorted({t for t, _ in ranks})
    pps = sorted({p for _, p in ranks})

    def load_rank(tp, pp):
        return torch.load(os.path.join(path, ranks[(tp, pp)],
                                       "model_optim_rng.pt"),
                          map_location="cpu", weights_only=False)

    full = {}
    common = {}
    layer_base = 0
    for pp in pps:
        shards = [load_rank(tp, pp) for tp in tps]
        sd0 = shards[0]
        model_keys = [k for k in sd0 if k == "model" or
                      (k.startswith("model") and isinstance(sd0[k], dict))]
        n_local_layers = 0
        for mk in model_keys:
            prefix = "model." if mk == "model" else f"{mk}."
            for name, t in sd0[mk].items():
                if not torch.is_tensor(t):
                    continue
                m = re.search(r"layers\.(\d+)\.", name)
                if m:
                    n_local_layers = max(n_local_layers, int(m.group(1)) + 1)
                gname = prefix + (re.sub(
                    r"layers\.(\d+)\.",
                    lambda mm: f"layers.{int(mm.group(1)) + layer_base}.",
                    name) if m else name)
                axis = _legacy_tp_axis(name)
                if axis is None or len(tps) == 1:
                    full[gname] = t
                else:
                    full[gname] = torch.cat(
                        [sh[mk][name] for sh in shards], dim=axis)
        layer_base += n_local_layers
        if pp == pps[0]:
            common = {k: v for k, v in sd0.items() if k not in model_keys
                      and k not in ("optimizer", "rng_state")}
    return full, common


def load_consolidated(path):
    blob = torch.load(path, map_location="cpu", weights_only=False)
    if "weights" in blob:
        return blob["weights"], blob.get("common", {})
    return {k: v for k, v in blob.items() if torch.is_tensor(v)}, {}


def save_torch_dist(full, common, path):
    os.makedirs(path, exist_ok=True)
    from megatronapp_amd.core.dist_checkpointing.mapping import ShardedTensor
    from megatronapp_amd.core.dist_checkpointing.torch_dcp import save_dcp
    sd = {k: ShardedTensor(k, t, tuple(t.shape), (0,) * t.dim(), 0)
          for k, t in full.items()}
    save_dcp(sd, path)
    torch.save(common, os.path.join(path, "common.pt"))
    # tracker so load_checkpoint finds it when pointed at the parent dir
    parent, leaf = os.path.split(os.path.normpath(path))
    if leaf.startswith("iter_"):
        with open(os.path.join(parent,
                               "latest_checkpointed_iteration.txt"), "w") as f:
            f.write(str(int(leaf[len("iter_"):])))


def save_consolidated(full, common, path):
    os.makedirs(os.path.dirname(os.path.abspath(path)), exist_ok=True)
    torch.save({"weights": full, "common": common}, path)


def save_hf_gpt2(full, common, path):
    from saver_hf_gpt2 import save_hf_gpt2 as impl
    impl(full, common, path)


def load_hf_gpt2(path):
    from saver_hf_gpt2 import load_hf_gpt2 as impl
    return impl(path)


def save_hf_llama(full, common, path):
    from saver_hf_llama import save_hf_llama as impl
    impl(full, common, path)


def load_hf_llama(path):
    from saver_hf_llama import load_hf_llama as impl
    return impl(path)


def save_hf_mixtral(full, common, path):
    from saver_hf_mixtral import save_hf_mixtral as impl
    impl(full, common, path)


def load_hf_mixtral(path):
    from saver_hf_mixtral import load_hf_mixtral as impl
    return impl(path)


LOADERS = {"torch_dist": load_torch_dist, "legacy": load_legacy,
           "consolidated": load_consolidated, "hf_gpt2": load_hf_gpt2,
           "hf_llama": load_hf_llama, "hf_mixtral": load_hf_mixtral}
def save_gguf(full, common, path):
    from saver_gguf import save_gguf as impl
    impl(full, common, path)


SAVERS = {"torch_dist": save_torch_dist, "consolidated": save_consolidated,
          "gguf": save_gguf,
          "hf_gpt2": save_hf_gpt2, "hf_llama": save_hf_llama,
          "hf_mixtral": save_hf_mixtral}


def main():
    p = argparse.ArgumentParser(description=__doc__,
                                formatter_class=argparse.RawDescriptionHelpFormatter)
    p.add_argument("--load", required=True,
                   help="checkpoint iter dir (or .pt for consolidated)")
    p.add_argument("--loader", choices=sorted(LOADERS), default="torch_dist")
    p.add_argument("--save", required=True)
    p.add_argument("--saver", choices=sorted(SAVERS), default="torch_dist")
    p.add_argument("--dtype", default=None,
                   choices=[None, "float32", "bfloat16", "float16"],
                   help="optionally cast all weights")
    p.add_argument("--inspect", action="store_true",
                   help="print every key/shape/dtype after loading")
    args = p.parse_args()

    full, common = LOADERS[args.loader](args.load)
    n_params = sum(t.numel() for t in full.values())
    print(f"loaded {len(full)} tensors, {n_params / 1e6:.1f} M elements")
    if args.inspect:
        for k in sorted(full):
            t = full[k]
            print(f"  {k:70s} {tuple(t.shape)!s:24s} {t.dtype}")
    if args.dtype:
        dt = getattr(torch, args.dtype)
        full = {k: (t.to(dt) if t.is_floating_point() else t)
                for k, t in full.items()}
    SAVERS[args.saver](full, common, args.save)
    print(f"wrote {args.saver} checkpoint to {args.save}")


if __name__ == "__main__":
    main()
