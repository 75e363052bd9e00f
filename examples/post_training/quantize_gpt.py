#!/usr/bin/env python3
"""Weight-only int8 PTQ of a GPT checkpoint (reference
examples/post_training): load -> quantize -> report error -> export."""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", ".."))

import torch

from megatronapp_amd.core import parallel_state
from megatronapp_amd.core.models.gpt import GPTModel
from megatronapp_amd.core.models.gpt.gpt_layer_specs import (
    get_gpt_layer_local_spec)
from megatronapp_amd.core.transformer_config import TransformerConfig
from megatronapp_amd.post_training import (
    export_int8_state_dict, quantize_model)


def main():
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29799")
    os.environ.setdefault("RANK", "0")
    os.environ.setdefault("WORLD_SIZE", "1")
    torch.distributed.init_process_group("gloo")
    parallel_state.initialize_model_parallel()
    cfg = TransformerConfig(
        num_layers=12, hidden_size=768, num_attention_heads=12,
        ffn_hidden_size=3072, hidden_dropout=0.0, attention_dropout=0.0,
        masked_softmax_fusion=False)
    model = GPTModel(
        config=cfg,
        transformer_layer_spec=get_gpt_layer_local_spec(use_flash=False),
        vocab_size=50304, max_sequence_length=2048).eval()
    n = quantize_model(model)
    exported = export_int8_state_dict(model)
    bytes_int8 = sum(q.numel() + s.numel() * 4
                     for q, s in exported.values())
    bytes_fp = sum(q.numel() * 2 for q, _ in exported.values())
    print(f"quantized {n} linears; exported int8 payload "
          f"{bytes_int8 / 1e6:.1f} MB vs bf16 {bytes_fp / 1e6:.1f} MB")
    torch.save(exported, "gpt_int8.pt")


if __name__ == "__main__":
    main()
