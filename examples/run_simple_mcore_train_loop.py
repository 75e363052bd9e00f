"""Minimal megatron-core training loop (reference
examples/run_simple_mcore_train_loop.py): build a tiny GPT from core APIs
only, run a few optimizer steps on mock data, then round-trip a
distributed checkpoint.  Runs on CPU (gloo, world_size=1) or GPU.

torchrun --nproc-per-node 1 --master-addr 127.0.0.1 \
    examples/run_simple_mcore_train_loop.py
"""

import os
import sys
from functools import partial
from pathlib import Path

import torch
from torch.optim import Adam
from torch.utils.data import DataLoader

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from megatronapp_amd.core import parallel_state
from megatronapp_amd.core import dist_checkpointing
from megatronapp_amd.core.pipeline_parallel import get_forward_backward_func
from megatronapp_amd.core.tensor_parallel.random import (
    model_parallel_cuda_manual_seed)
from megatronapp_amd.core.transformer_config import TransformerConfig
from megatronapp_amd.core.models.gpt import GPTModel
from megatronapp_amd.core.models.gpt.gpt_layer_specs import (
    get_gpt_layer_local_spec)
from megatronapp_amd.core.datasets.gpt_dataset import (
    GPTDatasetConfig, MockGPTDataset)

_SEQUENCE_LENGTH = 64


def initialize_distributed(tp=1, pp=1):
    parallel_state.destroy_model_parallel()
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29383")
    os.environ.setdefault("RANK", "0")
    os.environ.setdefault("WORLD_SIZE", "1")
    rank = int(os.environ.get("LOCAL_RANK", os.environ["RANK"]))
    backend = "nccl" if torch.cuda.is_available() else "gloo"
    if torch.cuda.is_available():
        torch.cuda.set_device(rank % torch.cuda.device_count())
    torch.distributed.init_process_group(backend=backend)
    parallel_state.initialize_model_parallel(tp, pp)


def model_provider():
    config = TransformerConfig(
        num_layers=2, hidden_size=64, num_attention_heads=4,
        hidden_dropout=0.0, attention_dropout=0.0,
        pipeline_dtype=torch.float32)
    return GPTModel(config=config,
                    transformer_layer_spec=get_gpt_layer_local_spec(
                        use_flash=False),
                    vocab_size=100, max_sequence_length=_SEQUENCE_LENGTH)


def get_train_data_iterator():
    config = GPTDatasetConfig(
        random_seed=0, sequence_length=_SEQUENCE_LENGTH, vocab_size=100)
    dataset = MockGPTDataset(config, num_samples=1000)
    return iter(DataLoader(dataset, batch_size=8, shuffle=True))


def forward_step_func(data_iterator, model):
    def loss_func(loss_mask, output_tensor):
        losses = output_tensor.float()
        loss_mask = loss_mask.view(-1).float()
        loss = torch.sum(losses.view(-1) * loss_mask) / loss_mask.sum()
        return loss, {"lm loss": loss}

    device = next(model.parameters()).device
    data = next(data_iterator)
    tokens = data["tokens"].to(device)
    position_ids = data["position_ids"].to(device)
    labels = data["labels"].to(device)
    loss_mask = data["loss_mask"].to(device)
    output_tensor = model(tokens, position_ids, None, labels=labels)
    return output_tensor, partial(loss_func, loss_mask)


def main():
    initialize_distributed(tp=1, pp=1)
    model_parallel_cuda_manual_seed(123)
    model = model_provider()
    if torch.cuda.is_available():
        model.cuda()
    optim = Adam(model.parameters(), lr=1e-3)
    it = get_train_data_iterator()
    fb = get_forward_backward_func()

    for step in range(5):
        optim.zero_grad()
        losses = fb(forward_step_func=forward_step_func, data_iterator=it,
                    model=model, num_microbatches=1,
                    seq_length=_SEQUENCE_LENGTH, micro_batch_size=8,
                    forward_only=False)
        optim.step()
        print(f"step {step}: {losses[0]['lm loss']:.4f}", flush=True)

    ckpt = Path(os.environ.get("CKPT_DIR", "/tmp/simple_mcore_ckpt"))
    dist_checkpointing.save(model.sharded_state_dict(prefix=""), str(ckpt))
    state = dist_checkpointing.load(model.sharded_state_dict(prefix=""),
                                    str(ckpt))
    model.load_state_dict(state)
    print("distributed checkpoint round trip OK", flush=True)
    torch.distributed.destroy_process_group()


if __name__ == "__main__":
    main()
