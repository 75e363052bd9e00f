#!/usr/bin/env python3
"""Text-generation server entry (reference tools/run_text_generation_server.py).

Builds the GPT model (optionally loading a checkpoint), wraps it in the
static inference engine and serves:
  * the MegaScope WebSocket protocol on --inference-ws-port
  * a Flask REST /api endpoint on --port (run with --rest)
Non-(tp0,pp0) ranks enter the broadcast follower loop.
"""

import os
import sys
import threading

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from megatronapp_amd.core import parallel_state
from megatronapp_amd.core.inference.static_engine import get_inference_engine
from megatronapp_amd.inference.text_generation_server import (
    InferenceWSServer,
    MegatronServer,
    follower_loop,
)
from megatronapp_amd.training.checkpointing import load_checkpoint
from megatronapp_amd.training.global_vars import get_args, get_tokenizer
from megatronapp_amd.training.initialize import initialize_megatron
from megatronapp_amd.training.training import get_model
from pretrain_gpt import model_provider


def _extra_args(parser):
    g = parser.add_argument_group("text generation server")
    g.add_argument("--rest", action="store_true",
                   help="serve Flask REST /api instead of WebSocket")
    return parser


def main():
    args = initialize_megatron(
        extra_args_provider=_extra_args,
        args_defaults={"tokenizer_type": "NullTokenizer",
                       "attention_dropout": 0.0, "hidden_dropout": 0.0})
    model = get_model(model_provider, wrap_with_ddp=False, args=args)
    for chunk in model:
        chunk.eval()
    if args.load is not None:
        load_checkpoint(model, None, None)
    assert len(model) == 1
    engine = get_inference_engine(model[0], get_tokenizer(),
                                  max_batch_size=8)

    is_server_rank = (parallel_state.get_tensor_model_parallel_rank() == 0 and
                      parallel_state.get_pipeline_model_parallel_rank() == 0)
    if is_server_rank:
        if args.rest:
            MegatronServer(engine, args.num_layers, args.port).run()
        else:
            InferenceWSServer(engine, args.inference_ws_port,
                              args.num_layers).run()
    else:
        follower_loop(engine, args.num_layers)


if __name__ == "__main__":
    main()
