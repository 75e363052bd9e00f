"""BERT / T5 model family tests (CPU)."""

import os
import subprocess
import sys

import torch

from .utils import destroy, initialize_model_parallel

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
COMMON = ["--num-layers", "2", "--hidden-size", "64",
          "--num-attention-heads", "4", "--seq-length", "32",
          "--micro-batch-size", "2", "--global-batch-size", "4",
          "--mock-data", "--train-iters", "2", "--lr", "1e-3",
          "--log-interval", "1", "--vocab-size", "128", "--eval-iters", "0",
          "--hidden-dropout", "0", "--attention-dropout", "0"]


def _run(entry, port):
    env = dict(os.environ, MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
               RANK="0", WORLD_SIZE="1", LOCAL_RANK="0")
    out = subprocess.run([sys.executable, os.path.join(REPO, entry)] + COMMON,
                         capture_output=True, text=True, cwd=REPO, env=env,
                         timeout=300)
    assert out.returncode == 0, out.stderr[-3000:]
    assert "lm loss" in out.stdout
    return out.stdout


def test_pretrain_bert_runs():
    out = _run("pretrain_bert.py", 29701)
    assert "sop loss" in out


def test_pretrain_t5_runs():
    _run("pretrain_t5.py", 29702)


def test_bert_padding_mask_blocks_pad_tokens():
    """Masked-out (padding) positions must not influence other tokens."""
    initialize_model_parallel()
    from megatronapp_amd.core.models.bert import BertModel
    from megatronapp_amd.core.models.bert.bert_layer_specs import (
        get_bert_layer_local_spec)
    from megatronapp_amd.core.transformer_config import TransformerConfig
    torch.manual_seed(0)
    config = TransformerConfig(num_layers=2, hidden_size=32,
                               num_attention_heads=4, hidden_dropout=0.0,
                               attention_dropout=0.0)
    m = BertModel(config=config,
                  transformer_layer_spec=get_bert_layer_local_spec(),
                  vocab_size=64, max_sequence_length=32,
                  add_binary_head=False)
    m.eval()
    ids = torch.randint(0, 61, (1, 16))
    mask = torch.ones(1, 16, dtype=torch.int64)
    mask[0, 12:] = 0   # pad tail
    with torch.no_grad():
        out1, _ = m(ids, mask)
        ids2 = ids.clone()
        ids2[0, 12:] = 7  # change ONLY the padded tokens
        out2, _ = m(ids2, mask)
    # non-pad positions unaffected by pad-token contents
    assert torch.allclose(out1[0, :12], out2[0, :12], atol=1e-5)
    destroy()
