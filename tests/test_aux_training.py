"""yaml_arguments + theoretical_memory_usage unit tests."""
import os
import sys

import pytest


def test_yaml_config_overlay(tmp_path, monkeypatch):
    cfg = tmp_path / "cfg.yml"
    cfg.write_text(
        "model:\n  num-layers: 6\n  hidden_size: 320\n"
        "training:\n  micro_batch_size: 4\n  save: ${AUX_SAVE}\n")
    monkeypatch.setenv("AUX_SAVE", "/tmp/aux_ck")
    monkeypatch.setattr(sys, "argv",
                        ["x", "--yaml-cfg", str(cfg), "--hidden-size", "640"])
    from megatronapp_amd.training.arguments import parse_args
    a = parse_args(ignore_unknown_args=True)
    assert a.num_layers == 6
    assert a.hidden_size == 640          # CLI overrides YAML
    assert a.micro_batch_size == 4
    assert a.save == "/tmp/aux_ck"       # env interpolation


def test_yaml_config_unknown_key(tmp_path):
    cfg = tmp_path / "bad.yml"
    cfg.write_text("model:\n  not_a_real_flag: 1\n")
    from megatronapp_amd.training.arguments import parse_args
    from megatronapp_amd.training.yaml_arguments import apply_yaml_config
    import argparse
    ns = argparse.Namespace(num_layers=2)
    with pytest.raises(ValueError):
        apply_yaml_config(ns, str(cfg))


def _gpt3_1p3b_args(**over):
    from types import SimpleNamespace
    d = dict(hidden_size=2048, num_attention_heads=16, kv_channels=128,
             num_layers=24, ffn_hidden_size=8192, swiglu=False,
             group_query_attention=False, num_query_groups=None,
             num_experts=None, padded_vocab_size=51200, vocab_size=50257,
             untie_embeddings_and_output_weights=False,
             pipeline_model_parallel_size=1, tensor_model_parallel_size=1,
             data_parallel_size=1, use_distributed_optimizer=False,
             seq_length=2048, micro_batch_size=2,
             virtual_pipeline_model_parallel_size=None)
    d.update(over)
    return SimpleNamespace(**d)


def test_theoretical_memory_gpt3_1p3b():
    from megatronapp_amd.training.theoretical_memory_usage import (
        compute_weight_and_optimizer_memory, compute_activation_memory,
        report_theoretical_memory)
    args = _gpt3_1p3b_args()
    w = compute_weight_and_optimizer_memory(args)
    # ~1.3B params to within 15%, 18 B/param
    assert 1.1e9 * 18 < w < 1.6e9 * 18, w
    # distributed optimizer across dp=8 shrinks state
    w8 = compute_weight_and_optimizer_memory(
        _gpt3_1p3b_args(use_distributed_optimizer=True, data_parallel_size=8))
    assert w8 < w * 0.55
    a = compute_activation_memory(args, num_microbatches=8)
    assert a > 0
    # TP halves activations
    a2 = compute_activation_memory(
        _gpt3_1p3b_args(tensor_model_parallel_size=2), num_microbatches=8)
    assert abs(a2 - a / 2) / a < 0.01
    total = report_theoretical_memory(args, num_microbatches=8)
    assert total == pytest.approx(w + a)


def test_checkpoint_converter(tmp_path):
    """torch_dist(TP2) -> converter merge -> torch_dist(single-shard) ->
    loadable back at TP1 via the overlap-window loader."""
    import subprocess
    import torch
    from megatronapp_amd.core.dist_checkpointing import (
        ShardedTensor, save as dist_save, load as dist_load)

    src = tmp_path / "src" / "iter_0000007"
    torch.manual_seed(11)
    w = torch.randn(8, 6)
    # fake a TP=2 save: two shards along dim 0, written in one process
    sd0 = {"model.w": ShardedTensor("model.w", w[:4].clone(), (8, 6), (0, 0))}
    from megatronapp_amd.core.dist_checkpointing.serialization import \
        _save_legacy
    _save_legacy(sd0, str(src), common_state={"iteration": 7})
    # second shard appended by hand (single-process test)
    import json, os
    shard = {"model.w": {"offset": (4, 0), "global_shape": (8, 6),
                         "tensor": w[4:].clone()}}
    torch.save(shard, str(src / "shards_rank00001.pt"))
    idx = json.load(open(src / "index.json"))
    idx["model.w"].append({"file": "shards_rank00001.pt", "offset": [4, 0],
                           "shape": [4, 6], "global_shape": [8, 6]})
    json.dump(idx, open(src / "index.json", "w"))

    dst = tmp_path / "dst" / "iter_0000007"
    r = subprocess.run(
        [os.sys.executable, "tools/checkpoint/convert.py",
         "--load", str(src), "--loader", "torch_dist",
         "--save", str(dst), "--saver", "torch_dist"],
        capture_output=True, text=True)
    assert r.returncode == 0, r.stderr
    assert (tmp_path / "dst" /
            "latest_checkpointed_iteration.txt").read_text() == "7"

    # load the converted checkpoint as if we were TP=1
    out = torch.zeros(8, 6)
    dist_load({"model.w": ShardedTensor("model.w", out, (8, 6), (0, 0))},
              str(dst))
    assert torch.equal(out, w)

    # consolidated export too
    flat = tmp_path / "full.pt"
    r = subprocess.run(
        [os.sys.executable, "tools/checkpoint/convert.py",
         "--load", str(dst), "--save", str(flat),
         "--saver", "consolidated", "--inspect"],
        capture_output=True, text=True)
    assert r.returncode == 0, r.stderr
    blob = torch.load(flat, weights_only=False)
    assert torch.equal(blob["weights"]["model.w"], w)


def test_async_save_and_config_logger(tmp_path):
    """--async-save writes a loadable checkpoint from a background thread;
    --config-logger-dir dumps resolved args; heartbeat files appear."""
    import subprocess
    save = str(tmp_path / "ck")
    cfgdir = str(tmp_path / "cfg")
    hbdir = str(tmp_path / "hb")
    from tests.test_checkpointing import _run
    out = _run(["--train-iters", "3", "--save", save, "--save-interval", "3",
                "--ckpt-format", "torch", "--async-save",
                "--config-logger-dir", cfgdir,
                "--ft-heartbeat-dir", hbdir], 29641)
    assert os.path.exists(os.path.join(
        save, "iter_0000003", "mp_rank_00", "model_optim_rng.pt"))
    assert os.path.exists(os.path.join(cfgdir, "args.json"))
    import json
    hb = json.load(open(os.path.join(hbdir, "heartbeat_rank0.json")))
    assert hb["section"] in ("idle", "checkpoint")
    out2 = _run(["--train-iters", "5", "--save", save, "--load", save,
                 "--ckpt-format", "torch", "--save-interval", "100"], 29642)
    assert "loaded checkpoint" in out2


def test_async_save_torch_dist(tmp_path):
    from tests.test_checkpointing import _run
    save = str(tmp_path / "ckd")
    _run(["--train-iters", "3", "--save", save, "--save-interval", "3",
          "--ckpt-format", "torch_dist", "--async-save"], 29643)
    assert os.path.exists(os.path.join(save, "iter_0000003", ".metadata"))
    out = _run(["--train-iters", "5", "--save", save, "--load", save,
                "--ckpt-format", "torch_dist", "--save-interval", "100"],
               29644)
    assert "loaded checkpoint (torch_dist)" in out


def test_eval_loop_and_exit_interval(tmp_path):
    from tests.test_checkpointing import _run
    out = _run(["--train-iters", "6", "--eval-iters", "2",
                "--eval-interval", "3", "--exit-interval", "4"], 29646)
    # eval ran at iter 3 and training stopped at the exit interval
    assert "validation loss" in out or "val loss" in out or "eval" in out.lower()
    assert "iteration        5" not in out and "iteration        4" in out


def test_train_samples_mode(tmp_path):
    """--train-samples derives train_iters from the global batch."""
    from tests.test_checkpointing import _run
    out = _run(["--train-samples", "32", "--train-iters-none"], 29647) \
        if False else None
    import subprocess, sys, os
    REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    env = dict(os.environ, MASTER_ADDR="127.0.0.1", MASTER_PORT="29648",
               RANK="0", WORLD_SIZE="1", LOCAL_RANK="0")
    r = subprocess.run(
        [sys.executable, os.path.join(REPO, "pretrain_gpt.py"),
         "--num-layers", "2", "--hidden-size", "64",
         "--num-attention-heads", "4", "--seq-length", "32",
         "--max-position-embeddings", "32", "--micro-batch-size", "2",
         "--global-batch-size", "8", "--vocab-size", "128", "--mock-data",
         "--train-samples", "32", "--lr", "1e-3", "--log-interval", "1",
         "--eval-iters", "0", "--hidden-dropout", "0",
         "--attention-dropout", "0"],
        capture_output=True, text=True, env=env, cwd=REPO, timeout=300)
    assert r.returncode == 0, r.stderr[-2000:]
    assert "4/4" in r.stdout        # 32 samples / gbs 8 = 4 iters


def test_tensorboard_jsonl_fallback(tmp_path):
    from tests.test_checkpointing import _run
    tb = str(tmp_path / "tb")
    _run(["--train-iters", "2", "--tensorboard-dir", tb], 29649)
    import json
    lines = [json.loads(l) for l in
             open(os.path.join(tb, "scalars.jsonl"))]
    tags = {l["tag"] for l in lines}
    assert "lm loss" in tags and "learning-rate" in tags


def test_log_params_norm(tmp_path):
    from tests.test_checkpointing import _run
    out = _run(["--train-iters", "2", "--log-params-norm"], 29650)
    assert "params norm:" in out


def test_use_checkpoint_args(tmp_path):
    """Resume with --use-checkpoint-args restores the architecture even if
    the CLI disagrees."""
    from tests.test_checkpointing import _run
    save = str(tmp_path / "ca")
    _run(["--train-iters", "2", "--save", save, "--save-interval", "2",
          "--ckpt-format", "torch_dist"], 29652)
    # resume with WRONG --hidden-size on the CLI; checkpoint args win
    out = _run(["--train-iters", "4", "--load", save, "--hidden-size",
                "128", "--use-checkpoint-args", "--ckpt-format",
                "torch_dist", "--save-interval", "100"], 29653)
    assert "architecture args restored" in out
    assert "loaded checkpoint (torch_dist)" in out
