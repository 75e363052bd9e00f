"""Weights & Biases helpers (reference training/wandb_utils.py).

wandb is not installed in this image; every hook degrades to a no-op and
``setup_wandb`` reports why, so launch scripts carrying --wandb-project
still run unchanged.
"""

from __future__ import annotations

import os


def _wandb():
    try:
        import wandb
        return wandb
    except ImportError:
        return None


def setup_wandb(args):
    wandb = _wandb()
    if wandb is None or not getattr(args, "wandb_project", None):
        if getattr(args, "wandb_project", None) and args.rank == 0:
            print("wandb requested but not installed; metrics stay local")
        return None
    if args.rank != 0:
        return None
    wandb.init(project=args.wandb_project,
               name=args.wandb_exp_name or None,
               config={k: v for k, v in vars(args).items()
                       if isinstance(v, (int, float, str, bool))})
    return wandb


def on_save_checkpoint_success(checkpoint_path, tracker_filename, save_dir,
                               iteration):
    wandb = _wandb()
    if wandb is None or wandb.run is None:
        return
    art = wandb.Artifact(os.path.basename(save_dir), type="model")
    art.add_reference(f"file://{checkpoint_path}")
    wandb.run.log_artifact(art)


def on_load_checkpoint_success(checkpoint_path, load_dir):
    pass
