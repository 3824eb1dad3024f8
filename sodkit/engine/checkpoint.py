"""Checkpoint save / resume with the reference's output-tree layout
(SURVEY §5.4 — a compatibility REQUIREMENT [BJ:5]):

    output/<exp_name>/
      pth/state_final.pth           # model state_dict only (deploy/test)
          checkpoint_final.pth.tar  # {'epoch','net_state','opti_state','scaler'}
      pre/<dataset>/<name>.png      # predicted saliency maps
      tb/                           # tensorboard events
      log_<exp>.txt, results.csv[/xlsx], cfg_<exp>.txt

Rank 0 writes; callers barrier after (C6)."""

from __future__ import annotations

import os
from typing import Optional

import torch


def exp_paths(output_root: str, exp_name: str) -> dict:
    root = os.path.join(output_root, exp_name)
    return {
        "root": root,
        "pth": os.path.join(root, "pth"),
        "pre": os.path.join(root, "pre"),
        "tb": os.path.join(root, "tb"),
        "log": os.path.join(root, f"log_{exp_name}.txt"),
        "cfg": os.path.join(root, f"cfg_{exp_name}.txt"),
        "state_final": os.path.join(root, "pth", "state_final.pth"),
        "checkpoint_final": os.path.join(root, "pth", "checkpoint_final.pth.tar"),
    }


def make_exp_tree(paths: dict) -> None:
    for key in ("root", "pth", "pre", "tb"):
        os.makedirs(paths[key], exist_ok=True)


def save_checkpoint(paths: dict, epoch: int, model: torch.nn.Module,
                    optimizer, scaler_state: Optional[dict] = None) -> None:
    torch.save(model.state_dict(), paths["state_final"])
    torch.save({
        "epoch": epoch,
        "net_state": model.state_dict(),
        "opti_state": optimizer.state_dict(),
        "scaler": scaler_state,
    }, paths["checkpoint_final"])


def load_checkpoint(paths: dict, model: torch.nn.Module, optimizer=None,
                    map_location="cpu") -> "tuple[int, Optional[dict]]":
    """Returns (next epoch to run, saved scaler state or None)."""
    ckpt = torch.load(paths["checkpoint_final"], map_location=map_location,
                      weights_only=False)
    model.load_state_dict(ckpt["net_state"])
    if optimizer is not None and ckpt.get("opti_state") is not None:
        optimizer.load_state_dict(ckpt["opti_state"])
    return int(ckpt["epoch"]) + 1, ckpt.get("scaler")
