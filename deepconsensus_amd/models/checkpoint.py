"""Checkpoint / resume contracts.

Parity with the reference's checkpoint subsystem (model_utils.py:511-618,
model_train_custom_loop.py:283-300): a checkpoint directory holds weights +
params.json; training appends to checkpoint_metrics.tsv, tracks the best
checkpoint by eval/per_example_accuracy in best_checkpoint.txt, and resumes
from checkpoint-N plus the eval_checkpoint.txt (name, epoch, step) sidecar.
Weight blobs are torch state_dicts (the documented format-compatible
directory contract; TF .index/.data files are out of scope — SURVEY.md
section 7 hard-part 5).
"""
from __future__ import annotations

import os
from typing import Dict, Optional, Tuple

import torch

from deepconsensus_amd.models.config import (
    Params,
    read_params_from_json,
    save_params_as_json,
)


def save_checkpoint(
    out_dir: str,
    step: int,
    epoch: int,
    model: torch.nn.Module,
    optimizer: Optional[torch.optim.Optimizer],
    params: Params,
    metrics: Optional[Dict[str, float]] = None,
) -> str:
    """Writes checkpoint-<n> plus sidecars; returns the checkpoint path."""
    os.makedirs(out_dir, exist_ok=True)
    name = f"checkpoint-{step}"
    path = os.path.join(out_dir, name + ".pt")
    payload = {
        "model": model.state_dict(),
        "optimizer": optimizer.state_dict() if optimizer else None,
        "step": step,
        "epoch": epoch,
    }
    torch.save(payload, path)
    save_params_as_json(out_dir, params)
    with open(os.path.join(out_dir, "checkpoint"), "w") as f:
        f.write(name)
    if metrics:
        tsv = os.path.join(out_dir, "checkpoint_metrics.tsv")
        write_header = not os.path.exists(tsv)
        with open(tsv, "a") as f:
            if write_header:
                f.write("checkpoint_name\t" + "\t".join(metrics) + "\n")
            f.write(name + "\t" + "\t".join(
                str(v) for v in metrics.values()) + "\n")
    with open(os.path.join(out_dir, "eval_checkpoint.txt"), "w") as f:
        f.write(f"{name}\t{epoch}\t{step}\n")
    return path


def update_best_checkpoint(
    out_dir: str, name: str, metric_value: float
) -> bool:
    """Tracks max eval/per_example_accuracy (dc_constants.py:130)."""
    best_file = os.path.join(out_dir, "best_checkpoint.txt")
    best_val = -1.0
    if os.path.exists(best_file):
        with open(best_file) as f:
            parts = f.read().strip().split("\t")
            if len(parts) == 2:
                best_val = float(parts[1])
    if metric_value > best_val:
        with open(best_file, "w") as f:
            f.write(f"{name}\t{metric_value}\n")
        return True
    return False


def latest_checkpoint(out_dir: str) -> Optional[str]:
    marker = os.path.join(out_dir, "checkpoint")
    if not os.path.exists(marker):
        return None
    with open(marker) as f:
        name = f.read().strip()
    path = os.path.join(out_dir, name + ".pt")
    return path if os.path.exists(path) else None


def get_checkpoint_and_initial_epoch(
    out_dir: str,
) -> Tuple[Optional[str], int, int]:
    """Resume point: (checkpoint path, epoch, step) (model_utils.py:511-540)."""
    path = latest_checkpoint(out_dir)
    if path is None:
        return None, 0, 0
    sidecar = os.path.join(out_dir, "eval_checkpoint.txt")
    epoch, step = 0, 0
    if os.path.exists(sidecar):
        with open(sidecar) as f:
            parts = f.read().strip().split("\t")
            if len(parts) == 3:
                epoch, step = int(parts[1]), int(parts[2])
    return path, epoch, step


def load_checkpoint(
    checkpoint: str,
    model: torch.nn.Module,
    optimizer: Optional[torch.optim.Optimizer] = None,
    strict: bool = True,
) -> Dict:
    """Restores model (+optimizer) from a checkpoint file or directory."""
    if os.path.isdir(checkpoint):
        path = latest_checkpoint(checkpoint)
        if path is None:
            raise FileNotFoundError(f"no checkpoint in {checkpoint}")
    else:
        path = checkpoint
        if not path.endswith(".pt") and os.path.exists(path + ".pt"):
            path = path + ".pt"
    payload = torch.load(path, map_location="cpu", weights_only=False)
    model.load_state_dict(payload["model"], strict=strict)
    if optimizer is not None and payload.get("optimizer"):
        optimizer.load_state_dict(payload["optimizer"])
    return payload


def load_params(checkpoint: str) -> Params:
    return read_params_from_json(checkpoint)
