"""Checkpoint -> serving bundle converter.

Parity with reference convert_to_saved_model.py:67-105 (checkpoint ->
SavedModel): rebuilds the model from params.json, restores the checkpoint
(running a forward pass first so a wrong checkpoint fails loudly, mirroring
assert_existing_objects_matched), exports a TorchScript trace of the
rows->probs forward plus params.json next to it. quick_inference accepts
either a checkpoint directory or an exported bundle.
"""
from __future__ import annotations

import argparse
import logging
import os
import shutil
from typing import List, Optional

import torch

from deepconsensus_amd.models import checkpoint as ckpt_lib
from deepconsensus_amd.models import config as cfg
from deepconsensus_amd.models.model import get_model

log = logging.getLogger(__name__)


class _ForwardWrapper(torch.nn.Module):
    def __init__(self, model):
        super().__init__()
        self.model = model

    def forward(self, rows: torch.Tensor) -> torch.Tensor:
        return self.model(rows, training=False)


def initialize_model(checkpoint_path: str, device: str = "cpu"):
    """Rebuild + restore, with a forward pass to validate the restore."""
    params = ckpt_lib.load_params(checkpoint_path)
    cfg.modify_params(params, is_training=False)
    model = get_model(params).to(device)
    # Forward pass before restore, mirroring the reference's requirement.
    dummy = torch.zeros(
        1, params.total_rows, params.max_length, device=device
    )
    with torch.no_grad():
        model(dummy)
    ckpt_lib.load_checkpoint(checkpoint_path, model, strict=True)
    model.eval()
    return model, params


def export(checkpoint_path: str, out_dir: str, device: str = "cpu") -> str:
    model, params = initialize_model(checkpoint_path, device)
    os.makedirs(out_dir, exist_ok=True)
    wrapper = _ForwardWrapper(model)
    example = torch.zeros(
        2, params.total_rows, params.max_length, device=device
    )
    with torch.no_grad():
        # torch.jit.trace is deprecated upstream in favor of torch.compile/
        # torch.export, but compile is Triton-backed (excluded on this
        # CDNA4-native build) and jit remains the dependency-free serialized
        # form loadable with plain torch.jit.load.
        traced = torch.jit.trace(wrapper, example)
    traced_path = os.path.join(out_dir, "serving_model.pt")
    traced.save(traced_path)
    cfg.save_params_as_json(out_dir, params)
    # Also copy raw weights so the bundle restores into training code.
    src = ckpt_lib.latest_checkpoint(checkpoint_path) if os.path.isdir(
        checkpoint_path
    ) else checkpoint_path
    if src and os.path.exists(src):
        shutil.copy(src, os.path.join(out_dir, os.path.basename(src)))
        with open(os.path.join(out_dir, "checkpoint"), "w") as f:
            f.write(os.path.basename(src)[:-3])
    log.info("exported serving bundle to %s", out_dir)
    return traced_path


def main(argv: Optional[List[str]] = None) -> None:
    ap = argparse.ArgumentParser("deepconsensus export")
    ap.add_argument("--checkpoint", required=True)
    ap.add_argument("--output_dir", required=True)
    args = ap.parse_args(argv)
    export(args.checkpoint, args.output_dir)


if __name__ == "__main__":
    logging.basicConfig(level=logging.INFO)
    main()
