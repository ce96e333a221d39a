#!/usr/bin/env python3
"""Bisect the ROCm 7.2 hipGraph train-step capture crash (ROADMAP r1).

Round 1: capturing the full autograd train step core-dumped natively
(not a python error). Suspects: dropout RNG-offset capture,
foreach-norm multi-tensor-apply in LAMB. This script captures
progressively larger stages in SEPARATE invocations (the driver shell
loops over --stage so a native crash only kills one probe):

  1 forward (training=True, dropout ON)
  2 forward with dropout OFF
  3 forward + AlignmentLoss
  4 forward + loss + backward
  5 full step incl. capturable LAMB
  6 full step, dropout OFF
  7 full step, dropout OFF, plain-loop (non-foreach) LAMB

Exit 0 = capture + 3 replays succeeded and replay output matched an
eager step within tolerance; nonzero/death = that stage triggers it.
"""
import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch

from deepconsensus_amd.models import config as cfg
from deepconsensus_amd.models import lamb as lamb_lib
from deepconsensus_amd.models import losses as losses_lib
from deepconsensus_amd.models.model import get_model


def zero_dropout(model):
    for name in ("dropout", "post_dropout"):
        pass
    for m in model.modules():
        for attr in ("dropout", "post_dropout", "attention_dropout",
                     "relu_dropout"):
            if hasattr(m, attr) and isinstance(getattr(m, attr), float):
                setattr(m, attr, 0.0)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--stage", type=int, required=True)
    ap.add_argument("--batch-size", type=int, default=1024)
    args = ap.parse_args()
    stage = args.stage

    torch.manual_seed(7)
    params = cfg.get_config("transformer_learn_values+custom")
    cfg.modify_params(params)
    device = "cuda"
    model = get_model(params).to(device)
    if stage in (2, 6, 7):
        zero_dropout(model)
    loss_fn = losses_lib.AlignmentLoss(
        del_cost=params.del_cost, loss_reg=params.loss_reg,
        reduction="sum",
    )
    B = args.batch_size
    rng = np.random.default_rng(42)
    mp = params.max_passes
    rows = np.zeros((B, params.total_rows, 100), dtype=np.float32)
    rows[:, 0:mp] = rng.integers(0, 5, size=(B, mp, 100))
    rows[:, mp:3 * mp] = rng.integers(0, 60, size=(B, 2 * mp, 100))
    rows[:, 3 * mp:4 * mp] = rng.integers(0, 3, size=(B, mp, 100))
    rows[:, 4 * mp] = rng.integers(0, 5, size=(B, 100))
    rows[:, -4:] = rng.integers(5, 30, size=(B, 4, 1))
    x = torch.from_numpy(rows).to(device)
    label = torch.from_numpy(
        rng.integers(0, 5, size=(B, 100)).astype(np.int64)
    ).to(device)

    use_loss = stage >= 3
    use_bwd = stage >= 4
    use_opt = stage >= 5
    optimizer = None
    if use_opt:
        optimizer = lamb_lib.LAMB(
            model.parameters(), lr=1e-3,
            weight_decay=params.get("weight_decay", 0.0) or 0.0,
            capturable=True, foreach=(stage != 7),
        )

    def body():
        with torch.autocast("cuda", dtype=torch.bfloat16):
            probs = model(x, training=True)
        if not use_loss:
            return probs.float().sum()
        loss = loss_fn(label, probs.float()) / B
        if use_bwd:
            loss.backward()
        if use_opt:
            optimizer.step()
        return loss

    def zero_grads():
        for p in model.parameters():
            if p.grad is None:
                p.grad = torch.zeros_like(p)
            else:
                p.grad.zero_()

    # Warmup on a side stream (allocator pool priming), standard recipe.
    s = torch.cuda.Stream()
    s.wait_stream(torch.cuda.current_stream())
    with torch.cuda.stream(s):
        for _ in range(3):
            if use_bwd:
                zero_grads()
            body()
    torch.cuda.current_stream().wait_stream(s)
    torch.cuda.synchronize()
    eager_out = body().detach().clone()
    torch.cuda.synchronize()
    print(f"stage {stage}: warmup ok, capturing...", flush=True)

    if use_bwd:
        zero_grads()
    g = torch.cuda.CUDAGraph()
    with torch.cuda.graph(g):
        out = body()
    torch.cuda.synchronize()
    print(f"stage {stage}: capture ok, replaying...", flush=True)
    for i in range(3):
        if use_bwd:
            zero_grads()
        g.replay()
    torch.cuda.synchronize()
    print(f"stage {stage}: replay ok; out={float(out):.4f} "
          f"eager={float(eager_out):.4f}", flush=True)


if __name__ == "__main__":
    main()
