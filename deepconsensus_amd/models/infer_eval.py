"""`deepconsensus eval`: evaluate a checkpoint over labeled TFRecords.

Parity with reference model_inference.py:79-145 /
model_utils.run_inference_and_write_results (:379-421): runs eval metrics
over a labeled dataset and writes inference.csv with one metric per row.
"""
from __future__ import annotations

import argparse
import logging
import os
from typing import List, Optional

import torch

from deepconsensus_amd.models import checkpoint as ckpt_lib
from deepconsensus_amd.models import config as cfg
from deepconsensus_amd.models import data as data_lib
from deepconsensus_amd.models import losses as losses_lib
from deepconsensus_amd.models.model import get_model
from deepconsensus_amd.models.train import _prepare_batch

log = logging.getLogger(__name__)


def run_inference(
    out_dir: str,
    checkpoint: str,
    eval_path: List[str],
    params: Optional[cfg.Params] = None,
    device: str = "cpu",
    limit: int = -1,
) -> dict:
    if params is None:
        params = ckpt_lib.load_params(checkpoint)
        cfg.modify_params(params)
    model = get_model(params).to(device)
    if checkpoint != "random":
        ckpt_lib.load_checkpoint(checkpoint, model)
    model.eval()

    ds = data_lib.DatasetIterator(
        eval_path, params, params.batch_size, shuffle=False, limit=limit,
    )
    loss_fn = losses_lib.AlignmentLoss(
        del_cost=params.del_cost, loss_reg=params.loss_reg,
        width=params.get("band_width"), reduction="mean",
    )
    acc = losses_lib.PerExampleAccuracy()
    per_class = {
        v: losses_lib.PerClassAccuracy(k)
        for k, v in enumerate(" ATCG")
    }
    total_loss, n = 0.0, 0
    with torch.no_grad():
        for batch in ds.iterate():
            rows, label = _prepare_batch(batch, device)
            probs = model(rows, training=False)
            total_loss += float(loss_fn(label, probs.float()))
            n += 1
            acc.update_state(label.cpu(), probs.cpu())
            for m in per_class.values():
                m.update_state(label.cpu(), probs.cpu())
    metrics = {"loss": total_loss / max(n, 1),
               "per_example_accuracy": acc.result()}
    for sym, m in per_class.items():
        key = "gap_or_pad" if sym == " " else sym.lower()
        metrics[f"per_class_accuracy_{key}"] = m.result()

    os.makedirs(out_dir, exist_ok=True)
    csv_path = os.path.join(out_dir, "inference.csv")
    with open(csv_path, "w") as f:
        f.write("metric,value\n")
        for k, v in metrics.items():
            f.write(f"{k},{v}\n")
    log.info("wrote %s: %s", csv_path, metrics)
    return metrics


def main(argv: Optional[List[str]] = None) -> None:
    ap = argparse.ArgumentParser("deepconsensus eval")
    ap.add_argument("--checkpoint", required=True)
    ap.add_argument("--eval_path", required=True)
    ap.add_argument("--out_dir", required=True)
    ap.add_argument("--limit", type=int, default=-1)
    ap.add_argument("--device", default=None)
    args = ap.parse_args(argv)
    device = args.device or (
        "cuda" if torch.cuda.is_available() else "cpu"
    )
    run_inference(args.out_dir, args.checkpoint, [args.eval_path],
                  device=device, limit=args.limit)


if __name__ == "__main__":
    logging.basicConfig(level=logging.INFO)
    main()
