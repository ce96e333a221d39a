"""Custom training loop with DP over RCCL/xGMI.

Parity with reference model_train_custom_loop.py:93-367: AlignmentLoss
scaled by the global batch (tf.nn.compute_average_loss semantics), LAMB with
polynomial decay + linear warmup, eval every N steps with checkpointing,
best-checkpoint tracking by eval/per_example_accuracy, checkpoint_metrics.tsv
and resume. Scale-out is one process per GPU with torch.distributed
(nccl=RCCL on ROCm, gloo on CPU) and a single fused flat-buffer gradient
all-reduce per step (parallel/comm.py) instead of the reference's
MirroredStrategy per-variable reduces.
"""
from __future__ import annotations

import argparse
import json
import logging
import os
import time
from typing import List, Optional

import numpy as np
import torch

from deepconsensus_amd.models import checkpoint as ckpt_lib
from deepconsensus_amd.models import config as cfg
from deepconsensus_amd.models import data as data_lib
from deepconsensus_amd.models import lamb as lamb_lib
from deepconsensus_amd.models import losses as losses_lib
from deepconsensus_amd.models.model import get_model
from deepconsensus_amd.parallel import comm
from deepconsensus_amd.utils import trace
from deepconsensus_amd.utils.events import EventWriter

log = logging.getLogger(__name__)


def _prepare_batch(batch, device):
    rows = torch.from_numpy(
        np.ascontiguousarray(batch["rows"][:, :, :, 0])
    ).to(device)
    label = torch.from_numpy(np.ascontiguousarray(batch["label"])).to(device)
    return rows, label


def train_model(
    out_dir: str,
    params: cfg.Params,
    device: str = "cpu",
    eval_every: int = 3000,
    limit_steps: int = 0,
    write_checkpoint_metrics: bool = True,
    warm_start: Optional[str] = None,
    use_bf16: bool = False,
) -> dict:
    """Runs the custom training loop; returns summary metrics."""
    from deepconsensus_amd.utils.tuned_gemm import enable_tuned_gemms

    enable_tuned_gemms()

    rank, world = comm.init_distributed()
    main = rank == 0
    os.makedirs(out_dir, exist_ok=True)
    if main:
        cfg.save_params_as_json(out_dir, params)

    torch.manual_seed(params.seed)
    model = get_model(params).to(device)
    if warm_start:
        ckpt_lib.load_checkpoint(warm_start, model)
        log.info("warm-started from %s", warm_start)
    comm.broadcast_parameters(model)

    n_train = params.get("n_examples_train") or 0
    global_batch = params.batch_size * world
    steps_per_epoch = max(n_train // global_batch, 1)
    decay_steps = steps_per_epoch * max(
        params.get("num_epochs_for_decay", params.num_epochs), 1
    )
    optimizer, schedule = lamb_lib.create_optimizer(
        params, decay_steps, model
    )
    reducer = comm.FlatGradAllreducer(model)

    loss_fn = losses_lib.AlignmentLoss(
        del_cost=params.del_cost,
        loss_reg=params.loss_reg,
        width=params.get("band_width"),
        reduction="sum",
    )

    # Resume.
    resume_path, initial_epoch, step = ckpt_lib.get_checkpoint_and_initial_epoch(
        out_dir
    )
    if resume_path:
        ckpt_lib.load_checkpoint(resume_path, model, optimizer)
        log.info("resumed from %s (epoch %d step %d)", resume_path,
                 initial_epoch, step)

    train_ds = data_lib.DatasetIterator(
        params.train_path, params, params.batch_size, rank=rank,
        world_size=world, seed=params.seed, limit=params.get("limit", -1),
    )
    eval_ds = data_lib.DatasetIterator(
        params.eval_path, params, params.batch_size, shuffle=False,
        rank=rank, world_size=world, limit=params.get("limit", -1),
    )

    def run_eval() -> dict:
        model.eval()
        acc = losses_lib.PerExampleAccuracy()
        align_metric = losses_lib.AlignmentMetric()
        yield_metric = losses_lib.YieldOverCCSMetric()
        total_loss, n_batches = 0.0, 0
        with torch.no_grad():
            for batch in eval_ds.iterate():
                rows, label = _prepare_batch(batch, device)
                probs = model(rows, training=False)
                loss = loss_fn(label, probs.float()) / max(
                    label.shape[0], 1
                )
                total_loss += float(loss)
                n_batches += 1
                # Metrics run on-device when the K14 HIP kernel is
                # available (AlignmentMetric dispatches); CPU otherwise.
                acc.update_state(label, probs)
                ccs_rows = rows[:, 4 * params.max_passes, :]
                ic, ip = losses_lib.get_batch_identity_ccs_pred(
                    ccs_rows, probs, label, align_metric
                )
                yield_metric.update_state(ic, ip)
        model.train()
        n = max(n_batches, 1)
        return {
            "eval/loss": total_loss / n,
            "eval/per_example_accuracy": acc.result(),
            "eval/yield_over_ccs": yield_metric.result(),
        }

    model.train()
    summary = {}
    t0 = time.time()
    steps_this_session = 0
    # TensorBoard-equivalent event files (model_utils.py:549-583):
    # train/ and eval/ writers under <out_dir>/summaries.
    train_writer = eval_writer = None
    if main:
        train_writer = EventWriter(os.path.join(out_dir, "summaries",
                                                "train"))
        eval_writer = EventWriter(os.path.join(out_dir, "summaries",
                                               "eval"))
    for epoch in range(initial_epoch, params.num_epochs):
        for batch in train_ds.iterate(epoch):
            lr = schedule.apply(optimizer, step)
            rows, label = _prepare_batch(batch, device)
            reducer.zero_()
            # bf16 autocast (BASELINE config #4): GEMMs/attention run bf16,
            # LN/softmax and the alignment loss stay fp32; fp32 master
            # weights and fp32 gradient all-reduce.
            with trace.range("train_step"):
                if use_bf16 and rows.is_cuda:
                    with torch.autocast("cuda", dtype=torch.bfloat16):
                        with trace.range("forward"):
                            probs = model(rows, training=True)
                else:
                    with trace.range("forward"):
                        probs = model(rows, training=True)
                # compute_average_loss: sum / global batch
                # (model_train_custom_loop.py:148-154).
                with trace.range("alignment_loss"):
                    loss = loss_fn(label, probs.float()) / global_batch
                with trace.range("backward"):
                    loss.backward()
                with trace.range("allreduce"):
                    reducer.reduce()
                with trace.range("lamb_step"):
                    optimizer.step()
            step += 1
            steps_this_session += 1
            if main and step % 10 == 0:
                steps_per_sec = steps_this_session / (time.time() - t0)
                log.info(
                    "epoch %d step %d loss %.4f lr %.2e (%.2f steps/s)",
                    epoch, step, float(loss) * world, lr, steps_per_sec,
                )
                train_writer.add_scalars(step, {
                    "train/loss": float(loss) * world,
                    "train/learning_rate": lr,
                    "train/steps_per_second": steps_per_sec,
                    "train/epoch": epoch + (step - epoch * steps_per_epoch)
                    / max(steps_per_epoch, 1),
                })
            if step % eval_every == 0 or (
                limit_steps and steps_this_session >= limit_steps
            ):
                metrics = run_eval()
                metrics["eval/loss"] = comm.allreduce_scalar(
                    metrics["eval/loss"]
                ) / world
                if main:
                    name = f"checkpoint-{step}"
                    ckpt_lib.save_checkpoint(
                        out_dir, step, epoch, model, optimizer, params,
                        metrics if write_checkpoint_metrics else None,
                    )
                    ckpt_lib.update_best_checkpoint(
                        out_dir, name,
                        metrics["eval/per_example_accuracy"],
                    )
                    log.info("eval @%d: %s", step, metrics)
                    eval_writer.add_scalars(step, metrics)
                    eval_writer.flush()
                    train_writer.flush()
                summary = metrics
            if limit_steps and steps_this_session >= limit_steps:
                break
        if limit_steps and steps_this_session >= limit_steps:
            break

    # Final eval + checkpoint.
    metrics = run_eval()
    if main:
        ckpt_lib.save_checkpoint(
            out_dir, step, params.num_epochs - 1, model, optimizer, params,
            metrics if write_checkpoint_metrics else None,
        )
        ckpt_lib.update_best_checkpoint(
            out_dir, f"checkpoint-{step}",
            metrics["eval/per_example_accuracy"],
        )
    summary = metrics
    summary["steps"] = step
    if main:
        eval_writer.add_scalars(step, metrics)
        train_writer.close()
        eval_writer.close()
        with open(os.path.join(out_dir, "training_summary.json"), "w") as f:
            json.dump(summary, f, indent=2)
    return summary


def main(argv: Optional[List[str]] = None) -> None:
    ap = argparse.ArgumentParser("deepconsensus train")
    ap.add_argument("--params", default="transformer_learn_values+test",
                    help="<model>+<dataset> config name")
    ap.add_argument("--out_dir", required=True)
    ap.add_argument("--train_path", default=None)
    ap.add_argument("--eval_path", default=None)
    ap.add_argument("--n_examples_train", type=int, default=None)
    ap.add_argument("--n_examples_eval", type=int, default=None)
    ap.add_argument("--batch_size", type=int, default=None)
    ap.add_argument("--epochs", type=int, default=None)
    ap.add_argument("--eval_every", type=int, default=3000)
    ap.add_argument("--limit_steps", type=int, default=0)
    ap.add_argument("--checkpoint", default=None,
                    help="warm-start checkpoint")
    ap.add_argument("--device", default=None)
    ap.add_argument("--bf16", action="store_true",
                    help="autocast forward to bf16 (fp32 loss/LN/softmax)")
    ap.add_argument("--write_checkpoint_metrics", type=lambda v: v != "false",
                    default=True,
                    help="append eval metrics to checkpoint_metrics.tsv")
    ap.add_argument("--eval_and_log_every_step", action="store_true",
                    help="debug: log every step and eval every step")
    args = ap.parse_args(argv)

    params = cfg.get_config(args.params)
    if args.train_path:
        params.train_path = [args.train_path]
    if args.eval_path:
        params.eval_path = [args.eval_path]
    if args.batch_size:
        params.batch_size = args.batch_size
    if args.epochs:
        params.num_epochs = args.epochs
    if args.n_examples_train:
        params.n_examples_train = args.n_examples_train
    if args.n_examples_eval:
        params.n_examples_eval = args.n_examples_eval
    cfg.modify_params(params)
    device = args.device or (
        "cuda" if torch.cuda.is_available() else "cpu"
    )
    # Parity with the reference's retry-on-UnavailableError loop
    # (model_train_custom_loop.py:333-347): transient device/communicator
    # failures resume from the latest checkpoint, bounded at 5 attempts.
    attempts = 0
    while True:
        try:
            train_model(
                args.out_dir, params, device=device,
                eval_every=(1 if args.eval_and_log_every_step
                            else args.eval_every),
                limit_steps=args.limit_steps,
                write_checkpoint_metrics=args.write_checkpoint_metrics,
                warm_start=args.checkpoint, use_bf16=args.bf16,
            )
            break
        except RuntimeError as e:
            attempts += 1
            transient = any(
                k in str(e).lower()
                for k in ("nccl", "rccl", "connection", "timed out",
                          "unavailable")
            )
            if not transient or attempts >= 5:
                raise
            log.warning("transient failure (%s); retrying (%d/5)",
                        e, attempts)


if __name__ == "__main__":
    logging.basicConfig(level=logging.INFO)
    main()
