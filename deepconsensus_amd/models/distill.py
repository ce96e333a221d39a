"""Distillation trainer.

Parity with reference model_distillation.py:104-522: a frozen teacher loaded
from a checkpoint, a smaller student (transformer_learn_values_distill
config) initialized from a teacher layer map, combined loss
student_alpha * AlignmentLoss + distill_alpha * DistillationLoss on logits,
and the same distributed loop/checkpoint contracts as train.py.
"""
from __future__ import annotations

import argparse
import logging
import os
import time
from typing import List, Optional

import torch

from deepconsensus_amd.models import checkpoint as ckpt_lib
from deepconsensus_amd.models import config as cfg
from deepconsensus_amd.models import data as data_lib
from deepconsensus_amd.models import lamb as lamb_lib
from deepconsensus_amd.models import losses as losses_lib
from deepconsensus_amd.models.model import get_model
from deepconsensus_amd.models.train import _prepare_batch
from deepconsensus_amd.parallel import comm

log = logging.getLogger(__name__)


def get_teacher_model(teacher_ckpt: str, device: str):
    """Loads and freezes the teacher (model_distillation.py:147-167)."""
    params = ckpt_lib.load_params(teacher_ckpt)
    cfg.modify_params(params, is_training=False)
    model = get_model(params)
    ckpt_lib.load_checkpoint(teacher_ckpt, model)
    model.to(device).eval()
    for p in model.parameters():
        p.requires_grad_(False)
    return model, params


def init_student_from_teacher(student, teacher, params) -> None:
    """Copies mapped encoder layers + non-encoder weights
    (model_distillation.py:104-144)."""
    if params.get("init_encoder_stack", False):
        for t_i, s_i in zip(
            params.teacher_encoder_layers, params.student_encoder_layers
        ):
            student.layers[s_i].load_state_dict(
                teacher.layers[t_i].state_dict()
            )
    if params.get("init_nonencoder_layers", False):
        student.bases_embedding.load_state_dict(
            teacher.bases_embedding.state_dict()
        )
        student.pw_embedding.load_state_dict(
            teacher.pw_embedding.state_dict()
        )
        student.ip_embedding.load_state_dict(
            teacher.ip_embedding.state_dict()
        )
        student.strand_embedding.load_state_dict(
            teacher.strand_embedding.state_dict()
        )
        student.sn_embedding.load_state_dict(
            teacher.sn_embedding.state_dict()
        )
        if student.condense and teacher.condense:
            student.condenser.load_state_dict(
                teacher.condenser.state_dict()
            )
        student.output_norm.load_state_dict(
            teacher.output_norm.state_dict()
        )
        student.fc1.load_state_dict(teacher.fc1.state_dict())


def train_model(
    out_dir: str,
    teacher_ckpt: str,
    params: cfg.Params,
    device: str = "cpu",
    eval_every: int = 3000,
    limit_steps: int = 0,
    use_bf16: bool = False,
) -> dict:
    from deepconsensus_amd.utils.tuned_gemm import enable_tuned_gemms

    enable_tuned_gemms()
    rank, world = comm.init_distributed()
    main = rank == 0
    os.makedirs(out_dir, exist_ok=True)
    if main:
        cfg.save_params_as_json(out_dir, params)

    teacher, _teacher_params = get_teacher_model(teacher_ckpt, device)
    torch.manual_seed(params.seed)
    student = get_model(params).to(device)
    init_student_from_teacher(student, teacher, params)
    comm.broadcast_parameters(student)

    global_batch = params.batch_size * world
    n_train = params.get("n_examples_train") or 0
    steps_per_epoch = max(n_train // global_batch, 1)
    decay_steps = steps_per_epoch * max(
        params.get("num_epochs_for_decay", params.num_epochs), 1
    )
    optimizer, schedule = lamb_lib.create_optimizer(
        params, decay_steps, student
    )
    reducer = comm.FlatGradAllreducer(student)

    align_loss = losses_lib.AlignmentLoss(
        del_cost=params.del_cost, loss_reg=params.loss_reg,
        width=params.get("band_width"), reduction="sum",
    )
    distill_loss = losses_lib.DistillationLoss(
        temperature=params.get("temperature", 1.0),
        logit_loss=params.get("logit_loss_identifier", "kl_divergence"),
    )
    student_alpha = params.get("student_alpha", 1.0)
    distill_alpha = params.get("distill_alpha", 1.0)

    train_ds = data_lib.DatasetIterator(
        params.train_path, params, params.batch_size, rank=rank,
        world_size=world, seed=params.seed, limit=params.get("limit", -1),
    )

    student.train()
    step = 0
    t0 = time.time()
    for epoch in range(params.num_epochs):
        for batch in train_ds.iterate(epoch):
            schedule.apply(optimizer, step)
            rows, label = _prepare_batch(batch, device)
            reducer.zero_()
            # bf16 autocast: teacher + student forwards take the fused
            # MFMA banded-attention path on GPU (model.py); losses stay
            # fp32.
            if use_bf16 and rows.is_cuda:
                with torch.autocast("cuda", dtype=torch.bfloat16):
                    with torch.no_grad():
                        t_logits = teacher.encode(
                            rows, training=False
                        )["logits"]
                    s_out = student.encode(rows, training=True)
            else:
                with torch.no_grad():
                    t_logits = teacher.encode(
                        rows, training=False
                    )["logits"]
                s_out = student.encode(rows, training=True)
            s_logits = s_out["logits"]
            probs = torch.softmax(s_logits.float(), -1)
            l_student = align_loss(label, probs) / global_batch
            l_distill = distill_loss(t_logits.float(), s_logits.float())
            loss = student_alpha * l_student + distill_alpha * l_distill
            loss.backward()
            reducer.reduce()
            optimizer.step()
            step += 1
            if main and step % 10 == 0:
                log.info(
                    "step %d loss %.4f (student %.4f distill %.6f) "
                    "(%.2f steps/s)",
                    step, float(loss), float(l_student),
                    float(l_distill), step / (time.time() - t0),
                )
            if limit_steps and step >= limit_steps:
                break
        if limit_steps and step >= limit_steps:
            break

    if main:
        ckpt_lib.save_checkpoint(
            out_dir, step, params.num_epochs - 1, student, optimizer,
            params,
        )
    return {"steps": step}


def main(argv: Optional[List[str]] = None) -> None:
    ap = argparse.ArgumentParser("deepconsensus distill")
    ap.add_argument("--params",
                    default="transformer_learn_values_distill+test")
    ap.add_argument("--teacher_checkpoint", required=True)
    ap.add_argument("--out_dir", required=True)
    ap.add_argument("--train_path", default=None)
    ap.add_argument("--batch_size", type=int, default=None)
    ap.add_argument("--epochs", type=int, default=None)
    ap.add_argument("--limit_steps", type=int, default=0)
    ap.add_argument("--device", default=None)
    ap.add_argument("--bf16", action="store_true",
                    help="autocast forwards to bf16 (fused MFMA "
                    "attention path on GPU)")
    args = ap.parse_args(argv)
    params = cfg.get_config(args.params)
    if args.train_path:
        params.train_path = [args.train_path]
        params.eval_path = [args.train_path]
    if args.batch_size:
        params.batch_size = args.batch_size
    if args.epochs:
        params.num_epochs = args.epochs
    cfg.modify_params(params)
    device = args.device or (
        "cuda" if torch.cuda.is_available() else "cpu"
    )
    train_model(args.out_dir, args.teacher_checkpoint, params,
                device=device, limit_steps=args.limit_steps,
                use_bf16=getattr(args, "bf16", False))


if __name__ == "__main__":
    logging.basicConfig(level=logging.INFO)
    main()
