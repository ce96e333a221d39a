#!/usr/bin/env python3
"""Training-step throughput benchmark (BASELINE config #4 evidence).

Times the full training step — forward, AlignmentLoss (HIP wavefront DP on
GPU), backward, fused-bucket all-reduce (when WORLD_SIZE > 1), LAMB — on
synthetic labeled windows. Reference anchor: ~4.1 h per 100 M examples at
global batch 8192 on a TPU v2-8 (docs/train_tpu_model.md:283-327)
= ~6780 examples/s.
"""
import argparse
import os
import sys
import time

import numpy as np
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from deepconsensus_amd.models import config as cfg
from deepconsensus_amd.models import lamb as lamb_lib
from deepconsensus_amd.models import losses as losses_lib
from deepconsensus_amd.models.model import get_model
from deepconsensus_amd.parallel import comm


def main():
    from deepconsensus_amd.utils.tuned_gemm import enable_tuned_gemms

    enable_tuned_gemms()
    ap = argparse.ArgumentParser()
    ap.add_argument("--batch-size", type=int, default=256)
    ap.add_argument("--steps", type=int, default=30)
    ap.add_argument("--warmup", type=int, default=5)
    ap.add_argument("--bf16", action="store_true")
    ap.add_argument("--graph", action="store_true",
                    help="hipGraph-capture the whole train step (fwd + "
                    "AlignmentLoss + bwd + capturable LAMB) and replay "
                    "it; see scripts/graph_bisect.py for the capture "
                    "validation ladder")
    args = ap.parse_args()

    rank, world = comm.init_distributed()
    have_cuda = torch.cuda.is_available()
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    device = f"cuda:{local_rank}" if have_cuda else "cpu"
    if have_cuda:
        torch.cuda.set_device(local_rank)

    params = cfg.get_config("transformer_learn_values+custom")
    cfg.modify_params(params)
    torch.manual_seed(7)
    model = get_model(params).to(device)
    comm.broadcast_parameters(model)
    optimizer, schedule = lamb_lib.create_optimizer(
        params, 10000, model, capturable=args.graph
    )
    reducer = comm.FlatGradAllreducer(model)
    loss_fn = losses_lib.AlignmentLoss(
        del_cost=params.del_cost, loss_reg=params.loss_reg, reduction="sum"
    )

    B = args.batch_size
    rng = np.random.default_rng(42 + rank)
    rows = np.zeros((B, params.total_rows, 100), dtype=np.float32)
    mp = params.max_passes
    rows[:, 0:mp] = rng.integers(0, 5, size=(B, mp, 100))
    rows[:, mp:2 * mp] = rng.integers(0, 256, size=(B, mp, 100))
    rows[:, 2 * mp:3 * mp] = rng.integers(0, 256, size=(B, mp, 100))
    rows[:, 3 * mp:4 * mp] = rng.integers(0, 3, size=(B, mp, 100))
    rows[:, 4 * mp] = rng.integers(0, 5, size=(B, 100))
    rows[:, -4:] = rng.integers(5, 30, size=(B, 4, 1))
    x = torch.from_numpy(rows).to(device)
    label = torch.from_numpy(
        rng.integers(0, 5, size=(B, 100)).astype(np.int64)
    ).to(device)
    global_batch = B * world

    def inner_step():
        reducer.zero_()
        if args.bf16 and have_cuda:
            with torch.autocast("cuda", dtype=torch.bfloat16):
                probs = model(x, training=True)
        else:
            probs = model(x, training=True)
        loss = loss_fn(label, probs.float()) / global_batch
        loss.backward()
        reducer.reduce()
        optimizer.step()
        return loss

    def step(i):
        schedule.apply(optimizer, i)
        return inner_step()

    graph = None
    if args.graph:
        assert have_cuda and world == 1, (
            "--graph currently captures the single-rank step"
        )
        # Warmup on a side stream, then capture the full step
        # (grad-zeroing included) and replay it thereafter.
        schedule.apply(optimizer, 0)
        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            for _ in range(3):
                inner_step()
        torch.cuda.current_stream().wait_stream(side)
        torch.cuda.synchronize()
        graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(graph):
            inner_step()
        torch.cuda.synchronize()

        def step(i):  # noqa: F811
            schedule.apply(optimizer, i)
            graph.replay()
            return None

    for i in range(args.warmup):
        step(i)
    if have_cuda:
        torch.cuda.synchronize()
    comm.barrier()
    t0 = time.perf_counter()
    for i in range(args.steps):
        step(i)
    if have_cuda:
        torch.cuda.synchronize()
    elapsed = time.perf_counter() - t0
    comm.barrier()
    if rank == 0:
        sps = args.steps / elapsed
        print(
            f"train step: {elapsed / args.steps * 1000:.1f} ms "
            f"({sps:.2f} steps/s, {sps * global_batch:.0f} examples/s, "
            f"global batch {global_batch}, world {world}, "
            f"graph={bool(graph)})"
        )


if __name__ == "__main__":
    main()
