#!/usr/bin/env python3
"""DeepConsensus-AMD flagship benchmark: windowed consensus inference.

Measures the BASELINE.json headline metric — ZMWs/sec of v1.2-architecture
model inference on chem2.2-shaped data (15 kb insert => 150 windows of 100 bp
per ZMW, max_passes=20, 85 feature rows) — on synthetic subread windows with
random-init weights, bf16 compute on MI355X.

One timed step = one batch of windows through the full serving path:
pinned-host H2D copy -> fused embed+condense (HIP) -> bf16 encoder ->
fused LN+head+QV (HIP) -> D2H of uint8 base/qual calls.

Reference CPU baseline: 0.76 ZMW/s on one 16-vCPU shard
(docs/quick_start.md:315-320).

Usage: python bench.py [--gpus N] [--steps K] [--warmup W] [--batch-size B]
Under torchrun, reads RANK/LOCAL_RANK/WORLD_SIZE from the environment.
"""
from __future__ import annotations

import argparse
import json
import os
import sys
import time

import numpy as np
import torch

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from deepconsensus_amd.models import config as cfg
from deepconsensus_amd.models.model import get_model
from deepconsensus_amd.models.runner import InferenceRunner

WINDOWS_PER_ZMW = 150  # chem2.2 15 kb insert / 100 bp windows
BASELINE_ZMW_PER_SEC = 0.76


def make_synthetic_windows(params, batch: int, seed: int) -> np.ndarray:
    """Random chem2.2-shaped window feature tensors [B, 85, 100] fp32."""
    rng = np.random.default_rng(seed)
    R, L, mp = params.total_rows, params.max_length, params.max_passes
    rows = np.zeros((batch, R, L), dtype=np.float32)
    # ~20% gaps in subread bases, like spaced-out alignments.
    bases = rng.integers(0, 5, size=(batch, mp, L))
    rows[:, 0:mp] = bases
    rows[:, mp : 2 * mp] = rng.integers(0, 256, size=(batch, mp, L))
    rows[:, 2 * mp : 3 * mp] = rng.integers(0, 256, size=(batch, mp, L))
    rows[:, 3 * mp : 4 * mp] = rng.integers(1, 3, size=(batch, mp, L))
    rows[:, 4 * mp] = rng.integers(0, 5, size=(batch, L))
    rows[:, -4:] = rng.integers(5, 30, size=(batch, 4, 1))
    return rows


used_graphs = False


def pipeline_main(args):
    """Whole-pipeline mode: BAM -> polished FASTQ.

    The product metric is whole-node `run` throughput
    (docs/quick_start.md:315-320), not the serving step alone: this
    drives BAM reading, parallel BGZF decompress, expand/spacing,
    windowing, featurization, batched native model execution, stitching
    and FASTQ writing. With --shards > 1 it runs one `deepconsensus
    run --shard i/N` process per shard over byte-range ZMW index
    sidecars, pinning shard i to GPU i % device_count — the production
    multi-GPU serving topology (embarrassingly parallel, no
    collectives).
    """
    import subprocess
    import tempfile

    from deepconsensus_amd.dcio import bam as bam_lib
    from deepconsensus_amd.utils.synth import make_synth_bams

    have_cuda = torch.cuda.is_available()
    n_gpus = max(torch.cuda.device_count(), 1) if have_cuda else 1
    shards = args.shards or n_gpus
    zmws = args.pipeline_zmws
    with tempfile.TemporaryDirectory() as td:
        t0 = time.perf_counter()
        sub, ccs, _ = make_synth_bams(
            td, zmws, args.pipeline_length, args.pipeline_subreads, 3
        )
        bam_lib.build_zmw_index(sub)
        bam_lib.build_zmw_index(ccs)
        gen_s = time.perf_counter() - t0
        print(f"# generated {zmws} ZMWs x {args.pipeline_subreads} "
              f"subreads x {args.pipeline_length} bp (+index) in "
              f"{gen_s:.1f}s", file=sys.stderr)

        cpus = min(os.cpu_count() or 4, 16)
        stage_s = {}
        if shards == 1:
            from deepconsensus_amd.inference import quick_inference as qi

            torch.manual_seed(1234)
            options = qi.InferenceOptions(
                batch_size=args.batch_size,
                batch_zmws=100, cpus=cpus, min_quality=0,
                skip_windows_above=0,
            )
            t0 = time.perf_counter()
            counter = qi.run(
                subreads_to_ccs=sub, ccs_bam=ccs, checkpoint="random",
                output=os.path.join(td, "out.fastq"), options=options,
            )
            elapsed = time.perf_counter() - t0
            success = counter.success
            stage_s = {}
            try:
                with open(os.path.join(td, "out.runtime.csv")) as f:
                    next(f)
                    for line in f:
                        parts = line.strip().split(",")
                        stage_s[parts[1]] = round(
                            stage_s.get(parts[1], 0.0) + float(parts[2]),
                            2,
                        )
            except OSError:
                pass
        else:
            per_shard_cpus = max(cpus // shards, 1)
            procs = []
            t0 = time.perf_counter()
            for i in range(shards):
                cmd = [
                    sys.executable, "-m", "deepconsensus_amd.cli", "run",
                    "--subreads_to_ccs", sub, "--ccs_bam", ccs,
                    "--checkpoint", "random",
                    "--output", os.path.join(td, f"out_{i}.fastq"),
                    "--batch_size", "8192", "--batch_zmws", "100",
                    "--cpus", str(per_shard_cpus), "--min_quality", "0",
                    "--skip_windows_above", "0",
                    "--shard", f"{i}/{shards}",
                ]
                if have_cuda:
                    cmd += ["--use_only_gpu_index", str(i % n_gpus)]
                procs.append(subprocess.Popen(
                    cmd, cwd=os.path.dirname(os.path.abspath(__file__))
                ))
            rcs = [p.wait() for p in procs]
            elapsed = time.perf_counter() - t0
            assert all(rc == 0 for rc in rcs), f"shard exit codes {rcs}"
            success = 0
            for i in range(shards):
                with open(os.path.join(td, f"out_{i}.fastq")) as f:
                    success += sum(1 for ln in f if ln.startswith("@"))

        result = {
            "metric": "zmw_per_sec_pipeline",
            "value": round(zmws / elapsed, 3),
            "unit": "ZMWs/sec (BAM->FASTQ whole pipeline)",
            "n_gpus": min(shards, n_gpus) if have_cuda else 0,
            "steps": 1,
            "warmup": 0,
            "ms_per_step": round(elapsed * 1000.0, 1),
            "higher_is_better": True,
            "scaling": "strong",
            "vs_baseline": round(zmws / elapsed / BASELINE_ZMW_PER_SEC, 2),
            "dtype": "bf16" if have_cuda else "fp32",
            "data": "synthetic BAMs (15 kb inserts, mutations+insertions)",
            "config": {
                "mode": "whole_pipeline",
                "zmws": zmws,
                "insert_len": args.pipeline_length,
                "subreads": args.pipeline_subreads,
                "shards": shards,
                "cpus": cpus,
                "reads_written": success,
                "stage_seconds": stage_s if shards == 1 else None,
            },
        }
        print(json.dumps(result))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=20)
    ap.add_argument("--warmup", type=int, default=5)
    ap.add_argument("--batch-size", type=int, default=16384)
    ap.add_argument("--pool-batches", type=int, default=4)
    ap.add_argument("--no-graphs", action="store_true",
                    help="disable hipGraph capture of the serving step")
    ap.add_argument("--max-passes", type=int, default=20,
                    help="subread stack depth (BASELINE config #5: 32)")
    ap.add_argument("--windows-per-zmw", type=int, default=None,
                    help="override (24 kb insert: 240)")
    ap.add_argument("--pipeline", action="store_true",
                    help="whole-pipeline mode: BAM -> polished FASTQ "
                    "(reading, spacing, windowing, model, stitch, write) "
                    "instead of the serving step")
    ap.add_argument("--pipeline-zmws", type=int, default=400)
    ap.add_argument("--pipeline-length", type=int, default=15000)
    ap.add_argument("--pipeline-subreads", type=int, default=8)
    ap.add_argument("--shards", type=int, default=0,
                    help="pipeline mode: worker processes (one per GPU "
                    "via modulo; default = visible GPU count)")
    args = ap.parse_args()

    if args.pipeline:
        return pipeline_main(args)

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    distributed = world > 1

    have_cuda = torch.cuda.is_available()
    # Modulo lets a smoke run place multiple ranks on one visible GPU
    # (RCCL permitting); on a full node device_count == nproc and this is
    # the identity mapping.
    dev_idx = local_rank % max(torch.cuda.device_count(), 1) if have_cuda else 0
    device = f"cuda:{dev_idx}" if have_cuda else "cpu"
    if have_cuda:
        torch.cuda.set_device(dev_idx)

    if distributed:
        import torch.distributed as dist

        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29571")
        # DC_BENCH_BACKEND=gloo: rendezvous/reduction over gloo so a
        # 1-GPU box can smoke-test the N>1 rank logic (RCCL refuses two
        # ranks on one device: "Duplicate GPU detected", gpurun_out/
        # rank0.log r2). The driver's 8-GPU run uses the default nccl.
        dist.init_process_group(
            backend=os.environ.get(
                "DC_BENCH_BACKEND", "nccl" if have_cuda else "gloo"
            ),
            rank=rank,
            world_size=world,
            device_id=torch.device(device) if have_cuda else None,
        )

    params = cfg.get_config("transformer_learn_values+custom")
    params.max_passes = args.max_passes
    cfg.modify_params(params, is_training=False)
    windows_per_zmw = args.windows_per_zmw or WINDOWS_PER_ZMW
    batch = args.batch_size if have_cuda else 32

    torch.manual_seed(1234)
    model = get_model(params)
    runner = InferenceRunner(
        params, model, device=device, calibration="0,1.197654,-0.99781"
    )

    # Pinned host pool of pre-featurized batches (the host preprocess stage is
    # pipelined/off-path in production; windows arrive pre-packed). Features
    # fit int16 (max value 500), halving H2D traffic vs float32.
    use_i16 = have_cuda and runner.native
    pool = []
    for i in range(args.pool_batches):
        arr = make_synthetic_windows(params, batch, 97 + i)
        t = torch.from_numpy(arr.astype(np.int16) if use_i16 else arr)
        if have_cuda:
            t = t.pin_memory()
        pool.append(t)

    if have_cuda:
        # Double-buffered H2D on a copy stream, overlapped with compute.
        copy_stream = torch.cuda.Stream()
        dev_bufs = [torch.empty_like(pool[0], device=device) for _ in range(2)]
        ready = [torch.cuda.Event(), torch.cuda.Event()]
        consumed = [torch.cuda.Event(), torch.cuda.Event()]

        def prefetch(i: int):
            buf = i % 2
            with torch.cuda.stream(copy_stream):
                # Don't overwrite a buffer a previous step is still reading.
                copy_stream.wait_event(consumed[buf])
                dev_bufs[buf].copy_(pool[i % len(pool)], non_blocking=True)
                ready[buf].record(copy_stream)

        prefetch(0)

        # hipGraph-capture the whole forward per device buffer: one replay
        # per step instead of ~50 eager launches.
        graphs = None
        global used_graphs
        if not args.no_graphs and runner.native:
            try:
                for b in range(2):  # allocator warmup outside capture
                    runner.forward_windows(dev_bufs[b])
                torch.cuda.synchronize()
                graphs, static_out = [], []
                for b in range(2):
                    g = torch.cuda.CUDAGraph()
                    with torch.cuda.graph(g):
                        static_out.append(runner.forward_windows(dev_bufs[b]))
                    graphs.append(g)
                used_graphs = True
            except Exception as e:  # pragma: no cover - graph support varies
                print(f"# hipGraph capture unavailable ({e}); eager path",
                      file=sys.stderr)
                graphs = None

        def one_step(i: int):
            buf = i % 2
            cur = torch.cuda.current_stream()
            cur.wait_event(ready[buf])
            if graphs is not None:
                graphs[buf].replay()
                bases, quals = static_out[buf]
            else:
                bases, quals = runner.forward_windows(dev_bufs[buf])
            consumed[buf].record(cur)
            prefetch(i + 1)
            # D2H of the uint8 calls (the serving step's output contract).
            return bases.to("cpu", non_blocking=True), quals.to(
                "cpu", non_blocking=True
            )
    else:
        def one_step(i: int):
            rows = pool[i % len(pool)]
            bases, quals = runner.forward_windows(rows)
            return bases.cpu(), quals.cpu()

    # Warmup.
    for i in range(args.warmup):
        one_step(i)
    if have_cuda:
        torch.cuda.synchronize()

    # Output-validity audit: the timed number only counts if the native
    # path computes the same calls as the torch reference AT THIS BATCH
    # (a large-batch corruption was observed on the native path —
    # tests/test_gpu_large_batch.py). Reported in the JSON, not timed.
    native_agree = None
    native_agree_confident = None
    if have_cuda and runner.native:
        with torch.no_grad():
            # Full-batch native forward (the corruption is batch-size
            # dependent), torch reference on head+tail slices. Raw
            # agreement counts bf16-vs-fp32 argmax flips at near-ties
            # (random-init logits are nearly uniform), so also report
            # agreement on positions where the fp32 top-2 probability
            # margin >= 1e-2 — corruption shows up there, rounding
            # doesn't.
            full = pool[0].to(device)
            bases_n, _ = runner.forward_windows(full)
            model_f = runner.model.float()
            agree, agree_conf = [], []
            for s in (slice(0, 128), slice(batch - 128, batch)):
                probs = model_f(full[s].float(), training=False)
                same = bases_n[s].long() == probs.argmax(-1)
                agree.append(same.float().mean())
                top2 = probs.topk(2, dim=-1).values
                confident = (top2[..., 0] - top2[..., 1]) >= 1e-2
                if confident.any():
                    agree_conf.append(same[confident].float().mean())
            native_agree = round(torch.stack(agree).mean().item(), 4)
            if agree_conf:
                native_agree_confident = round(
                    torch.stack(agree_conf).mean().item(), 6
                )
        torch.cuda.synchronize()
    if distributed:
        import torch.distributed as dist

        dist.barrier()

    t0 = time.perf_counter()
    for i in range(args.steps):
        one_step(i)
    if have_cuda:
        torch.cuda.synchronize()
    elapsed = time.perf_counter() - t0
    if distributed:
        import torch.distributed as dist

        dist.barrier()
        t = torch.tensor([elapsed], dtype=torch.float64, device=device if have_cuda else "cpu")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    windows_per_sec = args.steps * batch / elapsed * world
    zmw_per_sec = windows_per_sec / windows_per_zmw
    ms_per_step = elapsed / args.steps * 1000.0

    if rank == 0:
        result = {
            "metric": "zmw_per_sec",
            "value": round(zmw_per_sec, 3),
            "unit": "ZMWs/sec",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": round(zmw_per_sec / BASELINE_ZMW_PER_SEC, 2),
            "dtype": "bf16" if have_cuda else "fp32",
            "data": "synthetic",
            "config": {
                "model": "transformer_learn_values v1.2 (hidden 280, "
                f"6 layers, heads 2, band +-12, max_passes "
                f"{args.max_passes})",
                "global_batch": batch * world,
                "seq_len": 100,
                "windows_per_zmw": windows_per_zmw,
                "parallelism": f"dp{world}",
                "native_kernels": bool(runner.native),
                "native_vs_torch_agree": native_agree,
                "native_vs_torch_agree_confident": native_agree_confident,
                "hipgraph": used_graphs,
            },
        }
        print(json.dumps(result))

    if distributed:
        import torch.distributed as dist

        dist.destroy_process_group()


if __name__ == "__main__":
    main()
