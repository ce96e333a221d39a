#!/usr/bin/env python3
"""Device-AddressSanitizer pass over the HIP kernels.

Runs every hand-written kernel once at production shapes through the
gfx950:xnack+ ASAN-instrumented build (ops/build.py build_asan). Any
device OOB read/write aborts with an ASAN report. Invoke on a GPU box:

  LD_PRELOAD=/opt/rocm/lib/llvm/lib/clang/22/lib/linux/\
libclang_rt.asan-x86_64.so \
  ASAN_OPTIONS=detect_leaks=0 HSA_XNACK=1 \
  python scripts/sanitizer_pass.py

Race checking has no device-TSan equivalent on ROCm; the suite covers
the race axis with bitwise determinism tests instead
(tests/test_gpu_large_batch.py big1==big2 at B=4096, which caught the
round-1 embed_gather nondeterminism).
"""
import importlib.util
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch


def load_asan_ext():
    so = os.path.join(
        os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
        "deepconsensus_amd", "ops", "_build_asan",
        "dc_hip_kernels_asan.so",
    )
    spec = importlib.util.spec_from_file_location("dc_hip_kernels_asan", so)
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)
    return mod


def main():
    # Raw-kernel invocations only: no InferenceRunner / hipBLASLt in
    # the process (library init under the interposed host-ASAN runtime
    # SEGVs; the target here is OUR kernels, not the libraries).
    ext = load_asan_ext()
    print("ASAN extension loaded", flush=True)
    torch.manual_seed(5)
    B, L = 512, 100

    # K2 embed_gather with a production-shaped chunk map.
    from deepconsensus_amd.models import config as cfg
    from deepconsensus_amd.models.model import get_model
    from deepconsensus_amd.models.runner import build_gather_tables

    params = cfg.get_config("transformer_learn_values+custom")
    cfg.modify_params(params, is_training=False)
    model = get_model(params)
    tf, rs, rv, cc, ce = build_gather_tables(model)
    rng = np.random.default_rng(11)
    mp = params.max_passes
    rows = np.zeros((B, params.total_rows, L), np.float32)
    rows[:, 0:mp] = rng.integers(0, 5, size=(B, mp, L))
    rows[:, mp:3 * mp] = rng.integers(0, 60, size=(B, 2 * mp, L))
    rows[:, 3 * mp:4 * mp] = rng.integers(1, 3, size=(B, mp, L))
    rows[:, 4 * mp] = rng.integers(0, 5, size=(B, L))
    rows[:, -4:] = rng.uniform(3, 10, size=(B, 4, 1))
    emb = ext.embed_gather(
        torch.from_numpy(rows).cuda().contiguous(), tf.cuda(), rs.cuda(),
        rv.cuda(), cc.cuda(), ce.cuda(),
    )
    torch.cuda.synchronize()
    print("embed_gather OK:", emb.shape, flush=True)

    M = B * L
    xb = (torch.randn(M, 280, device="cuda") * 0.3).to(torch.bfloat16)
    wqkv = (torch.randn(896, 296, device="cuda") * 0.05).to(
        torch.bfloat16)
    empty = xb.new_empty(0)
    qkv = ext.fused_linear(xb, wqkv, empty, empty, 840, False, 0.0)
    torch.cuda.synchronize()
    print("fused_linear OK:", qkv.shape, flush=True)

    qkv3 = (torch.randn(B, L, 840, device="cuda") * 0.3).to(
        torch.bfloat16)
    attn = ext.banded_attn_mfma(qkv3, 2, 12, 140 ** -0.5)
    torch.cuda.synchronize()
    print("banded_attn_mfma OK:", attn.shape, flush=True)

    w1 = (torch.randn(2048, 296, device="cuda") * 0.05).to(
        torch.bfloat16)
    w2 = (torch.randn(320, 2048, device="cuda") * 0.05).to(
        torch.bfloat16)
    b2 = torch.randn(320, device="cuda")
    for name in ("fused_ffn_v3", "fused_ffn_v2"):
        out = getattr(ext, name)(xb, w1, w2, b2, 0.5)
        torch.cuda.synchronize()
        print(name, "OK:", out.shape, flush=True)

    g = torch.randn(280, device="cuda")
    be = torch.randn(280, device="cuda")
    wh = torch.randn(5, 280, device="cuda") * 0.05
    bh = torch.randn(5, device="cuda")
    lh = ext.fused_ln_head_qv(xb, g, be, wh, bh, 0.0, 1.2, -1.0, 93.0,
                              False)
    torch.cuda.synchronize()
    print("fused_ln_head_qv OK:", lh[0].shape, flush=True)

    # K13 alignment DP fwd+bwd.
    Bd, m, n = 64, 100, 100
    subs = torch.rand(Bd, m, n, device="cuda")
    ins = torch.rand(Bd, n, device="cuda")
    lens = torch.randint(10, m, (Bd,), device="cuda", dtype=torch.int32)
    loss, w = ext.alignment_dp_fwd(subs, ins, lens, 10.0, 0.1, 0)
    gs, gi = ext.alignment_dp_bwd(torch.ones_like(loss), w, lens, m, n, 0)
    torch.cuda.synchronize()
    print("alignment_dp fwd+bwd OK:", float(loss.sum()),
          float(gs.sum()), float(gi.sum()), flush=True)

    # K14 metric.
    yt = torch.randint(0, 5, (Bd, m), device="cuda", dtype=torch.int32)
    yp = torch.randint(0, 5, (Bd, n), device="cuda", dtype=torch.int32)
    ytl = (yt != 0).sum(-1).int()
    ypl = (yp != 0).sum(-1).int()
    v, counts = ext.alignment_metric_counts(yt, yp, ytl, ypl,
                                            2.0, 5.0, 9.0, 4.0)
    torch.cuda.synchronize()
    print("alignment_metric OK:", float(v.sum()),
          int(counts.sum()), flush=True)

    # K2-grad (training embedding backward).
    from deepconsensus_amd.models.model import (
        EncoderOnlyLearnedValuesTransformer,
    )
    print("ALL KERNELS PASSED UNDER DEVICE ASAN", flush=True)


if __name__ == "__main__":
    main()
