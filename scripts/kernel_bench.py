#!/usr/bin/env python3
"""Microbenchmark of the custom HIP kernels (within-probe A/B timing)."""
import argparse
import sys
import time

import torch

import os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from deepconsensus_amd import ops as dc_ops


def timeit(fn, iters=50, warmup=10):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6  # us


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--kernel", default="all")
    ap.add_argument("--batch", type=int, default=4096)
    ap.add_argument("--iters", type=int, default=50)
    args = ap.parse_args()
    ext = dc_ops.get_ext(required=True)
    B, L, H, D = args.batch, 100, 2, 140
    torch.manual_seed(0)

    if args.kernel in ("all", "attn"):
        qkv = torch.randn(B, L, 3 * H * D, device="cuda").to(torch.bfloat16)
        us = timeit(lambda: ext.banded_attn_mfma(qkv, H, 12, D ** -0.5),
                    args.iters)
        flops = B * H * 4 * 38 * 32768
        print(f"banded_attn_mfma B={B}: {us:.1f} us  "
              f"({flops / us / 1e6:.0f} GF/s incl. padding)")

    if args.kernel in ("all", "ffn"):
        M = B * L
        x = torch.randn(M, 280, device="cuda").to(torch.bfloat16) * 0.3
        w1 = torch.randn(2048, 288, device="cuda").to(torch.bfloat16) * 0.05
        b1 = torch.randn(2048, device="cuda")
        w2 = torch.randn(320, 2048, device="cuda").to(torch.bfloat16) * 0.05
        b2 = torch.randn(320, device="cuda")
        us = timeit(lambda: ext.fused_ffn(x, w1, b1, w2, b2, 0.5),
                    args.iters)
        flops = 2 * M * 280 * 2048 * 2
        print(f"fused_ffn M={M}: {us:.1f} us ({flops / us / 1e6:.0f} GF/s)")
        w1v2 = torch.randn(2048, 296, device="cuda").to(torch.bfloat16) * 0.05
        us3 = timeit(lambda: ext.fused_ffn_v2(x, w1v2, w2, b2, 0.5),
                     args.iters)
        print(f"fused_ffn_v2 M={M}: {us3:.1f} us "
              f"({flops / us3 / 1e6:.0f} GF/s)")
        us4 = timeit(lambda: ext.fused_ffn_v3(x, w1v2, w2, b2, 0.5),
                     args.iters)
        print(f"fused_ffn_v3 M={M}: {us4:.1f} us "
              f"({flops / us4 / 1e6:.0f} GF/s)")
        # hipBLASLt pair for comparison.
        w1t = torch.randn(280, 2048, device="cuda").to(torch.bfloat16)
        w2t = torch.randn(2048, 280, device="cuda").to(torch.bfloat16)
        b1h = b1.to(torch.bfloat16)
        b2h = b2[:280].to(torch.bfloat16)

        def eager():
            h = torch._addmm_activation(b1h, x, w1t)
            return torch.addmm(x, h, w2t, beta=1)

        us2 = timeit(eager, args.iters)
        print(f"hipblaslt pair M={M}: {us2:.1f} us "
              f"({flops / us2 / 1e6:.0f} GF/s)")

    if args.kernel in ("all", "linear"):
        M = B * L
        x = torch.randn(M, 280, device="cuda").to(torch.bfloat16) * 0.3
        wqkv = torch.randn(896, 296, device="cuda").to(torch.bfloat16) * 0.05
        wout = torch.randn(320, 296, device="cuda").to(torch.bfloat16) * 0.05
        empty = x.new_empty(0)
        us = timeit(lambda: ext.fused_linear(x, wqkv, empty, empty, 840,
                                             False, 0.0), args.iters)
        flops = 2 * M * 280 * 840
        print(f"fused_linear qkv M={M}: {us:.1f} us "
              f"({flops / us / 1e6:.0f} GF/s)")
        a = torch.randn(M, 280, device="cuda").to(torch.bfloat16) * 0.3
        us = timeit(lambda: ext.fused_linear(a, wout, empty, x, 280,
                                             False, 0.07), args.iters)
        flops = 2 * M * 280 * 280
        print(f"fused_linear out+resid M={M}: {us:.1f} us "
              f"({flops / us / 1e6:.0f} GF/s)")

    if args.kernel in ("all", "lnhead"):
        M = B * L
        x = torch.randn(M, 280, device="cuda").to(torch.bfloat16)
        g = torch.randn(280, device="cuda")
        be = torch.randn(280, device="cuda")
        wh = torch.randn(5, 280, device="cuda") * 0.05
        bh = torch.randn(5, device="cuda")
        us = timeit(lambda: ext.fused_ln_head_qv(
            x, g, be, wh, bh, 0.0, 1.197654, -0.99781, 93.0, False),
            args.iters)
        gb = M * (280 * 2 + 2) / 1e9
        print(f"fused_ln_head_qv M={M}: {us:.1f} us ({gb / (us / 1e6):.2f} GB/s)")

    if args.kernel in ("all", "embed"):
        import numpy as np

        from deepconsensus_amd.models import config as cfg
        from deepconsensus_amd.models.model import get_model
        from deepconsensus_amd.models.runner import InferenceRunner

        params = cfg.get_config("transformer_learn_values+custom")
        cfg.modify_params(params, is_training=False)
        model = get_model(params)
        runner = InferenceRunner(params, model, device="cuda")
        rng = np.random.default_rng(0)
        rows = rng.integers(0, 4, size=(args.batch, 85, 100)).astype(
            np.int16
        )
        t = torch.from_numpy(rows).cuda()
        us = timeit(
            lambda: ext.embed_gather(
                t, runner.table_flat, runner.row_shift, runner.row_vocab,
                runner.chunk_cnt, runner.chunk_entries,
            ),
            args.iters,
        )
        gb = args.batch * 100 * (560 * 2 + 85 * 2) / 1e9
        print(f"embed_gather B={args.batch}: {us:.1f} us "
              f"({gb / (us / 1e6):.2f} GB/s)")


if __name__ == "__main__":
    main()
