"""Micro-benchmark: fused_condense (K3+K4) vs hipBLASLt matmul + pos add.

Production serving shape: M = 16384*100 rows, 560 -> 280, pos [100, 280].
Run on a GPU box:  python scripts/condense_bench.py
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from deepconsensus_amd import ops as dc_ops


def timeit(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6


def main():
    ext = dc_ops.get_ext(required=True)
    torch.manual_seed(0)
    M, L = 16384 * 100, 100
    x = (torch.randn(M, 560, device="cuda") * 0.5).to(torch.bfloat16)
    w = torch.randn(280, 560, device="cuda") * 0.05
    w_img = torch.zeros(320, 568, device="cuda")
    w_img[:280, :560] = w
    w_img = w_img.to(torch.bfloat16).contiguous()
    wt = w.to(torch.bfloat16).t().contiguous()
    pos = (torch.randn(L, 280, device="cuda") * 0.3).contiguous()
    pos_bf = pos.to(torch.bfloat16)

    t_fused = timeit(lambda: ext.fused_condense(x, w_img, pos, 280, L))

    def plain():
        y = (x @ wt).view(-1, L, 280)
        return y + pos_bf

    t_plain = timeit(plain)
    t_mm = timeit(lambda: x @ wt)

    out = ext.fused_condense(x, w_img, pos, 280, L).float()
    idx = torch.arange(M, device="cuda") % L
    ref = x.float() @ w.t() + pos[idx]
    err = (out - ref).abs().max().item()
    print(
        f"fused_condense {t_fused:.0f} us | hipBLASLt+add {t_plain:.0f} us "
        f"(mm alone {t_mm:.0f} us) | max err {err:.4f}"
    )


if __name__ == "__main__":
    main()
