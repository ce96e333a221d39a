"""Micro-benchmark hipBLASLt formulations of the FFN wgrad GEMMs.

dw1 = dh^T @ x   ([2048, M] x [M, 280]), dw2 = dy^T @ hd ([280, M] x [M, 2048])
at M = 409,600 (train batch 4096). All variants compute the same values;
layouts differ (TN vs NN-with-copy vs NT-transposed-result).
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def timeit(fn, iters=10, warmup=3):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6


def main():
    torch.manual_seed(0)
    M = 409600
    bf16 = torch.bfloat16
    dh = (torch.randn(M, 2048, device="cuda") * 0.1).to(bf16)
    x = (torch.randn(M, 280, device="cuda") * 0.1).to(bf16)
    dy = (torch.randn(M, 280, device="cuda") * 0.1).to(bf16)
    hd = (torch.randn(M, 2048, device="cuda") * 0.1).to(bf16)

    rows = []
    rows.append(("dw1 TN  dh.t()@x", timeit(lambda: dh.t() @ x)))
    rows.append(("dw1 NT  (x.t()@dh).t()", timeit(lambda: (x.t() @ dh).t())))
    rows.append(("dw1 via mm out fp32",
                 timeit(lambda: torch.mm(dh.t(), x.float().to(bf16)))))
    rows.append(("dw2 TN  dy.t()@hd", timeit(lambda: dy.t() @ hd)))
    rows.append(("dw2 NT  (hd.t()@dy).t()", timeit(lambda: (hd.t() @ dy).t())))
    # fp32-accumulate output dtype
    rows.append(("dw1 TN out via addmm beta0",
                 timeit(lambda: torch.addmm(
                     torch.zeros(2048, 280, device="cuda", dtype=bf16),
                     dh.t(), x, beta=0))))
    # chunked K accumulation (4 chunks) — mimics coarse split-K
    def chunked(a, b, n):
        out = None
        step = M // n
        for i in range(n):
            p = a[i * step:(i + 1) * step].t() @ b[i * step:(i + 1) * step]
            out = p if out is None else out + p
        return out
    rows.append(("dw1 chunked x4", timeit(lambda: chunked(dh, x, 4))))
    rows.append(("dw2 chunked x4", timeit(lambda: chunked(dy, hd, 4))))
    for name, us in rows:
        print(f"{name:28s} {us:8.0f} us")
    # correctness spot check
    a = (dh.t() @ x).float()
    b = (x.t() @ dh).t().float()
    print("TN vs NT max diff:", (a - b).abs().max().item())


if __name__ == "__main__":
    main()
