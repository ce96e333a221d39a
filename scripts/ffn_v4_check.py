#!/usr/bin/env python3
"""fused_ffn_v4 numerics + timing vs v3 and an fp32 reference."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from deepconsensus_amd import ops as dc_ops

ext = dc_ops.get_ext(required=True)
torch.manual_seed(3)
for M in (256, 4096, 409600):
    x = (torch.randn(M, 280, device="cuda") * 0.3).to(torch.bfloat16)
    w1 = (torch.randn(2048, 296, device="cuda") * 0.05).to(torch.bfloat16)
    w1[:, 288:] = 0
    b1 = w1[:, 287].float()
    w2 = (torch.randn(320, 2048, device="cuda") * 0.05).to(torch.bfloat16)
    b2 = torch.randn(320, device="cuda")
    alpha = 0.37
    o3 = ext.fused_ffn_v3(x, w1, w2, b2, alpha)
    o4 = ext.fused_ffn_v4(x, w1, w2, b2, alpha)
    # fp32 reference
    xf = x.float()
    h = (xf @ w1[:, :280].float().t() + b1).relu()
    ref = xf + alpha * (h @ w2[:280].float().t() + b2[:280])
    e3 = (o3.float() - ref).abs().max().item()
    e4 = (o4.float() - ref).abs().max().item()
    d34 = (o3.float() - o4.float()).abs().max().item()
    print(f"M={M}: v3 maxerr {e3:.4f}  v4 maxerr {e4:.4f}  |v3-v4| {d34:.4f}")
    assert e4 < max(2.5 * e3, 0.05), "v4 numerics off"

def timeit(fn, iters=30, warmup=8):
    for _ in range(warmup): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6

M = 409600
x = (torch.randn(M, 280, device="cuda") * 0.3).to(torch.bfloat16)
w1 = (torch.randn(2048, 296, device="cuda") * 0.05).to(torch.bfloat16)
w2 = (torch.randn(320, 2048, device="cuda") * 0.05).to(torch.bfloat16)
b2 = torch.randn(320, device="cuda")
t3 = timeit(lambda: ext.fused_ffn_v3(x, w1, w2, b2, 0.5))
t4 = timeit(lambda: ext.fused_ffn_v4(x, w1, w2, b2, 0.5))
print(f"timing M={M}: v3 {t3:.0f} us   v4 {t4:.0f} us   ({t3/t4:.2f}x)")
