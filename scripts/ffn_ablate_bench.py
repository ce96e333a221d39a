#!/usr/bin/env python3
"""fused_ffn_v3 ablation timing (guide rule 8: ablate before optimize)."""
import sys, os, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from deepconsensus_amd import ops as dc_ops

ext = dc_ops.get_ext(required=True)
M = 4096 * 100
x = (torch.randn(M, 280, device="cuda") * 0.3).to(torch.bfloat16)
w1 = (torch.randn(2048, 296, device="cuda") * 0.05).to(torch.bfloat16)
w2 = (torch.randn(320, 2048, device="cuda") * 0.05).to(torch.bfloat16)
b2 = torch.randn(320, device="cuda")

def timeit(fn, iters=30, warmup=8):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6

full = timeit(lambda: ext.ffn_ablate(x, w1, w2, b2, 0.5, 0))
b1 = timeit(lambda: ext.ffn_ablate(x, w1, w2, b2, 0.5, 1))
b2o = timeit(lambda: ext.ffn_ablate(x, w1, w2, b2, 0.5, 2))
loads = timeit(lambda: ext.ffn_ablate(x, w1, w2, b2, 0.5, 3))
spin = timeit(lambda: ext.ffn_ablate(x, w1, w2, b2, 0.5, 4))
rot = timeit(lambda: ext.ffn_ablate(x, w1, w2, b2, 0.5, 6))
v3 = timeit(lambda: ext.fused_ffn_v3(x, w1, w2, b2, 0.5))
print(f"v3 shipped: {v3:.0f} us")
print(f"ablate full(0): {full:.0f} us")
print(f"B1-only (1):    {b1:.0f} us   (B2 marginal: {full-b1:.0f})")
print(f"B2-only (2):    {b2o:.0f} us  (B1 marginal: {full-b2o:.0f})")
print(f"loads-only (3): {loads:.0f} us (MFMA total marginal: {full-loads:.0f})")
print(f"loads+VALUspin (4): {spin:.0f} us (overlap test: ~loads means DMA progresses under compute)")
print(f"full+chunk-rotation (6): {rot:.0f} us (L2-stable weight set test)")
# correctness of rotation (accumulation order changes -> tolerance)
o0 = ext.ffn_ablate(x, w1, w2, b2, 0.5, 0)
o6 = ext.ffn_ablate(x, w1, w2, b2, 0.5, 6)
d = (o0.float() - o6.float()).abs().max().item()
print(f"rotation maxdiff vs in-order: {d:.5f}")
