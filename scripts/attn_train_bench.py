#!/usr/bin/env python3
"""Training attention segment: fused HIP vs torch chain (fwd+bwd)."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from deepconsensus_amd import ops as dc_ops
from deepconsensus_amd.models.model import _BandedAttnTrain

ext = dc_ops.get_ext(required=True)
torch.manual_seed(0)
B, H, T, D, win = 4096, 2, 100, 140, 12

q = (torch.randn(B, H, T, D, device="cuda") * 0.3).to(torch.bfloat16)
k = (torch.randn(B, H, T, D, device="cuda") * 0.3).to(torch.bfloat16)
v = (torch.randn(B, H, T, D, device="cuda") * 0.3).to(torch.bfloat16)
g = torch.randn(B, H, T, D, device="cuda").to(torch.bfloat16)
i = torch.arange(T, device="cuda")
band = (i[:, None] - i[None, :]).abs() <= win

def timeit(fn, iters=20, warmup=5):
    for _ in range(warmup): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6

def torch_seg():
    qq = q.detach().requires_grad_()
    kk = k.detach().requires_grad_()
    vv = v.detach().requires_grad_()
    logits = torch.matmul(qq * (D ** -0.5), kk.transpose(-1, -2))
    logits = logits.masked_fill(~band, -1e9)
    w = torch.softmax(logits.float(), dim=-1).to(qq.dtype)
    w = torch.nn.functional.dropout(w, p=0.1, training=True)
    ctx = torch.matmul(w, vv)
    ctx.backward(g)

def fused_seg():
    qq = q.detach().requires_grad_()
    kk = k.detach().requires_grad_()
    vv = v.detach().requires_grad_()
    mask = torch.rand(B, H, T, 2 * win + 1, device="cuda") >= 0.1
    ctx = _BandedAttnTrain.apply(qq, kk, vv, mask, win, 0.1)
    ctx.backward(g)

from deepconsensus_amd.models.model import _BandedAttnTrainPacked

qkv_p = (torch.randn(B, T, 3 * H * D, device="cuda") * 0.3).to(torch.bfloat16)
g_p = torch.randn(B, T, H * D, device="cuda").to(torch.bfloat16)

def packed_seg():
    qq = qkv_p.detach().requires_grad_()
    mask = torch.rand(B * H, T, 2 * win + 1, device="cuda") >= 0.1
    ctx = _BandedAttnTrainPacked.apply(qq, mask, H, win, 0.1)
    ctx.backward(g_p)

t_torch = timeit(torch_seg)
t_fused = timeit(fused_seg)
t_packed = timeit(packed_seg)
print(f"torch chain fwd+bwd: {t_torch:.0f} us")
print(f"fused HIP v1 fwd+bwd: {t_fused:.0f} us  ({t_torch/t_fused:.2f}x)")
print(f"fused HIP v2 packed:  {t_packed:.0f} us  ({t_torch/t_packed:.2f}x)")
# kernels alone
mask = torch.rand(B, H, T, 2 * win + 1, device="cuda") >= 0.1
tf = timeit(lambda: ext.banded_attn_train_fwd(q, k, v, mask, win, 0.1))
out, p = ext.banded_attn_train_fwd(q, k, v, mask, win, 0.1)
tb = timeit(lambda: ext.banded_attn_train_bwd(q, k, v, p, mask, g, win, 0.1))
print(f"kernel fwd: {tf:.0f} us   kernel bwd: {tb:.0f} us")
qkv_c = qkv_p.contiguous()
mask2 = torch.rand(B * H, T, 2 * win + 1, device="cuda") >= 0.1
tfm = timeit(lambda: ext.banded_attn_mfma_train_fwd(
    qkv_c, H, win, D ** -0.5, mask2, 0.1))
out2, p2 = ext.banded_attn_mfma_train_fwd(qkv_c, H, win, D ** -0.5,
                                          mask2, 0.1)
tbm = timeit(lambda: ext.banded_attn_bwd_mfma(
    qkv_c, p2, mask2, g_p, H, win, 0.1))
tb2 = timeit(lambda: ext.banded_attn_train_bwd2(
    qkv_c, p2, mask2, g_p, H, win, 0.1))
print(f"MFMA fwd: {tfm:.0f} us  MFMA bwd: {tbm:.0f} us  "
      f"(VALU bwd2: {tb2:.0f} us)")
