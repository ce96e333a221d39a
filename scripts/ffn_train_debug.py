"""Isolate the ffn_train_dgrad mismatch: compare dx and dh_pre piecewise."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from deepconsensus_amd import ops as dc_ops


def main():
    ext = dc_ops.get_ext(required=True)
    torch.manual_seed(3)
    bf16 = torch.bfloat16
    M, p = 512, 0.0
    x = (torch.randn(M, 280, device="cuda") * 0.5).to(bf16)
    w1 = torch.randn(2048, 280, device="cuda") * 0.05
    b1 = torch.randn(2048, device="cuda") * 0.1
    w2 = torch.randn(280, 2048, device="cuda") * 0.02
    b2 = torch.zeros(280, device="cuda")
    w1_img = torch.zeros(2048, 296, dtype=bf16, device="cuda")
    w1_img[:, :280] = w1.to(bf16)
    w1_img[:, 287] = b1.to(bf16)
    w2_img = torch.zeros(320, 2048, dtype=bf16, device="cuda")
    w2_img[:280] = w2.to(bf16)
    y, hd = ext.ffn_train_fwd(x, w1_img, w2_img, b2, p, 7)

    dy = (torch.randn(M, 280, device="cuda") * 0.5).to(bf16)
    w2t_img = torch.zeros(2048, 296, dtype=bf16, device="cuda")
    w2t_img[:, :280] = w2.to(bf16).t()
    w1t_img = torch.zeros(320, 2048, dtype=bf16, device="cuda")
    w1t_img[:280] = w1.to(bf16).t()
    dx, dh = ext.ffn_train_dgrad(dy, hd, w2t_img, w1t_img, p)

    dhd_ref = dy.float() @ w2  # [M, 2048]
    mask = (hd.float() > 0).float()
    dh_ref = dhd_ref * mask
    dx_ref = dh_ref @ w1

    e_dh = (dh.float() - dh_ref).abs()
    e_dx = (dx.float() - dx_ref).abs()
    print("dh err max/mean:", e_dh.max().item(), e_dh.mean().item(),
          " ref scale:", dh_ref.abs().max().item())
    print("dx err max/mean:", e_dx.max().item(), e_dx.mean().item(),
          " ref scale:", dx_ref.abs().max().item())
    # Where is dh wrong? column histogram of bad entries.
    bad = (e_dh > 0.05).nonzero()
    if bad.numel():
        print("bad dh entries:", bad.shape[0], "first:", bad[:8].tolist())
        cols = bad[:, 1]
        print("col range:", int(cols.min()), int(cols.max()),
              "col%64 hist:", torch.bincount(cols % 64, minlength=64)[:32].tolist())
        rows = bad[:, 0]
        print("row range:", int(rows.min()), int(rows.max()),
              "row%256 min/max:", int((rows % 256).min()), int((rows % 256).max()))
    # dhd WITHOUT mask: run dgrad against an all-positive hd.
    dx2, dh2 = ext.ffn_train_dgrad(dy, torch.ones_like(hd), w2t_img, w1t_img, p)
    e2 = (dh2.float() - dhd_ref).abs()
    print("unmasked dhd err max/mean:", e2.max().item(), e2.mean().item())
    n2 = torch.isnan(dh2.float())
    print("dh2 NaN count:", int(n2.sum()), "of", dh2.numel())
    if n2.any():
        locs = n2.nonzero()[:6].tolist()
        print("NaN locs:", locs)
    # sample values at a known-bad column
    print("dh2[0,18], ref:", dh2.float()[0, 18].item(), dhd_ref[0, 18].item())
    print("dh2[0,50], ref:", dh2.float()[0, 50].item(), dhd_ref[0, 50].item())
    print("dh2[0,17], ref:", dh2.float()[0, 17].item(), dhd_ref[0, 17].item())
    # Probe 1: zero B1 weights -> dh must be exactly 0.
    _, dh0 = ext.ffn_train_dgrad(dy, torch.ones_like(hd),
                                 torch.zeros_like(w2t_img), w1t_img, p)
    z = dh0.float()
    print("zero-w2t: dh nonzero:", int((z != 0).sum()), "NaN:",
          int(torch.isnan(z).sum()))
    if (z != 0).any():
        bl = (z != 0).nonzero()[:6].tolist()
        print("  nonzero locs:", bl, "vals:",
              [z[tuple(i)].item() for i in bl])
    # Probe 2: zero B2 weights -> dx must be exactly 0 (dh unaffected).
    dxz, _ = ext.ffn_train_dgrad(dy, torch.ones_like(hd), w2t_img,
                                 torch.zeros_like(w1t_img), p)
    zz = dxz.float()
    print("zero-w1t: dx nonzero:", int((zz != 0).sum()), "NaN:",
          int(torch.isnan(zz).sum()))
    # Probe 2b: MODE 2 (no mask at all) vs plain dhd.
    dx4, dh4 = ext.ffn_train_dgrad_nomask(dy, hd, w2t_img, w1t_img, p)
    e4 = (dh4.float() - dhd_ref).abs()
    print("MODE2 nomask dh err max/mean:", e4.max().item(), e4.mean().item(),
          "NaN:", int(torch.isnan(dh4.float()).sum()))
    dx4b, dh4b = ext.ffn_train_dgrad_nomask(dy, hd, w2t_img, w1t_img, p)
    print("MODE2 deterministic:", bool(torch.equal(dh4, dh4b)),
          bool(torch.equal(dx4, dx4b)))
    # Probe 3: is it deterministic?
    dx3, dh3 = ext.ffn_train_dgrad(dy, torch.ones_like(hd), w2t_img,
                                   w1t_img, p)
    print("dgrad deterministic:", bool(torch.equal(dh2, dh3)),
          bool(torch.equal(dx2, dx3)))


def probes2():
    """Decisive probes: same images as the WORKING forward, template 2."""
    ext = dc_ops.get_ext(required=True)
    torch.manual_seed(4)
    bf16 = torch.bfloat16
    M = 512
    x = (torch.randn(M, 280, device="cuda") * 0.5).to(bf16)
    w1 = torch.randn(2048, 280, device="cuda") * 0.05
    b1 = torch.randn(2048, device="cuda") * 0.1
    w2 = torch.randn(280, 2048, device="cuda") * 0.02
    w1_img = torch.zeros(2048, 296, dtype=bf16, device="cuda")
    w1_img[:, :280] = w1.to(bf16)
    w1_img[:, 287] = b1.to(bf16)
    w2_img = torch.zeros(320, 2048, dtype=bf16, device="cuda")
    w2_img[:280] = w2.to(bf16)
    ones = torch.ones(M, 2048, dtype=bf16, device="cuda")
    # A: MODE 2 with the EXACT images the working forward uses.
    _, dhA = ext.ffn_train_dgrad_nomask(x, ones, w1_img, w2_img, 0.0)
    refA = x.float() @ w1.t() + b1
    eA = (dhA.float() - refA).abs()
    print("A(M512, fwd-images, MODE2) err max:", eA.max().item(),
          "NaN:", int(torch.isnan(dhA.float()).sum()))
    _, dhA2 = ext.ffn_train_dgrad_nomask(x, ones, w1_img, w2_img, 0.0)
    print("A deterministic:", bool(torch.equal(dhA, dhA2)))
    # B: single block (M=256) — map the corruption precisely.
    for it in range(3):
        _, dhB = ext.ffn_train_dgrad_nomask(x[:256], ones[:256], w1_img,
                                            w2_img, 0.0)
        eB = (dhB.float() - refA[:256]).abs()
        bad = (eB > 0.05) | torch.isnan(dhB.float())
        locs = bad.nonzero()
        print(f"B run{it}: bad={int(bad.sum())} NaN="
              f"{int(torch.isnan(dhB.float()).sum())}")
        if it == 0:
            for r, cidx in locs[:6].tolist():
                got = dhB.float()[r, cidx].item()
                # which (m, h) does this value belong to?
                dm = (refA[:256, cidx].float() - got).abs()
                bm = int(dm.argmin())
                dh_ = (refA[r].float() - got).abs()
                bh = int(dh_.argmin())
                print(f"   (m={r},h={cidx}) got={got:.4g} "
                      f"want={refA[r,cidx].item():.4g} | best m={bm} "
                      f"(d={dm[bm].item():.2g}) best h={bh} "
                      f"(d={dh_[bh].item():.2g})")
    # C: MODE 0 with the dgrad transpose images (relu'd, but checks
    # whether the IMAGE data triggers anything in the working template).
    w2t_img = torch.zeros(2048, 296, dtype=bf16, device="cuda")
    w2t_img[:, :280] = w2.to(bf16).t()
    w1t_img = torch.zeros(320, 2048, dtype=bf16, device="cuda")
    w1t_img[:280] = w1.to(bf16).t()
    zb = torch.zeros(280, device="cuda")
    yC, hC = ext.ffn_train_fwd(x, w2t_img, w1t_img, zb, 0.0, 5)
    refC = torch.relu(x.float() @ w2)
    eC = (hC.float() - refC).abs()
    print("C(MODE0, dgrad-images) err max:", eC.max().item(),
          "NaN:", int(torch.isnan(hC.float()).sum()))
    yC2, hC2 = ext.ffn_train_fwd(x, w2t_img, w1t_img, zb, 0.0, 5)
    print("C deterministic:", bool(torch.equal(hC, hC2)))


if __name__ == "__main__":
    main()
    print("---- probes2 ----")
    probes2()
