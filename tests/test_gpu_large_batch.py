"""Large-batch bisection of the native inference chain (GPU).

Regression suite for the (fixed) round-1 large-batch corruption: each
test isolates one stage of the alpha=0 chain at B=4096 against a torch
reference, so any future failure names the broken kernel directly.
Root cause + forensics: profiles/r02_embed_gather_bug.md.

A fresh-box repro (profiles/r01_perf_journal.md, gap-regression section)
showed forward_windows producing ~87% gap calls at B=1908/4096 vs 6% at
B=64 with identical input distribution — a batch-size-dependent
corruption somewhere in the native chain. With random-init weights the
ReZero alphas are 0, so the residual stream reduces to
embed_gather -> condenser matmul -> +pos -> 6x identity passthrough
(fused_linear / fused_ffn) -> fused_ln_head_qv.

Each test below isolates ONE stage at large batch against a torch
reference, so a failure names the broken kernel directly.
"""
import numpy as np
import pytest
import torch

from deepconsensus_amd.models import config as cfg
from deepconsensus_amd.models.model import get_model
from deepconsensus_amd.models.runner import InferenceRunner

pytestmark = pytest.mark.gpu

B_LARGE = 4096


def _runner():
    params = cfg.get_config("transformer_learn_values+custom")
    cfg.modify_params(params, is_training=False)
    torch.manual_seed(1234)
    r = InferenceRunner(params, get_model(params), device="cuda")
    assert r.native
    if "wout_pad" not in r.layer_w[0]:
        pytest.skip("fused GEMM path disabled in this configuration")
    return r


def _rows(params, b, seed=7):
    rng = np.random.default_rng(seed)
    mp = params.max_passes
    L = params.max_length
    rows = np.zeros((b, params.total_rows, L), np.float32)
    rows[:, 0:mp] = rng.integers(0, 5, size=(b, mp, L))
    rows[:, mp : 3 * mp] = rng.integers(0, 60, size=(b, 2 * mp, L))
    rows[:, 3 * mp : 4 * mp] = rng.integers(1, 3, size=(b, mp, L))
    rows[:, 4 * mp] = rng.integers(0, 5, size=(b, L))
    rows[:, -4:] = rng.uniform(3, 10, size=(b, 4, 1))
    return rows


def test_embed_gather_large_batch_matches_small():
    """embed_gather at B=4096: every row must equal the same row run in
    a small (non-persistent-loop) batch."""
    r = _runner()
    rows = torch.from_numpy(_rows(r.params, B_LARGE)).cuda()
    big = r.ext.embed_gather(rows.contiguous(), r.table_flat, r.row_shift,
                             r.row_vocab, r.chunk_cnt, r.chunk_entries)
    # Reference: same kernel at B=64 (proven healthy) over slices.
    for start in (0, 1024, 2048, B_LARGE - 64):
        small = r.ext.embed_gather(
            rows[start : start + 64].contiguous(), r.table_flat,
            r.row_shift, r.row_vocab, r.chunk_cnt, r.chunk_entries)
        assert torch.equal(big[start : start + 64], small), (
            f"embed_gather diverges at rows {start}..{start + 64}"
        )


def test_condenser_matmul_large_m():
    """The hipBLASLt condenser GEMM at M=409600 vs fp32 reference and
    vs the same rows at small M (algorithm-selection differences)."""
    r = _runner()
    M = B_LARGE * 100
    emb = (torch.randn(M, r.cond_wt.shape[0], device="cuda") * 0.2).to(
        torch.bfloat16
    )
    big = emb @ r.cond_wt
    # Split-k algorithm changes with M can reorder accumulation, so
    # compare against an fp32 reference with tolerance, head and tail.
    for s in (slice(0, 4096), slice(M - 4096, M)):
        ref = emb[s].float() @ r.cond_wt.float()
        err = (big[s].float() - ref).abs().max().item()
        assert err < 0.5, (
            f"condenser GEMM rows {s} wrong (max err {err})"
        )


def test_fused_linear_passthrough_large_m():
    """alpha=0 + residual: out must be bitwise == resid at M=409600."""
    r = _runner()
    M = B_LARGE * 100
    x = (torch.randn(M, 280, device="cuda") * 0.1).to(torch.bfloat16)
    resid = (torch.randn(M, 280, device="cuda") * 0.1).to(torch.bfloat16)
    lw = r.layer_w[0]
    out = r.ext.fused_linear(x, lw["wout_pad"], x.new_empty(0), resid,
                             280, False, 0.0)
    bad = (out != resid).any(dim=1)
    assert not bad.any(), (
        f"fused_linear alpha=0 passthrough broke {int(bad.sum())} of {M} "
        f"rows; first bad row {int(bad.nonzero()[0])}"
    )


def test_fused_ffn_passthrough_large_m():
    """alpha=0 FFN: out must be bitwise == x at M=409600 (whichever
    fused FFN variant the runner selected)."""
    r = _runner()
    M = B_LARGE * 100
    x = (torch.randn(M, 280, device="cuda") * 0.1).to(torch.bfloat16)
    lw = r.layer_w[0]
    if getattr(r, "ffn_v3", False):
        out = r.ext.fused_ffn_v3(x, lw["w1_v2"], lw["w2_pad"],
                                 lw["b2_f32"], 0.0)
    elif getattr(r, "ffn_v2", False):
        out = r.ext.fused_ffn_v2(x, lw["w1_v2"], lw["w2_pad"],
                                 lw["b2_f32"], 0.0)
    else:
        out = r.ext.fused_ffn(x, lw["w1_pad"], lw["b1_f32"], lw["w2_pad"],
                              lw["b2_f32"], 0.0)
    bad = (out != x).any(dim=1)
    assert not bad.any(), (
        f"fused FFN alpha=0 passthrough broke {int(bad.sum())} of {M} "
        f"rows; first bad row {int(bad.nonzero()[0])}"
    )


def test_ln_head_large_n_matches_small():
    """fused_ln_head_qv at N=409600 equals itself at small N slices."""
    r = _runner()
    N = B_LARGE * 100
    x = (torch.randn(N, 280, device="cuda") * 0.5).to(torch.bfloat16)
    args = (r.ln_gamma, r.ln_beta, r.w_head, r.b_head,
            -1.0, 1.0, 0.0, 93.0, False)
    big_b, big_q = r.ext.fused_ln_head_qv(x, *args)
    for start in (0, 8192, 100000, N - 4096):
        sb, sq = r.ext.fused_ln_head_qv(x[start : start + 4096], *args)
        assert torch.equal(big_b[start : start + 4096], sb), (
            f"ln_head bases diverge at rows {start}.."
        )
        assert torch.equal(big_q[start : start + 4096], sq), (
            f"ln_head quals diverge at rows {start}.."
        )


def test_banded_attn_large_batch_matches_small():
    """banded_attn_mfma at B*H=8192 (persistent loop) equals B=64."""
    r = _runner()
    L, D, H = 100, 140, 2
    qkv = (torch.randn(B_LARGE, L, 3 * H * D, device="cuda") * 0.3).to(
        torch.bfloat16
    )
    scale = 1.0 / (D ** 0.5)
    big = r.ext.banded_attn_mfma(qkv, H, 12, scale)
    again = r.ext.banded_attn_mfma(qkv, H, 12, scale)
    assert torch.equal(big, again), (
        "banded_attn_mfma is nondeterministic at large batch (race)"
    )
    for start in (0, 256, 1024, B_LARGE - 64):
        small = r.ext.banded_attn_mfma(
            qkv[start : start + 64].contiguous(), H, 12, scale)
        assert torch.equal(big[start : start + 64], small), (
            f"banded_attn_mfma diverges at items {start}.."
        )


def test_encode_native_equals_condensed_input_at_alpha0():
    """With random-init ReZero alphas (all 0) the whole encoder stack is
    an identity over cond+pos: encode_native at B=4096 must equal the
    directly computed embed->condense->+pos tensor."""
    r = _runner()
    rows = torch.from_numpy(
        _rows(r.params, B_LARGE).astype(np.int16)
    ).cuda()
    x = r.encode_native(rows)
    emb = r.ext.embed_gather(rows.contiguous(), r.table_flat, r.row_shift,
                             r.row_vocab, r.chunk_cnt, r.chunk_entries)
    b, l, _ = emb.shape
    if r.cond_img is not None:
        # Same kernel as encode_native so the identity stays bitwise.
        pos = (
            r.pos_f32
            if r.pos_f32 is not None
            else emb.new_empty(0, dtype=torch.float32)
        )
        want = r.ext.fused_condense(
            emb.reshape(b * l, -1), r.cond_img, pos, r.cond_wt.shape[1], l
        ).view(b, l, -1)
    else:
        want = (emb.reshape(b * l, -1) @ r.cond_wt).view(b, l, -1)
        if r.pos is not None:
            want = want + r.pos[:l]
    bad = (x != want).any(dim=-1)
    assert not bad.any(), (
        f"alpha=0 encoder is not identity for {int(bad.sum())} of {b * l} "
        f"positions; first bad (window, pos) = "
        f"{tuple(int(v) for v in bad.nonzero()[0])}"
    )


def test_forward_windows_large_batch_matches_torch():
    """End-to-end regression: native forward at B=4096 must agree with
    the torch fp32 path on >=99% of base calls (random weights)."""
    r = _runner()
    rows_np = _rows(r.params, B_LARGE)
    bases_n, _ = r.forward_windows(
        torch.from_numpy(rows_np.astype(np.int16))
    )
    model = r.model.float()
    with torch.no_grad():
        agree = []
        for i in range(0, B_LARGE, 512):
            probs = model(
                torch.from_numpy(rows_np[i : i + 512]).cuda(),
                training=False,
            )
            bt = probs.argmax(-1)
            agree.append((bases_n[i : i + 512].long() == bt).float().mean())
        rate = torch.stack(agree).mean().item()
    assert rate > 0.99, (
        f"native vs torch base agreement {rate:.4f} at B={B_LARGE} "
        "(large-batch corruption)"
    )


def test_train_step_bitwise_deterministic_large_batch():
    """Fixed seeds => two identical fwd+bwd passes produce bitwise-equal
    losses and grads through the fully-fused bf16 training path
    (attention pair + ffn_train pair + embed_gather forward). This is
    the tripwire class that caught the r1 embed_gather corruption and
    the r2 dgrad allocator hazards."""
    from deepconsensus_amd.models import config as cfg
    from deepconsensus_amd.models.model import get_model

    params = cfg.get_config("transformer_learn_values+custom")
    cfg.modify_params(params, is_training=True)

    def one_pass():
        torch.manual_seed(123)
        model = get_model(params).cuda()
        rows = torch.from_numpy(
            _rows(params, 2048).astype(np.float32)
        ).cuda()
        labels = torch.randint(
            0, 5, (2048, params.max_length), device="cuda"
        )
        torch.manual_seed(77)
        with torch.autocast("cuda", dtype=torch.bfloat16):
            probs = model(rows, training=True)
            logp = torch.log(probs.float().clamp_min(1e-9))
            loss = torch.nn.functional.nll_loss(
                logp.reshape(-1, 5), labels.reshape(-1)
            )
        loss.backward()
        grads = {
            n: p.grad.detach().clone()
            for n, p in model.named_parameters()
            if p.grad is not None
        }
        return loss.detach().clone(), grads

    l1, g1 = one_pass()
    l2, g2 = one_pass()
    assert torch.equal(l1, l2), (l1.item(), l2.item())
    assert g1.keys() == g2.keys() and len(g1) > 10
    for n in g1:
        if "embedding" in n:
            # embed_grad merges per-block LDS accumulators with global
            # fp32 atomics — order-dependent rounding by design; bound
            # it tightly instead of bitwise.
            d = (g1[n] - g2[n]).abs().max().item()
            m = g1[n].abs().max().item() + 1e-9
            assert d < 1e-4 * m + 1e-6, (n, d, m)
        else:
            assert torch.equal(g1[n], g2[n]), n
