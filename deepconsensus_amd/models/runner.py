"""Device inference runner: the window -> (bases, QVs) hot path.

MI355X-native execution of the reference's run_model_on_examples
(quick_inference.py:341-415): on GPU the path is
  embed_gather (HIP, K2) -> condenser GEMM (hipBLASLt, K3)
  -> per layer: fused-QKV GEMM -> banded_attn (HIP, K5-K7) -> out-proj GEMM
     -> ReZero add -> FFN GEMMs (bias fused) -> ReZero add, all bf16
  -> fused_ln_head_qv (HIP, K10+K11+K12) emitting uint8 base ids + QVs.
On CPU it falls back to the fp32 torch reference (same numerics contract).

On a GPU machine the HIP extension is required (no silent eager fallback).
"""
from __future__ import annotations

import copy
import math
from typing import List, Optional, Tuple

import torch

from deepconsensus_amd import ops as dc_ops
from deepconsensus_amd.calibration.calibration import (
    parse_calibration_string,
)
from deepconsensus_amd.models.config import Params
from deepconsensus_amd.models.model import (
    EncoderOnlyLearnedValuesTransformer,
    get_model,
)
from deepconsensus_amd.utils import constants
from deepconsensus_amd.utils import trace


def build_gather_tables(model: EncoderOnlyLearnedValuesTransformer):
    """Builds the embed_gather kernel's scaled tables + 16-byte chunk map.

    Folds the sqrt(width) scale, the id-0 zero mask, and the ccs_bq +1 shift
    (networks.py:42-63,492-497) into flat bf16 tables; maps every 8-column
    output chunk to its source (input row, table slice) entries.
    """
    (bi, pwi, ipi, sti, ci, bqi, sni) = model.indices

    def scaled(emb):
        t = emb.table.detach().float() * math.sqrt(emb.width)
        t = t.clone()
        t[0].zero_()
        return t.to(torch.bfloat16)

    tables = {
        "bases": scaled(model.bases_embedding),
        "pw": scaled(model.pw_embedding),
        "ip": scaled(model.ip_embedding),
        "strand": scaled(model.strand_embedding),
        "sn": scaled(model.sn_embedding),
    }
    if model.use_ccs_bq:
        tables["ccs_bq"] = scaled(model.ccs_bq_embedding)

    order = ["bases", "pw", "ip", "strand", "sn"]
    if model.use_ccs_bq:
        order.append("ccs_bq")
    flat_parts, table_base = [], {}
    off = 0
    for name in order:
        t = tables[name]
        table_base[name] = off
        flat = t.reshape(-1)
        # Pad every table to an 8-element boundary so width-8 gathers stay
        # 16-B aligned (required for the kernel's LDS ds_read_b128 path).
        pad = (-flat.numel()) % 8
        if pad:
            flat = torch.cat([flat, flat.new_zeros(pad)])
        flat_parts.append(flat)
        off += flat.numel()
    table_flat = torch.cat(flat_parts).contiguous()

    # Per input row: (table name, shift).
    plan: List[Tuple[str, int]] = []
    plan += [("bases", 0)] * (bi[1] - bi[0])
    plan += [("pw", 0)] * (pwi[1] - pwi[0])
    plan += [("ip", 0)] * (ipi[1] - ipi[0])
    plan += [("strand", 0)] * (sti[1] - sti[0])
    plan += [("bases", 0)] * (ci[1] - ci[0])
    if model.use_ccs_bq:
        plan += [("ccs_bq", 1)]
    plan += [("sn", 0)] * (sni[1] - sni[0])

    row_shift = torch.tensor([s for _, s in plan], dtype=torch.int32)
    row_vocab = torch.tensor(
        [tables[n].shape[0] for n, _ in plan], dtype=torch.int32
    )

    # Column layout -> 8-col chunks -> entries (row, elem_base, width).
    cols: List[Tuple[int, int, int]] = []  # per col: (row, elem_base, width)
    for r, (name, _) in enumerate(plan):
        w = tables[name].shape[1]
        base = table_base[name]
        for j in range(w):
            cols.append((r, base, w))
    concat = len(cols)
    assert concat % 8 == 0, f"concat width {concat} not 8-aligned"
    nchunk = concat // 8
    chunk_cnt = torch.zeros(nchunk, dtype=torch.int32)
    chunk_entries = torch.zeros(nchunk * 4, 4, dtype=torch.int32)
    for c in range(nchunk):
        entries = []
        j = c * 8
        while j < (c + 1) * 8:
            r, base, w = cols[j]
            entries.append((r, base, w))
            j += w
        assert len(entries) <= 4, "chunk spans >4 table rows"
        # Kernel contract: a chunk is either ONE width-8 entry (16-B
        # gather) or up to four width-2 entries (one u32 gather per
        # output dword). Anything else means the row layout misaligns
        # a wide table against the 8-column grid — fail loudly here
        # rather than gather garbage on device.
        if len(entries) == 1:
            assert entries[0][2] == 8, (
                f"chunk {c}: single entry must be width 8, got "
                f"{entries[0][2]} (row layout misaligned)"
            )
        else:
            assert all(w == 2 for _, _, w in entries), (
                f"chunk {c}: multi-entry chunk must be all width-2 "
                f"entries, got {[w for _, _, w in entries]}"
            )
        chunk_cnt[c] = len(entries)
        for k, (r, base, w) in enumerate(entries):
            chunk_entries[c * 4 + k] = torch.tensor([r, base, w, 0])
    return table_flat, row_shift, row_vocab, chunk_cnt, chunk_entries.reshape(-1)


class InferenceRunner:
    """Runs window batches through the model, emitting base ids + QVs."""

    def __init__(
        self,
        params: Params,
        model: Optional[torch.nn.Module] = None,
        device: Optional[str] = None,
        calibration: str = "skip",
        max_qual: int = constants.MAX_QUAL,
    ):
        self.params = params
        if device is None:
            device = "cuda" if torch.cuda.is_available() else "cpu"
        self.device = torch.device(device)
        self.calib = parse_calibration_string(calibration)
        self.max_qual = max_qual
        if model is None:
            model = get_model(params)
        model.eval()
        self.model = model

        self.native = (
            self.device.type == "cuda"
            and isinstance(model, EncoderOnlyLearnedValuesTransformer)
            and model.condense
        )
        if self.device.type == "cuda":
            # The HIP extension is mandatory on-device for the flagship model.
            self.ext = dc_ops.get_ext(required=self.native)
        else:
            self.ext = None

        self.model = model.to(self.device)
        if self.native:
            self._prepare_native(model)

    def _prepare_native(self, model) -> None:
        dev, bf16 = self.device, torch.bfloat16
        tf, rs, rv, cc, ce = build_gather_tables(model)
        self.table_flat = tf.to(dev)
        self.row_shift = rs.to(dev)
        self.row_vocab = rv.to(dev)
        self.chunk_cnt = cc.to(dev)
        self.chunk_entries = ce.to(dev)
        self.win = model.params.get("attn_win_size")
        self.num_heads = model.params["num_heads"]
        # Condenser (no bias): store W^T contiguous for x @ Wt.
        self.cond_wt = (
            model.condenser.weight.detach().t().contiguous().to(bf16).to(dev)
        )
        self.pos = (
            model.pos_encoding.to(bf16).to(dev)
            if model.add_pos_encoding
            else None
        )
        # fused_condense (K3+K4): weight image [Npad, 568] (row n =
        # condenser.weight[n], LDS row stride) + fp32 pos table fused into
        # the epilogue. Production width only; DC_FUSED_CONDENSE=0 falls
        # back to hipBLASLt + separate add.
        import os as _os

        self.cond_img = None
        self.pos_f32 = None
        cw = model.condenser.weight.detach()
        if cw.shape[1] == 560 and _os.environ.get("DC_FUSED_CONDENSE") != "0":
            npad = -(-cw.shape[0] // 64) * 64
            img = torch.zeros(npad, 568, dtype=bf16)
            img[: cw.shape[0], :560] = cw.to(bf16)
            self.cond_img = img.contiguous().to(dev)
            if model.add_pos_encoding:
                self.pos_f32 = (
                    model.pos_encoding.detach().float().contiguous().to(dev)
                )
        # Packed per-layer weights (fast rezero path); generic fallback keeps
        # a bf16 deepcopy of the layer stack for non-rezero configs.
        self.rezero_fast = all(
            l.attn_wrap.rezero and l.ffn_wrap.rezero for l in model.layers
        ) and self.win is not None
        if self.rezero_fast:
            self.layer_w = []
            for l in model.layers:
                wq = l.attn.q_proj.weight.detach()
                wk = l.attn.k_proj.weight.detach()
                wv = l.attn.v_proj.weight.detach()
                wqkv_t = (
                    torch.cat([wq, wk, wv], 0).t().contiguous().to(bf16).to(dev)
                )
                self.layer_w.append(
                    dict(
                        wqkv_t=wqkv_t,
                        wout_t=l.attn.out_proj.weight.detach()
                        .t().contiguous().to(bf16).to(dev),
                        w1_t=l.ffn.filter_layer.weight.detach()
                        .t().contiguous().to(bf16).to(dev),
                        b1=l.ffn.filter_layer.bias.detach().to(bf16).to(dev),
                        w2_t=l.ffn.output_layer.weight.detach()
                        .t().contiguous().to(bf16).to(dev),
                        b2=l.ffn.output_layer.bias.detach().to(bf16).to(dev),
                        alpha_attn=float(l.attn_wrap.alpha.detach()),
                        alpha_ffn=float(l.ffn_wrap.alpha.detach()),
                    )
                )
            # Fold the ReZero alphas into the out-proj / FFN2 weights so the
            # residual adds fuse into the GEMM epilogues (addmm beta=1).
            for lw in self.layer_w:
                lw["wout_t_a"] = lw["wout_t"] * lw["alpha_attn"]
                lw["w2_t_a"] = lw["w2_t"] * lw["alpha_ffn"]
                lw["b2_a"] = lw["b2"] * lw["alpha_ffn"]
            # Padded weights for the fused FFN kernel (in-bounds 16B frags):
            # W1 [2048, 288], W2 [320, 2048], b1 fp32, b2 [320] fp32.
            # The hand-fused FFN kernels measure 1.29-1.32 ms/layer vs
            # 1.80 ms for the hipBLASLt pair with fused epilogues at batch
            # 4096 (profiles/r01_perf_journal.md) - default ON for the
            # production shape; DC_FUSED_FFN=0 falls back to hipBLASLt.
            import os as _os

            self.ffn_fused_ok = (
                model.params["hidden_size"] == 280
                and model.params["filter_size"] == 2048
                and _os.environ.get("DC_FUSED_FFN") != "0"
            )
            self.ffn_v2 = _os.environ.get("DC_FFN_V2", "1") != "0"
            # v3 (256-row tiles, register-resident h) measures fastest;
            # DC_FFN_V3=0 drops back to v2/v1.
            self.ffn_v3 = _os.environ.get("DC_FFN_V3", "1") != "0"

            if self.ffn_fused_ok:
                for i, l in enumerate(model.layers):
                    lw = self.layer_w[i]
                    # Padded projection weights for fused_linear (K5/K7):
                    # QKV rows [840->896, 296], out-proj [320, 296] — the
                    # 296-elem row stride matches the kernel's LDS image so
                    # weights stream by glds (raw row-major copy).
                    wq = l.attn.q_proj.weight.detach().float()
                    wk = l.attn.k_proj.weight.detach().float()
                    wv = l.attn.v_proj.weight.detach().float()
                    wqkv = torch.cat([wq, wk, wv], 0)  # [840, 280]
                    wqkv_pad = torch.zeros(896, 296)
                    wqkv_pad[:840, :280] = wqkv
                    lw["wqkv_pad"] = wqkv_pad.to(bf16).contiguous().to(dev)
                    wo = l.attn.out_proj.weight.detach().float()
                    wout_pad = torch.zeros(320, 296)
                    wout_pad[:280, :280] = wo
                    lw["wout_pad"] = wout_pad.to(bf16).contiguous().to(dev)
                    w1 = l.ffn.filter_layer.weight.detach().float()
                    w1p = torch.zeros(2048, 288)
                    w1p[:, :280] = w1
                    lw["w1_pad"] = w1p.to(bf16).contiguous().to(dev)
                    lw["b1_f32"] = (
                        l.ffn.filter_layer.bias.detach().float().to(dev)
                    )
                    w2 = l.ffn.output_layer.weight.detach().float()
                    w2p = torch.zeros(320, 2048)
                    w2p[:280] = w2
                    lw["w2_pad"] = w2p.to(bf16).contiguous().to(dev)
                    b2p = torch.zeros(320)
                    b2p[:280] = l.ffn.output_layer.bias.detach().float()
                    lw["b2_f32"] = b2p.to(dev)
                    # v2 (glds-pipelined) W1 layout: [2048, 296] with b1
                    # folded into column 287 (constant-1 input column).
                    w1v2 = torch.zeros(2048, 296)
                    w1v2[:, :280] = w1
                    w1v2[:, 287] = l.ffn.filter_layer.bias.detach().float()
                    lw["w1_v2"] = w1v2.to(bf16).contiguous().to(dev)
            # hipBLASLt fused bias+ReLU epilogue, when this torch exposes it.
            self._addmm_act = hasattr(torch, "_addmm_activation")
            if self._addmm_act:
                try:
                    torch._addmm_activation(
                        torch.zeros(4, device=dev, dtype=bf16),
                        torch.zeros(2, 4, device=dev, dtype=bf16),
                        torch.zeros(4, 4, device=dev, dtype=bf16),
                    )
                except Exception:
                    self._addmm_act = False
        else:
            layers = copy.deepcopy(model.layers).to(bf16)
            # LN stays fp32 (the model computes it on x.float(); a bf16
            # LN weight would make torch.layer_norm reject the mix).
            for m in layers.modules():
                if isinstance(m, torch.nn.LayerNorm):
                    m.float()
            self.layers_bf16 = layers.to(dev)
        self.ln_gamma = model.output_norm.weight.detach().float().to(dev)
        self.ln_beta = model.output_norm.bias.detach().float().to(dev)
        self.w_head = model.fc1.weight.detach().float().contiguous().to(dev)
        self.b_head = model.fc1.bias.detach().float().to(dev)

    def _attn(self, qkv: torch.Tensor) -> torch.Tensor:
        b, l, w = qkv.shape
        d = w // (3 * self.num_heads)
        if d == 140 and 32 <= l <= 104 and 1 <= self.win <= 12:
            return self.ext.banded_attn_mfma(
                qkv, self.num_heads, self.win, d ** -0.5
            )
        return self.ext.banded_attn(qkv, self.num_heads, self.win)

    @torch.no_grad()
    def encode_native(self, rows: torch.Tensor) -> torch.Tensor:
        """Native path up to (but excluding) the final LayerNorm: [B,L,H] bf16."""
        with trace.range("K2_embed_gather"):
            emb = self.ext.embed_gather(
                rows.contiguous(), self.table_flat, self.row_shift,
                self.row_vocab, self.chunk_cnt, self.chunk_entries,
            )  # [B, L, concat] bf16
        b, l, _ = emb.shape
        if self.cond_img is not None:
            with trace.range("K3_condenser"):
                h = self.cond_wt.shape[1]
                pos = (
                    self.pos_f32
                    if self.pos_f32 is not None
                    else emb.new_empty(0, dtype=torch.float32)
                )
                x = self.ext.fused_condense(
                    emb.reshape(b * l, -1), self.cond_img, pos, h, l
                )
            x = x.view(b, l, -1)
        else:
            with trace.range("K3_condenser"):
                x = emb.reshape(b * l, -1) @ self.cond_wt  # [B*L, H]
            x = x.view(b, l, -1)
            if self.pos is not None:
                x = x + self.pos[:l]
        if self.rezero_fast:
            h = x.shape[-1]
            flat = x.reshape(b * l, h)
            empty = flat.new_empty(0)
            for lw in self.layer_w:
                with trace.range("K5_qkv_proj"):
                    if self.ffn_fused_ok:
                        qkv = self.ext.fused_linear(
                            flat, lw["wqkv_pad"], empty, empty, 840,
                            False, 0.0
                        ).view(b, l, -1)
                    else:
                        qkv = (flat @ lw["wqkv_t"]).view(b, l, -1)
                with trace.range("K6_banded_attn"):
                    a = self._attn(qkv)
                if self.ffn_fused_ok:
                    with trace.range("K7_out_proj"):
                        flat = self.ext.fused_linear(
                            a.view(b * l, h), lw["wout_pad"], empty, flat,
                            280, False, lw["alpha_attn"],
                        )
                    with trace.range("K9_fused_ffn"):
                        if self.ffn_v3:
                            flat = self.ext.fused_ffn_v3(
                                flat, lw["w1_v2"], lw["w2_pad"],
                                lw["b2_f32"], lw["alpha_ffn"],
                            )
                        elif self.ffn_v2:
                            flat = self.ext.fused_ffn_v2(
                                flat, lw["w1_v2"], lw["w2_pad"],
                                lw["b2_f32"], lw["alpha_ffn"],
                            )
                        else:
                            flat = self.ext.fused_ffn(
                                flat, lw["w1_pad"], lw["b1_f32"],
                                lw["w2_pad"], lw["b2_f32"],
                                lw["alpha_ffn"],
                            )
                    continue
                # Residual fused into the GEMM epilogue (alpha pre-folded).
                flat = torch.addmm(flat, a.view(b * l, h), lw["wout_t_a"])
                if self._addmm_act:
                    ff = torch._addmm_activation(lw["b1"], flat, lw["w1_t"])
                    flat = torch.addmm(flat, ff, lw["w2_t_a"]).add_(lw["b2_a"])
                else:
                    ff = torch.addmm(lw["b1"], flat, lw["w1_t"]).relu_()
                    flat = torch.addmm(flat, ff, lw["w2_t_a"]).add_(lw["b2_a"])
            x = flat.view(b, l, h)
        else:
            for layer in self.layers_bf16:
                x, _ = layer(x, training=False)
        return x

    @torch.no_grad()
    def forward_windows_graphed(self, host_rows: torch.Tensor):
        """Fixed-shape hipGraph-captured serving step (native path).

        Captures H2D-target + embed/encoder/LN-head once per shape and
        replays it thereafter (the same capture bench.py uses; the
        whole-pipeline model stage ran eager and measured ~2x slower
        per window than the graphed bench step). Caller pads the batch
        to a fixed shape; returns device (bases, quals) views.
        Falls back to the eager path if capture fails.
        """
        if not self.native:
            return self.forward_windows(host_rows)
        shape = tuple(host_rows.shape)
        if getattr(self, "_graph_shape", None) != shape:
            try:
                g_in = torch.empty(
                    shape, dtype=host_rows.dtype, device=self.device
                )
                side = torch.cuda.Stream()
                side.wait_stream(torch.cuda.current_stream())
                with torch.cuda.stream(side):
                    for _ in range(2):
                        self.forward_windows(g_in)
                torch.cuda.current_stream().wait_stream(side)
                torch.cuda.synchronize()
                graph = torch.cuda.CUDAGraph()
                with torch.cuda.graph(graph):
                    g_bases, g_quals = self.forward_windows(g_in)
                torch.cuda.synchronize()
                self._graph = graph
                self._graph_in = g_in
                self._graph_out = (g_bases, g_quals)
                self._graph_shape = shape
            except Exception:  # pragma: no cover - capture unsupported
                self._graph_shape = None
                self._graph = None
                return self.forward_windows(host_rows)
        self._graph_in.copy_(host_rows, non_blocking=True)
        self._graph.replay()
        return self._graph_out

    @torch.no_grad()
    def forward_windows(
        self, rows: torch.Tensor, want_probs: bool = False
    ):
        """rows [B, R, L] float32 (device or host) -> (bases u8, quals u8[, probs])."""
        rows = rows.to(self.device, non_blocking=True)
        if self.native:
            x = self.encode_native(rows)
            with trace.range("K10_K12_ln_head_qv"):
                out = self.ext.fused_ln_head_qv(
                x.reshape(-1, x.shape[-1]),
                self.ln_gamma,
                self.ln_beta,
                self.w_head,
                self.b_head,
                float(self.calib.threshold) if self.calib.enabled else -1.0,
                float(self.calib.w) if self.calib.enabled else 1.0,
                float(self.calib.b) if self.calib.enabled else 0.0,
                float(self.max_qual),
                want_probs,
            )
            b, l = rows.shape[0], rows.shape[2]
            bases = out[0].view(b, l)
            quals = out[1].view(b, l)
            if want_probs:
                return bases, quals, out[2].view(b, l, 5)
            return bases, quals
        # Torch reference path.
        probs = self.model(rows)
        return self.probs_to_calls(probs, want_probs)

    def probs_to_calls(self, probs: torch.Tensor, want_probs: bool = False):
        """Reference QV math on softmax output (quick_inference.py:377-389)."""
        pmax, bases = probs.max(dim=-1)
        ep = (1.0 - pmax).clamp_min(1e-12)
        q = -10.0 * torch.log10(ep)
        if self.calib.enabled:
            if self.calib.threshold == 0:
                q = q * self.calib.w + self.calib.b
            else:
                mask = q > self.calib.threshold
                q = torch.where(mask, q * self.calib.w + self.calib.b, q)
        q = q.clamp_max(self.max_qual)
        # numpy round-half-even parity.
        q = torch.round(q).clamp_min(0)
        bases = bases.to(torch.uint8)
        quals = q.to(torch.uint8)
        if want_probs:
            return bases, quals, probs
        return bases, quals
