"""Device inference runner: the window -> (bases, QVs) hot path.

MI355X-native execution of the reference's run_model_on_examples
(quick_inference.py:341-415): on GPU the path is
  fused_embed_condense (HIP, K2+K3)
  -> bf16 encoder stack (rocBLAS GEMMs + banded attention)
  -> fused_ln_head_qv (HIP, K10+K11+K12) emitting uint8 base ids + QVs.
On CPU it falls back to the fp32 torch reference (same numerics contract).

On a GPU machine the HIP extension is required (no silent eager fallback).
"""
from __future__ import annotations

import math
from typing import Optional, Tuple

import numpy as np
import torch

from deepconsensus_amd import ops as dc_ops
from deepconsensus_amd.calibration.calibration import (
    QualityCalibrationValues,
    parse_calibration_string,
)
from deepconsensus_amd.models.config import Params, get_indices
from deepconsensus_amd.models.model import (
    EncoderOnlyLearnedValuesTransformer,
    get_model,
)
from deepconsensus_amd.utils import constants


def build_fused_tables(
    model: EncoderOnlyLearnedValuesTransformer,
) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor, torch.Tensor]:
    """Precomputes per-row fused embed+condense tables.

    T'_row[id] = (table_f[id] * sqrt(w) * (id != 0)) @ Wc[row_cols] so the
    device kernel reduces K2+K3 to R gather-adds of H-wide bf16 rows.

    Returns (fused_table [V_total, H] bf16, row_offset, row_shift, row_vocab).
    """
    p = model.params
    H = p["transformer_input_size"]
    W = model.condenser.weight.detach().float()  # [H, concat_dim]
    (bi, pwi, ipi, sti, ci, bqi, sni) = model.indices

    def scaled(emb):
        t = emb.table.detach().float() * math.sqrt(emb.width)
        t = t.clone()
        t[0].zero_()
        return t

    tables = {
        "bases": scaled(model.bases_embedding),
        "pw": scaled(model.pw_embedding),
        "ip": scaled(model.ip_embedding),
        "strand": scaled(model.strand_embedding),
        "sn": scaled(model.sn_embedding),
    }
    if model.use_ccs_bq:
        tables["ccs_bq"] = scaled(model.ccs_bq_embedding)

    # Row plan in input-row order with concat-column tracking.
    plan = []  # (table_name, shift)
    for _ in range(bi[0], bi[1]):
        plan.append(("bases", 0))
    for _ in range(pwi[0], pwi[1]):
        plan.append(("pw", 0))
    for _ in range(ipi[0], ipi[1]):
        plan.append(("ip", 0))
    for _ in range(sti[0], sti[1]):
        plan.append(("strand", 0))
    for _ in range(ci[0], ci[1]):
        plan.append(("bases", 0))
    if model.use_ccs_bq:
        plan.append(("ccs_bq", 1))
    for _ in range(sni[0], sni[1]):
        plan.append(("sn", 0))

    fused_rows = []
    row_offset, row_shift, row_vocab = [], [], []
    col = 0
    offset = 0
    for name, shift in plan:
        t = tables[name]
        w = t.shape[1]
        cols = W[:, col : col + w]  # [H, w]
        fused = t @ cols.T  # [vocab, H]
        fused_rows.append(fused)
        row_offset.append(offset)
        row_shift.append(shift)
        row_vocab.append(t.shape[0])
        offset += t.shape[0]
        col += w
    assert col == W.shape[1], (col, W.shape)
    fused_table = torch.cat(fused_rows, 0).to(torch.bfloat16).contiguous()
    return (
        fused_table,
        torch.tensor(row_offset, dtype=torch.int32),
        torch.tensor(row_shift, dtype=torch.int32),
        torch.tensor(row_vocab, dtype=torch.int32),
    )


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

        if self.native:
            ft, ro, rs, rv = build_fused_tables(model)
            self.fused_table = ft.to(self.device)
            self.row_offset = ro.to(self.device)
            self.row_shift = rs.to(self.device)
            self.row_vocab = rv.to(self.device)
            # Encoder stack in bf16 on device; LN/head params fp32. Deep-copied
            # so the fp32 reference model stays intact for validation.
            import copy

            self.model = model.to(self.device)
            self.layers_bf16 = copy.deepcopy(model.layers).to(
                torch.bfloat16
            ).to(self.device)
            self.pos = model.pos_encoding.to(self.device).to(torch.bfloat16) \
                if model.add_pos_encoding else None
            self.ln_gamma = model.output_norm.weight.detach().float().to(
                self.device
            )
            self.ln_beta = model.output_norm.bias.detach().float().to(
                self.device
            )
            self.w_head = model.fc1.weight.detach().float().contiguous().to(
                self.device
            )
            self.b_head = model.fc1.bias.detach().float().to(self.device)
        else:
            self.model = model.to(self.device)

    @torch.no_grad()
    def encode_native(self, rows: torch.Tensor) -> torch.Tensor:
        """Native path up to (but excluding) the final LayerNorm: [B,L,H] bf16."""
        x = self.ext.fused_embed_condense(
            rows.contiguous(),
            self.fused_table,
            self.row_offset,
            self.row_shift,
            self.row_vocab,
        )  # [B, L, H] bf16
        if self.pos is not None:
            x = x + self.pos[: x.shape[1]]
        for layer in self.layers_bf16:
            x, _ = layer(x, training=False)
        return x

    @torch.no_grad()
    def forward_windows(
        self, rows: torch.Tensor, want_probs: bool = False
    ):
        """rows [B, R, L] float32 (device or host) -> (bases u8, quals u8[, probs])."""
        rows = rows.to(self.device, non_blocking=True)
        if self.native:
            x = self.encode_native(rows)
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
