"""Fused-backward embedding stack for training.

The learned-values model embeds 85 input rows through 5-6 shared tables
(networks.py:368-520). Autograd's backward for that chain runs per-table
sort-based scatters plus the permute/concat backward — ~24% of the bf16
training step. This Function keeps the forward as plain torch ops (bit-
identical to EncoderOnlyLearnedValuesTransformer.embed) and replaces the
whole backward with one HIP kernel (ops/hip/embed_grad.hip) that
LDS-accumulates every table's gradient in a single pass. On CPU (or
without the extension) the backward falls back to index_add_ per block.
"""
from __future__ import annotations

from typing import List, Tuple

import torch
import torch.nn.functional as F


class EmbedMeta:
    """Per-block/per-row maps for the stack (built once per model)."""

    def __init__(self, model):
        # Unique tables in a fixed order; the CCS block reuses 'bases'.
        names = ["bases", "pw", "ip", "strand", "sn"]
        if model.use_ccs_bq:
            names.append("ccs_bq")
        self.table_attrs = [f"{n}_embedding" for n in names]
        embs = {n: getattr(model, f"{n}_embedding") for n in names}
        tbase, off = {}, 0
        for n in names:
            tbase[n] = off
            off += embs[n].table.numel()
        self.total_elems = off

        (bi, pwi, ipi, sti, ci, bqi, sni) = model.indices
        # blocks: (table_name, r0, r1, shift)
        blocks = [
            ("bases", bi[0], bi[1], 0),
            ("pw", pwi[0], pwi[1], 0),
            ("ip", ipi[0], ipi[1], 0),
            ("strand", sti[0], sti[1], 0),
            ("bases", ci[0], ci[1], 0),
        ]
        if model.use_ccs_bq:
            blocks.append(("ccs_bq", bqi[0], bqi[1], 1))
        blocks.append(("sn", sni[0], sni[1], 0))
        self.blocks: List[Tuple[int, int, int, int, int, float]] = []
        r_shift, r_vocab, r_tbase, r_width, r_col, r_scale = (
            [], [], [], [], [], []
        )
        col = 0
        for name, r0, r1, shift in blocks:
            e = embs[name]
            w = e.width
            scale = float(w) ** 0.5
            ti = names.index(name)
            self.blocks.append((ti, r0, r1, shift, col, scale))
            for _ in range(r0, r1):
                r_shift.append(shift)
                r_vocab.append(e.table.shape[0])
                r_tbase.append(tbase[name])
                r_width.append(w)
                r_col.append(col)
                r_scale.append(scale)
                col += w
        self.concat_width = col
        self.names = names
        self.table_slices = []
        for n in names:
            t = embs[n].table
            self.table_slices.append(
                (tbase[n], t.shape[0], t.shape[1])
            )
        i32 = torch.int32
        self.row_shift = torch.tensor(r_shift, dtype=i32)
        self.row_vocab = torch.tensor(r_vocab, dtype=i32)
        self.row_tbase = torch.tensor(r_tbase, dtype=i32)
        self.row_width = torch.tensor(r_width, dtype=i32)
        self.row_col = torch.tensor(r_col, dtype=i32)
        self.row_scale = torch.tensor(r_scale, dtype=torch.float32)
        self._dev_cache = {}

    def on(self, device):
        key = str(device)
        if key not in self._dev_cache:
            self._dev_cache[key] = tuple(
                t.to(device)
                for t in (self.row_shift, self.row_vocab, self.row_tbase,
                          self.row_width, self.row_col, self.row_scale)
            )
        return self._dev_cache[key]


def _forward_blocks(rows_f: torch.Tensor, meta: EmbedMeta, tables):
    """The model's embed math (scale + id-0 mask + permute/concat)."""
    b, r, l = rows_f.shape
    ids_all = rows_f.long()
    parts = []
    for ti, r0, r1, shift, _col, scale in meta.blocks:
        ids = ids_all[:, r0:r1] + shift
        nr = r1 - r0
        w = tables[ti].shape[1]
        e = F.embedding(ids.reshape(b, nr * l), tables[ti])
        e = e.reshape(b, nr, l, w) * scale
        e = e * (ids != 0).unsqueeze(-1).to(e.dtype)
        parts.append(e.permute(0, 2, 1, 3).reshape(b, l, nr * w))
    return torch.cat(parts, dim=-1)


def _gather_maps(meta: EmbedMeta, model, device):
    """Shape-only chunk maps for the embed_gather kernel (cached) — the
    table CONTENTS are rebuilt per call in _device_table_flat."""
    if getattr(meta, "_gmaps", None) is None:
        from deepconsensus_amd.models.runner import build_gather_tables

        tf, rs, rv, cc, ce = build_gather_tables(model)
        meta._gmaps = {
            "flat_len": tf.numel(),
            "row_shift": rs,
            "row_vocab": rv,
            "chunk_cnt": cc,
            "chunk_entries": ce,
        }
        meta._gmaps_dev = {}
    key = str(device)
    if key not in meta._gmaps_dev:
        g = meta._gmaps
        meta._gmaps_dev[key] = tuple(
            g[k].to(device)
            for k in ("row_shift", "row_vocab", "chunk_cnt",
                      "chunk_entries")
        )
    return meta._gmaps_dev[key], meta._gmaps["flat_len"]


def _device_table_flat(meta: EmbedMeta, tables, flat_len, device):
    """table_flat rebuilt from the LIVE fp32 tables (sqrt-width scale,
    id-0 row zeroed, 8-elem padding — the build_gather_tables layout)."""
    parts = []
    for ti, name in enumerate(meta.names):
        t = tables[ti]
        w = t.shape[1]
        s = t.detach().float() * (float(w) ** 0.5)
        s = torch.cat([torch.zeros_like(s[:1]), s[1:]], dim=0)
        flat = s.to(torch.bfloat16).reshape(-1)
        pad = (-flat.numel()) % 8
        if pad:
            flat = torch.cat([flat, flat.new_zeros(pad)])
        parts.append(flat)
    out = torch.cat(parts)
    assert out.numel() == flat_len, (out.numel(), flat_len)
    return out.to(device)


class EmbedStackFunction(torch.autograd.Function):
    @staticmethod
    def forward(ctx, rows_f: torch.Tensor, meta: EmbedMeta, use_gather,
                model_ref, *tables):
        ctx.meta = meta
        ctx.save_for_backward(rows_f)
        with torch.no_grad():
            if use_gather:
                from deepconsensus_amd import ops as dc_ops

                ext = dc_ops.get_ext(required=True)
                maps, flat_len = _gather_maps(meta, model_ref,
                                              rows_f.device)
                tf = _device_table_flat(meta, tables, flat_len,
                                        rows_f.device)
                return ext.embed_gather(rows_f.contiguous(), tf, *maps)
            return _forward_blocks(rows_f, meta, tables)

    @staticmethod
    def backward(ctx, grad_out: torch.Tensor):
        meta: EmbedMeta = ctx.meta
        (rows_f,) = ctx.saved_tensors
        grad_out = grad_out.float()
        flat = None
        if rows_f.is_cuda:
            from deepconsensus_amd import ops as dc_ops

            ext = dc_ops.get_ext(required=True)
            flat = torch.zeros(
                meta.total_elems, dtype=torch.float32,
                device=rows_f.device,
            )
            maps = meta.on(rows_f.device)
            ext.embed_grad(rows_f.contiguous(), grad_out.contiguous(),
                           *maps, flat)
            grads = [
                flat[o:o + v * w].view(v, w)
                for (o, v, w) in meta.table_slices
            ]
        else:
            grads = [
                torch.zeros(v, w, dtype=torch.float32)
                for (_o, v, w) in meta.table_slices
            ]
            b, r, l = rows_f.shape
            ids_all = rows_f.long()
            for ti, r0, r1, shift, col, scale in meta.blocks:
                nr = r1 - r0
                w = grads[ti].shape[1]
                ids = (ids_all[:, r0:r1] + shift).clamp_(
                    0, grads[ti].shape[0] - 1
                )
                g = grad_out[:, :, col:col + nr * w].reshape(b, l, nr, w)
                g = g.permute(0, 2, 1, 3).reshape(-1, w) * scale
                mask = (ids != 0).reshape(-1, 1).to(g.dtype)
                grads[ti].index_add_(0, ids.reshape(-1), g * mask)
        return (None, None, None, None) + tuple(grads)


_EMBED_FWD_GATHER = None


def _embed_fwd_gather_enabled() -> bool:
    """HIP embed_gather as the training FORWARD too (DC_EMBED_FWD=0
    falls back to the torch gather/concat chain). The table contents
    are rebuilt on-device from the live fp32 tables each call (~10k
    elements); the chunk maps are shape-only and cached."""
    global _EMBED_FWD_GATHER
    if _EMBED_FWD_GATHER is None:
        import os

        if os.environ.get("DC_EMBED_FWD", "1") == "0":
            _EMBED_FWD_GATHER = False
            return False
        try:
            from deepconsensus_amd import ops as dc_ops

            ext = dc_ops.get_ext()
            _EMBED_FWD_GATHER = bool(
                ext is not None and hasattr(ext, "embed_gather")
            )
        except Exception:  # pragma: no cover
            _EMBED_FWD_GATHER = False
    return _EMBED_FWD_GATHER


def embed_stack(model, rows_f: torch.Tensor) -> torch.Tensor:
    """Differentiable embedding stack with the fused backward."""
    if getattr(model, "_embed_meta", None) is None:
        model._embed_meta = EmbedMeta(model)
    meta = model._embed_meta
    tables = [getattr(model, a).table for a in meta.table_attrs]
    use_gather = (
        rows_f.is_cuda
        and torch.is_autocast_enabled()
        and _embed_fwd_gather_enabled()
    )
    return EmbedStackFunction.apply(rows_f, meta, use_gather, model,
                                    *tables)
