"""Losses and metrics.

Re-implements the reference's losses_and_metrics.py on torch/numpy:

* AlignmentLoss (:263-609) — differentiable Needleman-Wunsch-style loss:
  cross-entropy substitution costs, -log P(gap) insertion costs, constant
  deletion cost, run over anti-diagonal wavefronts with hard min or soft
  min (-reg * logsumexp(-t/reg)). The banded variant (:413-547) is realized
  by masking cells with |i - j| > width to +inf inside the same wavefront
  (path-equivalent to the reference's woven band) and fetching the result at
  (seq_len, min(n, seq_len + width)) (:549-560 index_ending_band).
  Torch autograd provides the backward pass (the HIP wavefront kernel
  replaces this on-device).
* AlignmentMetric (:666-1058) — hard NW with affine gaps (3-state wavefront
  with argmax direction tracking + vectorized backtrace), producing
  num_matches / insertions / deletions / correct_matches / PID. Numpy.
* PerExampleAccuracy / PerClassAccuracy (:37-89), batch identity +
  YieldOverCCSMetric (:1061-1167), DistillationLoss (:1170-1213).
"""
from __future__ import annotations

from typing import Mapping, Optional, Tuple

import numpy as np
import torch
import torch.nn.functional as F

from deepconsensus_amd.utils import constants

GAP = constants.GAP_INT
EPS = 1e-7
INF = 1e9


def left_shift_sequence(y_true: torch.Tensor) -> torch.Tensor:
    """Removes internal gaps, shifting left (losses_and_metrics.py:92-115)."""
    seq_length = y_true.shape[1]
    ixs = torch.arange(seq_length, device=y_true.device).expand_as(y_true)
    sort_order = torch.where(y_true != GAP, ixs, seq_length + ixs)
    sort_order = torch.sort(sort_order, dim=1).values
    sort_order = torch.where(
        sort_order < seq_length, sort_order, sort_order - seq_length
    )
    return torch.gather(y_true, 1, sort_order)


def xentropy_subs_cost_fn(
    y_true_oh: torch.Tensor, y_pred: torch.Tensor, eps: float = EPS
) -> torch.Tensor:
    """[B,m,n] pointwise cross-entropy (losses_and_metrics.py:123-143)."""
    y_pred = torch.clamp(y_pred, eps, 1 - eps)
    logp = torch.log(y_pred)  # [B, n, K]
    # out[b,i,j] = -sum_k y_true_oh[b,i,k] * log y_pred[b,j,k]
    return -torch.einsum("bik,bjk->bij", y_true_oh, logp)


def xentropy_ins_cost_fn(
    y_pred: torch.Tensor, eps: float = EPS
) -> torch.Tensor:
    """[B,n] insertion costs -log P(gap) (losses_and_metrics.py:191-207)."""
    ins_scores = torch.clamp(y_pred[..., GAP], eps, 1 - eps)
    return -torch.log(ins_scores)


class AlignmentLoss(torch.nn.Module):
    """Differentiable alignment loss (losses_and_metrics.py:263-609)."""

    def __init__(
        self,
        del_cost: float = 1.0,
        loss_reg: Optional[float] = 1.0,
        width: Optional[int] = None,
        reduction: str = "mean",
    ):
        super().__init__()
        self.del_cost = del_cost
        self.loss_reg = loss_reg
        self.width = width
        self.reduction = reduction

    @staticmethod
    def preprocess_y_true(y_true: torch.Tensor):
        y_true = y_true.to(torch.int64)
        y_true = left_shift_sequence(y_true)
        seq_lens = (y_true != GAP).sum(-1)
        y_true_oh = F.one_hot(
            y_true, num_classes=constants.SEQ_VOCAB_SIZE
        ).float()
        return y_true_oh, seq_lens

    @staticmethod
    def preprocess_y_pred(y_pred: torch.Tensor) -> torch.Tensor:
        return y_pred / y_pred.sum(-1, keepdim=True)

    def _minop(self, t: torch.Tensor) -> torch.Tensor:
        if self.loss_reg is None:
            return t.min(dim=0).values
        return -self.loss_reg * torch.logsumexp(-t / self.loss_reg, dim=0)

    def alignment(
        self,
        subs_costs: torch.Tensor,
        ins_costs: torch.Tensor,
        del_cost: float,
        seq_lens: torch.Tensor,
    ) -> torch.Tensor:
        """Wavefront recursion (losses_and_metrics.py:346-411)."""
        b, m, n = subs_costs.shape
        dev = subs_costs.device

        # Wavefrontified views: subs_w[k][i][b] = subs[b][i][k-i],
        # ins_w[k][i][b] = ins[b][k-i] (for i in 0..m).
        L = m + n - 1
        padded = F.pad(subs_costs, (m - 1, m - 1))  # [B, m, n+2m-2]
        gather_idx = (
            (m - 1) - torch.arange(m, device=dev)[:, None]
            + torch.arange(L, device=dev)[None, :]
        ).expand(b, m, L)
        subs_w = torch.gather(padded, 2, gather_idx).permute(2, 1, 0)

        L2 = (m + 1) + n - 1
        padded_i = F.pad(ins_costs, (m, m))  # [B, n+2m]
        gi = (
            m - torch.arange(m + 1, device=dev)[:, None]
            + torch.arange(L2, device=dev)[None, :]
        ).expand(b, m + 1, L2)
        ins_w = torch.gather(
            padded_i.unsqueeze(1).expand(b, m + 1, padded_i.shape[1]), 2, gi
        ).permute(2, 1, 0)

        inf = torch.tensor(INF, device=dev)
        v_opt = torch.full((b,), INF, device=dev)
        v_p2 = F.pad(
            torch.full((m - 1, b), INF, device=dev), (0, 0, 1, 0)
        )
        v_p1 = torch.cat(
            [
                ins_w[0][:1],
                torch.full((1, b), del_cost, device=dev),
                torch.full((m - 1, b), INF, device=dev),
            ],
            0,
        )

        i_range = torch.arange(m + 1, device=dev)
        n_idx = torch.arange(b, device=dev)
        if self.width is None:
            k_end = seq_lens + n
            fetch_i = seq_lens
        else:
            # Band end point: j = n - relu(n - seq_len - width).
            j_end = n - torch.clamp(n - seq_lens - self.width, min=0)
            k_end = seq_lens + j_end
            fetch_i = seq_lens

        for k in range(2, m + n + 1):
            j_range = k - i_range
            inv_mask = ((j_range >= 0) & (j_range <= n))[:, None]
            if self.width is not None:
                inv_mask = inv_mask & (
                    (j_range - i_range).abs() <= self.width
                )[:, None]

            o_m = v_p2 + subs_w[k - 2]
            o_i = v_p1 + ins_w[k - 1]
            v_p2 = v_p1[:-1]
            o_d = v_p2 + del_cost

            body = self._minop(torch.stack([o_m, o_i[1:], o_d]))
            v_p1 = torch.cat([o_i[:1], body], 0)
            v_p1 = torch.where(inv_mask, v_p1, inf)
            hit = k_end == k
            if hit.any():
                v_opt = torch.where(hit, v_p1[fetch_i, n_idx], v_opt)
        return v_opt

    def eval(self, y_true: torch.Tensor, y_pred: torch.Tensor):
        y_true_oh, seq_lens = self.preprocess_y_true(y_true)
        y_pred = self.preprocess_y_pred(y_pred)
        subs_costs = xentropy_subs_cost_fn(y_true_oh, y_pred)
        ins_costs = xentropy_ins_cost_fn(y_pred)
        if subs_costs.is_cuda:
            # HIP wavefront kernel with custom VJP (K13); fails loudly if the
            # extension is missing on a GPU machine.
            from deepconsensus_amd import ops as dc_ops

            ext = dc_ops.get_ext(required=True)
            return _AlignmentDP.apply(
                subs_costs.float(), ins_costs.float(), seq_lens,
                self.del_cost, self.loss_reg, self.width, ext,
            )
        return self.alignment(
            subs_costs, ins_costs, self.del_cost, seq_lens
        )

    def forward(self, y_true: torch.Tensor, y_pred: torch.Tensor):
        per_example = self.eval(y_true, y_pred)
        if self.reduction == "none":
            return per_example
        if self.reduction == "sum":
            return per_example.sum()
        return per_example.mean()


class _AlignmentDP(torch.autograd.Function):
    """HIP wavefront DP (ops/hip/alignment_dp.hip) with saved soft-min
    weights driving the backward gather recursion."""

    @staticmethod
    def forward(ctx, subs, ins, seq_lens, del_cost, loss_reg, width, ext):
        reg = 0.0 if loss_reg is None else float(loss_reg)
        w = 0 if width is None else int(width)
        loss, weights = ext.alignment_dp_fwd(
            subs, ins, seq_lens.to(torch.int32), float(del_cost), reg, w
        )
        ctx.save_for_backward(weights, seq_lens.to(torch.int32))
        ctx.dims = (subs.shape[1], subs.shape[2], w)
        ctx.ext = ext
        return loss

    @staticmethod
    def backward(ctx, grad_out):
        weights, seq_lens = ctx.saved_tensors
        m, n, w = ctx.dims
        grad_subs, grad_ins = ctx.ext.alignment_dp_bwd(
            grad_out.contiguous().float(), weights, seq_lens, m, n, w
        )
        return grad_subs, grad_ins, None, None, None, None, None


# ---------------------------------------------------------------------------
# AlignmentMetric (numpy)
# ---------------------------------------------------------------------------


def _left_shift_np(y: np.ndarray) -> np.ndarray:
    seq_length = y.shape[1]
    ixs = np.broadcast_to(np.arange(seq_length), y.shape)
    order = np.sort(np.where(y != GAP, ixs, seq_length + ixs), axis=1)
    order = np.where(order < seq_length, order, order - seq_length)
    return np.take_along_axis(y, order, axis=1)


class AlignmentMetric:
    """PBMM2-approximating NW alignment metric
    (losses_and_metrics.py:666-1058)."""

    def __init__(
        self,
        matching_score: float = 2.0,
        mismatch_penalty: float = 5.0,
        gap_open_penalty: float = 5.0,
        gap_extend_penalty: float = 4.0,
    ):
        self.matching_score = matching_score
        self.mismatch_penalty = mismatch_penalty
        # PBMM2 convention: open + (len-1)*extend.
        self.gap_open_penalty = gap_open_penalty + gap_extend_penalty
        self.gap_extend_penalty = gap_extend_penalty
        self._pid_sum = 0.0
        self._pid_n = 0

    def _device_alignment(self, y_true, y_pred):
        """K14 on device: affine-NW counts via the HIP wavefront kernel
        (ops/hip/alignment_metric.hip). Same mv contract as the numpy
        path; paths are not materialized (counts only)."""
        from deepconsensus_amd import ops as dc_ops

        ext = dc_ops.get_ext(required=True)
        yt = left_shift_sequence(y_true.long()).int()
        yp_tok = left_shift_sequence(y_pred.argmax(-1).long()).int()
        yt_len = (yt != GAP).sum(-1).int()
        yp_len = (yp_tok != GAP).sum(-1).int()
        v_opt, counts = ext.alignment_metric_counts(
            yt, yp_tok, yt_len, yp_len,
            self.matching_score, self.mismatch_penalty,
            self.gap_open_penalty, self.gap_extend_penalty,
        )
        c = counts.cpu().numpy().astype(np.int64)
        mv = {
            "num_matches": c[:, 0],
            "num_insertions": c[:, 1],
            "num_deletions": c[:, 2],
            "num_correct_matches": c[:, 3],
        }
        mv["alignment_length"] = (
            mv["num_matches"] + mv["num_insertions"] + mv["num_deletions"]
        )
        with np.errstate(divide="ignore", invalid="ignore"):
            unsafe = mv["num_correct_matches"] / mv["alignment_length"]
        mv["pid"] = np.where(mv["alignment_length"] > 0, unsafe, 1.0)
        return v_opt.cpu().numpy(), None, mv

    def alignment(self, y_true, y_pred):
        """Returns (v_opt, paths, metric_values).

        On CUDA inputs with the HIP extension available, runs the K14
        device kernel (paths returned as None); otherwise the numpy
        reference path."""
        if (
            isinstance(y_true, torch.Tensor)
            and y_true.is_cuda
            and isinstance(y_pred, torch.Tensor)
            and y_pred.is_cuda
        ):
            from deepconsensus_amd import ops as dc_ops

            if dc_ops.have_ext():
                return self._device_alignment(y_true, y_pred)
        y_true = np.asarray(
            y_true.detach().cpu().numpy()
            if isinstance(y_true, torch.Tensor)
            else y_true
        )
        y_pred = np.asarray(
            y_pred.detach().cpu().numpy()
            if isinstance(y_pred, torch.Tensor)
            else y_pred
        )
        b = y_true.shape[0]
        y_true = _left_shift_np(y_true.astype(np.int32))
        y_true_lens = (y_true != GAP).sum(-1)
        y_pred_tok = _left_shift_np(
            y_pred.argmax(-1).astype(np.int32)
        )
        y_pred_lens = (y_pred_tok != GAP).sum(-1)
        m, n = y_true.shape[1], y_pred_tok.shape[1]

        ms, mp = self.matching_score, self.mismatch_penalty
        go, ge = self.gap_open_penalty, self.gap_extend_penalty
        subs = np.where(
            y_true[:, :, None] == y_pred_tok[:, None, :], ms, -mp
        ).astype(np.float64)  # [B, m, n]

        # Wavefrontified substitution costs: sw[k][i][b] = subs[b][i][k-i].
        L = m + n - 1
        sw = np.zeros((L, m, b))
        for i in range(m):
            for_k = np.arange(n) + i
            sw[for_k, i, :] = subs[:, i, :].T

        gap_pens = np.array([go, go, ge])[:, None, None]

        # Init k=0 / k=1 (reference :655-712).
        v_all_p2 = np.full((3, m, b), -INF)
        v_all_p2[0, 0, :] = 0.0
        v_all_p1 = np.full((3, m + 1, b), -INF)
        v_all_p1[1, 0, :] = -go
        v_all_p1[2, 1, :] = -go

        dir_all = np.full((m + n + 1, 3, m + 1, b), -2, dtype=np.int32)
        # k=0 directions.
        dir_all[0, 0, 0, :] = -1
        # k=1 directions.
        dir_all[1, 1, 0, :] = 0
        dir_all[1, 2, 1, :] = 0

        v_opt = np.zeros(b)
        m_opt = np.full(b, -1, dtype=np.int32)
        k_end = y_true_lens + y_pred_lens
        samp = np.arange(b)
        i_range = np.arange(m + 1)

        def maybe_update(k, v_opt, m_opt, v_all_p1):
            v_k = v_all_p1.max(axis=0)
            m_k = v_all_p1.argmax(axis=0)
            cond = k_end == k
            v_opt = np.where(cond, v_k[y_true_lens, samp], v_opt)
            m_opt = np.where(cond, m_k[y_true_lens, samp], m_opt)
            return v_opt, m_opt

        v_opt, m_opt = maybe_update(1, v_opt, m_opt, v_all_p1)

        for k in range(2, m + n + 1):
            j_range = k - i_range
            inv = ((j_range >= 0) & (j_range <= n))[None, :, None]
            o_match = v_all_p2 + sw[k - 2][None, :, :] if k - 2 < L else (
                v_all_p2 - INF
            )
            o_ins = v_all_p1[:2] - gap_pens[1:]
            v_all_p2 = v_all_p1[:, :-1]
            o_del = v_all_p2 - gap_pens

            v_match = o_match.max(0)
            d_match = o_match.argmax(0)
            v_ins = o_ins.max(0)
            d_ins = o_ins.argmax(0)
            v_del = o_del.max(0)
            d_del = o_del.argmax(0)

            v_match = np.concatenate(
                [np.full((1, b), -INF), v_match], 0
            )
            v_del = np.concatenate([np.full((1, b), -INF), v_del], 0)
            d_match = np.concatenate(
                [np.full((1, b), -2, np.int32), d_match], 0
            )
            d_del = np.concatenate(
                [np.full((1, b), -2, np.int32), d_del], 0
            )

            v_all_p1 = np.where(
                inv, np.stack([v_match, v_ins, v_del]), -INF
            )
            dir_all[k] = np.stack([d_match, d_ins, d_del])
            v_opt, m_opt = maybe_update(k, v_opt, m_opt, v_all_p1)

        # Backtrace (reference :941-1006).
        steps_k = np.array([-2, -1, -1], dtype=np.int32)
        steps_i = np.array([-1, 0, -1], dtype=np.int32)
        trans_enc = np.array(
            [[1, 1, 1], [2, 3, 2], [4, 4, 5]], dtype=np.int32
        )
        k_opt = k_end.copy()
        i_opt = y_true_lens.copy()
        paths = np.zeros((b, m + 1, n + 1), dtype=np.int32)
        m_cur = m_opt.copy()
        for k in range(m + n, -1, -1):
            safe_m = np.maximum(m_cur, 0)
            safe_i = np.maximum(i_opt, 0)
            k_next = k_opt + steps_k[safe_m]
            i_next = i_opt + steps_i[safe_m]
            m_next = dir_all[k][safe_m, safe_i, samp]
            safe_m_next = np.maximum(m_next, 0)
            edges = trans_enc[safe_m, safe_m_next]
            reached_start = m_next == -1
            cond = (k_opt == k) & (~reached_start)
            j_here = k_opt - i_opt
            valid = cond & (i_opt >= 0) & (j_here >= 0) & (j_here <= n)
            paths[samp[valid], i_opt[valid], j_here[valid]] = edges[valid]
            k_opt = np.where(cond, k_next, k_opt)
            i_opt = np.where(cond, i_next, i_opt)
            m_cur = np.where(cond, m_next, m_cur)

        matches_mask = paths == 1
        insertions_mask = (paths == 2) | (paths == 3)
        deletions_mask = (paths == 4) | (paths == 5)
        correct = matches_mask[:, 1:, 1:] & (subs > 0)
        sum_pos = lambda t: t.reshape(b, -1).sum(-1).astype(np.int64)
        mv = {
            "num_matches": sum_pos(matches_mask),
            "num_insertions": sum_pos(insertions_mask),
            "num_deletions": sum_pos(deletions_mask),
            "num_correct_matches": sum_pos(correct),
        }
        mv["alignment_length"] = (
            mv["num_matches"] + mv["num_insertions"] + mv["num_deletions"]
        )
        with np.errstate(divide="ignore", invalid="ignore"):
            unsafe = mv["num_correct_matches"] / mv["alignment_length"]
        mv["pid"] = np.where(mv["alignment_length"] > 0, unsafe, 1.0)
        return v_opt, paths, mv

    def update_state(self, y_true, y_pred):
        _, _, mv = self.alignment(y_true, y_pred)
        self._pid_sum += float(mv["pid"].sum())
        self._pid_n += len(mv["pid"])

    def result(self) -> float:
        return self._pid_sum / max(self._pid_n, 1)

    def reset_states(self):
        self._pid_sum, self._pid_n = 0.0, 0


def per_batch_identity(mv: Mapping[str, np.ndarray]) -> float:
    tot = mv["alignment_length"].sum()
    if tot == 0:
        return 1.0
    return float(mv["num_correct_matches"].sum() / tot)


def get_batch_identity_ccs_pred(
    ccs, y_pred, y_true, alignment_metric: AlignmentMetric
) -> Tuple[float, float]:
    """(identity_ccs, identity_pred) (losses_and_metrics.py:1061-1098)."""
    _, _, mv_pred = alignment_metric.alignment(y_true, y_pred)
    identity_pred = per_batch_identity(mv_pred)
    if isinstance(ccs, torch.Tensor) and ccs.is_cuda:
        ccs_oh = torch.nn.functional.one_hot(
            ccs.long(), constants.SEQ_VOCAB_SIZE
        ).float()
    else:
        ccs_np = (
            ccs.detach().cpu().numpy()
            if isinstance(ccs, torch.Tensor) else ccs
        ).astype(np.int64)
        ccs_oh = np.eye(constants.SEQ_VOCAB_SIZE, dtype=np.float32)[ccs_np]
    _, _, mv_ccs = alignment_metric.alignment(y_true, ccs_oh)
    identity_ccs = per_batch_identity(mv_ccs)
    return identity_ccs, identity_pred


class PerExampleAccuracy:
    """Whole-window exact match after left shift
    (losses_and_metrics.py:37-65)."""

    def __init__(self):
        self.correct = 0
        self.total = 0

    def update_state(self, y_true, y_pred_scores):
        y_true = (y_true if isinstance(y_true, torch.Tensor)
                  else torch.as_tensor(np.asarray(y_true))).long()
        scores = (y_pred_scores
                  if isinstance(y_pred_scores, torch.Tensor)
                  else torch.as_tensor(np.asarray(y_pred_scores)))
        y_true = left_shift_sequence(y_true)
        y_pred = left_shift_sequence(scores.argmax(-1).long())
        matches = (y_true == y_pred).sum(-1)
        total = y_true.shape[-1]
        self.correct += int((matches == total).sum())
        self.total += y_true.shape[0]

    def result(self) -> float:
        return self.correct / max(self.total, 1)

    def reset_states(self):
        self.correct = self.total = 0


class PerClassAccuracy:
    """Per-position accuracy for one class (losses_and_metrics.py:68-89)."""

    def __init__(self, class_value: int):
        self.class_value = class_value
        self.correct = 0
        self.total = 0

    def update_state(self, y_true, y_pred_scores):
        y_true = (y_true if isinstance(y_true, torch.Tensor)
                  else torch.as_tensor(np.asarray(y_true))).long()
        y_pred = (y_pred_scores
                  if isinstance(y_pred_scores, torch.Tensor)
                  else torch.as_tensor(np.asarray(y_pred_scores))
                  ).argmax(-1)
        mask = y_true == self.class_value
        self.correct += int(((y_true == y_pred) & mask).sum())
        self.total += int(mask.sum())

    def result(self) -> float:
        return self.correct / max(self.total, 1)

    def reset_states(self):
        self.correct = self.total = 0


class YieldOverCCSMetric:
    """DC/CCS yield ratio at an identity threshold
    (losses_and_metrics.py:1114-1167)."""

    def __init__(self, quality_threshold: float = 0.997):
        self.quality_threshold = quality_threshold
        self.yield_dc = 0.0
        self.yield_ccs = 0.0

    def update_state(self, identity_ccs: float, identity_pred: float):
        if identity_pred >= self.quality_threshold:
            self.yield_dc += 1.0
        if identity_ccs >= self.quality_threshold:
            self.yield_ccs += 1.0

    def result(self) -> float:
        if self.yield_ccs == 0:
            return 0.0
        return self.yield_dc / self.yield_ccs

    def reset_state(self):
        self.yield_dc = self.yield_ccs = 0.0


class DistillationLoss(torch.nn.Module):
    """Temperature KL/MSE between teacher and student logits
    (losses_and_metrics.py:1170-1213)."""

    def __init__(self, temperature: float = 1.0,
                 logit_loss: str = "kl_divergence"):
        super().__init__()
        self.temperature = temperature
        self.logit_loss = logit_loss

    def forward(
        self, teacher_logits: torch.Tensor, student_logits: torch.Tensor
    ) -> torch.Tensor:
        t = torch.softmax(teacher_logits / self.temperature, dim=-1)
        s = torch.softmax(student_logits / self.temperature, dim=-1)
        if self.logit_loss == "mean_squared_error":
            per_pos = ((t - s) ** 2).mean(-1)
        else:  # kl_divergence
            per_pos = (t * (torch.log(t.clamp_min(1e-12))
                            - torch.log(s.clamp_min(1e-12)))).sum(-1)
        # Mean across positions, then batch.
        return per_pos.mean(-1).mean()
