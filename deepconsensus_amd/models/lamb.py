"""LAMB optimizer + polynomial-decay/linear-warmup schedule (torch).

Parity with the reference's optimizer stack (model_utils.py:621-669, built on
tf-models' optimization package): LAMB (You et al. 2020) with bias-corrected
moments, weight decay excluded for LayerNorm/bias/norm parameters, per-tensor
trust ratio; polynomial LR decay from initial to end over decay_steps with
linear warmup over warmup_steps.
"""
from __future__ import annotations

from typing import Tuple

import torch


class LAMB(torch.optim.Optimizer):
    """LAMB with a batched (_foreach) step.

    ``capturable=True`` keeps the step count, bias corrections and
    learning rate on-device (0-dim tensors) so the whole step can live
    inside a hipGraph capture; the LR schedule then updates each group's
    ``lr_t`` tensor in place between replays (PolynomialWarmupSchedule
    .apply does this automatically when the tensor exists).
    """

    def __init__(
        self,
        params,
        lr: float = 1e-3,
        betas: Tuple[float, float] = (0.9, 0.999),
        eps: float = 1e-6,
        weight_decay: float = 0.0,
        capturable: bool = False,
        foreach: bool = True,
    ):
        defaults = dict(lr=lr, betas=betas, eps=eps,
                        weight_decay=weight_decay)
        super().__init__(params, defaults)
        self.capturable = capturable
        # foreach=False: per-tensor loop (diagnostic — the hipGraph
        # bisect isolates multi-tensor-apply as a capture suspect).
        self.foreach = foreach

    @torch.no_grad()
    def step(self, closure=None):
        """Batched (_foreach) LAMB step.

        The per-tensor loop it replaced issued hundreds of tiny kernels
        per step plus one GPU->CPU sync per parameter (the float() on the
        trust ratio); this version runs a handful of foreach launches and
        keeps the trust ratios on-device (profiles/r01_train_top_kernels
        .txt showed per-tensor norms as a top train-step cost).
        """
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()
        for group in self.param_groups:
            beta1, beta2 = group["betas"]
            eps = group["eps"]
            wd = group["weight_decay"]
            lr = group["lr"]
            ps = [p for p in group["params"] if p.grad is not None]
            if not ps:
                continue
            grads = [p.grad for p in ps]
            ms, vs = [], []
            if self.capturable and "step_t" not in group:
                dev = ps[0].device
                group["step_t"] = torch.zeros((), device=dev)
                group["lr_t"] = torch.full((), lr, device=dev)
            for p in ps:
                state = self.state[p]
                if len(state) == 0:
                    state["step"] = 0
                    state["exp_avg"] = torch.zeros_like(p)
                    state["exp_avg_sq"] = torch.zeros_like(p)
                state["step"] += 1
                ms.append(state["exp_avg"])
                vs.append(state["exp_avg_sq"])
            if not self.foreach:
                if self.capturable:
                    group["step_t"] += 1
                    t = group["step_t"]
                    neg_lr = -group["lr_t"]
                for idx, p in enumerate(ps):
                    g_, m_, v_ = grads[idx], ms[idx], vs[idx]
                    m_.mul_(beta1).add_(g_, alpha=1 - beta1)
                    v_.mul_(beta2).addcmul_(g_, g_, value=1 - beta2)
                    if self.capturable:
                        bc1_ = 1.0 - beta1**t
                        bc2_ = 1.0 - beta2**t
                    else:
                        bc1_ = 1.0 - beta1 ** self.state[p]["step"]
                        bc2_ = 1.0 - beta2 ** self.state[p]["step"]
                        neg_lr = -lr
                    upd = (m_ / bc1_) / ((v_ / bc2_).sqrt() + eps)
                    if wd != 0:
                        upd = upd.add(p, alpha=wd)
                    w_n = p.norm()
                    u_n = upd.norm()
                    ratio = torch.where(
                        (w_n > 0) & (u_n > 0), w_n / u_n,
                        torch.ones_like(w_n),
                    ) * neg_lr
                    p.add_(upd * ratio)
                continue
            torch._foreach_mul_(ms, beta1)
            torch._foreach_add_(ms, grads, alpha=1 - beta1)
            torch._foreach_mul_(vs, beta2)
            torch._foreach_addcmul_(vs, grads, grads, value=1 - beta2)
            if self.capturable:
                group["step_t"] += 1
                t = group["step_t"]
                bc1 = [1.0 - beta1**t] * len(ps)
                bc2 = [1.0 - beta2**t] * len(ps)
                neg_lr = -group["lr_t"]
            else:
                bc1 = [1.0 - beta1 ** self.state[p]["step"] for p in ps]
                bc2 = [1.0 - beta2 ** self.state[p]["step"] for p in ps]
                neg_lr = -lr
            m_hat = torch._foreach_div(ms, bc1)
            denom = torch._foreach_div(vs, bc2)
            denom = torch._foreach_sqrt(denom)
            torch._foreach_add_(denom, eps)
            update = torch._foreach_div(m_hat, denom)
            if wd != 0:
                torch._foreach_add_(update, ps, alpha=wd)
            w_norm = torch.stack(torch._foreach_norm(ps))
            u_norm = torch.stack(torch._foreach_norm(update))
            ratio = torch.where(
                (w_norm > 0) & (u_norm > 0),
                w_norm / u_norm,
                torch.ones_like(w_norm),
            ) * neg_lr
            torch._foreach_mul_(update, list(ratio.unbind()))
            torch._foreach_add_(ps, update)
        return loss


_NO_DECAY_MARKERS = ("layer_norm", "norm", "bias", "alpha", "batchnorm")


def build_param_groups(model: torch.nn.Module, weight_decay: float):
    """Splits params: LayerNorm/bias/alpha excluded from weight decay
    (model_utils.py:634-665)."""
    decay, no_decay = [], []
    for name, p in model.named_parameters():
        if not p.requires_grad:
            continue
        lname = name.lower()
        if any(m in lname for m in _NO_DECAY_MARKERS):
            no_decay.append(p)
        else:
            decay.append(p)
    return [
        {"params": decay, "weight_decay": weight_decay},
        {"params": no_decay, "weight_decay": 0.0},
    ]


class PolynomialWarmupSchedule:
    """poly decay init->end over decay_steps; linear warmup over
    warmup_steps."""

    def __init__(
        self,
        initial_learning_rate: float,
        end_learning_rate: float,
        decay_steps: int,
        warmup_steps: int,
        power: float = 1.0,
    ):
        self.init = initial_learning_rate
        self.end = end_learning_rate
        self.decay_steps = max(decay_steps, 1)
        self.warmup_steps = warmup_steps
        self.power = power

    def __call__(self, step: int) -> float:
        frac = min(step, self.decay_steps) / self.decay_steps
        lr = (self.init - self.end) * (1 - frac) ** self.power + self.end
        if self.warmup_steps and step < self.warmup_steps:
            lr = lr * (step + 1) / self.warmup_steps
        return lr

    def apply(self, optimizer: torch.optim.Optimizer, step: int) -> float:
        lr = self(step)
        for g in optimizer.param_groups:
            g["lr"] = lr
            if "lr_t" in g:  # capturable mode: graph reads this tensor
                g["lr_t"].fill_(lr)
        return lr


def create_optimizer(params, decay_steps: int, model: torch.nn.Module,
                     capturable: bool = False):
    """Factory (model_utils.py:621-669): returns (optimizer, schedule)."""
    groups = build_param_groups(model, params.weight_decay_rate)
    opt = LAMB(
        groups,
        lr=params.initial_learning_rate,
        betas=(params.beta_1, params.beta_2),
        eps=params.epsilon,
        capturable=capturable,
    )
    sched = PolynomialWarmupSchedule(
        initial_learning_rate=params.initial_learning_rate,
        end_learning_rate=params.end_learning_rate,
        decay_steps=decay_steps,
        warmup_steps=params.warmup_steps,
    )
    return opt, sched
