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
    def __init__(
        self,
        params,
        lr: float = 1e-3,
        betas: Tuple[float, float] = (0.9, 0.999),
        eps: float = 1e-6,
        weight_decay: float = 0.0,
    ):
        defaults = dict(lr=lr, betas=betas, eps=eps,
                        weight_decay=weight_decay)
        super().__init__(params, defaults)

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
            ms, vs, bc1, bc2 = [], [], [], []
            for p in ps:
                state = self.state[p]
                if len(state) == 0:
                    state["step"] = 0
                    state["exp_avg"] = torch.zeros_like(p)
                    state["exp_avg_sq"] = torch.zeros_like(p)
                state["step"] += 1
                t = state["step"]
                ms.append(state["exp_avg"])
                vs.append(state["exp_avg_sq"])
                bc1.append(1.0 - beta1**t)
                bc2.append(1.0 - beta2**t)
            torch._foreach_mul_(ms, beta1)
            torch._foreach_add_(ms, grads, alpha=1 - beta1)
            torch._foreach_mul_(vs, beta2)
            torch._foreach_addcmul_(vs, grads, grads, value=1 - beta2)
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
            ) * (-lr)
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
        return lr


def create_optimizer(params, decay_steps: int, model: torch.nn.Module):
    """Factory (model_utils.py:621-669): returns (optimizer, schedule)."""
    groups = build_param_groups(model, params.weight_decay_rate)
    opt = LAMB(
        groups,
        lr=params.initial_learning_rate,
        betas=(params.beta_1, params.beta_2),
        eps=params.epsilon,
    )
    sched = PolynomialWarmupSchedule(
        initial_learning_rate=params.initial_learning_rate,
        end_learning_rate=params.end_learning_rate,
        decay_steps=decay_steps,
        warmup_steps=params.warmup_steps,
    )
    return opt, sched
