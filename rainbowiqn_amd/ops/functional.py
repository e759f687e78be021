"""Autograd wrappers over the HIP kernels (GPU) with torch_ref semantics
(reference: rainbowiqn/agent.py learn() loss math, SURVEY.md section 4.1;
used by the eager-autograd fallback path -- the fast path calls the raw
kernels from runtime/gpu_learner.py instead)."""

from __future__ import annotations

from typing import Optional, Tuple

import torch

from . import get_extension, torch_ref


class _QuantileHuberFn(torch.autograd.Function):
    """Fused n-step target + quantile-Huber loss + grad (kernels K10/K11).

    forward returns (loss, per_sample); backward scales the precomputed
    d(loss)/d(theta) by the incoming scalar gradient.
    """

    @staticmethod
    def forward(ctx, theta, target_q, taus, returns, nonterminals, weights,
                gamma_n: float, kappa: float):
        ext = get_extension()
        B, N = theta.shape
        loss = torch.zeros(1, device=theta.device)
        per_sample = torch.empty(B, device=theta.device)
        grad_theta = torch.empty(B, N, device=theta.device)
        ext.quantile_huber(
            theta.contiguous().float(), target_q.contiguous().float(),
            taus.contiguous().float(), returns.contiguous().float(),
            nonterminals.contiguous().float(),
            None if weights is None else weights.contiguous().float(),
            gamma_n, kappa, loss, per_sample, grad_theta,
        )
        ctx.save_for_backward(grad_theta)
        return loss[0], per_sample

    @staticmethod
    def backward(ctx, grad_loss, _grad_per_sample):
        (grad_theta,) = ctx.saved_tensors
        return grad_theta * grad_loss, None, None, None, None, None, None, None


def quantile_huber_loss_fused(
    theta: torch.Tensor,  # [B, N] online quantiles (requires grad)
    target_q: torch.Tensor,  # [B, Np] target-net quantiles Z'(s', a*)
    taus: torch.Tensor,  # [B, N]
    returns: torch.Tensor,  # [B]
    nonterminals: torch.Tensor,  # [B]
    weights: Optional[torch.Tensor],  # [B]
    discount: float,
    n_step: int,
    kappa: float = 1.0,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """GPU: one fused kernel; CPU: torch_ref composition (same semantics)."""
    gamma_n = discount ** n_step
    if theta.is_cuda:
        return _QuantileHuberFn.apply(
            theta, target_q, taus, returns, nonterminals, weights, gamma_n, kappa
        )
    t_theta = torch_ref.n_step_target(target_q, returns, nonterminals, discount, n_step)
    return torch_ref.quantile_huber_loss(theta, t_theta, taus, weights, kappa)


def preprocess_fused(frames_u8: torch.Tensor, out: Optional[torch.Tensor] = None):
    """u8 -> f32/255 (K15)."""
    if frames_u8.is_cuda:
        ext = get_extension()
        if out is None:
            out = torch.empty(frames_u8.shape, dtype=torch.float32,
                              device=frames_u8.device)
        ext.preprocess(frames_u8.contiguous(), out)
        return out
    return torch_ref.preprocess_frames(frames_u8)
