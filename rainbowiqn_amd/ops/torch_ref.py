"""Pure-PyTorch reference implementations of the hot ops.

These define the semantics that the hand-written CDNA4 HIP kernels
(rainbowiqn_amd/ops/csrc) must reproduce; kernel parity tests compare against
these in fp32.  They are also the CPU execution path.
"""

from __future__ import annotations

from typing import Tuple

import torch


def quantile_huber_loss(
    online_quantiles: torch.Tensor,  # [B, N]   theta_i  (requires grad)
    target_quantiles: torch.Tensor,  # [B, N']  T theta_j (no grad)
    taus: torch.Tensor,  # [B, N] quantile fractions of the online samples
    weights: torch.Tensor | None = None,  # [B] PER importance weights
    kappa: float = 1.0,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """IQN quantile-Huber loss (Dabney et al. 2018, eq. 3).

    rho_tau(d) = |tau - 1{d<0}| * L_kappa(d) / kappa, d_ij = Ttheta_j - theta_i
    per-sample loss = sum_i mean_j rho ; returns (scalar weighted mean loss,
    per-sample loss detached -- the new PER priorities).

    Reference call site: agent.learn (kernel K11).
    """
    b, n = online_quantiles.shape
    d = target_quantiles.unsqueeze(1) - online_quantiles.unsqueeze(2)  # [B, N, N']
    abs_d = d.abs()
    huber = torch.where(abs_d <= kappa, 0.5 * d * d, kappa * (abs_d - 0.5 * kappa))
    asym = (taus.unsqueeze(2) - (d.detach() < 0).float()).abs()
    rho = asym * huber / kappa
    per_sample = rho.mean(dim=2).sum(dim=1)  # mean over j, sum over i
    if weights is not None:
        loss = (weights * per_sample).mean()
    else:
        loss = per_sample.mean()
    return loss, per_sample.detach()


def n_step_target(
    target_quantiles_next: torch.Tensor,  # [B, N'] Z_tau'(s', a*)
    returns: torch.Tensor,  # [B] n-step discounted returns
    nonterminals: torch.Tensor,  # [B] 1.0 if s' valid
    discount: float,
    n: int,
) -> torch.Tensor:
    """T theta_j = R^(n) + gamma^n * nonterminal * Z_tau'(s', a*)  (kernel K10)."""
    gamma_n = discount ** n
    return returns.unsqueeze(1) + gamma_n * nonterminals.unsqueeze(1) * target_quantiles_next


def preprocess_frames(frames_u8: torch.Tensor, dtype=torch.float32) -> torch.Tensor:
    """uint8 [.., 84, 84] -> dtype scaled to [0, 1]  (kernel K15)."""
    return frames_u8.to(dtype) / 255.0


def argmax_mean_quantiles(z: torch.Tensor) -> torch.Tensor:
    """z: [B, K, A] -> greedy action per sample  (kernel K9)."""
    return z.mean(dim=1).argmax(dim=1)
