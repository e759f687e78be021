"""Rainbow-IQN network.

Architecture parity with the reference model (rainbowiqn/model.py, SURVEY.md
section 3.1 "Network" row; Kaixhin/Rainbow lineage):

  * Nature CNN encoder: 32x8x8s4 -> 64x4x4s2 -> 64x3x3s1, ReLU, flatten to 3136
  * factorised-Gaussian NoisyLinear (sigma0=0.1, f(x)=sign(x)*sqrt(|x|))
  * IQN cosine quantile embedding: 64 cosines cos(pi*i*tau), Linear(64->3136),
    ReLU, elementwise product with the conv features
  * dueling NoisyLinear heads 3136->512->{1, |A|}; Q = V + A - mean_a(A)

State-dict key parity ([MED] until the reference mount
is readable): ``convs.{0,2,4}.{weight,bias}``, ``iqn_fc.{weight,bias}``,
``fc_h_v.*``, ``fc_h_a.*``, ``fc_z_v.*``, ``fc_z_a.*`` with NoisyLinear
parameters ``weight_mu/weight_sigma/bias_mu/bias_sigma`` and epsilon buffers.
``load_reference_state_dict`` additionally accepts a remapping table for any
divergent names (utils/checkpoint.py).

The module is the *semantic reference*: hand-written CDNA4 HIP kernel paths
(rainbowiqn_amd/ops) are validated against this forward in fp32.
"""

from __future__ import annotations

import math
from typing import Optional, Tuple

import torch
from torch import nn
from torch.nn import functional as F

CONV_OUT = 3136  # 64 * 7 * 7 for 84x84 input


class NoisyLinear(nn.Module):
    """Factorised-Gaussian noisy linear layer (Fortunato et al. 2018).

    y = (w_mu + w_sigma * eps_out x eps_in) x + (b_mu + b_sigma * eps_out)
    with eps ~ f(N(0,1)), f(x) = sign(x) * sqrt(|x|).
    """

    def __init__(self, in_features: int, out_features: int, std_init: float = 0.1):
        super().__init__()
        self.in_features = in_features
        self.out_features = out_features
        self.std_init = std_init
        self.weight_mu = nn.Parameter(torch.empty(out_features, in_features))
        self.weight_sigma = nn.Parameter(torch.empty(out_features, in_features))
        self.register_buffer("weight_epsilon", torch.empty(out_features, in_features))
        self.bias_mu = nn.Parameter(torch.empty(out_features))
        self.bias_sigma = nn.Parameter(torch.empty(out_features))
        self.register_buffer("bias_epsilon", torch.empty(out_features))
        self.reset_parameters()
        self.reset_noise()

    def reset_parameters(self) -> None:
        mu_range = 1.0 / math.sqrt(self.in_features)
        self.weight_mu.data.uniform_(-mu_range, mu_range)
        self.weight_sigma.data.fill_(self.std_init / math.sqrt(self.in_features))
        self.bias_mu.data.uniform_(-mu_range, mu_range)
        self.bias_sigma.data.fill_(self.std_init / math.sqrt(self.out_features))

    @staticmethod
    def _scale_noise(size: int, device, generator=None) -> torch.Tensor:
        x = torch.randn(size, device=device, generator=generator)
        return x.sign().mul_(x.abs().sqrt_())

    def reset_noise(self, generator=None) -> None:
        eps_in = self._scale_noise(self.in_features, self.weight_mu.device, generator)
        eps_out = self._scale_noise(self.out_features, self.weight_mu.device, generator)
        self.weight_epsilon.copy_(eps_out.outer(eps_in))
        self.bias_epsilon.copy_(eps_out)

    def enable_fused_noise(self) -> None:
        """GPU fast path: `weight_noisy`/`bias_noisy` buffers are refreshed by
        the riqn noisy_material kernel once per noise reset, so every forward
        is a single GEMM (no per-forward w = mu + sigma*eps materialization);
        backward routes grads to mu/sigma through _NoisyLinearFn."""
        self.register_buffer("weight_noisy", self.weight_mu.detach().clone(),
                             persistent=False)
        self.register_buffer("bias_noisy", self.bias_mu.detach().clone(),
                             persistent=False)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if self.training:
            if getattr(self, "weight_noisy", None) is not None:
                return _NoisyLinearFn.apply(
                    x, self.weight_mu, self.weight_sigma, self.bias_mu,
                    self.bias_sigma, self.weight_noisy, self.bias_noisy,
                    self.weight_epsilon, self.bias_epsilon,
                )
            w = self.weight_mu + self.weight_sigma * self.weight_epsilon
            b = self.bias_mu + self.bias_sigma * self.bias_epsilon
        else:
            w, b = self.weight_mu, self.bias_mu
        return F.linear(x, w, b)


class _NoisyLinearFn(torch.autograd.Function):
    """F.linear on the pre-materialized noisy weight; backward applies the
    noise chain rule: dmu = dW, dsigma = dW * eps (same for bias)."""

    @staticmethod
    def forward(ctx, x, w_mu, w_sigma, b_mu, b_sigma, w_noisy, b_noisy,
                w_eps, b_eps):
        ctx.save_for_backward(x, w_noisy, w_eps, b_eps)
        return F.linear(x, w_noisy, b_noisy)

    @staticmethod
    def backward(ctx, g):
        x, w_noisy, w_eps, b_eps = ctx.saved_tensors
        g2 = g.reshape(-1, g.shape[-1])
        x2 = x.reshape(-1, x.shape[-1])
        grad_x = (g2 @ w_noisy).view_as(x)
        gw = g2.t() @ x2
        gb = g2.sum(0)
        return (grad_x, gw, gw * w_eps, gb, gb * b_eps,
                None, None, None, None)


class RainbowIQN(nn.Module):
    def __init__(self, cfg, action_space: int):
        super().__init__()
        self.action_space = action_space
        self.quantile_embedding_dim = cfg.quantile_embedding_dim
        self.history = cfg.history_length
        self.convs = nn.Sequential(
            nn.Conv2d(cfg.history_length, 32, 8, stride=4),
            nn.ReLU(),
            nn.Conv2d(32, 64, 4, stride=2),
            nn.ReLU(),
            nn.Conv2d(64, 64, 3, stride=1),
            nn.ReLU(),
        )
        self.conv_output_size = CONV_OUT
        self.iqn_fc = nn.Linear(self.quantile_embedding_dim, self.conv_output_size)
        self.fc_h_v = NoisyLinear(self.conv_output_size, cfg.hidden_size, std_init=cfg.noisy_std)
        self.fc_h_a = NoisyLinear(self.conv_output_size, cfg.hidden_size, std_init=cfg.noisy_std)
        self.fc_z_v = NoisyLinear(cfg.hidden_size, 1, std_init=cfg.noisy_std)
        self.fc_z_a = NoisyLinear(cfg.hidden_size, action_space, std_init=cfg.noisy_std)
        # cosine basis multipliers pi * i, i in [0, embedding_dim)
        self.register_buffer(
            "cos_basis",
            math.pi * torch.arange(self.quantile_embedding_dim, dtype=torch.float32),
            persistent=False,
        )

    # -- pieces (exposed so kernel-parity tests can target each stage) -----
    def encode(self, x: torch.Tensor) -> torch.Tensor:
        """x: [B, history, 84, 84] in [0,1] -> [B, 3136]."""
        w_dtype = self.convs[0].weight.dtype
        if x.dtype != w_dtype:
            x = x.to(w_dtype)
        return self.convs(x).flatten(1)

    def embed_taus(self, taus: torch.Tensor) -> torch.Tensor:
        """taus: [R, 1] -> [R, 3136] cosine embedding through iqn_fc."""
        basis = self.cos_basis
        if basis.dtype != torch.float32:
            # .to(bf16) on the module must not degrade the fp32 basis: a
            # 0.5% rounding of pi*i is ~0.5 rad of phase at i=50
            basis = math.pi * torch.arange(
                self.quantile_embedding_dim, dtype=torch.float32,
                device=taus.device)
            self.cos_basis = basis  # cache fp32 again
        cos = torch.cos(taus.float() * basis)  # [R, 64]
        return F.relu(self.iqn_fc(cos.to(self.iqn_fc.weight.dtype)))

    def head(self, h: torch.Tensor) -> torch.Tensor:
        """h: [R, 3136] -> quantile Q values [R, A] (dueling combine)."""
        v = self.fc_z_v(F.relu(self.fc_h_v(h)))
        a = self.fc_z_a(F.relu(self.fc_h_a(h)))
        return v + a - a.mean(dim=1, keepdim=True)

    def forward(
        self,
        x: torch.Tensor,
        num_quantiles: int,
        taus: Optional[torch.Tensor] = None,
    ) -> Tuple[torch.Tensor, torch.Tensor]:
        """Return (quantile_values [B, Nq, A], taus [B*Nq, 1]).

        Quantile fractions are sampled U(0,1) unless supplied (tests /
        hipGraph capture provide counter-RNG taus explicitly).
        """
        b = x.shape[0]
        feat = self.encode(x)  # [B, F]
        if taus is None:
            taus = torch.rand(b * num_quantiles, 1, device=x.device, dtype=feat.dtype)
        phi = self.embed_taus(taus)  # [B*Nq, F]
        # reference order: feature vector of sample i repeats Nq times
        h = feat.repeat_interleave(num_quantiles, dim=0) * phi
        q = self.head(h)  # [B*Nq, A]
        return q.view(b, num_quantiles, self.action_space), taus

    def q_values(self, x: torch.Tensor, num_quantiles: int) -> torch.Tensor:
        """Mean over quantiles -> [B, A]; used for greedy action selection."""
        z, _ = self.forward(x, num_quantiles)
        return z.mean(dim=1)

    def reset_noise(self, generator=None) -> None:
        for m in (self.fc_h_v, self.fc_h_a, self.fc_z_v, self.fc_z_a):
            m.reset_noise(generator)
