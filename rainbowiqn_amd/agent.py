"""Agent: act / learn / target sync / checkpoint.

API parity with the reference Agent (rainbowiqn/agent.py, SURVEY.md section
3.1): ``act(state)``, ``learn(mem)``, ``update_target_net()``, ``save(path)``,
plus flat-weight (de)serialization used by the Ape-X weight flow (reference:
Redis SET/GET of the state_dict blob; here: an RCCL broadcast buffer).

Algorithm:
  act    : argmax_a (1/K) sum_k Z_tau_k(s, a); exploration via NoisyNets only.
  learn  : sample batch -> online Z with N taus; double-Q greedy action on s'
           via the ONLINE net (K taus); target net Z' with N' taus;
           T theta = R^(n) + gamma^n (1-done) Z'(s', a*); quantile-Huber loss
           x PER weights -> Adam; new priorities = per-sample loss.
"""

from __future__ import annotations

import os
from typing import Optional

import torch
from torch import nn

from .models.model import RainbowIQN
from .ops import torch_ref


class Agent:
    def __init__(self, cfg, action_space: int, device: Optional[str] = None):
        self.cfg = cfg
        self.action_space = action_space
        self.device = torch.device(device if device is not None else cfg.resolve_device())
        self.n = cfg.multi_step
        self.discount = cfg.discount
        self.batch_size = cfg.batch_size
        self.N = cfg.num_tau_samples
        self.Np = cfg.num_tau_prime_samples
        self.K = cfg.num_quantile_samples
        self.kappa = cfg.kappa

        # bf16 compute via autocast on GPU (fp32 master weights + Adam state)
        self.autocast_dtype = (
            torch.bfloat16
            if (cfg.dtype == "bf16" and self.device.type == "cuda")
            else None
        )

        self.online_net = RainbowIQN(cfg, action_space).to(self.device)
        self.target_net = RainbowIQN(cfg, action_space).to(self.device)
        if cfg.model:
            self.load(cfg.model)
        self.update_target_net()
        self.target_net.eval()
        for p in self.target_net.parameters():
            p.requires_grad_(False)
        self.online_net.train()

        self.optimiser = torch.optim.Adam(
            self.online_net.parameters(), lr=cfg.lr, eps=cfg.adam_eps
        )

    # -- acting ------------------------------------------------------------
    def act(self, state: torch.Tensor) -> int:
        """Greedy action from mean of K quantile samples (NoisyNet explore)."""
        with torch.no_grad():
            x = state.to(self.device).unsqueeze(0)
            return int(self.online_net.q_values(x, self.K).argmax(dim=1).item())

    def act_batch(self, states: torch.Tensor) -> torch.Tensor:
        with torch.no_grad(), self._autocast():
            return self.online_net.q_values(states.to(self.device), self.K).argmax(dim=1)

    def _autocast(self):
        if self.autocast_dtype is not None:
            return torch.autocast("cuda", dtype=self.autocast_dtype)
        import contextlib

        return contextlib.nullcontext()

    def act_e_greedy(self, state: torch.Tensor, epsilon: float = 0.001) -> int:
        """Evaluation-time safety epsilon (reference eval behavior)."""
        if torch.rand(1).item() < epsilon:
            return int(torch.randint(0, self.action_space, (1,)).item())
        return self.act(state)

    # -- learning ----------------------------------------------------------
    def learn(self, mem) -> float:
        idxs, states, actions, returns, next_states, nonterminals, weights = mem.sample(
            self.batch_size
        )
        loss, new_priorities = self.learn_from_batch(
            states.to(self.device),
            actions.to(self.device),
            returns.to(self.device),
            next_states.to(self.device),
            nonterminals.to(self.device),
            weights.to(self.device),
        )
        mem.update_priorities(idxs, new_priorities)
        return float(loss)

    def learn_from_batch(self, states, actions, returns, next_states, nonterminals, weights):
        b = states.shape[0]
        net = self.online_net
        with self._autocast():
            # online quantiles for the taken actions: theta_i  [B, N]
            z, taus = net(states, self.N)  # [B, N, A], [B*N, 1]
            theta = z.gather(2, actions.view(b, 1, 1).expand(b, self.N, 1)).squeeze(2)
            taus = taus.view(b, self.N)

            with torch.no_grad():
                # double-Q action selection on s' via the ONLINE net (K samples)
                a_star = net.q_values(next_states, self.K).argmax(dim=1)  # [B]
                # target quantiles Z'_tau'(s', a*)  [B, N']
                zp, _ = self.target_net(next_states, self.Np)
                theta_p = zp.gather(2, a_star.view(b, 1, 1).expand(b, self.Np, 1)).squeeze(2)
                t_theta = torch_ref.n_step_target(
                    theta_p, returns, nonterminals, self.discount, self.n
                )

        # loss in fp32 for numerical fidelity of priorities/gradient scale
        loss, per_sample = torch_ref.quantile_huber_loss(
            theta.float(), t_theta.float(), taus.float(), weights, self.kappa
        )
        self.optimiser.zero_grad(set_to_none=True)
        loss.backward()
        self.optimiser.step()
        return loss.detach(), per_sample

    def reset_noise(self) -> None:
        self.online_net.reset_noise()

    def update_target_net(self) -> None:
        self.target_net.load_state_dict(self.online_net.state_dict())

    # -- weight flow (Ape-X: learner -> actors) ----------------------------
    def flat_weights(self) -> torch.Tensor:
        """Online-net parameters flattened into one buffer (RCCL broadcast payload)."""
        with torch.no_grad():
            return torch.cat([p.detach().reshape(-1) for p in self.online_net.parameters()])

    def load_flat_weights(self, flat: torch.Tensor) -> None:
        with torch.no_grad():
            off = 0
            for p in self.online_net.parameters():
                k = p.numel()
                p.copy_(flat[off : off + k].view_as(p))
                off += k

    # -- evaluation helpers (reference test.py) ----------------------------
    def evaluate_q(self, states: torch.Tensor) -> float:
        with torch.no_grad():
            return float(
                self.online_net.q_values(states.to(self.device), self.K).max(dim=1)[0].mean()
            )

    def train(self) -> None:
        self.online_net.train()

    def eval(self) -> None:
        self.online_net.eval()

    # -- checkpointing ------------------------------------------------------
    def save(self, path: str, name: str = "model.pth") -> None:
        os.makedirs(path, exist_ok=True)
        torch.save(self.online_net.state_dict(), os.path.join(path, name))

    def load(self, path: str) -> None:
        if os.path.isdir(path):
            path = os.path.join(path, "model.pth")
        sd = torch.load(path, map_location=self.device, weights_only=True)
        from .utils.checkpoint import remap_reference_state_dict

        self.online_net.load_state_dict(remap_reference_state_dict(sd, self.online_net))
