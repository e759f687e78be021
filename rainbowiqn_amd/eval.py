"""Evaluation: run E greedy episodes in eval mode (reference test.py-style
module): no life-loss terminals, fixed K tau samples,
best-model checkpointing and reward-curve plotting."""

from __future__ import annotations

import os
from typing import List, Tuple


from .env import make_env
from .utils.logging import log, plot_rewards


class Evaluator:
    def __init__(self, cfg, device="cpu"):
        self.cfg = cfg
        self.env = make_env(cfg, device)
        self.env.eval()
        self.best_avg_reward = -float("inf")
        self.steps_hist: List[int] = []
        self.rewards_hist: List[List[float]] = []

    def evaluate(self, agent, T: int, results_dir: str | None = None) -> Tuple[float, float]:
        agent.eval()
        rewards = []
        for _ in range(self.cfg.evaluation_episodes):
            state, total, done = self.env.reset(), 0.0, False
            steps = 0
            while not done and steps < self.cfg.max_episode_length:
                action = agent.act_e_greedy(state)
                state, reward, done = self.env.step(action)
                total += reward
                steps += 1
            rewards.append(total)
        agent.train()
        avg = sum(rewards) / len(rewards)
        std = (sum((r - avg) ** 2 for r in rewards) / len(rewards)) ** 0.5
        self.steps_hist.append(T)
        self.rewards_hist.append(rewards)
        if results_dir:
            os.makedirs(results_dir, exist_ok=True)
            plot_rewards(self.steps_hist, self.rewards_hist,
                         os.path.join(results_dir, "rewards.html"),
                         title=f"{self.cfg.game} reward")
            if avg > self.best_avg_reward:
                self.best_avg_reward = avg
                agent.save(results_dir, "best_model.pth")
        log(f"eval @ T={T}: avg reward {avg:.2f} +- {std:.2f}")
        return avg, std
