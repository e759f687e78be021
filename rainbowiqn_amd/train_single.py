"""Single-process trainer: 1 actor + 1 learner interleaved (BASELINE.json
config 1; the reference's Kaixhin-lineage main loop, SURVEY.md section 4.1/4.2
collapsed into one process -- useful for plumbing tests and CPU CI).

Usage: python -m rainbowiqn_amd.train_single --game pong --memory-capacity 10000 ...
"""

from __future__ import annotations

import torch

from .agent import Agent
from .config import Config, parse_args
from .env import make_env
from .eval import Evaluator
from .replay import ReplayMemory
from .utils.logging import log


def train(cfg: Config, max_steps: int | None = None, results_dir: str | None = None):
    device = cfg.resolve_device()
    torch.manual_seed(cfg.seed)
    env = make_env(cfg, "cpu")
    env.train()
    agent = Agent(cfg, env.action_space(), device)
    mem = ReplayMemory(cfg)
    T_max = max_steps if max_steps is not None else cfg.T_max
    # anneal beta -> 1 over the run
    beta_increase = (1.0 - cfg.priority_weight) / max(1, T_max - cfg.learn_start)

    evaluator = Evaluator(cfg) if cfg.evaluation_interval <= T_max else None
    state, done = env.reset(), False
    n_episodes, ep_reward, losses = 0, 0.0, []
    for T in range(1, T_max + 1):
        if done:
            state, done = env.reset(), False
            n_episodes += 1
            ep_reward = 0.0
        if T % cfg.replay_frequency == 0:
            agent.reset_noise()
        action = agent.act(state)
        next_state, reward, done = env.step(action)
        if cfg.reward_clip_enabled:
            reward = max(-cfg.reward_clip, min(cfg.reward_clip, reward))
        mem.append(state, action, reward, done)
        ep_reward += reward
        state = next_state

        if T >= cfg.learn_start:
            mem.priority_weight = min(1.0, mem.priority_weight + beta_increase)
            if T % cfg.replay_frequency == 0:
                losses.append(agent.learn(mem))
            if T % cfg.target_update == 0:
                agent.update_target_net()
            if evaluator and T % cfg.evaluation_interval == 0:
                evaluator.evaluate(agent, T, results_dir)
        if cfg.checkpoint_interval and T % cfg.checkpoint_interval == 0 and results_dir:
            agent.save(results_dir)
        if T % 10000 == 0:
            avg_loss = sum(losses[-100:]) / max(1, len(losses[-100:]))
            log(f"T={T} episodes={n_episodes} loss={avg_loss:.4f}")
    return agent, mem, losses


def main(argv=None):
    cfg = parse_args(argv)
    train(cfg, results_dir=f"results/{cfg.id}")


if __name__ == "__main__":
    main()
