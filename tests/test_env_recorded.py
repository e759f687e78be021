"""Recorded-trajectory env: the REAL AtariEnv wrapper (env_ale.py --
reference rainbowiqn/env.py semantics) driven by a deterministic tape
(env_recorded.FakeALE), so the non-synthetic env code path is exercised
end-to-end without an ALE wheel."""

import numpy as np
import pytest
import torch

from rainbowiqn_amd.config import Config
from rainbowiqn_amd.env_recorded import (FakeALE, make_recorded_env,
                                         record_trajectory)


@pytest.fixture(scope="module")
def tape(tmp_path_factory):
    p = str(tmp_path_factory.mktemp("tape") / "traj.npz")
    record_trajectory(p, n_frames=3000, seed=1)
    return p


def test_tape_roundtrip(tape):
    ale = FakeALE(tape)
    assert ale.frames.shape == (3000, 210, 160)
    assert ale.getMinimalActionSet() == list(range(6))
    r_total = 0.0
    for _ in range(100):
        r_total += ale.act(0)
    assert ale.t == 100 and ale.acts == 100
    assert np.isfinite(r_total)


def test_interface_parity_with_synthetic(tape):
    """Same Env interface contract as the synthetic env: reset/step shapes,
    [0,1] range, train/eval toggles, action_space."""
    cfg = Config(game="pong", seed=1)
    env = make_recorded_env(cfg, tape)
    assert env.action_space() == 6
    state = env.reset()
    assert state.shape == (cfg.history_length, 84, 84)
    assert float(state.min()) >= 0.0 and float(state.max()) <= 1.0
    for _ in range(10):
        state, reward, done = env.step(1)
        assert state.shape == (cfg.history_length, 84, 84)
        assert isinstance(reward, float) and np.isfinite(reward)
        assert isinstance(done, bool)
        if done:
            env.reset()
    env.eval()
    env.train()


def test_noop_starts(tape):
    """reset() performs 1..30 no-op emulator frames (reference behavior)."""
    cfg = Config(game="pong", seed=2)
    env = make_recorded_env(cfg, tape)
    np.random.seed(5)
    before = env.ale.acts
    env.reset()
    noops = env.ale.acts - before
    assert 1 <= noops <= 30


def test_life_loss_pseudo_terminal_training_only(tape):
    """A life loss during TRAINING surfaces done=True without restarting the
    emulator (the next reset continues the episode); in EVAL mode the same
    tape position does not end the episode (SURVEY.md env row)."""
    cfg = Config(game="pong", seed=3)
    env = make_recorded_env(cfg, tape)
    env.train()
    env.reset()
    done_at, t_at_done = None, None
    for i in range(200):
        _, _, done = env.step(0)
        if done:
            done_at, t_at_done = i, env.ale.t
            break
    assert done_at is not None, "tape must contain a life loss in 200 steps"
    assert env.life_termination, "life loss, not game over"
    # reset must CONTINUE the episode: exactly one ack frame, no tape jump
    env.reset()
    assert env.ale.t == t_at_done + 1

    # eval mode from the same tape region: no pseudo-terminal
    env2 = make_recorded_env(cfg, tape)
    env2.eval()
    env2.reset()
    env2.ale.t = 0
    dones_eval = 0
    for _ in range(done_at + 50):
        _, _, done = env2.step(0)
        if done:
            dones_eval += 1
            env2.reset()
    # the only eval done in this window would be a real game-over (lives==0)
    train_window_has_gameover = bool(
        env2.ale.terminal[: 4 * (done_at + 50)].any())
    if not train_window_has_gameover:
        assert dones_eval == 0


def test_max_pool_over_last_two_raw_frames(tape):
    """The stacked observation's newest frame equals max(resize(f_t2),
    resize(f_t3)) of the repeat window -- the wrapper's pooling contract."""
    from rainbowiqn_amd.env_ale import _resize_84

    cfg = Config(game="pong", seed=4)
    env = make_recorded_env(cfg, tape)
    env.reset()
    t0 = env.ale.t
    state, _, _ = env.step(0)
    f2 = _resize_84(env.ale.frames[t0 + 3])
    f3 = _resize_84(env.ale.frames[t0 + 4])
    expect = np.maximum(f2, f3).astype(np.float32) / 255.0
    got = state[-1].cpu().numpy()
    assert np.allclose(got, expect, atol=1e-6)


def test_agent_runs_on_recorded_env(tape):
    """A few act/learn plumbing steps through the recorded env (the
    make_env dispatch the reference's train loop would use)."""
    from rainbowiqn_amd.agent import Agent
    from rainbowiqn_amd.env import make_env
    from rainbowiqn_amd.replay import ReplayMemory

    cfg = Config(game="pong", seed=5, recorded_env=tape, hidden_size=32,
                 num_tau_samples=2, num_tau_prime_samples=2,
                 num_quantile_samples=3, batch_size=8, learn_start=32,
                 memory_capacity=1024, device="cpu", dtype="fp32")
    env = make_env(cfg)
    assert type(env).__name__ == "AtariEnv"
    agent = Agent(cfg, env.action_space(), "cpu")
    mem = ReplayMemory(cfg)
    state = env.reset()
    for _ in range(64):
        a = agent.act(state)
        ns, r, done = env.step(a)
        mem.append(state, a, r, done)
        state = env.reset() if done else ns
    loss = agent.learn(mem)
    assert np.isfinite(loss)
