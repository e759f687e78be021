import numpy as np
import pytest
import torch

from rainbowiqn_amd.agent import Agent
from rainbowiqn_amd.env import SyntheticAtariEnv, make_env
from rainbowiqn_amd.replay import ReplayMemory


def test_env_interface(cfg):
    env = make_env(cfg)
    assert isinstance(env, SyntheticAtariEnv)  # no ALE on this box
    s = env.reset()
    assert s.shape == (cfg.history_length, 84, 84)
    assert 0.0 <= s.min() and s.max() <= 1.0
    s2, r, d = env.step(0)
    assert s2.shape == s.shape and isinstance(r, float) and isinstance(d, bool)
    assert env.action_space() >= 2


def test_env_episode_terminates(cfg):
    env = SyntheticAtariEnv(cfg, mean_episode_len=32)
    env.reset()
    done = False
    for _ in range(10000):
        _, _, done = env.step(0)
        if done:
            break
    assert done


def test_act_returns_valid_action(cfg):
    env = make_env(cfg)
    agent = Agent(cfg, env.action_space(), "cpu")
    s = env.reset()
    for _ in range(3):
        a = agent.act(s)
        assert 0 <= a < env.action_space()


def test_learn_step_updates_weights(cfg):
    env = make_env(cfg)
    agent = Agent(cfg, env.action_space(), "cpu")
    mem = ReplayMemory(cfg, capacity=512)
    s = env.reset()
    for _ in range(80):
        a = agent.act(s)
        s2, r, d = env.step(a)
        mem.append(s, a, r, d)
        s = env.reset() if d else s2
    before = agent.online_net.fc_h_v.weight_mu.clone()
    loss = agent.learn(mem)
    assert np.isfinite(loss)
    assert not torch.equal(before, agent.online_net.fc_h_v.weight_mu)


def test_priorities_updated_after_learn(cfg):
    env = make_env(cfg)
    agent = Agent(cfg, env.action_space(), "cpu")
    mem = ReplayMemory(cfg, capacity=512)
    s = env.reset()
    for _ in range(80):
        a = 0
        s2, r, d = env.step(a)
        mem.append(s, a, r, d)
        s = env.reset() if d else s2
    t0 = mem.tree.total()
    agent.learn(mem)
    assert mem.tree.total() != pytest.approx(t0)  # priorities rewritten


def test_target_net_update(cfg):
    agent = Agent(cfg, 4, "cpu")
    with torch.no_grad():
        agent.online_net.fc_h_v.weight_mu.add_(1.0)
    assert not torch.equal(
        agent.online_net.fc_h_v.weight_mu, agent.target_net.fc_h_v.weight_mu
    )
    agent.update_target_net()
    assert torch.equal(
        agent.online_net.fc_h_v.weight_mu, agent.target_net.fc_h_v.weight_mu
    )


def test_flat_weights_roundtrip(cfg):
    a = Agent(cfg, 4, "cpu")
    b = Agent(cfg, 4, "cpu")
    flat = a.flat_weights()
    assert flat.dim() == 1
    b.load_flat_weights(flat)
    for p, q in zip(a.online_net.parameters(), b.online_net.parameters()):
        assert torch.equal(p, q)


def test_train_single_smoke(cfg, tmp_path):
    from rainbowiqn_amd.train_single import train

    agent, mem, losses = train(cfg, max_steps=300, results_dir=str(tmp_path))
    assert len(mem) == 300
    assert len(losses) > 0
    assert all(np.isfinite(l) for l in losses)


def test_evaluator(cfg, tmp_path):
    from rainbowiqn_amd.eval import Evaluator

    cfg.evaluation_episodes = 2
    cfg.max_episode_length = 64
    agent = Agent(cfg, 6, "cpu")
    ev = Evaluator(cfg)
    avg, std = ev.evaluate(agent, T=100, results_dir=str(tmp_path))
    assert np.isfinite(avg) and std >= 0
    assert (tmp_path / "rewards.html").exists()
    assert (tmp_path / "best_model.pth").exists()


def test_reference_flag_surface_complete():
    """Every reference argparse flag parses."""
    from rainbowiqn_amd.config import parse_args

    argv = [
        "--id", "x", "--game", "breakout", "--seed", "7", "--disable-cuda",
        "--T-max", "50000000", "--max-episode-length", "108000",
        "--history-length", "4", "--hidden-size", "512", "--noisy-std", "0.1",
        "--atoms", "51", "--quantile-embedding-dim", "64",
        "--num-tau-samples", "8", "--num-tau-prime-samples", "8",
        "--num-quantile-samples", "32", "--kappa", "1.0",
        "--memory-capacity", "1000000", "--replay-frequency", "4",
        "--priority-exponent", "0.5", "--priority-weight", "0.4",
        "--multi-step", "3", "--learn-start", "80000", "--discount", "0.99",
        "--target-update", "32000", "--reward-clip", "1", "--lr", "0.0000625",
        "--adam-eps", "0.00015", "--batch-size", "32",
        "--evaluation-interval", "100000", "--evaluation-episodes", "10",
        "--evaluation-size", "500", "--render",
        "--checkpoint-interval", "1000", "--memory-save-path", "/tmp/m.pt",
        "--nb-actor", "4", "--id-actor", "1",
        "--transition-chunk-size", "64", "--weight-sync-period", "25",
        "--host-redis", "localhost", "--port-redis", "6379",
    ]
    cfg = parse_args(argv)
    assert cfg.game == "breakout" and cfg.nb_actor == 4
    assert cfg.lr == 0.0000625 and cfg.target_update == 32000
    assert cfg.memory_save_path == "/tmp/m.pt"
    # reference defaults preserved when unset
    d = parse_args([])
    assert d.T_max == int(50e6) and d.learn_start == int(80e3)
    assert d.priority_exponent == 0.5 and d.multi_step == 3
    assert d.adam_eps == 1.5e-4 and d.batch_size == 32
