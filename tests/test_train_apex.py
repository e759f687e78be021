"""End-to-end tests of the Ape-X training entry point (CPU, single process)
and of replay persistence (reference: Redis RDB snapshot -- SURVEY.md 6.4)."""

import os

import numpy as np
import torch

from rainbowiqn_amd.config import Config
from rainbowiqn_amd.parallel.apex import ApexRunner, ShardedReplay
from rainbowiqn_amd.parallel.comm import Communicator
from rainbowiqn_amd.train_apex import train_apex


def tiny_cfg(**kw):
    base = dict(
        game="pong", seed=7, device="cpu", dtype="fp32", T_max=4000,
        learn_start=256, memory_capacity=2048, batch_size=8,
        num_tau_samples=4, num_tau_prime_samples=4, num_quantile_samples=8,
        learn_per_round=2, evaluation_interval=10**9, hidden_size=32,
        max_episode_length=200,
    )
    base.update(kw)
    return Config(**base)


def test_train_apex_single_process(tmp_path):
    cfg = tiny_cfg(checkpoint_interval=1000)
    runner = train_apex(cfg, n_envs=8, results_dir=str(tmp_path),
                        max_rounds=40, log_interval=20)
    assert runner.grad_steps > 0
    assert os.path.exists(tmp_path / "model.pth")
    # loss was finite
    assert np.isfinite(runner.losses[-1])


def test_train_apex_resume(tmp_path):
    cfg = tiny_cfg(checkpoint_interval=500,
                   memory_save_path=str(tmp_path / "mem.pt"))
    r1 = train_apex(cfg, n_envs=8, results_dir=str(tmp_path), max_rounds=30)
    assert os.path.exists(tmp_path / "mem.pt")
    w1 = r1.agent.flat_weights().clone()

    cfg2 = tiny_cfg(model=str(tmp_path),
                    memory_save_path=str(tmp_path / "mem.pt"))
    comm = Communicator(backend="gloo")
    runner2 = ApexRunner(cfg2, comm, n_envs=8, device="cpu")
    from rainbowiqn_amd.train_apex import _load_memory, _load_model

    _load_model(runner2, cfg2.model)
    _load_memory(runner2, cfg2.memory_save_path)
    w2 = runner2.agent.flat_weights()
    assert torch.allclose(w1, w2)
    assert len(runner2.replay) == len(r1.replay)
    # restored replay is sampleable and trains
    runner2.run_round()
    assert runner2.grad_steps > 0


def test_sharded_replay_state_dict_roundtrip():
    cfg = tiny_cfg()
    rep = ShardedReplay(cfg, n_actors=2, capacity=512)
    rng = np.random.default_rng(0)
    for a in range(2):
        frames = rng.integers(0, 255, size=(300, 84, 84), dtype=np.uint8)
        actions = rng.integers(0, 6, size=300)
        rewards = rng.random(300).astype(np.float32)
        dones = rng.random(300) < 0.05
        rep.append_actor_chunk(a, frames, actions, rewards, dones)
    sd = rep.state_dict()

    rep2 = ShardedReplay(cfg, n_actors=2, capacity=512)
    rep2.load_state_dict(sd)
    assert len(rep2) == len(rep)
    for r1, r2 in zip(rep.rings, rep2.rings):
        assert np.array_equal(r1.frames, r2.frames)
        assert abs(r1.tree.total() - r2.tree.total()) < 1e-4
    # sampling from the restored replay works
    torch.manual_seed(0)
    np.random.seed(0)
    _idx, s, a, r, s2, nt, w = rep2.sample(8)
    assert s.shape == (8, cfg.history_length, 84, 84)
    assert torch.isfinite(w).all()


def test_svg_plot_fallback(tmp_path):
    from rainbowiqn_amd.utils.logging import _plot_rewards_svg

    p = tmp_path / "r.html"
    _plot_rewards_svg([100, 200, 300], [[1, 2], [2, 3], [3, 4]], str(p), "t")
    body = p.read_text()
    assert "<svg" in body and "polyline" in body


def test_metrics_exporter(tmp_path):
    import urllib.request

    cfg = tiny_cfg()
    runner = train_apex(cfg, n_envs=8, results_dir=str(tmp_path),
                        max_rounds=25, log_interval=10, metrics_port=18123)
    body = urllib.request.urlopen(
        "http://127.0.0.1:18123/metrics", timeout=5).read().decode()
    assert "riqn_grad_steps_total" in body
    assert "riqn_env_steps_per_sec" in body


def test_multiseed_launcher_smoke(tmp_path):
    """scripts/launch_multiseed.py (reference multi-seed run scripts,
    SURVEY 3.1 plotting/multi-seed row): two sequential tiny seeds run and
    produce per-seed results dirs + a collation line."""
    import json
    import subprocess
    import sys

    out = subprocess.run(
        [sys.executable, "scripts/launch_multiseed.py", "--seeds", "2",
         "--game", "cue", "--id", "mstest-tmp", "--",
         "--max-rounds", "2", "--n-envs", "4", "--chunk-steps", "4",
         "--hidden-size", "32", "--memory-capacity", "2048",
         "--learn-start", "64", "--learn-per-round", "1",
         "--num-tau-samples", "2", "--num-tau-prime-samples", "2",
         "--num-quantile-samples", "3",
         "--evaluation-interval", "1000000000",
         "--device", "cpu", "--dtype", "fp32"],
        capture_output=True, text=True, timeout=600, check=True,
    )
    lines = [l for l in out.stdout.splitlines() if l.startswith("{")]
    assert lines, out.stdout
    runs = json.loads(lines[-1])["runs"]
    assert len(runs) == 2
    import os
    import shutil

    for r in runs:
        assert os.path.isdir(r)
        shutil.rmtree(r, ignore_errors=True)
