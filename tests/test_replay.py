import numpy as np
import pytest
import torch

from rainbowiqn_amd.replay import ReplayMemory


def _fill(mem, n, reward_fn=lambda i: float(i % 5), term_every=None):
    h = mem.history
    for i in range(n):
        frame = np.full((84, 84), i % 256, dtype=np.uint8)
        state = np.repeat(frame[None], h, axis=0).astype(np.float32) / 255.0
        terminal = term_every is not None and (i + 1) % term_every == 0
        mem.append(state, action=i % 4, reward=reward_fn(i), terminal=terminal)


def test_append_and_len(cfg):
    mem = ReplayMemory(cfg, capacity=128)
    _fill(mem, 50)
    assert len(mem) == 50
    _fill(mem, 100)
    assert len(mem) == 128  # capped at capacity


def test_sample_shapes(cfg):
    mem = ReplayMemory(cfg, capacity=256)
    _fill(mem, 200)
    idxs, s, a, r, s2, nt, w = mem.sample(16)
    assert s.shape == (16, cfg.history_length, 84, 84)
    assert s2.shape == (16, cfg.history_length, 84, 84)
    assert a.shape == (16,) and r.shape == (16,) and nt.shape == (16,) and w.shape == (16,)
    assert s.dtype == torch.float32
    assert (w > 0).all() and (w <= 1.0).all()
    assert ((nt == 0) | (nt == 1)).all()


def test_sampled_windows_never_cross_write_head(cfg):
    mem = ReplayMemory(cfg, capacity=64)
    _fill(mem, 500)  # ring wrapped many times
    for _ in range(20):
        idxs, *_ = mem.sample(16)
        fwd = (mem.index - idxs) % mem.capacity
        assert (fwd > mem.n).all()
        bwd = (idxs - mem.index) % mem.capacity
        assert (bwd >= mem.history - 1).all()


def test_nstep_return_values(cfg):
    """Deterministic check of the n-step return: distinct rewards, no terminals."""
    mem = ReplayMemory(cfg, capacity=256)
    _fill(mem, 200, reward_fn=lambda i: float(i))
    idxs, _, _, r, _, nt, _ = mem.sample(32)
    g = cfg.discount
    for k, t in enumerate(idxs):
        expect = sum(g**s * float((t + s) % 256 if (t + s) < 200 else (t + s)) for s in range(mem.n))
        # reward at slot j was j (by construction, no wraparound for t<200-n)
        expect = sum(g**s * (t + s) for s in range(mem.n))
        assert r[k].item() == pytest.approx(expect, rel=1e-5)
        assert nt[k] == 1.0


def test_terminal_cutoff(cfg):
    """Episodes of length 8: samples near terminals must cut off the return."""
    mem = ReplayMemory(cfg, capacity=256)
    _fill(mem, 200, reward_fn=lambda i: 1.0, term_every=8)
    found_cut = False
    for _ in range(50):
        idxs, _, _, r, _, nt, _ = mem.sample(16)
        for k, t in enumerate(idxs):
            # terminal at slots 7, 15, 23, ...
            dist_to_term = (7 - t) % 8  # steps until the next terminal slot
            g = cfg.discount
            if dist_to_term < mem.n:  # a terminal inside [t, t+n) cuts
                assert nt[k] == 0.0
                found_cut = True
                n_rew = dist_to_term + 1
                expect = sum(g**s for s in range(n_rew))
                assert r[k].item() == pytest.approx(expect, rel=1e-5)
            else:
                assert nt[k] == 1.0
    assert found_cut


def test_pre_episode_frames_zeroed(cfg):
    """First frames of an episode must have zeroed history before them."""
    mem = ReplayMemory(cfg, capacity=256)
    _fill(mem, 100, term_every=10)
    # slot 10 is the first of an episode (timestep 0)
    assert mem.timesteps[10] == 0
    states, _, _, _ = mem._assemble(np.array([10]))
    # history frames before the episode start are zero
    assert (states[0, : cfg.history_length - 1] == 0).all()
    assert states[0, -1].max() > 0


def test_update_priorities_changes_sampling(cfg):
    mem = ReplayMemory(cfg, capacity=256)
    _fill(mem, 200)
    # crush all priorities except slot 50
    all_idx = np.arange(150)
    mem.update_priorities(all_idx, np.full(150, 1e-12))
    mem.update_priorities(np.array([50]), np.array([1e6]))
    idxs, *_ = mem.sample(32)
    assert (idxs == 50).sum() >= 24  # dominates sampling


def test_priority_exponent_applied(cfg):
    mem = ReplayMemory(cfg, capacity=64)
    _fill(mem, 32)
    mem.update_priorities(np.array([3]), np.array([4.0]))
    assert mem.tree.get(np.array([3]))[0] == pytest.approx(4.0**cfg.priority_exponent)


def test_replay_checkpoint_roundtrip(cfg):
    mem = ReplayMemory(cfg, capacity=128)
    _fill(mem, 100, term_every=13)
    sd = mem.state_dict()
    mem2 = ReplayMemory(cfg, capacity=128)
    mem2.load_state_dict(sd)
    assert len(mem2) == len(mem)
    assert (mem2.frames[:100] == mem.frames[:100]).all()
    assert mem2.tree.total() == pytest.approx(mem.tree.total())
    idxs, *_ = mem2.sample(8)
    assert len(idxs) == 8


def test_chunk_append_matches_single(cfg):
    m1 = ReplayMemory(cfg, capacity=128)
    m2 = ReplayMemory(cfg, capacity=128)
    rng = np.random.default_rng(0)
    frames = rng.integers(0, 255, size=(40, 84, 84), dtype=np.uint8)
    actions = rng.integers(0, 4, size=40)
    rewards = rng.random(40).astype(np.float32)
    terms = rng.random(40) < 0.1
    for i in range(40):
        m1.append(frames[i], int(actions[i]), float(rewards[i]), bool(terms[i]))
    m2.append_chunk(frames, actions, rewards, terms)
    assert (m1.frames[:40] == m2.frames[:40]).all()
    assert (m1.timesteps[:40] == m2.timesteps[:40]).all()
    assert m1.tree.total() == pytest.approx(m2.tree.total())
