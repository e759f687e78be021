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


def test_randomized_differential_vs_bruteforce(cfg):
    """Randomized episodes: every sampled transition's frame stack, n-step
    return, nonterminal flag and next-state stack must match a brute-force
    replayer of the raw stream (reference memory.py semantics)."""
    rng = np.random.default_rng(42)
    mem = ReplayMemory(cfg, capacity=512)
    N = 400
    frames = rng.integers(0, 256, size=(N, 84, 84)).astype(np.uint8)
    actions = rng.integers(0, 6, size=N)
    rewards = rng.normal(size=N).astype(np.float64)
    # random episode ends, ~1/12 of steps
    terminals = rng.random(N) < 1.0 / 12
    t_in_ep = np.zeros(N, dtype=int)
    t = 0
    for i in range(N):
        t_in_ep[i] = t
        t = 0 if terminals[i] else t + 1
    h, n, g = cfg.history_length, cfg.multi_step, cfg.discount
    for i in range(N):
        state = np.repeat(frames[i][None], h, 0).astype(np.float32) / 255.0
        mem.append(state, int(actions[i]), float(rewards[i]),
                   bool(terminals[i]))

    def brute_stack(i):
        """Frame stack ending at slot i, zeroed before the episode start."""
        out = np.zeros((h, 84, 84), dtype=np.float32)
        for j in range(h):
            back = h - 1 - j
            if t_in_ep[i] >= back and i - back >= 0:
                out[j] = frames[i - back].astype(np.float32) / 255.0
        return out

    for _ in range(30):
        idxs, s, a, r, s2, nt, w = mem.sample(16)
        for k, i in enumerate(np.asarray(idxs)):
            i = int(i)
            assert a[k].item() == actions[i]
            # n-step return with terminal cutoff
            expect_r, done = 0.0, False
            steps = 0
            for sct in range(n):
                expect_r += (g ** sct) * rewards[i + sct]
                steps = sct + 1
                if terminals[i + sct]:
                    done = True
                    break
            assert r[k].item() == pytest.approx(expect_r, rel=1e-4, abs=1e-5)
            assert nt[k].item() == (0.0 if done else 1.0)
            np.testing.assert_allclose(s[k].numpy(), brute_stack(i),
                                       rtol=0, atol=1e-6)
            if not done:
                np.testing.assert_allclose(s2[k].numpy(),
                                           brute_stack(i + n),
                                           rtol=0, atol=1e-6)


def test_remote_replay_states_cat_contract(cfg):
    """RemoteReplay (sharded mode learner facade) must expose the GpuReplay
    states_cat contract the manual fused backward reads: sample() fills one
    contiguous [s; s'] buffer and returns views into it."""
    import torch

    from rainbowiqn_amd.parallel.sharded import RemoteReplay

    class _Comm:
        n_actors = 3

    cfg.batch_size = 8
    rep = RemoteReplay(cfg, _Comm(), steps=4, device="cpu",
                       dtype=torch.float32)
    B = cfg.batch_size
    rep.bst.normal_()
    rep.bs2.normal_()
    for s in range(6):  # wraps past S to check the cycling counter
        _, st, a, r, s2, nt, w = rep.sample()
        assert st.data_ptr() == rep.states_cat.data_ptr()
        assert torch.equal(st, rep.bst[s % 4])
        assert torch.equal(s2, rep.bs2[s % 4])
        assert torch.equal(rep.states_cat[:B], rep.bst[s % 4])
        assert torch.equal(rep.states_cat[B:], rep.bs2[s % 4])
        rep.update_priorities(None, torch.full((B,), float(s)))
        assert torch.equal(rep.bpri[s % 4], torch.full((B,), float(s)))
