"""HIP kernel parity tests vs the pure-PyTorch fp32 references (torch_ref /
CPU replay).  All run on an MI355X via gpurun; numerics compare against eager
fp32 at fp32 tolerances."""

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def ext():
    from rainbowiqn_amd.ops import get_extension

    return get_extension()


@pytest.fixture
def gcfg():
    from rainbowiqn_amd.config import Config

    return Config(batch_size=32, device="cuda")


# ---------------------------------------------------------------------------
# quantile-Huber loss kernel
# ---------------------------------------------------------------------------
@pytest.mark.parametrize("kappa", [1.0, 2.0])
@pytest.mark.parametrize("shapes", [(32, 8, 8), (5, 3, 7), (64, 64, 32)])
def test_quantile_huber_matches_ref(ext, kappa, shapes):
    from rainbowiqn_amd.ops import torch_ref
    from rainbowiqn_amd.ops.functional import quantile_huber_loss_fused

    torch.manual_seed(0)
    B, N, Np = shapes
    theta = (torch.randn(B, N, device="cuda") * 2).requires_grad_()
    target_q = torch.randn(B, Np, device="cuda") * 2
    taus = torch.rand(B, N, device="cuda")
    returns = torch.randn(B, device="cuda")
    nonterm = (torch.rand(B, device="cuda") > 0.3).float()
    weights = torch.rand(B, device="cuda") + 0.5
    discount, n_step = 0.99, 3

    loss, per = quantile_huber_loss_fused(
        theta, target_q, taus, returns, nonterm, weights, discount, n_step, kappa
    )
    loss.backward()
    g_fused = theta.grad.clone()

    theta2 = theta.detach().clone().requires_grad_()
    t_theta = torch_ref.n_step_target(target_q, returns, nonterm, discount, n_step)
    loss_ref, per_ref = torch_ref.quantile_huber_loss(theta2, t_theta, taus, weights, kappa)
    loss_ref.backward()

    assert torch.allclose(loss, loss_ref, rtol=1e-5, atol=1e-6), (loss, loss_ref)
    assert torch.allclose(per, per_ref, rtol=1e-4, atol=1e-5)
    assert torch.allclose(g_fused, theta2.grad, rtol=1e-4, atol=1e-6)


def test_quantile_huber_no_weights(ext):
    from rainbowiqn_amd.ops import torch_ref
    from rainbowiqn_amd.ops.functional import quantile_huber_loss_fused

    torch.manual_seed(1)
    theta = torch.randn(8, 4, device="cuda", requires_grad=True)
    target_q = torch.randn(8, 5, device="cuda")
    taus = torch.rand(8, 4, device="cuda")
    r = torch.zeros(8, device="cuda")
    nt = torch.ones(8, device="cuda")
    loss, _ = quantile_huber_loss_fused(theta, target_q, taus, r, nt, None, 0.99, 1)
    t_theta = torch_ref.n_step_target(target_q, r, nt, 0.99, 1)
    loss_ref, _ = torch_ref.quantile_huber_loss(theta.detach(), t_theta, taus, None)
    assert torch.allclose(loss, loss_ref, rtol=1e-5, atol=1e-6)


def test_grad_scale_by_upstream(ext):
    from rainbowiqn_amd.ops.functional import quantile_huber_loss_fused

    theta = torch.randn(4, 3, device="cuda", requires_grad=True)
    args = (torch.randn(4, 5, device="cuda"), torch.rand(4, 3, device="cuda"),
            torch.zeros(4, device="cuda"), torch.ones(4, device="cuda"), None,
            0.99, 3, 1.0)
    loss, _ = quantile_huber_loss_fused(theta, *args)
    (loss * 2.5).backward()
    g1 = theta.grad.clone()
    theta.grad = None
    loss, _ = quantile_huber_loss_fused(theta, *args)
    loss.backward()
    assert torch.allclose(g1, theta.grad * 2.5, rtol=1e-5, atol=1e-7)


# ---------------------------------------------------------------------------
# preprocess + Adam kernels
# ---------------------------------------------------------------------------
def test_preprocess_matches_ref(ext):
    from rainbowiqn_amd.ops.functional import preprocess_fused

    x = torch.randint(0, 256, (3, 4, 84, 84), dtype=torch.uint8, device="cuda")
    out = preprocess_fused(x)
    assert torch.allclose(out, x.float() / 255.0)


def test_fused_adam_matches_torch(ext):
    from rainbowiqn_amd.ops.flat_params import FusedAdam

    torch.manual_seed(0)
    n = 100_003
    p_ref = torch.randn(n, device="cuda")
    p = p_ref.clone()
    g = torch.randn(n, device="cuda")
    opt_ref = torch.optim.Adam([p_ref.requires_grad_()], lr=1e-3, eps=1.5e-4)
    fused = FusedAdam(p, g, lr=1e-3, eps=1.5e-4)
    for step in range(5):
        gi = torch.randn(n, device="cuda")
        g.copy_(gi)
        p_ref.grad = gi.clone()
        opt_ref.step()
        fused.step()
        assert torch.allclose(p, p_ref.detach(), rtol=1e-5, atol=1e-6), step


def test_flat_params_backward_into_flat_grad(ext, gcfg):
    from rainbowiqn_amd.models import RainbowIQN
    from rainbowiqn_amd.ops.flat_params import attach_flat_grads, flatten_module

    m = RainbowIQN(gcfg, 6).cuda()
    flat = flatten_module(m)
    fg = attach_flat_grads(m)
    z, _ = m(torch.rand(2, 4, 84, 84, device="cuda"), 4)
    z.sum().backward()
    assert fg.abs().sum() > 0
    # parameter mutation through the flat buffer reaches the module
    with torch.no_grad():
        flat.add_(1.0)
    assert float(m.fc_h_v.bias_mu[0].detach()) != 0.0


# ---------------------------------------------------------------------------
# GPU replay
# ---------------------------------------------------------------------------
def _mk_gpu_replay(gcfg, capacity=1024, n_rings=1, batch=32):
    from rainbowiqn_amd.replay.gpu_replay import GpuReplay

    gcfg.memory_capacity = capacity
    return GpuReplay(gcfg, n_rings=n_rings, batch_size=batch)


def _rand_chunk(rng, c, n_act=6, p_term=0.08):
    frames = rng.integers(0, 255, size=(c, 84, 84), dtype=np.uint8)
    actions = rng.integers(0, n_act, size=c)
    rewards = rng.standard_normal(c).astype(np.float32)
    terms = rng.random(c) < p_term
    return frames, actions, rewards, terms


def test_replay_append_state(ext, gcfg):
    rep = _mk_gpu_replay(gcfg, capacity=256)
    rng = np.random.default_rng(0)
    f, a, r, t = _rand_chunk(rng, 40)
    rep.append_actor_chunk(0, f, a, r, t)
    torch.cuda.synchronize()
    rs = rep.ring_state.cpu().numpy()
    assert rs[0, 0] == 40 and rs[0, 1] == 40  # head, count
    assert (rep.frames[:40].cpu().numpy() == f).all()
    assert (rep.actions[:40].cpu().numpy() == a).all()
    np.testing.assert_allclose(rep.rewards[:40].cpu().numpy(), r, rtol=1e-6)
    assert (rep.terminals[:40].cpu().numpy() == t.astype(np.uint8)).all()
    # timesteps match the CPU replay's episode bookkeeping
    from rainbowiqn_amd.replay import ReplayMemory

    cmem = ReplayMemory(gcfg, capacity=256)
    cmem.append_chunk(f, a, r, t)
    assert (rep.timesteps[:40].cpu().numpy() == cmem.timesteps[:40]).all()
    # leaves got max_priority=1, tree total consistent
    assert rep.tree_total() == pytest.approx(40.0)


def test_replay_ring_wraparound(ext, gcfg):
    rep = _mk_gpu_replay(gcfg, capacity=128)
    rng = np.random.default_rng(1)
    for _ in range(5):  # 5*40 = 200 > 128: wraps
        f, a, r, t = _rand_chunk(rng, 40)
        rep.append_actor_chunk(0, f, a, r, t)
    torch.cuda.synchronize()
    rs = rep.ring_state.cpu().numpy()
    assert rs[0, 0] == 200 % 128 and rs[0, 1] == 128
    assert rep.tree_total() == pytest.approx(128.0)


def test_replay_sample_valid_and_weighted(ext, gcfg):
    rep = _mk_gpu_replay(gcfg, capacity=512, batch=32)
    rng = np.random.default_rng(2)
    for _ in range(4):
        f, a, r, t = _rand_chunk(rng, 100)
        rep.append_actor_chunk(0, f, a, r, t)
    idx, s, a_, ret, s2, nt, w = rep.sample()
    torch.cuda.synchronize()
    idxs = idx.cpu().numpy()
    head = int(rep.ring_state[0, 0])
    fwd = (head - idxs) % rep.ring_cap
    assert (fwd > rep.n).all()  # never samples the invalid head window
    assert (idxs < 400).all()
    assert s.shape == (32, 4, 84, 84) and s.dtype == torch.float32
    assert s.max() <= 1.0 and s.min() >= 0.0
    assert ((nt == 0) | (nt == 1)).cpu().all()
    assert w.max().item() == pytest.approx(1.0)  # normalized by batch max
    assert (w > 0).cpu().all()
    # uniform priorities -> all weights 1
    assert torch.allclose(w, torch.ones_like(w))


def test_replay_sample_assembly_matches_cpu(ext, gcfg):
    """Force the sampler onto one specific slot; assembled batch must equal
    the CPU replay's assembly of the same data."""
    from rainbowiqn_amd.replay import ReplayMemory

    rep = _mk_gpu_replay(gcfg, capacity=256, batch=32)
    cmem = ReplayMemory(gcfg, capacity=256)
    rng = np.random.default_rng(3)
    f, a, r, t = _rand_chunk(rng, 200, p_term=0.1)
    rep.append_actor_chunk(0, f, a, r, t)
    cmem.append_chunk(f, a, r, t)
    for target in [7, 63, 120, int(np.argmax(t[:150])) + 1]:
        # crush all priorities, then spike `target`
        rep.update_priorities(
            torch.arange(200, dtype=torch.int32, device="cuda"),
            torch.full((200,), 1e-7, device="cuda"),
        )
        rep.update_priorities(
            torch.tensor([target], dtype=torch.int32, device="cuda"),
            torch.tensor([1e6], device="cuda"),
        )
        idx, s, a_, ret, s2, nt, w = rep.sample()
        torch.cuda.synchronize()
        assert (idx.cpu().numpy() == target).all(), idx.cpu().numpy()
        states, next_states, returns, nonterms = cmem._assemble(np.array([target]))
        np.testing.assert_allclose(
            s[0].cpu().numpy(), states[0], rtol=0, atol=1e-6
        )
        np.testing.assert_allclose(
            s2[0].cpu().numpy(), next_states[0], rtol=0, atol=1e-6
        )
        assert ret[0].item() == pytest.approx(returns[0], abs=1e-5)
        assert nt[0].item() == nonterms[0]
        assert a_[0].item() == a[target]


def test_replay_priority_sampling_distribution(ext, gcfg):
    rep = _mk_gpu_replay(gcfg, capacity=512, batch=64)
    rng = np.random.default_rng(4)
    f, a, r, t = _rand_chunk(rng, 400, p_term=0.0)
    rep.append_actor_chunk(0, f, a, r, t)
    # spike slot 100 to ~90% of the mass
    rep.update_priorities(
        torch.arange(400, dtype=torch.int32, device="cuda"),
        torch.full((400,), 1e-4, device="cuda"),
    )
    rep.update_priorities(
        torch.tensor([100], dtype=torch.int32, device="cuda"),
        torch.tensor([1.0], device="cuda"),
    )
    # p^0.5: slot100 = 1.0, others = 1e-2 -> mass 1/(1+399*.01) = 0.2
    hits = 0
    for _ in range(10):
        idx, *_ = rep.sample()
        hits += int((idx == 100).sum().item())
    frac = hits / (10 * 64)
    assert 0.1 < frac < 0.35, frac  # expect ~0.2


def test_replay_update_priorities_and_rebuild(ext, gcfg):
    rep = _mk_gpu_replay(gcfg, capacity=256)
    rng = np.random.default_rng(5)
    f, a, r, t = _rand_chunk(rng, 128)
    rep.append_actor_chunk(0, f, a, r, t)
    pri = torch.rand(128, device="cuda") + 0.1
    rep.update_priorities(torch.arange(128, dtype=torch.int32, device="cuda"), pri)
    torch.cuda.synchronize()
    expect = (pri.cpu().numpy() ** rep.priority_exponent).sum()
    assert rep.tree_total() == pytest.approx(expect, rel=1e-4)
    # full rebuild agrees with the atomically-maintained tree
    drift_before = rep.tree_total()
    rep.rebuild_tree()
    torch.cuda.synchronize()
    assert rep.tree_total() == pytest.approx(drift_before, rel=1e-4)
    leaf_sum = rep.leaf_priorities().sum().item()
    assert rep.tree_total() == pytest.approx(leaf_sum, rel=1e-5)


def test_replay_multi_ring(ext, gcfg):
    rep = _mk_gpu_replay(gcfg, capacity=512, n_rings=4, batch=64)
    rng = np.random.default_rng(6)
    for ring in range(4):
        f, a, r, t = _rand_chunk(rng, 100)
        rep.append_actor_chunk(ring, f, a, r, t)
    idx, *_ = rep.sample()
    torch.cuda.synchronize()
    rings_hit = set((idx.cpu().numpy() // rep.ring_cap).tolist())
    assert rings_hit == {0, 1, 2, 3}  # all rings represented in a 64-batch
    rs = rep.ring_state.cpu().numpy()
    assert (rs[:, 1] == 100).all()


def test_replay_sample_inside_hipgraph(ext, gcfg):
    """The whole sample op must be graph-capturable and produce fresh draws
    on each replay (device-side philox offset)."""
    rep = _mk_gpu_replay(gcfg, capacity=512, batch=32)
    rng = np.random.default_rng(7)
    f, a, r, t = _rand_chunk(rng, 400)
    rep.append_actor_chunk(0, f, a, r, t)
    rep.sample()  # warmup
    torch.cuda.synchronize()
    g = torch.cuda.CUDAGraph()
    with torch.cuda.graph(g):
        rep.sample()
    g.replay()
    torch.cuda.synchronize()
    i1 = rep.out_idx.cpu().numpy().copy()
    g.replay()
    torch.cuda.synchronize()
    i2 = rep.out_idx.cpu().numpy().copy()
    assert (i1 != i2).any()  # RNG advanced across replays
    head = int(rep.ring_state[0, 0])
    for ii in (i1, i2):
        fwd = (head - ii) % rep.ring_cap
        assert (fwd > rep.n).all()


# ---------------------------------------------------------------------------
# device env + graph-captured actor/learner
# ---------------------------------------------------------------------------
def test_gpu_actor_loop_round(ext, gcfg):
    from rainbowiqn_amd.models import RainbowIQN
    from rainbowiqn_amd.runtime.gpu_actor import GpuActorLoop

    net = RainbowIQN(gcfg, 6).cuda()
    loop = GpuActorLoop(gcfg, net, n_envs=8, chunk_steps=8, use_graph=True)
    loop.reset_noise()
    loop.run_round()
    loop.run_round()
    torch.cuda.synchronize()
    meta = loop.chunk_meta.cpu().numpy()
    assert ((meta[:, 0] >= 0) & (meta[:, 0] < 6)).all()  # actions in range
    assert np.isin(np.unique(meta[:, 2]), [0.0, 1.0]).all()
    assert loop.chunk_frames.cpu().numpy().max() > 0  # frames synthesized
    assert loop.total_steps == 3 * 8 * 8  # warmup + 2 replays
    es = loop.env_state.cpu().numpy()
    assert (es[:, 0] >= 0).all() and (es[:, 1] >= 0).all()


def test_gpu_actor_rewards_sparse(ext, gcfg):
    from rainbowiqn_amd.models import RainbowIQN
    from rainbowiqn_amd.runtime.gpu_actor import GpuActorLoop

    net = RainbowIQN(gcfg, 6).cuda()
    loop = GpuActorLoop(gcfg, net, n_envs=16, chunk_steps=16, use_graph=False)
    for _ in range(8):
        loop.run_round()
    torch.cuda.synchronize()
    meta = loop.chunk_meta.cpu().numpy()
    # reward sparsity ~2%, clipped to [-1, 1]
    assert (np.abs(meta[:, 1]) <= 1.0).all()


def test_gpu_learner_graph_learns(ext, gcfg):
    from rainbowiqn_amd.runtime.gpu_learner import GpuLearner

    gcfg.learn_start = 128
    rng = np.random.default_rng(0)
    lrn = GpuLearner(gcfg, 6, n_rings=1, use_graph=True)
    f, a, r, t = _rand_chunk(rng, 512)
    lrn.replay.append_actor_chunk(0, f, a, r, t)
    w0 = lrn.flat_online.clone()
    lrn.learn_steps(8)
    torch.cuda.synchronize()
    assert lrn.graph is not None  # captured
    assert lrn.grad_steps == 8
    assert not torch.equal(w0, lrn.flat_online)  # weights moved
    loss1 = lrn.last_loss()
    assert np.isfinite(loss1)
    # replays keep advancing RNG/adam state
    lrn.learn_steps(4)
    torch.cuda.synchronize()
    assert int(lrn.opt.step_t.item()) == 12
    lrn.update_target()
    assert torch.equal(lrn.flat_target, lrn.flat_online)


def test_gpu_learner_priorities_written_back(ext, gcfg):
    from rainbowiqn_amd.runtime.gpu_learner import GpuLearner

    gcfg.learn_start = 128
    rng = np.random.default_rng(1)
    lrn = GpuLearner(gcfg, 6, n_rings=1, use_graph=False)
    f, a, r, t = _rand_chunk(rng, 512)
    lrn.replay.append_actor_chunk(0, f, a, r, t)
    torch.cuda.synchronize()
    t0 = lrn.replay.tree_total()
    lrn.learn_steps(2)
    torch.cuda.synchronize()
    assert lrn.replay.tree_total() != pytest.approx(t0)
