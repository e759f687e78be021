"""GPU smoke tests (eager torch path; kernel parity tests live in
test_kernels_gpu.py once the HIP extension exists)."""

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture
def gcfg():
    from rainbowiqn_amd.config import Config

    return Config(
        memory_capacity=8192, learn_start=256, batch_size=32, device="cuda",
        dtype="bf16",
    )


def test_agent_learn_on_gpu(gcfg):
    from rainbowiqn_amd.agent import Agent

    agent = Agent(gcfg, 6, "cuda")
    b = gcfg.batch_size
    loss, pri = agent.learn_from_batch(
        torch.rand(b, 4, 84, 84, device="cuda"),
        torch.randint(0, 6, (b,), device="cuda"),
        torch.rand(b, device="cuda"),
        torch.rand(b, 4, 84, 84, device="cuda"),
        torch.ones(b, device="cuda"),
        torch.ones(b, device="cuda"),
    )
    torch.cuda.synchronize()
    assert torch.isfinite(loss)
    assert pri.shape == (b,) and torch.isfinite(pri).all() and (pri >= 0).all()


def test_act_batch_on_gpu(gcfg):
    from rainbowiqn_amd.agent import Agent

    agent = Agent(gcfg, 6, "cuda")
    a = agent.act_batch(torch.rand(16, 4, 84, 84, device="cuda"))
    assert a.shape == (16,)
    assert (a >= 0).all() and (a < 6).all()


def test_apex_single_gpu_rounds(gcfg):
    from rainbowiqn_amd.parallel.apex import ApexRunner
    from rainbowiqn_amd.parallel.comm import Communicator

    gcfg.learn_start = 64
    gcfg.learn_per_round = 2
    comm = Communicator(backend="auto")
    runner = ApexRunner(gcfg, comm, n_envs=8, chunk_steps=8)
    for _ in range(3):
        runner.run_round()
    torch.cuda.synchronize()
    # overlap mode appends with a one-round staging lag (+ the graph-capture
    # warmup round's chunk is recorded but round 0's append lands in round 1)
    expect = 3 * 64 if not runner.overlap else 2 * 64
    assert len(runner.replay) == expect, len(runner.replay)
    assert runner.grad_steps > 0
    assert all(np.isfinite(l) for l in runner.losses)


def test_bf16_autocast_matches_fp32_roughly(gcfg):
    """bf16 act decisions should mostly agree with fp32 on the same weights."""
    from rainbowiqn_amd.agent import Agent

    torch.manual_seed(0)
    a16 = Agent(gcfg, 6, "cuda")
    gcfg32 = type(gcfg)(**{**gcfg.__dict__, "dtype": "fp32"})
    a32 = Agent(gcfg32, 6, "cuda")
    a32.online_net.load_state_dict(a16.online_net.state_dict())
    x = torch.rand(64, 4, 84, 84, device="cuda")
    with torch.no_grad():
        taus = torch.rand(64 * 8, 1, device="cuda")
        z16, _ = a16.online_net(x, 8, taus=taus)
        z32, _ = a32.online_net(x, 8, taus=taus)
    # same taus + same weights: values close at bf16 tolerance
    assert (z16.float() - z32).abs().max() < 0.15


@pytest.mark.gpu
def test_apex_multi_actor_one_gpu():
    """BASELINE config 3: several independent actors (own nets/noise/rings)
    plus the learner colocated on ONE GPU (reference: 4 actor processes)."""
    from rainbowiqn_amd.config import Config
    from rainbowiqn_amd.parallel.apex import ApexRunner
    from rainbowiqn_amd.parallel.comm import Communicator

    cfg = Config(game="breakout", device="cuda", dtype="bf16",
                 learn_start=512, memory_capacity=16384, nb_actor=4,
                 learn_per_round=2)
    comm = Communicator(backend="auto")
    runner = ApexRunner(cfg, comm, n_envs=16, chunk_steps=8)
    assert len(runner.gactors) == 4
    assert runner.replay.n_rings == 4
    for _ in range(8):
        runner.run_round()
    torch.cuda.synchronize()
    assert runner.grad_steps > 0
    # every actor's ring received transitions
    counts = runner.replay.ring_state[:, 1].cpu()
    assert (counts > 0).all(), counts
    # M * (rounds + 1 capture-warmup round) * E * steps
    assert runner.actor.total_steps == 4 * 9 * 16 * 8


@pytest.mark.gpu
def test_replay_large_capacity_hbm():
    """BASELINE config 5 (capacity leg): a multi-million-transition PER ring
    resident in HBM3E -- 10M transitions is ~71 GB of frames on one GPU;
    append and stratified sampling stay kernel-bound."""
    from rainbowiqn_amd.config import Config
    from rainbowiqn_amd.replay.gpu_replay import GpuReplay

    free, total = torch.cuda.mem_get_info()
    if free < 90 * 2**30:
        pytest.skip("needs ~90 GB free HBM")
    cfg = Config(device="cuda", dtype="bf16")
    rep = GpuReplay(cfg, n_rings=8, capacity=10_000_000, device="cuda",
                    out_dtype=torch.bfloat16)
    assert rep.capacity >= 10_000_000 - 8
    cf = torch.randint(0, 255, (4096, 84, 84), dtype=torch.uint8, device="cuda")
    cm = torch.rand(4096, 3, device="cuda")
    for ring in range(8):
        for _ in range(2):
            rep.append_actor_chunk(ring, cf, meta=cm)
    torch.cuda.synchronize()
    assert len(rep) == 8 * 2 * 4096
    idx, s, a, r, s2, nt, w = rep.sample()
    torch.cuda.synchronize()
    assert torch.isfinite(w).all() and (w > 0).all()
    assert s.shape == (cfg.batch_size, 4, 84, 84)
    tot = rep.tree_total()
    assert abs(tot - len(rep)) / len(rep) < 0.01  # all priorities 1.0
    del rep
    torch.cuda.empty_cache()


@pytest.mark.gpu
def test_checkpoint_roundtrip_gpu(tmp_path):
    """Reference-format checkpoint parity on the real path: bf16 GpuLearner
    (fp32 Adam master) -> torch.save -> eager fp32 Agent -> same greedy
    Q-values in eval mode (mu weights only)."""
    from rainbowiqn_amd.agent import Agent
    from rainbowiqn_amd.config import Config
    from rainbowiqn_amd.runtime.gpu_learner import GpuLearner

    cfg = Config(device="cuda", dtype="bf16", learn_start=128,
                 memory_capacity=4096, batch_size=32)
    lrn = GpuLearner(cfg, 6, n_rings=1, use_graph=False)
    rng = np.random.default_rng(3)
    frames = rng.integers(0, 255, size=(512, 84, 84), dtype=np.uint8)
    lrn.replay.append_actor_chunk(0, frames, rng.integers(0, 6, 512),
                                  rng.random(512).astype(np.float32),
                                  rng.random(512) < 0.02)
    lrn.learn_steps(5)
    torch.cuda.synchronize()
    lrn.save(str(tmp_path))

    cfg2 = Config(device="cuda", dtype="fp32", model=str(tmp_path))
    agent = Agent(cfg2, 6, "cuda")
    agent.eval()  # mu weights only -> deterministic across impls
    # fp32 reference net from the saved fp32 state_dict vs the learner's
    # fp32 master: parameters must match exactly
    sd = lrn.state_dict()
    for k, v in agent.online_net.state_dict().items():
        if k in sd and "epsilon" not in k:
            assert torch.allclose(v, sd[k].to(v.device)), k
    # and a forward produces finite, matching greedy actions vs the bf16 net
    x = torch.rand(8, 4, 84, 84, device="cuda")
    taus = torch.rand(8 * 32, 1, device="cuda")
    agent.online_net.eval()
    lrn.online.eval()
    with torch.no_grad():
        q32 = agent.online_net(x, 32, taus=taus)[0].mean(dim=1)
        q16 = lrn.online(x.to(torch.bfloat16), 32, taus=taus)[0].float().mean(dim=1)
    err = (q32 - q16).abs().max().item()
    assert err < 0.05 * max(q32.abs().max().item(), 1.0) + 0.02, err
    agent.train()
    lrn.online.train()


@pytest.mark.gpu
def test_colocated_overlap_double_buffered_weights():
    """Colocated OVERLAP mode: actor and learner run concurrently on
    separate streams, so the actor must have its OWN net/FusedForward packs
    (a one-round-stale snapshot copied at round boundaries) -- sharing the
    learner's live buffers would tear mid-forward."""
    from rainbowiqn_amd.config import Config
    from rainbowiqn_amd.parallel.apex import ApexRunner
    from rainbowiqn_amd.parallel.comm import Communicator

    cfg = Config(device="cuda", dtype="bf16", learn_start=512,
                 memory_capacity=8192, learn_per_round=2)
    runner = ApexRunner(cfg, Communicator(backend="auto"), n_envs=16,
                        chunk_steps=8)
    assert runner.overlap
    assert runner.gactor.fused is not runner.glearner.fused_online
    assert runner.gactor.net is not runner.glearner.online
    assert runner._actor_flats[0] is not None
    wcat_before = runner.glearner.fused_online.wcat.clone()
    for _ in range(6):
        runner.run_round()
    torch.cuda.synchronize()
    # noise regenerated + Adam steps: the learner's GEMM operand moved
    assert not torch.equal(wcat_before, runner.glearner.fused_online.wcat)
    assert runner.grad_steps > 0
    # bounded staleness: after a final boundary copy the actor's flat
    # weights equal the learner's current flat weights
    runner.run_round(learn=False)
    torch.cuda.synchronize()
    assert torch.equal(runner._actor_flats[0], runner.glearner.flat_online)


@pytest.mark.gpu
def test_colocated_sequential_shares_fused_forward():
    """With overlap OFF the colocated round is strictly sequential and the
    actor shares the learner's online net + FusedForward (no copy cost)."""
    from rainbowiqn_amd.config import Config
    from rainbowiqn_amd.parallel.apex import ApexRunner
    from rainbowiqn_amd.parallel.comm import Communicator

    cfg = Config(device="cuda", dtype="bf16", learn_start=512,
                 memory_capacity=8192, learn_per_round=2,
                 overlap_colocated=False)
    runner = ApexRunner(cfg, Communicator(backend="auto"), n_envs=16,
                        chunk_steps=8)
    assert not runner.overlap
    assert runner.gactor.fused is runner.glearner.fused_online
    for _ in range(6):
        runner.run_round()
    torch.cuda.synchronize()
    assert runner.grad_steps > 0


@pytest.mark.gpu
def test_replay_tree_invariants_after_traffic():
    """Sum-tree node = sum(children) after interleaved appends, samples and
    priority updates (SURVEY.md 6.2 race-check row)."""
    from rainbowiqn_amd.config import Config
    from rainbowiqn_amd.replay.gpu_replay import GpuReplay

    cfg = Config(device="cuda", dtype="bf16", batch_size=32)
    rep = GpuReplay(cfg, n_rings=2, capacity=16384, device="cuda",
                    out_dtype=torch.bfloat16)
    cf = torch.randint(0, 255, (1024, 84, 84), dtype=torch.uint8, device="cuda")
    cm = torch.rand(1024, 3, device="cuda")
    for it in range(6):
        rep.append_actor_chunk(it % 2, cf, meta=cm)
        if len(rep) > 2048:
            idx, *_ = rep.sample()
            rep.update_priorities(idx, torch.rand(32, device="cuda") + 0.1)
    torch.cuda.synchronize()
    rep.check_invariants()


@pytest.mark.gpu
def test_nan_guard_fast_rollback():
    """Failure recovery on the fast path: corrupt the fp32 master + feed a
    non-finite loss -> NanGuard restores master, bf16 weights and Adam
    moments, and the runner keeps training."""
    from rainbowiqn_amd.config import Config
    from rainbowiqn_amd.parallel.apex import ApexRunner
    from rainbowiqn_amd.parallel.comm import Communicator
    from rainbowiqn_amd.utils.failsafe import NanGuard

    cfg = Config(device="cuda", dtype="bf16", learn_start=256,
                 memory_capacity=8192, learn_per_round=2)
    runner = ApexRunner(cfg, Communicator(backend="auto"), n_envs=16,
                        chunk_steps=8)
    for _ in range(4):
        runner.run_round()
    torch.cuda.synchronize()
    guard = NanGuard(runner, interval=1)
    assert not guard.check(runner.losses[-1])  # snapshot
    gl = runner.glearner
    healthy = gl.opt.master.detach().clone()
    with torch.no_grad():
        gl.opt.master.add_(float("inf"))
        gl.opt.m.fill_(float("nan"))
    assert guard.check(float("nan"))
    torch.cuda.synchronize()
    assert torch.equal(gl.opt.master, healthy)
    assert torch.isfinite(gl.opt.m).all()
    assert torch.equal(gl.flat_online.float(),
                       healthy.to(torch.bfloat16).float())
    # training continues with finite losses
    for _ in range(2):
        runner.run_round()
    torch.cuda.synchronize()
    assert np.isfinite(runner.losses[-1])


@pytest.mark.gpu
def test_actor_shard_host_gpu():
    """Config-5 actor-side shard on GPU: HBM-resident PER shard samples its
    learner-batch quota with local frame-stack/n-step assembly."""
    from types import SimpleNamespace

    from rainbowiqn_amd.config import Config
    from rainbowiqn_amd.parallel.sharded import ActorShardHost

    cfg = Config(device="cuda", dtype="bf16", memory_capacity=65536,
                 batch_size=32, learn_per_round=4)
    comm = SimpleNamespace(n_actors=7, rank=3)
    host = ActorShardHost(cfg, comm, quota=5, steps=4, device="cuda",
                          fast=True)
    cf = torch.randint(0, 255, (2048, 84, 84), dtype=torch.uint8,
                       device="cuda")
    cm = torch.rand(2048, 3, device="cuda")
    host.append_chunk(cf, meta=cm)
    host.append_chunk(cf, meta=cm)
    host.fill_buffers()
    torch.cuda.synchronize()
    assert host.st.shape == (4, 5, 4, 84, 84)
    assert torch.isfinite(host.ww).all() and (host.ww > 0).all()
    assert float(host.counts()[0]) == 4096
    # priority write-back path
    host.pri.uniform_(0.1, 1.0)
    for s in range(4):
        host.replay.update_priorities(host._idxs[s], host.pri[s])
    torch.cuda.synchronize()
    host.replay.check_invariants()


@pytest.mark.gpu
def test_actor_loop_fp8_heads():
    """--fp8-heads actor routing (act_gather keeps the unfused
    tau_embed -> quant -> MX GEMM chain): valid actions, finite state."""
    import torch as t

    from rainbowiqn_amd.config import Config
    from rainbowiqn_amd.models import RainbowIQN
    from rainbowiqn_amd.runtime.gpu_actor import GpuActorLoop

    cfg = Config(device="cuda", dtype="bf16", fp8_heads=True)
    net = RainbowIQN(cfg, 6).cuda().to(t.bfloat16).train()
    for layer in (net.fc_h_v, net.fc_h_a, net.fc_z_v, net.fc_z_a):
        layer.enable_fused_noise()
    loop = GpuActorLoop(cfg, net, n_envs=64, chunk_steps=8, use_graph=True)
    assert loop.fused is not None and loop.fused.fp8_heads
    for _ in range(2):
        loop.run_round()
    t.cuda.synchronize()
    meta = loop.chunk_meta.cpu().numpy()
    assert ((meta[:, 0] >= 0) & (meta[:, 0] < 6)).all()
    assert loop.chunk_frames.cpu().numpy().max() > 0


@pytest.mark.gpu
def test_round_tracer_cuda_events(tmp_path):
    """GPU tracer path: hipEvent spans resolve to a valid Chrome trace."""
    import json

    from rainbowiqn_amd.utils.trace import RoundTracer

    tr = RoundTracer(str(tmp_path / "t.json"), use_cuda=True)
    x = torch.rand(512, 512, device="cuda")
    for _ in range(3):
        with tr.span("matmul"):
            x = x @ x * 1e-3
    torch.cuda.synchronize()
    ev = json.load(open(tr.dump()))["traceEvents"]
    assert len(ev) == 3 and all(e["dur"] > 0 for e in ev)
    # spans are ordered and non-overlapping on one stream
    starts = [e["ts"] for e in ev]
    assert starts == sorted(starts)


@pytest.mark.gpu
def test_colocated_two_actor_streams():
    """--nb-actor 2 colocated overlap: two act chains on their own streams,
    chunks landing in their own replay rings, learner training from both
    (the latency-overlap config; ADVICE staleness contract holds per loop)."""
    from rainbowiqn_amd.config import Config
    from rainbowiqn_amd.parallel.apex import ApexRunner
    from rainbowiqn_amd.parallel.comm import Communicator

    cfg = Config(device="cuda", dtype="bf16", learn_start=512,
                 memory_capacity=16384, learn_per_round=2, nb_actor=2)
    runner = ApexRunner(cfg, Communicator(backend="auto"), n_envs=16,
                        chunk_steps=8)
    assert runner.overlap and runner.n_local_actors == 2
    assert len(runner.gactors) == 2
    assert runner.gactors[0].net is not runner.gactors[1].net
    assert runner.glearner.replay.n_rings == 2
    for _ in range(8):
        runner.run_round()
    torch.cuda.synchronize()
    # both rings received chunks (ring counts live in ring_state[:, 1])
    counts = runner.glearner.replay.ring_state[:, 1].cpu()
    assert (counts > 0).all(), counts
    assert runner.grad_steps > 0
    # 8 rounds x 2 loops x 16 envs x 8 steps, plus each loop's graph-capture
    # warmup round
    assert runner.actor.total_steps >= 2 * 16 * 8 * 8
    # both actors' weights equal the learner's after the boundary copy
    runner.run_round(learn=False)
    torch.cuda.synchronize()
    for flat in runner._actor_flats:
        assert torch.equal(flat, runner.glearner.flat_online)


@pytest.mark.gpu
def test_replay_at_reference_scale():
    """The bench's replay config exactly: 1M-transition ring (2^20-leaf
    tree), 80k prefill.  Sampling must return valid indices, finite
    weights, and tree invariants must hold -- the correctness side of the
    headline configuration."""
    from rainbowiqn_amd.config import Config
    from rainbowiqn_amd.replay.gpu_replay import GpuReplay

    cfg = Config(device="cuda", dtype="bf16", batch_size=32)
    rep = GpuReplay(cfg, n_rings=1, capacity=1_000_000, device="cuda",
                    out_dtype=torch.bfloat16)
    assert rep.tree_size == 1 << 20
    g = torch.Generator(device="cpu").manual_seed(0)
    for i in range(5):
        f = torch.randint(0, 256, (16384, 84, 84), dtype=torch.uint8,
                          generator=g).cuda()
        m = torch.rand(16384, 3, generator=g).cuda()
        rep.append_actor_chunk(0, f, meta=m)
    assert len(rep) == 5 * 16384
    for _ in range(20):
        idx, s, a, r, s2, nt, w = rep.sample()
        rep.update_priorities(idx, torch.rand(32, device="cuda") + 0.05)
    torch.cuda.synchronize()
    assert int(idx.max()) < len(rep)
    assert int(idx.min()) >= 0
    assert torch.isfinite(w).all() and float(w.max()) <= 1.0 + 1e-5
    assert torch.isfinite(r).all()
    rep.check_invariants()


@pytest.mark.gpu
def test_single_env_actor_reference_shape():
    """n_envs=1: the reference's one-env-per-actor shape
    must run through the same fused kernels (grid-of-1 edge cases)."""
    from rainbowiqn_amd.config import Config
    from rainbowiqn_amd.parallel.apex import ApexRunner
    from rainbowiqn_amd.parallel.comm import Communicator

    cfg = Config(device="cuda", dtype="bf16", learn_start=64,
                 memory_capacity=4096, learn_per_round=1)
    runner = ApexRunner(cfg, Communicator(backend="auto"), n_envs=1,
                        chunk_steps=8)
    assert runner.fast
    for _ in range(16):
        runner.run_round()
    torch.cuda.synchronize()
    assert runner.grad_steps > 0
    assert all(l == l for l in runner.losses)  # finite (not NaN)
    acts = runner.gactor.actions
    assert 0 <= int(acts[0]) < 6
