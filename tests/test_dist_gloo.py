"""Multi-process CPU tests of the Ape-X distributed path (gloo backend).

The same Communicator/ApexRunner code runs over RCCL on the GPU node; gloo at
world_size 3 (1 learner + 2 actors) exercises the collectives and role logic
here without a GPU (SURVEY.md section 5 test pyramid, distributed tier).
"""

import os

import numpy as np
import pytest
import torch
import torch.multiprocessing as mp


def _find_free_port():
    import socket

    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def _worker_apex(rank, world, port, results):
    os.environ.update(
        RANK=str(rank), WORLD_SIZE=str(world), LOCAL_RANK=str(rank),
        MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
    )
    torch.manual_seed(rank)
    from rainbowiqn_amd.config import Config
    from rainbowiqn_amd.parallel.apex import ApexRunner
    from rainbowiqn_amd.parallel.comm import Communicator

    cfg = Config(
        memory_capacity=4096, learn_start=64, batch_size=8,
        num_tau_samples=3, num_tau_prime_samples=3, num_quantile_samples=4,
        hidden_size=32, learn_per_round=2, device="cpu",
    )
    comm = Communicator(backend="gloo", device=torch.device("cpu"))
    runner = ApexRunner(cfg, comm, n_envs=4, chunk_steps=8)
    for _ in range(6):
        runner.run_round()
    out = {"rank": rank, "is_learner": comm.is_learner}
    if comm.is_learner:
        out["replay_len"] = len(runner.replay)
        out["grad_steps"] = runner.grad_steps
        out["losses_finite"] = all(np.isfinite(l) for l in runner.losses)
        out["ring_lens"] = [len(r) for r in runner.replay.rings]
    else:
        # actor weights must equal the broadcast learner weights
        out["flat_sum"] = float(runner.agent.flat_weights().sum())
    out["bcast_sum"] = float(runner.flat_weights.sum())
    results[rank] = out
    comm.close()


@pytest.mark.parametrize("world", [3, 8])
def test_apex_round_gloo(world):
    """world 8 = the target 7+1 topology (BASELINE config 4) on gloo."""
    port = _find_free_port()
    mgr = mp.Manager()
    results = mgr.dict()
    mp.spawn(_worker_apex, args=(world, port, results), nprocs=world, join=True)
    res = dict(results)
    assert len(res) == world
    learner = next(r for r in res.values() if r["is_learner"])
    actors = [r for r in res.values() if not r["is_learner"]]
    # (world-1) actors x 6 rounds x 4 envs x 8 steps
    assert learner["replay_len"] == (world - 1) * 6 * 4 * 8
    assert all(l == 192 for l in learner["ring_lens"])
    assert len(learner["ring_lens"]) == world - 1
    assert learner["grad_steps"] > 0
    assert learner["losses_finite"]
    # weight flow: every rank ends with identical broadcast weights
    sums = {round(r["bcast_sum"], 4) for r in res.values()}
    assert len(sums) == 1
    for a in actors:
        assert a["flat_sum"] == pytest.approx(a["bcast_sum"], rel=1e-5)


def _worker_comm(rank, world, port, results):
    os.environ.update(
        RANK=str(rank), WORLD_SIZE=str(world),
        MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
    )
    from rainbowiqn_amd.parallel.comm import Communicator

    comm = Communicator(backend="gloo", device=torch.device("cpu"))
    t = torch.full((4,), float(rank))
    chunks = comm.gather_chunks(t)
    out = {}
    if comm.is_learner:
        out["gathered"] = [c.tolist() for c in chunks]
    w = torch.full((3,), 42.0) if rank == 0 else torch.zeros(3)
    comm.broadcast_weights(w)
    out["w"] = w.tolist()
    c = torch.ones(1)
    comm.all_reduce_sum(c)
    out["count"] = c.item()
    results[rank] = out
    comm.close()


def test_comm_primitives_gloo():
    world = 3
    port = _find_free_port()
    mgr = mp.Manager()
    results = mgr.dict()
    mp.spawn(_worker_comm, args=(world, port, results), nprocs=world, join=True)
    res = dict(results)
    assert res[0]["gathered"] == [[1.0] * 4, [2.0] * 4]  # actor ranks only
    for r in res.values():
        assert r["w"] == [42.0] * 3
        assert r["count"] == world


def test_single_process_runner(cfg):
    """World size 1: both roles in one process, no process group."""
    from rainbowiqn_amd.parallel.apex import ApexRunner
    from rainbowiqn_amd.parallel.comm import Communicator

    cfg.learn_start = 32
    cfg.learn_per_round = 2
    comm = Communicator(backend="gloo", device=torch.device("cpu"))
    assert not comm.initialized and comm.is_learner and comm.is_actor
    runner = ApexRunner(cfg, comm, n_envs=4, chunk_steps=4)
    for _ in range(4):
        runner.run_round()
    assert len(runner.replay) == 64
    assert runner.grad_steps > 0


def _worker_sharded(rank, world, port, results):
    os.environ.update(
        RANK=str(rank), WORLD_SIZE=str(world), LOCAL_RANK=str(rank),
        MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
    )
    torch.manual_seed(rank)
    from rainbowiqn_amd.config import Config
    from rainbowiqn_amd.parallel.apex import ApexRunner
    from rainbowiqn_amd.parallel.comm import Communicator

    cfg = Config(
        memory_capacity=8192, learn_start=64, batch_size=8,
        num_tau_samples=3, num_tau_prime_samples=3, num_quantile_samples=4,
        hidden_size=32, learn_per_round=2, device="cpu",
        sharded_replay=True,
    )
    comm = Communicator(backend="gloo", device=torch.device("cpu"))
    runner = ApexRunner(cfg, comm, n_envs=4, chunk_steps=8)
    assert runner.sharded
    for _ in range(8):
        runner.run_round()
    out = {"rank": rank, "is_learner": comm.is_learner}
    if comm.is_learner:
        out["grad_steps"] = runner.grad_steps
        out["losses_finite"] = all(np.isfinite(l) for l in runner.losses)
        out["remote_len"] = len(runner.replay)
        # priorities flowed back: batch weights are positive and finite
        out["w_ok"] = bool(torch.isfinite(runner.replay.bww).all()
                           and (runner.replay.bww >= 0).all())
    else:
        out["shard_len"] = len(runner.shard.replay)
        # learner's priority updates landed in the local tree
        out["tree_total"] = float(runner.shard.replay.tree.total())
    out["bcast_sum"] = float(runner.flat_weights.sum())
    results[rank] = out
    comm.close()


@pytest.mark.parametrize("world", [3, 8])
def test_apex_sharded_replay_gloo(world):
    """Config-5 flow on CPU/gloo: actor-resident replay shards, P2P batch
    service to the learner, priorities returned to the owning shards."""
    port = _find_free_port()
    mgr = mp.Manager()
    results = mgr.dict()
    mp.spawn(_worker_sharded, args=(world, port, results), nprocs=world,
             join=True)
    rs = dict(results)
    learner = next(r for r in rs.values() if r["is_learner"])
    actors = [r for r in rs.values() if not r["is_learner"]]
    assert learner["grad_steps"] > 0
    assert learner["losses_finite"] and learner["w_ok"]
    assert learner["remote_len"] == sum(a["shard_len"] for a in actors)
    for a in actors:
        assert a["shard_len"] > 0
        assert np.isfinite(a["tree_total"]) and a["tree_total"] > 0
        # weights broadcast reached everyone identically
        assert a["bcast_sum"] == pytest.approx(learner["bcast_sum"])


def _worker_entry(rank, world, port, results, tmpdir):
    os.environ.update(
        RANK=str(rank), WORLD_SIZE=str(world), LOCAL_RANK=str(rank),
        MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
    )
    from rainbowiqn_amd.config import Config
    from rainbowiqn_amd.parallel.comm import Communicator
    from rainbowiqn_amd.train_apex import train_apex

    cfg = Config(
        memory_capacity=4096, learn_start=64, batch_size=8,
        num_tau_samples=3, num_tau_prime_samples=3, num_quantile_samples=4,
        hidden_size=32, learn_per_round=2, device="cpu",
        evaluation_interval=10**9, checkpoint_interval=0, T_max=10**9,
    )
    comm = Communicator(backend="gloo", device=torch.device("cpu"))
    runner = train_apex(cfg, comm, n_envs=4, results_dir=tmpdir,
                        max_rounds=5,
                        trace_file=os.path.join(tmpdir, "trace.json"))
    results[rank] = {
        "is_learner": comm.is_learner,
        "grad_steps": runner.grad_steps,
        "model_saved": os.path.exists(os.path.join(tmpdir, "model.pth")),
        "trace": os.path.exists(os.path.join(
            tmpdir, f"trace.json.rank{rank}")),
    }
    comm.close()


def test_train_apex_entry_gloo(tmp_path):
    """Full distributed entry path (train_apex: runner + failsafe + tracer +
    final checkpoint) at world 2 over gloo."""
    world = 2
    port = _find_free_port()
    mgr = mp.Manager()
    results = mgr.dict()
    mp.spawn(_worker_entry, args=(world, port, results, str(tmp_path)),
             nprocs=world, join=True)
    res = dict(results)
    assert len(res) == world
    learner = next(r for r in res.values() if r["is_learner"])
    assert learner["grad_steps"] > 0
    assert learner["model_saved"]
    assert all(r["trace"] for r in res.values())


def _worker_tags(rank, world, port, results):
    os.environ.update(
        RANK=str(rank), WORLD_SIZE=str(world),
        MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
    )
    from rainbowiqn_amd.parallel.comm import Communicator

    comm = Communicator(backend="gloo", device=torch.device("cpu"))
    out = {}
    # two back-to-back batched-P2P gathers of DIFFERENT shapes per round,
    # several rounds: exactly the frames+meta pattern of the Ape-X round.
    # Payloads are (round, rank)-stamped so any tag/order mix-up between the
    # two gathers or between rounds is detected by content, deterministically.
    for rnd in range(4):
        f = torch.full((16, 7), float(rank * 1000 + rnd))
        m = torch.full((5,), float(-(rank * 1000 + rnd)))
        fl = comm.gather_chunks(f)
        ml = comm.gather_chunks(m)
        if comm.is_learner:
            for a in range(comm.n_actors):
                src_rank = a + 1
                assert fl[a].shape == (16, 7)
                assert (fl[a] == src_rank * 1000 + rnd).all(), (rnd, a)
                assert (ml[a] == -(src_rank * 1000 + rnd)).all(), (rnd, a)
    out["ok"] = True
    results[rank] = out
    comm.close()


def test_gather_tag_ordering_world8():
    """P2P tag ordering of the chunk gather at the target world size: frames
    and meta land in the right per-actor slots across consecutive rounds
    (the batched_isend_irecv ordering risk flagged in round 1)."""
    world = 8
    port = _find_free_port()
    mgr = mp.Manager()
    results = mgr.dict()
    mp.spawn(_worker_tags, args=(world, port, results), nprocs=world,
             join=True)
    assert all(r["ok"] for r in dict(results).values())


def _worker_ring_interleave(rank, world, port, results):
    os.environ.update(
        RANK=str(rank), WORLD_SIZE=str(world), LOCAL_RANK=str(rank),
        MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
    )
    torch.manual_seed(rank)
    from rainbowiqn_amd.config import Config
    from rainbowiqn_amd.parallel.apex import ApexRunner
    from rainbowiqn_amd.parallel.comm import Communicator

    cfg = Config(
        memory_capacity=8192, learn_start=10**9, batch_size=8,
        num_tau_samples=3, num_tau_prime_samples=3, num_quantile_samples=4,
        hidden_size=32, learn_per_round=2, device="cpu",
    )
    comm = Communicator(backend="gloo", device=torch.device("cpu"))
    runner = ApexRunner(cfg, comm, n_envs=2, chunk_steps=8)
    # stamp each actor's generated rewards with its rank so the learner can
    # verify which ring each actor's chunk landed in
    if comm.is_actor:
        orig = runner.actor.run_chunk

        def stamped():
            orig()
            runner.actor.chunk_meta[:, 1] = float(comm.rank)

        runner.actor.run_chunk = stamped
    for _ in range(3):
        runner.run_round(learn=False)
    out = {"rank": rank}
    if comm.is_learner:
        ring_rewards = []
        for ring in runner.replay.rings:
            vals = set(float(v) for v in ring.rewards[: len(ring)])
            ring_rewards.append(vals)
        out["ring_rewards"] = ring_rewards
    results[rank] = out
    comm.close()


def test_ring_id_assignment_world8():
    """Each actor rank's chunks must land in ITS OWN learner replay ring
    (ring id = actor index), never mixed -- the frame-stack/n-step
    reconstruction correctness depends on per-actor stream contiguity."""
    world = 8
    port = _find_free_port()
    mgr = mp.Manager()
    results = mgr.dict()
    mp.spawn(_worker_ring_interleave, args=(world, port, results),
             nprocs=world, join=True)
    res = dict(results)
    learner = res[0]
    for a, vals in enumerate(learner["ring_rewards"]):
        assert vals == {float(a + 1)}, (a, vals)


def _worker_fastflow(rank, world, port, results):
    """Drive the REAL _run_round_fast control flow (the RCCL multi-rank
    path: gather -> broadcast -> learn) over gloo with mocked GPU pieces --
    validates collective order, ring-id assignment and the broadcast-
    before-learn staleness reorder without GPU hardware."""
    os.environ.update(
        RANK=str(rank), WORLD_SIZE=str(world), LOCAL_RANK=str(rank),
        MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
    )
    from rainbowiqn_amd.config import Config
    from rainbowiqn_amd.parallel.apex import ApexRunner
    from rainbowiqn_amd.parallel.comm import Communicator

    cfg = Config(device="cpu", learn_start=64, memory_capacity=4096,
                 learn_per_round=2, batch_size=8, hidden_size=32,
                 num_tau_samples=2, num_tau_prime_samples=2,
                 num_quantile_samples=3)
    comm = Communicator(backend="gloo", device=torch.device("cpu"))
    runner = ApexRunner.__new__(ApexRunner)
    runner.cfg, runner.comm = cfg, comm
    runner.n_envs, runner.chunk_steps = 4, 4
    runner.n_local_actors = 1
    runner.grad_steps, runner.losses = 0, []
    runner.tracer = None
    runner.sharded = False
    runner.overlap = False
    runner.fast = True
    runner.gactors = []
    runner.glearner = None
    runner.flat_weights = torch.zeros(8)
    c = runner.n_envs * runner.chunk_steps

    class _FakeLoop:
        def __init__(self):
            self.chunk_frames = torch.full((c, 84, 84), float(rank),
                                           dtype=torch.uint8)
            self.chunk_meta = torch.full((c, 3), float(rank))
            self.total_steps = 0

        def run_round(self):
            self.total_steps += c

    class _FakeReplay(dict):
        def __len__(self):
            return 10**6  # past learn_start

    class _FakeLearner:
        def __init__(self):
            self.replay = _FakeReplay()
            self.appended = []
            self.grad_steps = 0
            self.flat_online = torch.zeros(8)

        def append_actor_chunk(self, ring, f, meta=None):
            self.appended.append((ring, float(f.float().mean()),
                                  float(meta.mean())))

        def learn_steps(self, k):
            self.grad_steps += k
            # weights change AFTER the broadcast already happened
            self.flat_online += 1.0

        def last_loss(self):
            return 0.5

        def update_target(self):
            pass

        def maintenance(self):
            pass

    if comm.is_learner:
        runner.glearner = _FakeLearner()
        runner.flat_weights = runner.glearner.flat_online
        runner.actor = None
    else:
        loop = _FakeLoop()
        runner.gactors = [loop]
        runner._actor_flats = [runner.flat_weights]
        from rainbowiqn_amd.parallel.apex import _ActorSteps

        runner.actor = _ActorSteps([loop])
        runner.gactor = loop
    out = {"rank": rank, "bcast": []}
    for rnd in range(3):
        runner._run_round_fast(learn=True)
        out["bcast"].append(float(runner.flat_weights[0]))
    if comm.is_learner:
        out["appended"] = runner.glearner.appended
        out["grad_steps"] = runner.glearner.grad_steps
    results[rank] = out
    comm.close()


def test_fast_multirank_flow_gloo():
    """The fast-path multi-rank round (the path the 8-GPU RCCL run takes):
    chunks land in per-actor rings, and actors receive the PRE-learn
    weights each round (bounded one-round staleness: broadcast precedes
    the learn phase)."""
    world = 3
    port = _find_free_port()
    mgr = mp.Manager()
    results = mgr.dict()
    mp.spawn(_worker_fastflow, args=(world, port, results), nprocs=world,
             join=True)
    res = dict(results)
    learner = res[0]
    # ring id == actor index; payload identifies the sender rank (= a+1)
    rings = sorted({(ring, int(fmean)) for ring, fmean, _ in
                    learner["appended"]})
    assert rings == [(0, 1), (1, 2)], rings
    assert learner["grad_steps"] == 6
    # actors saw weights 0, 1, 2 at rounds 0, 1, 2: each round's broadcast
    # carries the weights BEFORE that round's learn (one-round stale)
    for r in (1, 2):
        assert res[r]["bcast"] == [0.0, 1.0, 2.0], res[r]["bcast"]
