"""Ape-X actor/replay/learner split over RCCL/xGMI.

Reference architecture (SURVEY.md sections 4.1/4.2): N actor processes
generate experience with near-latest weights and push transition chunks to a
shared prioritized replay; one learner samples and trains; weights flow back.
The reference decouples processes through a Redis server over TCP; here the
whole split runs inside one MI355X node, one process per GPU, and the
exchange is a fixed-cadence RCCL round:

  round := actors run `chunk` env steps and fill a device-resident chunk
           (frames u8 + meta) -> all_gather to the learner (each actor's
           chunk rides its own xGMI link) -> learner drains chunks into its
           per-actor replay rings and runs `learn_per_round` grad steps ->
           learner broadcasts the flat online weights -> actors load them.

Bounded staleness (one round) replaces Redis polling; per-actor replay rings
keep episode streams contiguous so frame-stack/n-step reconstruction stays
correct (the reference gets this by per-actor redis keys).

World size 1 collapses both roles into one process on one GPU (BASELINE.json
configs 1-3); CPU + gloo runs the identical code path for CI.
"""

from __future__ import annotations

from typing import List, Optional

import numpy as np
import torch

from ..agent import Agent
from ..env import make_vector_env
from ..replay import ReplayMemory
from .comm import Communicator


class ShardedReplay:
    """Learner-side replay: one ring per actor, one logical PER distribution.

    Sampling draws each batch element from an actor ring with probability
    proportional to that ring's priority total (stratified within rings), so
    the joint distribution equals a single global proportional-PER tree.
    The on-GPU replay (ops/gpu_replay) implements the same contract with a
    single physical tree over per-actor ring segments.
    """

    def __init__(self, cfg, n_actors: int, capacity: Optional[int] = None):
        cap = int(capacity if capacity is not None else cfg.memory_capacity)
        per = max(64, cap // n_actors)
        self.rings: List[ReplayMemory] = [ReplayMemory(cfg, per) for _ in range(n_actors)]
        self.n_actors = n_actors

    def __len__(self) -> int:
        return sum(len(r) for r in self.rings)

    @property
    def priority_weight(self) -> float:
        return self.rings[0].priority_weight

    @priority_weight.setter
    def priority_weight(self, v: float) -> None:
        for r in self.rings:
            r.priority_weight = v

    def append_actor_chunk(self, actor: int, frames, actions, rewards, terminals) -> None:
        self.rings[actor].append_chunk(frames, actions, rewards, terminals)

    def sample(self, batch_size: int):
        totals = np.array([r.tree.total() if len(r) > r.history + r.n + 1 else 0.0
                           for r in self.rings])
        assert totals.sum() > 0, "sample() from empty sharded replay"
        counts = np.random.multinomial(batch_size, totals / totals.sum())
        parts = []
        for a, c in enumerate(counts):
            if c == 0:
                continue
            idxs, s, act, r, s2, nt, w = self.rings[a].sample(int(c))
            parts.append((np.full(c, a), idxs, s, act, r, s2, nt, w))
        ring_ids = np.concatenate([p[0] for p in parts])
        idxs = np.concatenate([p[1] for p in parts])
        cat = lambda i: torch.cat([p[i] for p in parts])
        return (ring_ids, idxs), cat(2), cat(3), cat(4), cat(5), cat(6), cat(7)

    def update_priorities(self, handle, priorities) -> None:
        ring_ids, idxs = handle
        p = priorities.detach().cpu().numpy() if isinstance(priorities, torch.Tensor) else np.asarray(priorities)
        for a in np.unique(ring_ids):
            m = ring_ids == a
            self.rings[int(a)].update_priorities(idxs[m], p[m])

    # -- persistence (reference: Redis RDB snapshot) ----------
    def state_dict(self) -> dict:
        return {"kind": "sharded_replay",
                "rings": [r.state_dict() for r in self.rings]}

    def load_state_dict(self, sd: dict) -> None:
        assert len(sd["rings"]) == len(self.rings), "actor-ring count mismatch"
        for r, rsd in zip(self.rings, sd["rings"]):
            r.load_state_dict(rsd)


class ActorState:
    """Device-side actor loop state: E env streams with frame-stack buffers."""

    def __init__(self, cfg, comm: Communicator, agent: Agent, n_envs: int, chunk_steps: int):
        self.cfg = cfg
        self.comm = comm
        self.agent = agent
        self.n_envs = n_envs
        self.chunk_steps = chunk_steps
        self.env = make_vector_env(cfg, n_envs, seed_offset=comm.rank * 1009)
        self.device = agent.device
        h = cfg.history_length
        self.stack = torch.zeros(n_envs, h, 84, 84, device=self.device)
        first = torch.from_numpy(self.env.reset_all().astype(np.float32) / 255.0)
        self.stack[:, -1] = first.to(self.device)
        c = n_envs * chunk_steps
        self.chunk_frames = torch.zeros(c, 84, 84, dtype=torch.uint8, device=self.device)
        self.chunk_meta = torch.zeros(c, 3, device=self.device)  # action, reward, done
        self._fill = 0
        self.total_steps = 0

    def run_chunk(self) -> None:
        """Run chunk_steps vectorized env steps, filling the chunk buffers."""
        self._fill = 0
        clip = self.cfg.reward_clip
        for _ in range(self.chunk_steps):
            actions = self.agent.act_batch(self.stack)  # [E] on device
            a_np = actions.cpu().numpy()
            frames, rewards, dones = self.env.step(a_np)
            if self.cfg.reward_clip_enabled:
                rewards = np.clip(rewards, -clip, clip)
            f_dev = torch.from_numpy(frames).to(self.device, non_blocking=True)
            lo = self._fill * self.n_envs
            hi = lo + self.n_envs
            # the stored frame is the LAST frame of the state acted on
            self.chunk_frames[lo:hi] = (self.stack[:, -1] * 255).to(torch.uint8)
            self.chunk_meta[lo:hi, 0] = actions.float()
            self.chunk_meta[lo:hi, 1] = torch.from_numpy(rewards).to(self.device)
            self.chunk_meta[lo:hi, 2] = torch.from_numpy(dones.astype(np.float32)).to(self.device)
            # roll the stack, reset done streams
            self.stack = torch.roll(self.stack, -1, dims=1)
            self.stack[:, -1] = f_dev.float() / 255.0
            if dones.any():
                d = torch.from_numpy(dones).to(self.device)
                self.stack[d] = 0.0
                self.stack[d, -1] = f_dev[d].float() / 255.0
            self._fill += 1
            self.total_steps += self.n_envs


class _ActorSteps:
    """total_steps across the rank's local actor loops."""

    def __init__(self, loops):
        self._loops = loops

    @property
    def total_steps(self) -> int:
        return sum(l.total_steps for l in self._loops)


class ApexRunner:
    """Fixed-cadence Ape-X round loop shared by the launch scripts and bench.

    Two execution paths, same round semantics:
      * fast path (GPU + HIP extension): hipGraph-captured GpuLearner /
        GpuActorLoop with on-GPU replay and device-resident synthetic env;
      * reference path (CPU or --no-hip-kernels): eager Agent + CPU sharded
        replay + numpy env -- also what the gloo CI tests exercise.
    """

    def __init__(self, cfg, comm: Communicator, n_envs: int = 16,
                 chunk_steps: Optional[int] = None, replay_capacity: Optional[int] = None,
                 device: Optional[str] = None, n_local_actors: Optional[int] = None):
        self.cfg = cfg
        self.comm = comm
        dev = device or str(comm.device)
        self.n_envs = n_envs
        self.chunk_steps = chunk_steps or max(1, cfg.transition_chunk_size // n_envs)
        # BASELINE config 3: several independent actors (own nets, own noise,
        # own env streams, own replay rings) colocated on each actor rank.
        # Defaults to the reference's --nb-actor flag in single-process mode.
        if n_local_actors is None:
            n_local_actors = max(1, cfg.nb_actor) if comm.world_size == 1 else 1
        self.n_local_actors = n_local_actors
        self.grad_steps = 0
        self.losses = []
        self.tracer = None  # RoundTracer (utils/trace.py) via set_tracer()
        self.actor = None
        self.replay = None
        self.glearner = None
        self.gactor = None
        self.gactors = []

        from ..env import n_actions_for

        n_act = n_actions_for(cfg.game)
        self.fast = False
        if torch.device(dev).type == "cuda" and cfg.use_hip_kernels:
            from ..ops import extension_available

            self.fast = extension_available()
            if not self.fast:
                from ..ops import get_extension

                get_extension()  # raises: native path is mandatory on GPU

        if not self.fast:
            self.n_local_actors = 1  # multi-local-actor is a fast-path feature
        self.sharded = bool(getattr(cfg, "sharded_replay", False)) and \
            comm.world_size > 1
        if self.sharded:
            assert self.n_local_actors == 1, "sharded mode uses one shard/rank"
        self.shard = None
        if self.fast:
            self._init_fast(cfg, comm, dev, n_act, replay_capacity)
        else:
            self.agent = Agent(cfg, n_act, dev)
            self.flat_weights = self.agent.flat_weights()
            if comm.is_actor:
                self.actor = ActorState(cfg, comm, self.agent, n_envs, self.chunk_steps)
                if self.sharded:
                    from .sharded import ActorShardHost, actor_quota

                    self.shard = ActorShardHost(
                        cfg, comm,
                        actor_quota(cfg.batch_size, comm.n_actors,
                                    comm.actor_index()),
                        cfg.learn_per_round, "cpu", fast=False)
            if comm.is_learner:
                if self.sharded:
                    from .sharded import RemoteReplay

                    self.replay = RemoteReplay(cfg, comm, cfg.learn_per_round,
                                               "cpu", dtype=torch.float32)
                else:
                    self.replay = ShardedReplay(cfg, comm.n_actors,
                                                replay_capacity)

    def _init_fast(self, cfg, comm, dev, n_act, replay_capacity):
        from ..models.model import RainbowIQN
        from ..ops.flat_params import flatten_module
        from ..runtime.gpu_actor import GpuActorLoop
        from ..runtime.gpu_learner import GpuLearner

        # chunk_steps must be a multiple of history for the actor ring phase
        h = cfg.history_length
        self.chunk_steps = max(h, (self.chunk_steps // h) * h)
        M = self.n_local_actors
        if comm.is_learner:
            if replay_capacity is not None:
                cfg_cap = cfg
                cfg_cap.memory_capacity = replay_capacity
            ext_replay = None
            if self.sharded:
                from .sharded import RemoteReplay

                dtt = torch.bfloat16 if cfg.dtype == "bf16" else torch.float32
                ext_replay = RemoteReplay(cfg, comm, cfg.learn_per_round, dev,
                                          dtype=dtt)
            # colocated (world 1): the learner shares the chip with the
            # actor streams under a CU partition -- shallow fork (level 1);
            # a dedicated learner GPU (world > 1) takes the full fork
            self.glearner = GpuLearner(cfg, n_act, n_rings=comm.n_actors * M,
                                       device=dev, use_graph=cfg.hipgraph,
                                       replay=ext_replay,
                                       fork_level=1 if comm.world_size == 1
                                       else 2)
            self.replay = self.glearner.replay
            self.flat_weights = self.glearner.flat_online
        if comm.is_actor:
            dt = torch.bfloat16 if cfg.dtype == "bf16" else torch.float32
            self._actor_flats = []
            # Colocated overlap (world 1, M 1, overlap_colocated on): actor
            # and learner rounds run CONCURRENTLY on separate streams, so the
            # actor must NOT read the learner's live net -- Adam/pack/noise
            # writes on the learner stream would tear mid-forward.  The actor
            # gets its own net (own noise, own packs); _run_round_overlap
            # copies flat weights at the round boundary on the outer stream,
            # making the actor's view a consistent one-round-stale snapshot
            # (the same bounded-staleness contract as the multi-GPU split).
            # Only the strictly sequential non-overlap colocated mode shares
            # the online net directly.
            seq_colocated = (comm.world_size == 1 and M == 1
                             and not getattr(cfg, "overlap_colocated", True))
            for j in range(M):
                if seq_colocated:
                    net = self.glearner.online  # sequential: safe to share
                    flat = None
                else:
                    # independent actor: own net copy (own NoisyNet noise),
                    # refreshed from the learner's flat weights each round
                    net = RainbowIQN(cfg, n_act).to(dev).to(dt)
                    flat = flatten_module(net)
                    if comm.world_size == 1:
                        flat.copy_(self.glearner.flat_online)
                    if comm.world_size > 1 and j == 0:
                        self.flat_weights = flat
                self._actor_flats.append(flat)
                self.gactors.append(
                    GpuActorLoop(cfg, net, self.n_envs, self.chunk_steps,
                                 device=dev, use_graph=cfg.hipgraph,
                                 seed_offset=comm.rank * 7717 + j * 131071,
                                 external_weights=seq_colocated))
            self.gactor = self.gactors[0]
            self.actor = _ActorSteps(self.gactors)
            if self.sharded:
                from .sharded import ActorShardHost, actor_quota

                self.shard = ActorShardHost(
                    cfg, comm,
                    actor_quota(cfg.batch_size, comm.n_actors,
                                comm.actor_index()),
                    cfg.learn_per_round, dev, fast=True)
        # colocated overlap: each actor round runs on its own HIP stream,
        # concurrent with the learner's grad steps AND with the other local
        # actor loops (one-round bounded staleness -- the same contract as
        # the multi-GPU split).  With --nb-actor M > 1 the M half-size act
        # chains interleave on M queues: the act-path kernels are ~70%
        # latency waits (r02 PMC), so co-resident independent chains fill
        # the stall slots.  Outgoing chunks are double-buffered so the
        # learner appends round k-1's chunks while round k records.
        self.overlap = (comm.world_size == 1 and bool(self.gactors)
                        and self.glearner is not None
                        and getattr(cfg, "overlap_colocated", True))
        if self.overlap:
            import os as _os

            d = self.gactors[0].device
            c = self.n_envs * self.chunk_steps
            self._astreams = [torch.cuda.Stream(device=d) for _ in range(M)]
            # RIQN_LPRIO=1: learner work on a high-priority HSA queue so its
            # small grids (batch 32) dispatch ahead of the actor's queued
            # chip-filling WGs instead of waiting behind them (opt-in until
            # the bench decides a default).
            self._lstream = (torch.cuda.Stream(device=d, priority=-1)
                             if _os.environ.get("RIQN_LPRIO", "0") == "1"
                             else None)
            # Hard CU partition (default on): learner queue gets CUs
            # [0,cu_split) = 3 XCDs, actor queue the rest (per-queue CU
            # mask) — free-running contention measured 5.23M frames/s+1278
            # grad/s; the 96/160 split 5.58M+1362 (both metrics up 6%, and
            # each role keeps its own XCDs' L2).  RIQN_CU_SPLIT overrides;
            # 0 disables.
            cus = int(_os.environ.get("RIQN_CU_SPLIT",
                                      str(getattr(cfg, "cu_split", 96))))
            if cus > 0:
                try:
                    from ..ops import get_extension

                    ext = get_extension()
                    total = torch.cuda.get_device_properties(
                        d).multi_processor_count
                    cus = min(cus, max(total - 32, 0))

                    def _mask(lo, hi):
                        w = [0] * ((total + 31) // 32)
                        for cu in range(lo, hi):
                            w[cu // 32] |= 1 << (cu % 32)
                        return w

                    # RIQN_LFREE=1: leave the learner on free high-
                    # priority queues (no mask) while the actor keeps its
                    # CU share -- the learn step's small grids fit in far
                    # fewer CUs than the partition grants, so the mask may
                    # cost more latency than the isolation buys (A/B knob).
                    lfree = _os.environ.get("RIQN_LFREE", "0") == "1"
                    if lfree:
                        self._lstream = torch.cuda.Stream(device=d,
                                                          priority=-1)
                        self.glearner._bstream = torch.cuda.Stream(
                            device=d, priority=-1)
                    else:
                        lp = ext.create_cu_masked_stream(_mask(0, cus))
                        self._lstream = torch.cuda.ExternalStream(lp,
                                                                  device=d)
                    # destroy at exit: a leaked HSA queue crashes profiler
                    # teardown (__cxa_finalize segfault under rocprofv3)
                    import atexit

                    if not lfree:
                        atexit.register(ext.destroy_stream, lp)
                    # all actor queues share the actor CU partition
                    self._astreams = []
                    for _ in range(M):
                        ap = ext.create_cu_masked_stream(_mask(cus, total))
                        atexit.register(ext.destroy_stream, ap)
                        self._astreams.append(
                            torch.cuda.ExternalStream(ap, device=d))
                    # unmasked streams for the pre-learn_start warmup phase
                    self._astreams_full = [torch.cuda.Stream(device=d)
                                           for _ in range(M)]
                except Exception:  # no CU-mask support: plain streams
                    pass
            self._staging_f = [[torch.zeros(c, 84, 84, dtype=torch.uint8,
                                            device=d) for _ in range(2)]
                               for _ in range(M)]
            self._staging_m = [[torch.zeros(c, 3, device=d) for _ in range(2)]
                               for _ in range(M)]
            self._round_idx = 0

    # -- one Ape-X round ----------------------------------------------------
    def set_tracer(self, tracer) -> None:
        """Attach a utils.trace.RoundTracer; round phases get named spans."""
        self.tracer = tracer

    def _sp(self, name: str):
        from contextlib import nullcontext

        return self.tracer.span(name) if self.tracer is not None \
            else nullcontext()

    def run_round(self, learn: bool = True) -> None:
        if self.sharded:
            self._run_round_sharded(learn)
        elif self.fast:
            self._run_round_fast(learn)
        else:
            self._run_round_ref(learn)

    def _run_round_sharded(self, learn: bool) -> None:
        """Config-5 round: replay shards live on the actor ranks; the
        learner receives assembled batches and returns priorities
        (parallel/sharded.py).  Collective order is identical on every
        rank: counts all_gather -> batch P2P -> priorities P2P -> weight
        broadcast."""
        import torch.distributed as dist

        cfg, comm = self.cfg, self.comm
        self._sh_rounds = getattr(self, "_sh_rounds", 0) + 1
        # 1. actors: generate experience, append to the LOCAL shard
        if comm.is_actor:
            if self.fast:
                loop = self.gactors[0]
                with self._sp("actor"):
                    if self._actor_flats[0] is not None and \
                            self._actor_flats[0] is not self.flat_weights:
                        self._actor_flats[0].copy_(self.flat_weights)
                    loop.run_round()
                    self.shard.append_chunk(loop.chunk_frames,
                                            meta=loop.chunk_meta)
            else:
                self.agent.reset_noise()
                self.actor.run_chunk()
                mc = self.actor.chunk_meta.cpu().numpy()
                self.shard.append_chunk(
                    self.actor.chunk_frames.cpu().numpy().astype("uint8"),
                    actions=mc[:, 0].astype("int64"), rewards=mc[:, 1],
                    terminals=mc[:, 2] > 0.5)
            # local beta anneal (actors own the IS weights)
            T = self._sh_rounds * self.chunk_steps * self.n_envs * comm.n_actors
            span = max(1, cfg.T_max - cfg.learn_start)
            self.shard.anneal_beta(
                cfg.priority_weight + (1.0 - cfg.priority_weight)
                * max(0, T - cfg.learn_start) / span)
        # 2. shard counts
        cnt = (self.shard.counts() if comm.is_actor
               else torch.zeros(2)).to(comm.device)
        gathered = [torch.empty_like(cnt) for _ in range(comm.world_size)]
        dist.all_gather(gathered, cnt)
        total = int(sum(float(g[0]) for g in gathered))
        learning = learn and total >= cfg.learn_start
        # 3. batches -> learner; grad steps; priorities -> actors
        if learning:
            if comm.is_actor:
                with self._sp("batch-service"):
                    self.shard.sample_and_send()
                    self.shard.recv_priorities()
            else:
                self.replay.set_len(total)
                with self._sp("gather"):
                    self.replay.recv_batches()
                if self.fast:
                    with self._sp("learn"):
                        self.glearner.learn_steps(cfg.learn_per_round)
                    self.grad_steps = self.glearner.grad_steps
                    self.losses.append(self.glearner.last_loss())
                    tgt = max(1, cfg.target_update
                              // max(1, cfg.replay_frequency))
                    if self.grad_steps % tgt < cfg.learn_per_round:
                        self.glearner.update_target()
                else:
                    for _ in range(cfg.learn_per_round):
                        self.losses.append(self.agent.learn(self.replay))
                        self.grad_steps += 1
                        if self.grad_steps % max(
                                1, cfg.target_update
                                // max(1, cfg.replay_frequency)) == 0:
                            self.agent.update_target_net()
                self.replay.send_priorities()
        elif comm.is_learner:
            self.replay.set_len(total)
        # 4. weight flow
        if not self.fast and comm.is_learner:
            self.flat_weights = self.agent.flat_weights()
        comm.broadcast_weights(self.flat_weights, src=0)
        if not self.fast and comm.is_actor:
            self.agent.load_flat_weights(self.flat_weights)

    def _run_round_fast(self, learn: bool) -> None:
        cfg = self.cfg
        comm = self.comm
        if getattr(self, "overlap", False):
            return self._run_round_overlap(learn)
        M = self.n_local_actors
        local_f, local_m = [], []
        if self.gactors:
            with self._sp("actor"):
                for j, loop in enumerate(self.gactors):
                    if self._actor_flats[j] is not None:
                        # refresh this actor's net from the latest weights
                        src = (self.glearner.flat_online
                               if comm.world_size == 1
                               else self.flat_weights)
                        if self._actor_flats[j] is not src:
                            self._actor_flats[j].copy_(src)
                    loop.run_round()
                    local_f.append(loop.chunk_frames)
                    local_m.append(loop.chunk_meta)
        else:
            # learner-only rank: shape templates for the P2P receives
            if not hasattr(self, "_dummy_chunks"):
                c = self.n_envs * self.chunk_steps
                self._dummy_chunks = (
                    [torch.zeros(c, 84, 84, dtype=torch.uint8,
                                 device=comm.device) for _ in range(M)],
                    [torch.zeros(c, 3, device=comm.device) for _ in range(M)])
            local_f, local_m = self._dummy_chunks

        frames_list, meta_list = [], []
        if comm.world_size > 1:
            with self._sp("gather"):
                for j in range(M):
                    fl = comm.gather_chunks(local_f[j])
                    ml = comm.gather_chunks(local_m[j])
                    if comm.is_learner:
                        frames_list.extend(fl)
                        meta_list.extend(ml)
        else:
            frames_list, meta_list = local_f, local_m
        # weight flow BEFORE the learn phase: every rank broadcasts the
        # learner's current (post-last-round) weights, then actor ranks run
        # their NEXT round concurrently with the learner's grad steps --
        # the same one-round bounded staleness as the colocated overlap.
        # Collective order (gather -> broadcast) is identical on all ranks.
        if comm.world_size > 1:
            with self._sp("broadcast"):
                comm.broadcast_weights(self.flat_weights, src=0)
        if comm.is_learner:
            # ring id = actor_rank * M + local actor (interleaved per j-gather:
            # frames_list order is [j0: ranks 1..W-1][j1: ...] when W > 1)
            with self._sp("append"):
                for i, (f, m) in enumerate(zip(frames_list, meta_list)):
                    if comm.world_size > 1:
                        j, a = divmod(i, comm.n_actors)
                        ring = a * M + j
                    else:
                        ring = i
                    self.glearner.append_actor_chunk(ring, f, meta=m)
            if learn and len(self.glearner.replay) >= cfg.learn_start:
                with self._sp("learn"):
                    self.glearner.learn_steps(cfg.learn_per_round)
                self.grad_steps = self.glearner.grad_steps
                self.losses.append(self.glearner.last_loss())
                tgt = max(1, cfg.target_update // max(1, cfg.replay_frequency))
                if self.grad_steps % tgt < cfg.learn_per_round:
                    self.glearner.update_target()
                    self.glearner.maintenance()

    def _run_round_overlap(self, learn: bool) -> None:
        """Colocated round: M actor streams + learner stream, concurrent."""
        cfg = self.cfg
        gl = self.glearner
        M = self.n_local_actors
        j = self._round_idx % 2
        prev = (self._round_idx - 1) % 2
        outer = torch.cuda.current_stream()
        # round-boundary weight snapshot: the previous round's learner work
        # is complete w.r.t. the outer stream here, so the actors' flat
        # copies see consistent (one-round-stale) weights -- never torn
        for flat in self._actor_flats:
            if flat is not None:
                flat.copy_(gl.flat_online)
        # learner work optionally rides a high-priority queue (RIQN_LPRIO):
        # its small grids then dispatch ahead of the actor's queued WGs
        from contextlib import nullcontext

        lctx = (lambda: torch.cuda.stream(self._lstream)) \
            if self._lstream is not None else nullcontext
        if self._lstream is not None:
            self._lstream.wait_stream(outer)
        # learner stream: append LAST round's chunks, then grad steps; these
        # small-grid kernels fill the CUs the actor's big grids leave idle
        with lctx():
            if self._round_idx > 0:
                with self._sp("append"):
                    for a in range(M):
                        gl.append_actor_chunk(a, self._staging_f[a][prev],
                                              meta=self._staging_m[a][prev])
        # actor streams: each loop's round + chunk stage-out, concurrent
        # with the learner and each other.  Phase-aware partition: before
        # learn_start the learner queue is idle, so actors run on UNMASKED
        # streams (all 256 CUs) instead of their partition share.
        astreams = self._astreams
        if (getattr(self, "_astreams_full", None) is not None
                and len(gl.replay) < cfg.learn_start):
            astreams = self._astreams_full
        for a, loop in enumerate(self.gactors):
            astreams[a].wait_stream(outer)
            with torch.cuda.stream(astreams[a]):
                with self._sp(f"actor{a}" if M > 1 else "actor"):
                    loop.run_round()
                    self._staging_f[a][j].copy_(loop.chunk_frames,
                                                non_blocking=True)
                    self._staging_m[a][j].copy_(loop.chunk_meta,
                                                non_blocking=True)
        with lctx():
            if learn and len(gl.replay) >= cfg.learn_start:
                with self._sp("learn"):
                    gl.learn_steps(cfg.learn_per_round)
                self.grad_steps = gl.grad_steps
                self.losses.append(gl.last_loss())
                tgt = max(1, cfg.target_update
                          // max(1, cfg.replay_frequency))
                if self.grad_steps % tgt < cfg.learn_per_round:
                    gl.update_target()
                    gl.maintenance()
        if self._lstream is not None:
            outer.wait_stream(self._lstream)
        for a in range(M):
            outer.wait_stream(astreams[a])
        self._round_idx += 1

    def _run_round_ref(self, learn: bool) -> None:
        cfg = self.cfg
        # 1. actors generate a chunk of experience
        if self.actor is not None:
            with self._sp("actor"):
                self.agent.reset_noise()
                self.actor.run_chunk()
            chunk_f, chunk_m = self.actor.chunk_frames, self.actor.chunk_meta
        else:
            c = self.n_envs * self.chunk_steps
            chunk_f = torch.zeros(c, 84, 84, dtype=torch.uint8, device=self.comm.device)
            chunk_m = torch.zeros(c, 3, device=self.comm.device)

        # 2. transition flow: actors -> learner replay (reference: redis push)
        with self._sp("gather"):
            frames_list = self.comm.gather_chunks(chunk_f)
            meta_list = self.comm.gather_chunks(chunk_m)
        if self.comm.is_learner:
            if self.comm.world_size == 1:
                frames_list, meta_list = [chunk_f], [chunk_m]
            with self._sp("append"):
                for a, (f, m) in enumerate(zip(frames_list, meta_list)):
                    mc = m.cpu().numpy()
                    self.replay.append_actor_chunk(
                        a, f.cpu().numpy(), mc[:, 0].astype(np.int64),
                        mc[:, 1], mc[:, 2] > 0.5,
                    )

        # 3. learner trains
        if self.comm.is_learner and learn and len(self.replay) >= cfg.learn_start:
            with self._sp("learn"):
                for _ in range(cfg.learn_per_round):
                    loss = self.agent.learn(self.replay)
                    self.losses.append(loss)
                    self.grad_steps += 1
                    if self.grad_steps % max(1, cfg.target_update // max(1, cfg.replay_frequency)) == 0:
                        self.agent.update_target_net()

        # 4. weight flow: learner -> actors (reference: redis weight blob)
        if self.comm.world_size > 1:
            with self._sp("broadcast"):
                if self.comm.is_learner:
                    self.flat_weights = self.agent.flat_weights()
                self.comm.broadcast_weights(self.flat_weights, src=0)
                if not self.comm.is_learner:
                    self.agent.load_flat_weights(self.flat_weights)
