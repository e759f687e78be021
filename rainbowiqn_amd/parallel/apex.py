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
from ..env import VectorSyntheticEnv
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


class ActorState:
    """Device-side actor loop state: E env streams with frame-stack buffers."""

    def __init__(self, cfg, comm: Communicator, agent: Agent, n_envs: int, chunk_steps: int):
        self.cfg = cfg
        self.comm = comm
        self.agent = agent
        self.n_envs = n_envs
        self.chunk_steps = chunk_steps
        self.env = VectorSyntheticEnv(cfg, n_envs, seed_offset=comm.rank * 1009)
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


class ApexRunner:
    """Fixed-cadence Ape-X round loop shared by the launch scripts and bench."""

    def __init__(self, cfg, comm: Communicator, n_envs: int = 16,
                 chunk_steps: Optional[int] = None, replay_capacity: Optional[int] = None,
                 device: Optional[str] = None):
        self.cfg = cfg
        self.comm = comm
        dev = device or str(comm.device)
        self.agent = Agent(cfg, VectorSyntheticEnv(cfg, 1).action_space(), dev)
        self.chunk_steps = chunk_steps or max(1, cfg.transition_chunk_size // n_envs)
        self.n_envs = n_envs
        self.actor: Optional[ActorState] = None
        self.replay: Optional[ShardedReplay] = None
        self.flat_weights = self.agent.flat_weights()
        if comm.is_actor:
            self.actor = ActorState(cfg, comm, self.agent, n_envs, self.chunk_steps)
        if comm.is_learner:
            self.replay = ShardedReplay(cfg, comm.n_actors, replay_capacity)
        self.grad_steps = 0
        self.losses: List[float] = []

    # -- one Ape-X round ----------------------------------------------------
    def run_round(self, learn: bool = True) -> None:
        cfg = self.cfg
        # 1. actors generate a chunk of experience
        if self.actor is not None:
            self.agent.reset_noise()
            self.actor.run_chunk()
            chunk_f, chunk_m = self.actor.chunk_frames, self.actor.chunk_meta
        else:
            c = self.n_envs * self.chunk_steps
            chunk_f = torch.zeros(c, 84, 84, dtype=torch.uint8, device=self.comm.device)
            chunk_m = torch.zeros(c, 3, device=self.comm.device)

        # 2. transition flow: actors -> learner replay (reference: redis push)
        frames_list = self.comm.gather_chunks(chunk_f)
        meta_list = self.comm.gather_chunks(chunk_m)
        if self.comm.is_learner:
            if self.comm.world_size == 1:
                frames_list, meta_list = [chunk_f], [chunk_m]
            for a, (f, m) in enumerate(zip(frames_list, meta_list)):
                mc = m.cpu().numpy()
                self.replay.append_actor_chunk(
                    a, f.cpu().numpy(), mc[:, 0].astype(np.int64), mc[:, 1],
                    mc[:, 2] > 0.5,
                )

        # 3. learner trains
        if self.comm.is_learner and learn and len(self.replay) >= cfg.learn_start:
            for _ in range(cfg.learn_per_round):
                loss = self.agent.learn(self.replay)
                self.losses.append(loss)
                self.grad_steps += 1
                if self.grad_steps % max(1, cfg.target_update // max(1, cfg.replay_frequency)) == 0:
                    self.agent.update_target_net()

        # 4. weight flow: learner -> actors (reference: redis weight blob)
        if self.comm.world_size > 1:
            if self.comm.is_learner:
                self.flat_weights = self.agent.flat_weights()
            self.comm.broadcast_weights(self.flat_weights, src=0)
            if not self.comm.is_learner:
                self.agent.load_flat_weights(self.flat_weights)
