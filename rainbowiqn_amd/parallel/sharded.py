"""Sharded replay across GPUs (BASELINE config 5: 100M-transition PER across
8 x 288 GB HBM3E; the reference's closest analog is Redis cluster sharding
of the transition DB, SURVEY.md sections 3.5/6.8).

The single-learner replay tops out at ~35M transitions per GPU (frames are
~7 KB each).  In sharded mode each ACTOR rank keeps its own HBM-resident
shard of the prioritized replay -- transitions never travel to the learner
wholesale.  Per Ape-X round:

  1. all ranks all_gather shard (count, priority-total) -- 2 floats each;
  2. each actor samples its fixed quota q_a of the learner's batch (q_a sums
     to B, remainder spread over the first actors; shards fill at the same
     rate so fixed quotas match the proportional-allocation ideal), for each
     of the round's learn steps, assembling frame-stacks / n-step returns
     LOCALLY, and P2P-sends the batch tensors to the learner;
  3. the learner scatters them into the static step-indexed buffers of a
     :class:`RemoteReplay` facade and replays its captured learn graph;
  4. new per-sample priorities are P2P-returned to the owning actors, which
     update their local trees (indices never left the actor).

IS weights are computed shard-locally ((N_a p_i / total_a)^-beta, max-
normalized per shard); with balanced shards this matches the global-replay
weights up to a per-shard normalization (documented approximation).

Enabled with --sharded-replay (world_size > 1).  The gloo/CPU path runs the
identical flow over ReplayMemory shards (tests/test_dist_gloo.py).
"""

from __future__ import annotations

from typing import List, Optional

import torch
import torch.distributed as dist


def actor_quota(batch: int, n_actors: int, actor_index: int) -> int:
    q = batch // n_actors
    if actor_index < batch % n_actors:
        q += 1
    return q


class ActorShardHost:
    """Actor-side: local PER shard + per-round batch service."""

    def __init__(self, cfg, comm, quota: int, steps: int, device, fast: bool):
        self.cfg = cfg
        self.comm = comm
        self.q = quota
        self.S = steps
        self.device = torch.device(device)
        self.fast = fast
        cap = max(1024, cfg.memory_capacity // comm.n_actors)
        if fast:
            from ..replay.gpu_replay import GpuReplay

            self.replay = GpuReplay(cfg, n_rings=1, capacity=cap,
                                    device=device, batch_size=quota,
                                    seed=cfg.seed * 31 + comm.rank,
                                    out_dtype=torch.bfloat16
                                    if cfg.dtype == "bf16"
                                    else torch.float32)
        else:
            from ..replay.cpu_replay import ReplayMemory

            self.replay = ReplayMemory(cfg, cap)
        h = cfg.history_length
        dt = (torch.bfloat16 if (fast and cfg.dtype == "bf16")
              else torch.float32)
        S, q = steps, quota
        d = self.device
        self.st = torch.zeros(S, q, h, 84, 84, dtype=dt, device=d)
        self.s2 = torch.zeros(S, q, h, 84, 84, dtype=dt, device=d)
        self.aa = torch.zeros(S, q, dtype=torch.int64, device=d)
        self.rr = torch.zeros(S, q, device=d)
        self.nt = torch.zeros(S, q, device=d)
        self.ww = torch.zeros(S, q, device=d)
        self.pri = torch.zeros(S, q, device=d)
        self._idxs: List = [None] * steps

    def append_chunk(self, frames, meta=None, **kw) -> None:
        if self.fast:
            self.replay.append_actor_chunk(0, frames, meta=meta, **kw)
        else:
            self.replay.append_chunk(frames, kw["actions"], kw["rewards"],
                                     kw["terminals"])

    def counts(self) -> torch.Tensor:
        if self.fast:
            tot = self.replay.tree_total()
        else:
            tot = self.replay.tree.total()
        return torch.tensor([float(len(self.replay)), float(tot)])

    def fill_buffers(self) -> None:
        """Sample S x q transitions from the local shard into the send
        buffers (frame-stack + n-step assembly happen in the shard)."""
        for s in range(self.S):
            idx, st, a, r, s2, nt, w = self.replay.sample(self.q)
            self._idxs[s] = idx if self.fast else idx.copy()
            if self.fast:
                # GpuReplay returns its static buffers: copy out
                self.st[s].copy_(st)
                self.s2[s].copy_(s2)
                self.aa[s].copy_(a)
                self.rr[s].copy_(r)
                self.nt[s].copy_(nt)
                self.ww[s].copy_(w)
                self._idxs[s] = idx.clone()
            else:
                self.st[s].copy_(st)
                self.s2[s].copy_(s2)
                self.aa[s].copy_(a)
                self.rr[s].copy_(r)
                self.nt[s].copy_(nt)
                self.ww[s].copy_(w)

    def sample_and_send(self) -> None:
        self.fill_buffers()
        ops = [dist.P2POp(dist.isend, t.contiguous(), 0)
               for t in (self.st, self.s2, self.aa.float(), self.rr, self.nt,
                         self.ww)]
        for wk in dist.batch_isend_irecv(ops):
            wk.wait()

    def recv_priorities(self) -> None:
        for wk in dist.batch_isend_irecv(
                [dist.P2POp(dist.irecv, self.pri, 0)]):
            wk.wait()
        for s in range(self.S):
            self.replay.update_priorities(
                self._idxs[s],
                self.pri[s] if self.fast else self.pri[s].cpu().numpy())

    def anneal_beta(self, beta: float) -> None:
        self.replay.priority_weight = min(1.0, beta)


class RemoteReplay:
    """Learner-side replay facade over the actors' shards.

    Exposes the GpuReplay contract the learner's captured graph uses:
    sample() hands out the next step-indexed slice of the static recv
    buffers (a cycling counter, so capture order is just a permutation of
    the round's steps), and update_priorities() stashes the step's
    priorities for the end-of-round P2P return.
    """

    def __init__(self, cfg, comm, steps: int, device,
                 dtype: torch.dtype = torch.bfloat16):
        self.cfg = cfg
        self.comm = comm
        self.S = steps
        self.B = cfg.batch_size
        self.device = torch.device(device)
        h = cfg.history_length
        d = self.device
        n = comm.n_actors
        self.quotas = [actor_quota(self.B, n, a) for a in range(n)]
        # per-actor recv buffers
        self.rst = [torch.zeros(steps, q, h, 84, 84, dtype=dtype, device=d)
                    for q in self.quotas]
        self.rs2 = [torch.zeros(steps, q, h, 84, 84, dtype=dtype, device=d)
                    for q in self.quotas]
        self.raa = [torch.zeros(steps, q, device=d) for q in self.quotas]
        self.rrr = [torch.zeros(steps, q, device=d) for q in self.quotas]
        self.rnt = [torch.zeros(steps, q, device=d) for q in self.quotas]
        self.rww = [torch.zeros(steps, q, device=d) for q in self.quotas]
        # assembled step-indexed batch (static: the learn graph reads these)
        self.bst = torch.zeros(steps, self.B, h, 84, 84, dtype=dtype, device=d)
        self.bs2 = torch.zeros(steps, self.B, h, 84, 84, dtype=dtype, device=d)
        # GpuReplay contract for the manual fused backward (gpu_learner
        # reads replay.states_cat after sample()): one contiguous [s; s']
        # buffer refreshed per sample() call.  Static, so graph-capturable:
        # each unrolled learn step records its own slice copy into it.
        self.states_cat = torch.zeros(2 * self.B, h, 84, 84, dtype=dtype,
                                      device=d)
        self.baa = torch.zeros(steps, self.B, dtype=torch.int64, device=d)
        self.brr = torch.zeros(steps, self.B, device=d)
        self.bnt = torch.zeros(steps, self.B, device=d)
        self.bww = torch.zeros(steps, self.B, device=d)
        self.bpri = torch.zeros(steps, self.B, device=d)
        self._step = 0
        self._len = 0
        self.beta = torch.zeros(1, device=d)  # anneal target (actor-side too)

    def __len__(self) -> int:
        return self._len

    def recv_batches(self) -> None:
        ops = []
        for a in range(self.comm.n_actors):
            for t in (self.rst[a], self.rs2[a], self.raa[a], self.rrr[a],
                      self.rnt[a], self.rww[a]):
                ops.append(dist.P2POp(dist.irecv, t, a + 1))
        for wk in dist.batch_isend_irecv(ops):
            wk.wait()
        # scatter into the contiguous per-step batch buffers
        off = 0
        for a, q in enumerate(self.quotas):
            self.bst[:, off:off + q].copy_(self.rst[a])
            self.bs2[:, off:off + q].copy_(self.rs2[a])
            self.baa[:, off:off + q].copy_(self.raa[a].long())
            self.brr[:, off:off + q].copy_(self.rrr[a])
            self.bnt[:, off:off + q].copy_(self.rnt[a])
            self.bww[:, off:off + q].copy_(self.rww[a])
            off += q

    def send_priorities(self) -> None:
        ops = []
        off = 0
        for a, q in enumerate(self.quotas):
            ops.append(dist.P2POp(
                dist.isend, self.bpri[:, off:off + q].contiguous(), a + 1))
            off += q
        for wk in dist.batch_isend_irecv(ops):
            wk.wait()

    # -- GpuLearner replay contract -----------------------------------------
    def sample(self, batch_size: Optional[int] = None):
        s = self._step % self.S
        self._cur = s
        self._step += 1
        # refresh the contiguous [s; s'] view (device copy, capture-safe)
        self.states_cat[: self.B].copy_(self.bst[s])
        self.states_cat[self.B:].copy_(self.bs2[s])
        return (None, self.states_cat[: self.B], self.baa[s], self.brr[s],
                self.states_cat[self.B:], self.bnt[s], self.bww[s])

    def update_priorities(self, idx, priorities) -> None:
        self.bpri[self._cur].copy_(priorities.view(-1))

    def rebuild_tree(self) -> None:  # maintenance: trees live on the actors
        pass

    def set_len(self, n: int) -> None:
        self._len = n

    def state_dict(self) -> dict:
        # persistence of sharded replay is actor-side (each shard saves its
        # own GpuReplay.state_dict); nothing to snapshot on the learner
        return {"kind": "remote_replay"}

    def load_state_dict(self, sd: dict) -> None:
        pass
