"""HBM3E-resident prioritized replay (BASELINE config 2+).

Same sampling/priority semantics as the CPU ReplayMemory/ShardedReplay, but
every structure -- uint8 frame ring, priority sum-tree, ring heads, RNG state
-- is a device tensor and every operation is a handful of HIP kernels on the
current stream, so the entire learner step (sample -> forward -> loss ->
backward -> Adam -> priority update) is hipGraph-capturable with zero host
round-trips.

Memory: a 1M-transition ring is ~7 GB of frames + 8 MB of tree; 288 GB of
HBM3E admits ~35M transitions per GPU (config 5's 100M-across-8-GPUs shards
by actor ring).
"""

from __future__ import annotations

from typing import Optional

import torch

from ..ops import get_extension


def _next_pow2(n: int) -> int:
    p = 1
    while p < n:
        p *= 2
    return p


class GpuReplay:
    def __init__(self, cfg, n_rings: int = 1, capacity: Optional[int] = None,
                 device: str = "cuda", batch_size: Optional[int] = None,
                 seed: Optional[int] = None, out_dtype: torch.dtype = torch.float32):
        self.ext = get_extension()
        self.cfg = cfg
        self.device = torch.device(device)
        cap_req = int(capacity if capacity is not None else cfg.memory_capacity)
        self.n_rings = n_rings
        self.ring_cap = max(64, cap_req // n_rings)
        self.capacity = self.ring_cap * n_rings
        self.tree_size = _next_pow2(self.capacity)
        self.history = cfg.history_length
        self.n = cfg.multi_step
        self.discount = cfg.discount
        self.priority_exponent = cfg.priority_exponent
        self.batch_size = int(batch_size if batch_size is not None else cfg.batch_size)

        d = self.device
        self.frames = torch.zeros(self.capacity, 84, 84, dtype=torch.uint8, device=d)
        self.actions = torch.zeros(self.capacity, dtype=torch.int32, device=d)
        self.rewards = torch.zeros(self.capacity, dtype=torch.float32, device=d)
        self.terminals = torch.zeros(self.capacity, dtype=torch.uint8, device=d)
        self.timesteps = torch.zeros(self.capacity, dtype=torch.int32, device=d)
        self.tree = torch.zeros(2 * self.tree_size, dtype=torch.float32, device=d)
        self.ring_state = torch.zeros(n_rings, 4, dtype=torch.int32, device=d)
        self.max_priority = torch.ones(1, dtype=torch.float32, device=d)
        self.beta = torch.full((1,), cfg.priority_weight, device=d)
        s = seed if seed is not None else cfg.seed
        self.rng_state = torch.tensor([s * 2654435761 % (2**63), 0],
                                      dtype=torch.int64, device=d)
        b = self.batch_size
        # static output buffers (graph-capture targets)
        self.out_idx = torch.zeros(b, dtype=torch.int32, device=d)
        # bf16 output feeds the native-bf16 model path with no cast kernels
        # states and next_states are halves of ONE contiguous buffer so the
        # learner can run a single conv_encode over [s; s'] (B -> 2B batch)
        self.states_cat = torch.zeros(2 * b, self.history, 84, 84,
                                      dtype=out_dtype, device=d)
        self.out_states = self.states_cat[:b]
        self.out_next_states = self.states_cat[b:]
        self.out_actions = torch.zeros(b, dtype=torch.int64, device=d)
        self.out_returns = torch.zeros(b, device=d)
        self.out_nonterminals = torch.zeros(b, device=d)
        self.out_weights = torch.zeros(b, device=d)
        self._len_host = 0  # host mirror of total count (len() without sync)

    # -- API parity with ShardedReplay --------------------------------------
    def __len__(self) -> int:
        return self._len_host

    @property
    def priority_weight(self) -> float:
        return float(self.beta.item())

    @priority_weight.setter
    def priority_weight(self, v: float) -> None:
        self.beta.fill_(v)

    def append_actor_chunk(self, ring_id: int, frames, actions=None,
                           rewards=None, terminals=None,
                           meta: Optional[torch.Tensor] = None) -> None:
        """frames: u8 [C,84,84] (device); either meta f32 [C,3] (action,
        reward, done) or separate arrays."""
        if meta is None:
            import numpy as np

            c = len(actions)
            m = torch.empty(c, 3, dtype=torch.float32)
            m[:, 0] = torch.as_tensor(np.asarray(actions), dtype=torch.float32)
            m[:, 1] = torch.as_tensor(np.asarray(rewards), dtype=torch.float32)
            m[:, 2] = torch.as_tensor(
                np.asarray(terminals).astype("float32"), dtype=torch.float32
            )
            meta = m.to(self.device)
        if not isinstance(frames, torch.Tensor):
            frames = torch.from_numpy(frames)
        frames = frames.to(self.device).contiguous()
        self.ext.replay_append(
            self.frames, self.actions, self.rewards, self.terminals,
            self.timesteps, self.tree, self.tree_size, self.capacity,
            self.ring_state.view(-1), self.n_rings, self.ring_cap,
            self.max_priority, frames, meta.contiguous(), ring_id,
        )
        self._len_host = min(self._len_host + frames.shape[0], self.capacity)

    def sample(self, batch_size: Optional[int] = None):
        b = batch_size or self.batch_size
        assert b == self.batch_size, "GpuReplay uses fixed-size static buffers"
        self.ext.replay_sample(
            self.frames, self.actions, self.rewards, self.terminals,
            self.timesteps, self.tree, self.tree_size, self.capacity,
            self.ring_state.view(-1), self.n_rings, self.ring_cap,
            self.history, self.n, self.discount, self.beta, self.rng_state, b,
            self.out_idx, self.out_states, self.out_next_states,
            self.out_actions, self.out_returns, self.out_nonterminals,
            self.out_weights,
        )
        return (
            self.out_idx,
            self.out_states,
            self.out_actions,
            self.out_returns,
            self.out_next_states,
            self.out_nonterminals,
            self.out_weights,
        )

    def update_priorities(self, idxs: torch.Tensor, priorities: torch.Tensor) -> None:
        self.ext.replay_update_priorities(
            self.tree, self.tree_size, self.max_priority,
            idxs.to(torch.int32), priorities.float().contiguous(),
            self.priority_exponent,
        )

    def rebuild_tree(self) -> None:
        """Full level-wise rebuild: clears fp32 drift from delta propagation.
        Called outside graph capture (e.g. at target-update cadence)."""
        self.ext.tree_rebuild(self.tree, self.tree_size)

    # -- persistence (reference: Redis RDB snapshot) ----------
    _PERSIST = ("frames", "actions", "rewards", "terminals", "timesteps",
                "tree", "ring_state", "max_priority", "beta", "rng_state")

    def state_dict(self) -> dict:
        sd = {k: getattr(self, k).cpu() for k in self._PERSIST}
        sd["len"] = self._len_host
        sd["kind"] = "gpu_replay"
        return sd

    def load_state_dict(self, sd: dict) -> None:
        assert sd["frames"].shape[0] == self.capacity, (
            f"replay checkpoint capacity {sd['frames'].shape[0]} != {self.capacity}"
        )
        for k in self._PERSIST:
            getattr(self, k).copy_(sd[k].to(self.device))
        self._len_host = int(sd["len"])

    # -- introspection for tests --------------------------------------------
    def tree_total(self) -> float:
        return float(self.tree[1].item())

    def leaf_priorities(self) -> torch.Tensor:
        return self.tree[self.tree_size : self.tree_size + self.capacity]

    def check_invariants(self, atol: float = 1e-2) -> None:
        """Debug assertion pass over the sum-tree (the on-GPU
        shared replay needs its own race checks): every internal node must
        equal the sum of its children and leaves must be non-negative.
        Host-side; call outside captured graphs."""
        t = self.tree.cpu()
        leaves = t[self.tree_size:]
        assert (leaves >= 0).all(), "negative leaf priority"
        import torch as _t

        node = t[1 : self.tree_size]
        kids = t[2 : 2 * self.tree_size].view(-1, 2).sum(dim=1)
        bad = (node - kids).abs() > atol * (1.0 + node.abs())
        assert not bool(bad.any()), (
            f"sum-tree invariant violated at {int(bad.nonzero()[0])}"
        )
