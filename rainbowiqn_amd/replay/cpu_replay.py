"""CPU prioritized n-step replay (plumbing config + behavioral reference).

Semantics follow the reference ReplayMemory (rainbowiqn/memory.py, SURVEY.md
sections 3.1 and 4.4, Kaixhin lineage):

  * stores SINGLE 84x84 uint8 frames (not 4-stacks) in a ring buffer and
    reconstructs the 4-frame stacks at sample time (4x memory saving);
  * slot t holds (last frame of s_t, a_t, r_{t+1}, done_t);
  * proportional PER: leaf priorities are p^omega, sampled by segment,
    importance weights (N * P)^-beta normalized by the batch max;
  * n-step returns assembled at sample time with early-terminal cutoff;
  * samples whose frame windows cross the ring write head are rejected and
    redrawn (the reference's trickiest logic -- see test_replay.py);
  * frames from before an episode start are zeroed in the stacks.

The on-GPU replay (gpu_replay.py + ops/csrc/replay.hip) implements the same
contract; tests compare the two.
"""

from __future__ import annotations


import numpy as np
import torch

from .sumtree import SumTree


class ReplayMemory:
    def __init__(self, cfg, capacity: int | None = None):
        self.capacity = int(capacity if capacity is not None else cfg.memory_capacity)
        self.history = cfg.history_length
        self.n = cfg.multi_step
        self.discount = cfg.discount
        self.priority_exponent = cfg.priority_exponent
        self.priority_weight = cfg.priority_weight  # beta; annealed by caller
        self.tree = SumTree(self.capacity)
        h, w = 84, 84
        self.frames = np.zeros((self.capacity, h, w), dtype=np.uint8)
        self.actions = np.zeros(self.capacity, dtype=np.int64)
        self.rewards = np.zeros(self.capacity, dtype=np.float32)
        self.terminals = np.zeros(self.capacity, dtype=bool)
        self.timesteps = np.zeros(self.capacity, dtype=np.int64)
        self.index = 0  # next write position
        self.count = 0  # number of valid slots
        self._t = 0  # episode-relative timestep of next append
        self._discount_pows = self.discount ** np.arange(self.n)

    def __len__(self) -> int:
        return self.count

    # -- append ------------------------------------------------------------
    def append(self, state, action: int, reward: float, terminal: bool,
               priority: float | None = None) -> None:
        """state: [history,84,84] tensor/array in [0,1] or a single uint8 frame."""
        frame = self._last_frame_u8(state)
        i = self.index
        self.frames[i] = frame
        self.actions[i] = action
        self.rewards[i] = reward
        self.terminals[i] = terminal
        self.timesteps[i] = self._t
        p = self.tree.max_priority if priority is None else priority
        self.tree.update(np.array([i]), np.array([p]))
        self.index = (i + 1) % self.capacity
        self.count = min(self.count + 1, self.capacity)
        self._t = 0 if terminal else self._t + 1

    def append_chunk(self, frames: np.ndarray, actions, rewards, terminals,
                     priorities=None) -> None:
        """Bulk append (the Ape-X actor->replay path).  frames: [C,84,84] u8."""
        c = len(actions)
        idxs = (self.index + np.arange(c)) % self.capacity
        self.frames[idxs] = frames
        self.actions[idxs] = actions
        self.rewards[idxs] = rewards
        term = np.asarray(terminals, dtype=bool)
        self.terminals[idxs] = term
        ts = np.empty(c, dtype=np.int64)
        t = self._t
        for k in range(c):  # episode-relative timestep bookkeeping
            ts[k] = t
            t = 0 if term[k] else t + 1
        self._t = t
        self.timesteps[idxs] = ts
        if priorities is None:
            p = np.full(c, self.tree.max_priority)
        else:
            p = np.asarray(priorities, dtype=np.float64) ** self.priority_exponent
        self.tree.update(idxs, p)
        self.index = int((self.index + c) % self.capacity)
        self.count = min(self.count + c, self.capacity)

    @staticmethod
    def _last_frame_u8(state) -> np.ndarray:
        if isinstance(state, torch.Tensor):
            if state.dim() == 3:
                state = state[-1]
            return (state.detach().cpu().numpy() * 255).astype(np.uint8)
        state = np.asarray(state)
        if state.ndim == 3:
            state = state[-1]
        if state.dtype == np.uint8:
            return state
        return (state * 255).astype(np.uint8)

    # -- sampling ----------------------------------------------------------
    def _uniform_valid_slots(self, k: int) -> np.ndarray:
        """Uniform draw over slots whose [t-h+1, t+n] windows are in-bounds."""
        if self.count < self.capacity:
            span = max(1, self.count - self.n - 1)
            return np.random.randint(0, span, k)
        lo = (self.index + self.history - 1) % self.capacity
        span = max(1, self.capacity - (self.n + self.history))
        return (lo + np.random.randint(0, span, k)) % self.capacity

    def _valid(self, idxs: np.ndarray) -> np.ndarray:
        """A sampled slot t is valid iff the windows [t-h+1, t+n] do not cross
        the ring write head and the slot has nonzero priority."""
        head = self.index
        # distance from idx forward to head must exceed n (need idx..idx+n)
        fwd = (head - idxs) % self.capacity
        ok = fwd > self.n
        # distance from head forward to the history start must be >= 0, i.e.
        # idx-h+1 must not be in the not-yet-overwritten-window before head
        bwd = (idxs - head) % self.capacity
        ok &= (bwd >= self.history - 1) | (self.count < self.capacity)
        if self.count < self.capacity:  # ring not full: need idx >= 0 slots
            ok &= idxs < self.count
            ok &= (idxs + self.n) < self.count
        ok &= self.tree.get(idxs) > 0
        return ok

    def sample(self, batch_size: int):
        total = self.tree.total()
        assert total > 0, "sample() from empty replay"
        seg = total / batch_size
        rng = np.random
        idxs = np.zeros(batch_size, dtype=np.int64)
        pending = np.arange(batch_size)
        for attempt in range(64):  # bounded rejection loop
            if attempt < 8:
                # stratified: redraw within each element's own segment
                u = (pending + rng.random(len(pending))) * seg
            else:
                # a segment can lie entirely inside the invalid window near
                # the write head -- redraw stragglers over the full range
                u = rng.random(len(pending)) * total
            cand = self.tree.find(np.minimum(u, total * (1 - 1e-12)))
            idxs[pending] = cand
            good = self._valid(cand)
            pending = pending[~good]
            if len(pending) == 0:
                break
        else:
            # last resort: uniform over the definitely-valid index region
            idxs[pending] = self._uniform_valid_slots(len(pending))

        states, next_states, returns, nonterminals = self._assemble(idxs)
        # floor: last-resort fallback slots can carry (near-)zero priority
        probs = np.maximum(self.tree.get(idxs), 1e-12 * max(total, 1.0)) / total
        n_valid = self.count
        weights = (n_valid * probs) ** -self.priority_weight
        weights = weights / weights.max()
        return (
            idxs,
            torch.from_numpy(states),
            torch.from_numpy(self.actions[idxs]),
            torch.from_numpy(returns),
            torch.from_numpy(next_states),
            torch.from_numpy(nonterminals),
            torch.from_numpy(weights.astype(np.float32)),
        )

    def _assemble(self, idxs: np.ndarray):
        b = len(idxs)
        h = self.history
        states = np.zeros((b, h, 84, 84), dtype=np.float32)
        next_states = np.zeros((b, h, 84, 84), dtype=np.float32)
        returns = np.zeros(b, dtype=np.float32)
        nonterminals = np.zeros(b, dtype=np.float32)
        for k, t in enumerate(idxs):
            # state stack: frames t-h+1 .. t, zeroing pre-episode frames
            for j in range(h):
                src = (t - (h - 1 - j)) % self.capacity
                steps_back = h - 1 - j
                if self.timesteps[t] >= steps_back:
                    states[k, j] = self.frames[src].astype(np.float32) / 255.0
            # n-step return with early-terminal cutoff
            r = 0.0
            alive = True
            for s in range(self.n):
                j = (t + s) % self.capacity
                r += self._discount_pows[s] * self.rewards[j]
                if self.terminals[j]:
                    alive = False
                    break
            returns[k] = r
            nonterminals[k] = 1.0 if alive else 0.0
            if alive:
                tn = (t + self.n) % self.capacity
                for j in range(h):
                    src = (tn - (h - 1 - j)) % self.capacity
                    steps_back = h - 1 - j
                    if self.timesteps[tn] >= steps_back:
                        next_states[k, j] = self.frames[src].astype(np.float32) / 255.0
        return states, next_states, returns, nonterminals

    # -- priority maintenance ----------------------------------------------
    def update_priorities(self, idxs: np.ndarray, priorities) -> None:
        p = np.asarray(
            priorities.detach().cpu().numpy() if isinstance(priorities, torch.Tensor) else priorities,
            dtype=np.float64,
        )
        # floor keeps every stored transition reachable and the tree finite
        # (a zero/NaN priority would otherwise poison totals and IS weights)
        p = np.where(np.isfinite(p), np.abs(p), 1.0)
        p = np.maximum(p ** self.priority_exponent, 1e-8)
        self.tree.update(np.asarray(idxs, dtype=np.int64), p)

    # -- checkpoint (reference: redis RDB persistence) ----------------------
    def state_dict(self) -> dict:
        return {
            "frames": self.frames[: self.count],
            "actions": self.actions[: self.count],
            "rewards": self.rewards[: self.count],
            "terminals": self.terminals[: self.count],
            "timesteps": self.timesteps[: self.count],
            "priorities": self.tree.get(np.arange(self.count)),
            "index": self.index,
            "count": self.count,
            "episode_t": self._t,
        }

    def load_state_dict(self, sd: dict) -> None:
        c = int(sd["count"])
        self.frames[:c] = sd["frames"]
        self.actions[:c] = sd["actions"]
        self.rewards[:c] = sd["rewards"]
        self.terminals[:c] = sd["terminals"]
        self.timesteps[:c] = sd["timesteps"]
        self.count = c
        self.index = int(sd["index"])
        self._t = int(sd["episode_t"])
        self.tree.update(np.arange(c), np.asarray(sd["priorities"], dtype=np.float64))
