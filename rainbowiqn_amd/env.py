"""Environment layer.

The reference (rainbowiqn/env.py) wraps the raw ALE:
grayscale 84x84 frames, 4-frame stack kept as device tensors scaled to [0,1],
action-repeat 4 with a max-pool over the last two raw frames, up to 30 random
no-ops at reset, life-loss pseudo-terminals during training only, and a
108k-frame episode cap.

This box has no ALE wheel and no network, so the
default source is :class:`SyntheticAtariEnv` -- same interface, same frame
shape/dtype, deterministic procedural frames -- which is also what the driver's
benchmark configs specify ("synthetic 84x84x4 frames").  If ``atari_py`` or
``ale_py`` is importable, :func:`make_env` returns the real wrapper instead.
"""

from __future__ import annotations

from collections import deque
from typing import Deque, Tuple

import numpy as np
import torch

FRAME_H = 84
FRAME_W = 84

_GAME_ACTIONS = {
    # minimal action-set sizes of common benchmark games
    "pong": 6,
    "breakout": 4,
    "space_invaders": 6,
    "seaquest": 18,
    "qbert": 6,
    # learnable synthetic task (reward a deterministic function of the
    # frame): 4 cue slots -> 4 actions.  See CueEnv below.
    "cue": 4,
}

CUE_SLOTS = 4  # cue task: frame shows a bright block in one of these slots


def n_actions_for(game: str) -> int:
    g = game.lower().replace("noframeskip-v4", "").replace("-", "_").rstrip("_")
    for k, v in _GAME_ACTIONS.items():
        if k in g:
            return v
    return 18  # full Atari action set


class SyntheticAtariEnv:
    """Drop-in synthetic replacement for the reference Env.

    Interface parity with the reference:
      reset() -> state [history,84,84] float tensor in [0,1]
      step(a) -> (state, reward, done)
      action_space() -> int
      train() / eval() mode toggle

    Frames are procedurally generated uint8 images: cheap (a few numpy ops),
    deterministic given the seed, and with enough structure that preprocessing
    and the CNN see realistic data statistics.  Episode lengths are sampled
    geometrically with mean ``mean_episode_len``; rewards are sparse +-1.
    """

    def __init__(self, cfg, device: str | torch.device = "cpu", mean_episode_len: int = 512):
        self.cfg = cfg
        self.device = torch.device(device)
        self.n_actions = n_actions_for(cfg.game)
        self.history = cfg.history_length
        self.max_episode_length = cfg.max_episode_length
        self.mean_episode_len = mean_episode_len
        self.rng = np.random.default_rng(cfg.seed * 7919 + 17)
        self.training = True
        self._t = 0  # frames this episode
        self._phase = self.rng.integers(0, 1 << 30)
        self.state_buffer: Deque[torch.Tensor] = deque([], maxlen=self.history)
        self.last_frame_u8: np.ndarray = np.zeros((FRAME_H, FRAME_W), dtype=np.uint8)

    # -- mode toggles (reference: life-loss terminals in train only) -------
    def train(self) -> None:
        self.training = True

    def eval(self) -> None:
        self.training = False

    def action_space(self) -> int:
        return self.n_actions

    # -- frame synthesis ---------------------------------------------------
    def _raw_frame(self) -> np.ndarray:
        """One 84x84 uint8 frame: moving gradient + a few sprites + noise."""
        t = int(self._t + self._phase) & 0xFFFF
        y = np.arange(FRAME_H, dtype=np.int64)[:, None]
        x = np.arange(FRAME_W, dtype=np.int64)[None, :]
        img = ((x * 3 + y * 2 + t * 5) & 0xFF).astype(np.uint8)
        # sprite: an 8x8 bright square orbiting
        cy = int(FRAME_H / 2 + (FRAME_H / 2 - 6) * np.sin(t * 0.11))
        cx = int(FRAME_W / 2 + (FRAME_W / 2 - 6) * np.cos(t * 0.07))
        img[max(0, cy - 4) : cy + 4, max(0, cx - 4) : cx + 4] = 255
        noise = self.rng.integers(0, 32, size=(FRAME_H, FRAME_W), dtype=np.uint8)
        return img | noise

    def _get_state(self) -> torch.Tensor:
        return torch.stack(list(self.state_buffer), 0)

    def _reset_buffer(self) -> None:
        zero = torch.zeros(FRAME_H, FRAME_W, device=self.device)
        self.state_buffer.clear()
        for _ in range(self.history):
            self.state_buffer.append(zero)

    def _observe(self) -> torch.Tensor:
        f = self._raw_frame()
        self.last_frame_u8 = f
        obs = torch.from_numpy(f.astype(np.float32) / 255.0).to(self.device)
        self.state_buffer.append(obs)
        return self._get_state()

    # -- env API -----------------------------------------------------------
    def reset(self) -> torch.Tensor:
        self._t = 0
        self._phase = int(self.rng.integers(0, 1 << 30))
        self._reset_buffer()
        # up-to-30 random no-ops at reset (reference behavior)
        for _ in range(int(self.rng.integers(1, 31))):
            self._t += 1
        return self._observe()

    def step(self, action: int) -> Tuple[torch.Tensor, float, bool]:
        assert 0 <= action < self.n_actions
        # action-repeat 4 (reference: max-pool over last 2 raw frames --
        # synthetic frames make pooling irrelevant; we keep the 4x frame cost
        # accounting by advancing t by 4)
        self._t += 4
        state = self._observe()
        # sparse reward: +-1 with small prob, slightly action-dependent
        u = self.rng.random()
        reward = 0.0
        if u < 0.02:
            reward = 1.0 if (action + self._t) % 3 else -1.0
        done = self.rng.random() < (4.0 / self.mean_episode_len)
        if self._t >= self.max_episode_length:
            done = True
        return state, reward, done


class CueEnv(SyntheticAtariEnv):
    """Learnable synthetic task: the newest frame
    shows a bright 20x20 block in one of CUE_SLOTS horizontal slots on a dim
    background; acting with the matching slot index yields +1, anything else
    -1.  Reward is a deterministic function of the frame, so a correct
    learner's average reward climbs from ~(2/A - 1) (random, -0.5 at A=4)
    toward +1 -- the end-to-end learning signal the random-reward synthetic
    env cannot provide.  Same Env interface as the reference wrapper.
    """

    def __init__(self, cfg, device="cpu", mean_episode_len: int = 512):
        super().__init__(cfg, device, mean_episode_len)
        self.n_actions = CUE_SLOTS
        self._cue = int(self.rng.integers(0, CUE_SLOTS))

    def _raw_frame(self) -> np.ndarray:
        t = int(self._t + self._phase) & 0xFFFF
        y = np.arange(FRAME_H, dtype=np.int64)[:, None]
        x = np.arange(FRAME_W, dtype=np.int64)[None, :]
        img = ((x * 3 + y * 2 + t * 5) & 0x3F).astype(np.uint8)  # dim bg
        x0 = self._cue * (FRAME_W // CUE_SLOTS) + 1
        img[32:52, x0 : x0 + 19] = 255
        noise = self.rng.integers(0, 32, size=(FRAME_H, FRAME_W), dtype=np.uint8)
        return img | noise

    def step(self, action: int) -> Tuple[torch.Tensor, float, bool]:
        assert 0 <= action < self.n_actions
        reward = 1.0 if action == self._cue else -1.0
        self._cue = int(self.rng.integers(0, CUE_SLOTS))
        self._t += 4
        state = self._observe()  # renders the NEW cue
        done = self.rng.random() < (4.0 / self.mean_episode_len)
        if self._t >= self.max_episode_length:
            done = True
        return state, reward, done


class VectorSyntheticEnv:
    """Batched synthetic env: E independent episode streams, numpy-vectorized.

    Used by the Ape-X actor loop to amortize the per-step act() inference over
    several env streams per actor GPU (capability superset of the reference's
    one-env-per-actor-process; set n_envs=1 for reference behavior).
    Frames are produced directly as uint8 [E,84,84]; the caller preprocesses
    on-GPU (ops kernel K15).
    """

    def __init__(self, cfg, n_envs: int, seed_offset: int = 0):
        self.cfg = cfg
        self.n_envs = n_envs
        self.n_actions = n_actions_for(cfg.game)
        self.history = cfg.history_length
        self.max_episode_length = cfg.max_episode_length
        self.rng = np.random.default_rng(cfg.seed * 100003 + seed_offset)
        self._t = np.zeros(n_envs, dtype=np.int64)
        self._phase = self.rng.integers(0, 1 << 16, size=n_envs)
        self._y = np.arange(FRAME_H, dtype=np.int64)[:, None]
        self._x = np.arange(FRAME_W, dtype=np.int64)[None, :]
        self.p_done = 4.0 / 512.0

    def action_space(self) -> int:
        return self.n_actions

    def frames(self) -> np.ndarray:
        """Current raw frames [E,84,84] uint8."""
        t = ((self._t + self._phase) & 0xFFFF)[:, None, None]
        img = ((self._x * 3 + self._y * 2 + t * 5) & 0xFF).astype(np.uint8)
        cy = (FRAME_H / 2 + (FRAME_H / 2 - 6) * np.sin(t[:, 0, 0] * 0.11)).astype(np.int64)
        cx = (FRAME_W / 2 + (FRAME_W / 2 - 6) * np.cos(t[:, 0, 0] * 0.07)).astype(np.int64)
        for e in range(self.n_envs):  # sprite stamp (tiny loop, 8x8 region)
            img[e, max(0, cy[e] - 4) : cy[e] + 4, max(0, cx[e] - 4) : cx[e] + 4] = 255
        noise = self.rng.integers(0, 32, size=img.shape, dtype=np.uint8)
        return img | noise

    def reset_all(self) -> np.ndarray:
        self._t[:] = 0
        self._phase = self.rng.integers(0, 1 << 16, size=self.n_envs)
        return self.frames()

    def step(self, actions: np.ndarray):
        """actions [E] -> (frames u8 [E,84,84], rewards f32 [E], dones bool [E])."""
        self._t += 4
        u = self.rng.random(self.n_envs)
        rewards = np.where(
            u < 0.02, np.where((actions + self._t) % 3 != 0, 1.0, -1.0), 0.0
        ).astype(np.float32)
        dones = self.rng.random(self.n_envs) < self.p_done
        dones |= self._t >= self.max_episode_length
        frames = self.frames()
        if dones.any():
            self._t[dones] = 0
            self._phase[dones] = self.rng.integers(0, 1 << 16, size=int(dones.sum()))
        return frames, rewards, dones


class VectorCueEnv(VectorSyntheticEnv):
    """Batched CueEnv: E independent streams of the learnable cue task."""

    def __init__(self, cfg, n_envs: int, seed_offset: int = 0):
        super().__init__(cfg, n_envs, seed_offset)
        self.n_actions = CUE_SLOTS
        self._cue = self.rng.integers(0, CUE_SLOTS, size=n_envs)

    def frames(self) -> np.ndarray:
        t = ((self._t + self._phase) & 0xFFFF)[:, None, None]
        img = ((self._x * 3 + self._y * 2 + t * 5) & 0x3F).astype(np.uint8)
        w = FRAME_W // CUE_SLOTS
        for e in range(self.n_envs):
            x0 = int(self._cue[e]) * w + 1
            img[e, 32:52, x0 : x0 + 19] = 255
        noise = self.rng.integers(0, 32, size=img.shape, dtype=np.uint8)
        return img | noise

    def step(self, actions: np.ndarray):
        rewards = np.where(actions == self._cue, 1.0, -1.0).astype(np.float32)
        self._cue = self.rng.integers(0, CUE_SLOTS, size=self.n_envs)
        self._t += 4
        dones = self.rng.random(self.n_envs) < self.p_done
        dones |= self._t >= self.max_episode_length
        frames = self.frames()
        if dones.any():
            self._t[dones] = 0
            self._phase[dones] = self.rng.integers(0, 1 << 16, size=int(dones.sum()))
        return frames, rewards, dones


def make_vector_env(cfg, n_envs: int, seed_offset: int = 0):
    """Vector env for the actor loop: the learnable cue task when
    cfg.game == 'cue', else the throughput-oriented random synthetic env."""
    if cfg.game.lower() == "cue":
        return VectorCueEnv(cfg, n_envs, seed_offset)
    return VectorSyntheticEnv(cfg, n_envs, seed_offset)


def make_env(cfg, device: str | torch.device = "cpu"):
    """Return the real ALE wrapper when available, else the synthetic env."""
    if getattr(cfg, "recorded_env", None):
        from .env_recorded import make_recorded_env

        return make_recorded_env(cfg, cfg.recorded_env, device)
    if cfg.game.lower() == "cue":  # learnable synthetic task, never ALE
        return CueEnv(cfg, device)
    try:  # pragma: no cover - no ALE wheel on this box
        try:
            import atari_py  # noqa: F401
        except ImportError:
            import ale_py  # noqa: F401

        from .env_ale import AtariEnv

        return AtariEnv(cfg, device)
    except ImportError:
        return SyntheticAtariEnv(cfg, device)
