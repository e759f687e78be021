"""Recorded-trajectory environment.

The boxes have no ALE wheel, so `env_ale.AtariEnv` (the reference
rainbowiqn/env.py rebuild) was unexercisable.  This module closes that gap
without ALE: :class:`FakeALE` implements the slice of the ALEInterface API
the wrapper uses, backed by a DETERMINISTIC recorded trajectory from disk
(raw 210x160 grayscale frames + per-frame reward/lives/terminal).  Driving
`AtariEnv(cfg, ale=FakeALE(path))` runs the real wrapper code -- area
resize to 84x84, 4-frame stacking, action-repeat-4 with max-pool over the
last two raw frames, up to 30 no-op starts, life-loss pseudo-terminals in
training mode only, episode cap -- against reproducible data.

`record_trajectory` synthesizes and saves a trajectory (procedural frames;
no network, no ROMs); a deployment can equally record a real ALE run to the
same .npz schema and replay it here bit-for-bit.
"""

from __future__ import annotations

import numpy as np

RAW_H, RAW_W = 210, 160


def record_trajectory(path: str, n_frames: int = 2000, n_lives: int = 3,
                      seed: int = 0, n_actions: int = 6) -> dict:
    """Synthesize a deterministic episode tape and save it as .npz.

    Schema (all length n_frames, one entry per RAW emulator frame):
      frames   u8 [T, 210, 160]  raw grayscale screens
      rewards  f32 [T]           reward surfaced at that frame
      lives    i32 [T]           lives remaining AFTER that frame
      terminal u8 [T]            1 = game over at that frame
    """
    rng = np.random.default_rng(seed)
    y = np.arange(RAW_H, dtype=np.int64)[:, None]
    x = np.arange(RAW_W, dtype=np.int64)[None, :]
    frames = np.empty((n_frames, RAW_H, RAW_W), dtype=np.uint8)
    rewards = np.zeros(n_frames, dtype=np.float32)
    lives = np.zeros(n_frames, dtype=np.int32)
    terminal = np.zeros(n_frames, dtype=np.uint8)
    life, next_death = n_lives, int(rng.integers(300, 600))
    for t in range(n_frames):
        img = ((x * 2 + y * 3 + t * 7) & 0xFF).astype(np.uint8)
        cy = int(RAW_H / 2 + (RAW_H / 2 - 12) * np.sin(t * 0.05))
        cx = int(RAW_W / 2 + (RAW_W / 2 - 12) * np.cos(t * 0.03))
        img[max(0, cy - 6): cy + 6, max(0, cx - 6): cx + 6] = 255
        frames[t] = img
        if rng.random() < 0.01:
            rewards[t] = float(rng.integers(1, 4))
        if t == next_death:
            life -= 1
            next_death = t + int(rng.integers(300, 600))
            if life <= 0:
                terminal[t] = 1
                life = n_lives  # tape continues with a fresh game
        lives[t] = life
    data = dict(frames=frames, rewards=rewards, lives=lives,
                terminal=terminal, n_actions=np.int64(n_actions))
    np.savez_compressed(path, **data)
    return data


class FakeALE:
    """ALEInterface-compatible playback of a recorded trajectory.

    act(a) advances ONE raw frame and returns its recorded reward (the
    action only selects the tape -- playback is deterministic, which is the
    point: the wrapper logic above it is what's under test/use).
    reset_game() jumps to the frame after the current episode's terminal.
    """

    def __init__(self, path_or_data):
        if isinstance(path_or_data, dict):
            d = path_or_data
        else:
            d = np.load(path_or_data)
        self.frames = d["frames"]
        self.rewards = d["rewards"]
        self._lives = d["lives"]
        self.terminal = d["terminal"]
        self.n_actions = int(d["n_actions"])
        self.T = len(self.frames)
        self.t = 0
        self.acts = 0  # total act() calls (test introspection)

    # -- ALEInterface surface used by AtariEnv ------------------------------
    def getMinimalActionSet(self):
        return list(range(self.n_actions))

    def act(self, action: int) -> float:
        del action
        self.acts += 1
        if self.t + 1 < self.T:
            self.t += 1
        return float(self.rewards[self.t])

    def game_over(self) -> bool:
        return bool(self.terminal[self.t])

    def lives(self) -> int:
        return int(self._lives[self.t])

    def getScreenGrayscale(self) -> np.ndarray:
        return self.frames[self.t]

    def reset_game(self) -> None:
        # jump past the current episode's terminal (wrap at tape end)
        t = self.t
        while t + 1 < self.T and not self.terminal[t]:
            t += 1
        self.t = (t + 1) % self.T


def make_recorded_env(cfg, path: str, device="cpu"):
    """The real AtariEnv wrapper over a recorded tape (reference env
    semantics, deterministic data)."""
    from .env_ale import AtariEnv

    return AtariEnv(cfg, device, ale=FakeALE(path))
