"""Real-ALE environment wrapper (reference rainbowiqn/env.py semantics,
SURVEY.md section 3.1 "Atari env" row).

Not exercised on the benchmark boxes (no ALE wheel, no network -- SURVEY.md
section 3.2); `make_env` falls back to the synthetic source there.  Kept so a
deployment WITH `atari_py`/`ale_py` gets the reference's exact behavior:

  * raw ALE (not gym), minimal action set
  * grayscale 84x84 observations (area-interpolated resize; cv2 if present)
  * 4-frame stack as device tensors in [0,1]
  * action-repeat 4 with a max-pool over the last two raw frames
  * up to 30 random no-ops at reset
  * life-loss pseudo-terminals during TRAINING only
  * 108k-frame (30-minute) episode cap
"""

from __future__ import annotations

from collections import deque
from typing import Deque, Tuple

import numpy as np
import torch


def _resize_84(gray: np.ndarray) -> np.ndarray:
    """210x160 (or other) grayscale -> 84x84 uint8."""
    try:
        import cv2

        return cv2.resize(gray, (84, 84), interpolation=cv2.INTER_AREA)
    except ImportError:
        h, w = gray.shape
        ys = (np.arange(84) * h // 84).clip(0, h - 1)
        xs = (np.arange(84) * w // 84).clip(0, w - 1)
        return gray[ys][:, xs]


class AtariEnv:
    def __init__(self, cfg, device: str | torch.device = "cpu", ale=None):
        # `ale` injection: any object with the ALEInterface surface this
        # class uses (act/reset_game/game_over/lives/getScreenGrayscale/
        # getMinimalActionSet).  env_recorded.FakeALE drives this wrapper
        # from a recorded trajectory on disk, so the REAL env code path
        # (resize, max-pool, life-loss pseudo-terminals, no-op starts) is
        # exercised end-to-end on boxes without an ALE wheel.
        if ale is not None:
            self.ale = ale
        else:
            try:
                import atari_py as ale_mod

                self.ale = ale_mod.ALEInterface()
                rom = ale_mod.get_game_path(cfg.game.lower().split("noframeskip")[0])
            except ImportError:  # ale_py fallback
                import ale_py

                self.ale = ale_py.ALEInterface()
                rom = ale_py.roms.get_rom_path(cfg.game)
            self.ale.setInt("random_seed", cfg.seed)
            self.ale.setInt("max_num_frames_per_episode", cfg.max_episode_length)
            self.ale.setFloat("repeat_action_probability", 0.0)
            self.ale.setInt("frame_skip", 0)
            self.ale.setBool("color_averaging", False)
            self.ale.loadROM(rom)

        actions = self.ale.getMinimalActionSet()
        self.actions = {i: int(a) for i, a in enumerate(actions)}
        self.device = torch.device(device)
        self.history = cfg.history_length
        self.training = True
        self.lives = 0
        self.life_termination = False
        self.state_buffer: Deque[torch.Tensor] = deque([], maxlen=self.history)

    # -- reference Env API --------------------------------------------------
    def action_space(self) -> int:
        return len(self.actions)

    def train(self) -> None:
        self.training = True

    def eval(self) -> None:
        self.training = False

    def _get_frame(self) -> torch.Tensor:
        gray = self.ale.getScreenGrayscale().squeeze()
        small = _resize_84(gray.astype(np.uint8))
        return torch.from_numpy(small.astype(np.float32) / 255.0).to(self.device)

    def _reset_buffer(self) -> None:
        self.state_buffer.clear()
        zero = torch.zeros(84, 84, device=self.device)
        for _ in range(self.history):
            self.state_buffer.append(zero)

    def reset(self) -> torch.Tensor:
        if self.life_termination:
            # pseudo-terminal: don't restart the emulator, just acknowledge
            self.life_termination = False
            self.ale.act(0)
        else:
            self._reset_buffer()
            self.ale.reset_game()
            # up to 30 random no-ops (reference behavior)
            for _ in range(np.random.randint(1, 31)):
                self.ale.act(0)
                if self.ale.game_over():
                    self.ale.reset_game()
        self.state_buffer.append(self._get_frame())
        self.lives = self.ale.lives()
        return torch.stack(list(self.state_buffer), 0)

    def step(self, action: int) -> Tuple[torch.Tensor, float, bool]:
        # action-repeat 4, max-pool over the last two raw frames
        frame_buffer = torch.zeros(2, 84, 84, device=self.device)
        reward, done = 0.0, False
        for t in range(4):
            reward += self.ale.act(self.actions[action])
            if t == 2:
                frame_buffer[0] = self._get_frame()
            elif t == 3:
                frame_buffer[1] = self._get_frame()
            done = self.ale.game_over()
            if done:
                break
        self.state_buffer.append(frame_buffer.max(0)[0])
        if self.training:
            lives = self.ale.lives()
            if lives < self.lives and lives > 0:
                # life lost: pseudo-terminal for the learner, emulator keeps
                # running (reference: training only)
                self.life_termination = not done
                done = True
            self.lives = lives
        return torch.stack(list(self.state_buffer), 0), reward, done
