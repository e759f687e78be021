"""Failure detection and recovery.

The reference gets fault tolerance by architecture: Redis decouples actor /
learner lifetimes, and a learner restart reloads the model checkpoint.  The
MI355X rebuild runs over a static in-node RCCL communicator, so the
equivalents are:

* :class:`GracefulShutdown` -- SIGTERM/SIGINT request a clean stop at the
  next round boundary; the training loop checkpoints the model (and replay
  snapshot) before exiting, so a preempted job resumes exactly like the
  reference's learner-restart path.
* :class:`NanGuard` -- divergence detection.  A rolling "last good" weight
  snapshot (fp32 master + Adam moments where the fast learner is active) is
  refreshed every ``interval`` healthy rounds; when a learn round produces a
  non-finite loss the guard restores the snapshot, zeroes the Adam moments
  (they are poisoned by the same inf/NaN gradient), and training continues.

Both are host-side and backend-agnostic: the CPU/gloo path exercises the
identical code in tests/test_failsafe.py.
"""

from __future__ import annotations

import math
import signal
from typing import Optional

import torch

from .logging import log


class GracefulShutdown:
    """Installs SIGTERM/SIGINT handlers that request (not force) a stop."""

    def __init__(self, signals=(signal.SIGTERM, signal.SIGINT)):
        self.requested = False
        self._prev = {}
        for s in signals:
            try:
                self._prev[s] = signal.signal(s, self._handler)
            except (ValueError, OSError):  # non-main thread: stay inert
                pass

    def _handler(self, signum, frame) -> None:
        self.requested = True
        log(f"shutdown requested (signal {signum}); "
            "checkpointing at round boundary")

    def restore(self) -> None:
        for s, h in self._prev.items():
            signal.signal(s, h)


class NanGuard:
    """Detects non-finite losses and rolls the learner back to the last
    healthy weight snapshot."""

    def __init__(self, runner, interval: int = 200):
        self.runner = runner
        self.interval = max(1, interval)
        self._since = 0
        self._snap: Optional[dict] = None
        self.recoveries = 0

    # -- snapshot ------------------------------------------------------------
    def _take(self) -> None:
        gl = self.runner.glearner
        if gl is not None:
            self._snap = {
                "master": (gl.opt.master.detach().clone()
                           if hasattr(gl.opt, "master")
                           else gl.flat_online.detach().clone()),
                "m": gl.opt.m.detach().clone(),
                "v": gl.opt.v.detach().clone(),
                "step": gl.opt.step_t.detach().clone()
                if hasattr(gl.opt, "step_t") else None,
            }
        else:
            agent = self.runner.agent
            self._snap = {
                "net": {k: v.detach().clone()
                        for k, v in agent.online_net.state_dict().items()},
            }

    def _restore(self) -> None:
        gl = self.runner.glearner
        if gl is not None:
            if hasattr(gl.opt, "master"):
                gl.opt.master.copy_(self._snap["master"])
                gl.flat_online.copy_(self._snap["master"].to(
                    gl.flat_online.dtype))
            else:
                gl.flat_online.copy_(self._snap["master"])
            # moments from the snapshot predate the divergence; restoring
            # them (not zeroing) keeps Adam's bias correction consistent
            gl.opt.m.copy_(self._snap["m"])
            gl.opt.v.copy_(self._snap["v"])
            if self._snap["step"] is not None:
                gl.opt.step_t.copy_(self._snap["step"])
            gl.opt.zero_grad()
            if getattr(gl, "fused_online", None) is not None:
                gl.fused_online.pack_weights()
        else:
            agent = self.runner.agent
            agent.online_net.load_state_dict(self._snap["net"])
            # torch Adam state may hold inf moments: drop it (fresh moments)
            agent.optimiser = type(agent.optimiser)(
                agent.online_net.parameters(),
                **agent.optimiser.defaults)

    # -- per-round hook ------------------------------------------------------
    def check(self, loss: Optional[float]) -> bool:
        """Call once per learn round with the round's loss.  Returns True if
        a rollback happened."""
        if loss is not None and not math.isfinite(loss):
            if self._snap is None:
                log("non-finite loss before first snapshot; skipping round")
                return False
            self._restore()
            self.recoveries += 1
            self._since = 0
            log(f"non-finite loss detected: rolled back to last healthy "
                f"snapshot (recovery #{self.recoveries})")
            return True
        self._since += 1
        if self._since >= self.interval or self._snap is None:
            self._take()
            self._since = 0
        return False
