"""Fully device-resident Ape-X actor round (kernels K1-K9/K15
plus the synthetic env source).

One round = chunk_steps x { stack-gather -> tau draw -> RainbowIQN forward ->
greedy action -> env-step kernel (reward/done + next frame + chunk record) },
unrolled into a single hip graph.  The frame-stack ring has `history` slots;
chunk_steps is a multiple of `history` so the ring head returns to its start
position every round and the captured graph's hardcoded slot indices stay
valid across replays.  Zero host round-trips inside a round; the outgoing
chunk (u8 frames + f32 meta) feeds the RCCL gather directly.
"""

from __future__ import annotations

from typing import Optional

import torch

from ..models.model import RainbowIQN
from ..ops import get_extension

FRAME_PX = 84 * 84


class GpuActorLoop:
    def __init__(self, cfg, net: RainbowIQN, n_envs: int, chunk_steps: int,
                 device: str = "cuda", use_graph: bool = True,
                 seed_offset: int = 0, p_done: float = 4.0 / 512.0,
                 external_weights: bool = False):
        # external_weights: the owner of `net` (the colocated learner)
        # regenerates noise and weight packs; this loop only READS them --
        # required when the actor graph runs CONCURRENTLY with the learner
        # stream (writes from two streams to the shared buffers would race).
        assert chunk_steps % cfg.history_length == 0, (
            "chunk_steps must be a multiple of history_length (ring phase)"
        )
        self.cfg = cfg
        self.ext = get_extension()
        self.net = net
        self.external_weights = external_weights
        self.E = n_envs
        self.chunk_steps = chunk_steps
        self.K = cfg.num_quantile_samples
        self.h = cfg.history_length
        self.p_done = p_done
        self.device = torch.device(device)
        d = self.device
        e = n_envs
        self.ring = torch.zeros(self.h, e, 84, 84, dtype=torch.uint8, device=d)
        self.env_state = torch.zeros(e, 4, dtype=torch.int32, device=d)
        # device env task: 1 = learnable cue task (env.py::CueEnv semantics
        # in the env_step kernel), 0 = random-reward throughput env
        self.task = 1 if cfg.game.lower() == "cue" else 0
        if self.task == 1:
            g = torch.Generator().manual_seed(cfg.seed + seed_offset)
            self.env_state[:, 3] = torch.randint(0, 4, (e,), generator=g,
                                                 dtype=torch.int32)
        self.rng_state = torch.tensor(
            [cfg.seed * 48271 + 101 + seed_offset, 0], dtype=torch.int64, device=d
        )
        self.dtype = torch.bfloat16 if cfg.dtype == "bf16" else torch.float32
        self.stack_f32 = torch.zeros(e, self.h, 84, 84, dtype=self.dtype, device=d)
        self.taus = torch.empty(e * self.K, 1, device=d)
        self.actions = torch.zeros(e, dtype=torch.int64, device=d)
        c = chunk_steps * e
        self.chunk_frames = torch.zeros(c, 84, 84, dtype=torch.uint8, device=d)
        self.chunk_meta = torch.zeros(c, 3, device=d)
        self.use_graph = use_graph
        self.graph: Optional[torch.cuda.CUDAGraph] = None
        self.total_steps = 0
        self._newest = 0  # ring slot of the most recent frame (round-invariant)
        # hand-written fused forward (4 kernels/step) on the bf16 path
        self.fused = None
        if (self.dtype == torch.bfloat16
                and net.fc_h_v.weight_mu.dtype == torch.bfloat16
                and getattr(cfg, "fused_forward", True)):
            from .fused_forward import FusedForward

            self.fused = FusedForward.for_net(
                net, d, fp8_heads=getattr(cfg, 'fp8_heads', False))

    def _round_body(self) -> None:
        cfg = self.cfg
        if not self.external_weights:
            self.reset_noise()  # in-graph: device philox noise regeneration
            if self.fused is not None:
                self.fused.pack_weights()
        newest = self._newest
        for s in range(self.chunk_steps):
            self.ext.rand_uniform(self.taus, self.rng_state)
            if self.fused is not None:
                # conv1 gathers the frame stack from the u8 ring itself
                # (fused K15); the bf16 stack tensor never materializes
                self.fused.act_gather(self.ring.view(self.h, -1, 7056),
                                      self.env_state.view(-1), self.h,
                                      newest, self.taus.view(-1), self.K,
                                      self.actions)
            else:
                self.ext.stack_gather(
                    self.ring.view(self.h, -1), self.env_state.view(-1),
                    self.h, newest, self.E, self.stack_f32,
                )
                with torch.no_grad():
                    z, _ = self.net(self.stack_f32, self.K, taus=self.taus)
                    self.actions.copy_(z.float().mean(dim=1).argmax(dim=1))
            newest = (newest + 1) % self.h
            self.ext.env_step(
                self.actions, self.env_state.view(-1), self.rng_state,
                self.ring.view(self.h, -1), self.h, newest,
                cfg.reward_clip if cfg.reward_clip_enabled else 0.0,
                self.p_done, cfg.max_episode_length,
                self.chunk_frames.view(-1, 84, 84), self.chunk_meta, s,
                self.task,
            )

    def reset_noise(self) -> None:
        # batched: all 4 NoisyLinear layers in 2 kernels
        if getattr(self, "_noisy_pack", None) is None:
            from .fused_forward import noisy_pack

            self._noisy_pack = noisy_pack(self.net, self.device)
        self.ext.noisy_material4(self._noisy_pack, self.rng_state)

    def capture(self) -> None:
        torch.cuda.synchronize()
        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            self._round_body()  # warmup (MIOpen find, workspace allocs)
        torch.cuda.current_stream().wait_stream(side)
        torch.cuda.synchronize()
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g):
            self._round_body()
        self.graph = g

    def run_round(self) -> None:
        if self.use_graph:
            if self.graph is None:
                self.capture()
                self.total_steps += self.chunk_steps * self.E  # warmup round counted
            self.graph.replay()
        else:
            self._round_body()
        self.total_steps += self.chunk_steps * self.E
