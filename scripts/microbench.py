#!/usr/bin/env python3
"""Per-component timing of the fast path: learner graph replay, actor round
replay, chunk append, weight sync.  Run on a GPU box."""

import time

import numpy as np
import torch

from rainbowiqn_amd.config import Config
from rainbowiqn_amd.models.model import RainbowIQN
from rainbowiqn_amd.runtime.gpu_actor import GpuActorLoop
from rainbowiqn_amd.runtime.gpu_learner import GpuLearner


def timeit(fn, iters=50, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3  # ms


def learner_only():
    """Loop learner graph replays only (for rocprofv3 kernel attribution)."""
    cfg = Config(device="cuda", dtype="bf16", learn_start=2048,
                 memory_capacity=200_000, learn_per_round=8)
    rng = np.random.default_rng(0)
    lrn = GpuLearner(cfg, 6, n_rings=1, use_graph=True)
    f = rng.integers(0, 255, size=(4096, 84, 84), dtype=np.uint8)
    lrn.replay.append_actor_chunk(0, f, rng.integers(0, 6, 4096),
                                  rng.random(4096).astype(np.float32),
                                  rng.random(4096) < 0.01)
    lrn.learn_steps(16)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(100):
        lrn.graph.replay()
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    print(f"100 replays x {lrn.steps_per_replay} steps: {dt*1000:.1f} ms "
          f"-> {100*lrn.steps_per_replay/dt:.0f} grad-steps/s")


def main():
    import sys
    if len(sys.argv) > 1 and sys.argv[1] == "learner":
        return learner_only()
    cfg = Config(device="cuda", dtype="bf16", learn_start=2048,
                 memory_capacity=200_000, learn_per_round=8)
    torch.manual_seed(0)
    rng = np.random.default_rng(0)

    lrn = GpuLearner(cfg, 6, n_rings=1, use_graph=True)
    f = rng.integers(0, 255, size=(4096, 84, 84), dtype=np.uint8)
    a = rng.integers(0, 6, size=4096)
    r = rng.random(4096).astype(np.float32)
    t = rng.random(4096) < 0.01
    lrn.replay.append_actor_chunk(0, f, a, r, t)
    lrn.learn_steps(16)  # capture + run
    torch.cuda.synchronize()

    ms = timeit(lambda: lrn.graph.replay(), iters=50)
    print(f"learner graph replay ({lrn.steps_per_replay} grad steps): "
          f"{ms:.3f} ms -> {lrn.steps_per_replay / ms * 1000:.0f} grad-steps/s")

    ms1 = timeit(lambda: lrn._step_body(), iters=20)
    print(f"learner eager step: {ms1:.3f} ms")

    net = lrn.online
    actor = GpuActorLoop(cfg, net, n_envs=16, chunk_steps=4, use_graph=True)
    actor.run_round()
    torch.cuda.synchronize()
    ms2 = timeit(lambda: actor.graph.replay(), iters=50)
    steps = actor.chunk_steps * actor.E
    print(f"actor round replay ({actor.chunk_steps} steps x {actor.E} envs): "
          f"{ms2:.3f} ms -> {steps / ms2 * 1000:.0f} env-steps/s")

    # larger actor batch
    actor2 = GpuActorLoop(cfg, net, n_envs=64, chunk_steps=8, use_graph=True)
    actor2.run_round()
    torch.cuda.synchronize()
    ms3 = timeit(lambda: actor2.graph.replay(), iters=50)
    steps = actor2.chunk_steps * actor2.E
    print(f"actor round replay (8 x 64 envs): {ms3:.3f} ms -> "
          f"{steps / ms3 * 1000:.0f} env-steps/s")

    chunk_f = actor.chunk_frames
    chunk_m = actor.chunk_meta
    ms4 = timeit(lambda: lrn.replay.append_actor_chunk(0, chunk_f, meta=chunk_m),
                 iters=50)
    print(f"replay append (chunk {chunk_f.shape[0]}): {ms4:.3f} ms")

    ms5 = timeit(lambda: lrn.update_target(), iters=50)
    print(f"target update (flat copy): {ms5:.3f} ms")

    # forward-only pieces at learner shapes
    s = lrn.replay.out_states
    taus = lrn.taus_online
    with torch.no_grad():
        ms6 = timeit(lambda: net(s, cfg.num_tau_samples, taus=taus), iters=50)
    print(f"eager online fwd (B=32,N=8): {ms6:.3f} ms")
    with torch.no_grad():
        ms7 = timeit(lambda: net(s, cfg.num_quantile_samples,
                                 taus=lrn.taus_select), iters=50)
    print(f"eager online fwd (B=32,K=32): {ms7:.3f} ms")


if __name__ == "__main__":
    main()
