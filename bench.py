#!/usr/bin/env python3
"""Flagship benchmark: Ape-X Rainbow-IQN throughput on MI355X.

Measures the metric named by BASELINE.json -- learner grad-steps/sec and actor
env-frames/sec on Atari Pong (synthetic 84x84x4 frames, random-init weights)
-- for N GPUs of one node, one process per GPU over RCCL/xGMI.

Topology: N=1 colocates actor+learner on one GPU
(BASELINE config 2/3); N>1 runs rank 0 as the learner and ranks 1..N-1 as
actors (config 4: Ape-X split over xGMI).  Scaling is weak: per-GPU work per
bench step is fixed (an actor rank always runs chunk_steps x n_envs env steps
per step; the learner always runs learn_per_round grad steps per step).

One bench step = one Ape-X round (actor chunk -> RCCL gather -> learner grad
steps -> RCCL weight broadcast).  `value` = aggregate actor env-frames/sec in
raw Atari frames (env steps x action-repeat 4 -- the convention by which the
reference's 50e6 agent steps are its "200M frames").  Learner grad-steps/sec
is reported alongside in the same JSON line.

Launch: python bench.py --gpus N --steps K --warmup W
  (N>1 via torchrun --nnodes=1 --nproc-per-node N --master-addr 127.0.0.1)
"""

from __future__ import annotations

import argparse
import json
import os
import time

import numpy as np
import torch

from rainbowiqn_amd.config import Config
from rainbowiqn_amd.parallel.apex import ApexRunner
from rainbowiqn_amd.parallel.comm import Communicator

ACTION_REPEAT = 4  # raw Atari frames per env step (reference env semantics)


def build_runner(cfg, comm, n_envs, chunk_steps):
    runner = ApexRunner(cfg, comm, n_envs=n_envs, chunk_steps=chunk_steps)
    # Pre-fill the learner replay to learn_start with synthetic transitions of
    # the benchmark shape so grad steps run from the first timed round (the
    # timed region must contain full learner work; data is synthetic either way).
    if comm.is_learner:
        from rainbowiqn_amd.env import n_actions_for

        n_act = n_actions_for(cfg.game)
        rng = np.random.default_rng(0)
        per_ring = cfg.learn_start // comm.n_actors + 256
        for a in range(comm.n_actors):
            frames = rng.integers(0, 255, size=(per_ring, 84, 84), dtype=np.uint8)
            actions = rng.integers(0, n_act, size=per_ring)
            rewards = rng.random(per_ring).astype(np.float32)
            dones = rng.random(per_ring) < 0.01
            runner.replay.append_actor_chunk(a, frames, actions, rewards, dones)
    return runner


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=20)
    p.add_argument("--warmup", type=int, default=5)
    p.add_argument("--n-envs", type=int, default=1024, help="env streams per actor rank")
    p.add_argument("--chunk-steps", type=int, default=8, help="env steps per round")
    p.add_argument("--learn-per-round", type=int, default=8)
    p.add_argument("--device", type=str, default=None)
    p.add_argument("--fp8-heads", action="store_true",
                   help="MX-fp8 NoisyLinear heads GEMM (BASELINE config 5)")
    p.add_argument("--no-hipgraph", dest="hipgraph", action="store_false")
    p.add_argument("--no-fused-forward", dest="fused_forward",
                   action="store_false")
    p.add_argument("--no-fused-backward", dest="fused_backward",
                   action="store_false")
    args = p.parse_args()

    on_gpu = torch.cuda.is_available() and args.device != "cpu"
    if on_gpu:
        torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", 0)))

    cfg = Config(
        game="pong",
        learn_start=2048,
        memory_capacity=200_000,
        learn_per_round=args.learn_per_round,
        batch_size=32,
        dtype="bf16" if on_gpu else "fp32",
        device="cuda" if on_gpu else "cpu",
        seed=int(os.environ.get("RANK", 0)) + 1,
        fp8_heads=args.fp8_heads,
        hipgraph=args.hipgraph,
        fused_forward=args.fused_forward,
        fused_backward=args.fused_backward,
    )
    comm = Communicator(backend="auto")
    torch.manual_seed(cfg.seed)

    runner = build_runner(cfg, comm, args.n_envs, args.chunk_steps)

    def sync():
        if on_gpu:
            torch.cuda.synchronize()
        comm.barrier()

    # -- warmup ------------------------------------------------------------
    for _ in range(args.warmup):
        runner.run_round()
    sync()

    # -- timed region ------------------------------------------------------
    grad0, act0 = runner.grad_steps, (runner.actor.total_steps if runner.actor else 0)
    sync()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        runner.run_round()
    sync()
    elapsed = time.perf_counter() - t0

    # max elapsed over ranks; sum work over ranks
    stats = torch.tensor(
        [elapsed, float(runner.grad_steps - grad0),
         float((runner.actor.total_steps if runner.actor else 0) - act0)]
    )
    if comm.initialized:
        el = stats[0:1].clone()
        torch.distributed.all_reduce(el, op=torch.distributed.ReduceOp.MAX)
        work = stats[1:3].clone()
        torch.distributed.all_reduce(work, op=torch.distributed.ReduceOp.SUM)
        elapsed_max, grad_steps, env_steps = float(el[0]), float(work[0]), float(work[1])
    else:
        elapsed_max, grad_steps, env_steps = elapsed, float(stats[1]), float(stats[2])

    n = comm.world_size
    frames_per_sec = env_steps * ACTION_REPEAT / elapsed_max
    grad_per_sec = grad_steps / elapsed_max
    if comm.rank == 0:
        out = {
            "metric": "actor_env_frames_per_sec (Ape-X; learner grad-steps/sec alongside)",
            "value": round(frames_per_sec, 1),
            "unit": "frames/s",
            "n_gpus": n,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed_max * 1000 / args.steps, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,  # reference publishes no throughput numbers
            "dtype": "bf16+fp8-heads" if cfg.fp8_heads else cfg.dtype,
            "data": "synthetic",
            "learner_grad_steps_per_sec": round(grad_per_sec, 2),
            "env_steps_per_sec": round(env_steps / elapsed_max, 1),
            "config": {
                "model": "RainbowIQN NatureCNN+IQN (~6.8M params), Pong (6 actions)",
                "global_batch": cfg.batch_size,
                "seq_len": cfg.history_length,
                "parallelism": (
                    "apex-colocated(actor+learner, 1 GPU)" if n == 1
                    else f"apex(1 learner + {n-1} actor GPUs, RCCL/xGMI)"
                ),
                "n_envs_per_actor": args.n_envs,
                "chunk_steps_per_round": args.chunk_steps,
                "learn_per_round": args.learn_per_round,
                "N_taus": cfg.num_tau_samples,
                "Np_taus": cfg.num_tau_prime_samples,
                "K_taus": cfg.num_quantile_samples,
                "frame_accounting": "raw Atari frames = env steps x 4 (action repeat)",
            },
        }
        print(json.dumps(out))
    comm.close()


if __name__ == "__main__":
    main()
