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

Replay config is the REFERENCE'S Atari config (SURVEY.md 3.4 / config.py
defaults): memory_capacity 1e6, prefilled to learn_start 80e3 before the
timed region, so every timed grad step samples a full-depth (2^20-leaf)
sum-tree at the reference's fill point; the ring keeps filling during the
run, exactly as in training.

One bench step = one Ape-X round (actor chunk -> RCCL gather -> learner grad
steps -> RCCL weight broadcast).  `value` = aggregate actor env-frames/sec in
raw Atari frames (env steps x action-repeat 4 -- the convention by which the
reference's 50e6 agent steps are its "200M frames").  Learner grad-steps/sec
is reported alongside in the same JSON line.

--learner-only measures the 7+1 split's learner rank in isolation: grad
steps with the whole GPU to itself (no actor work), the real situation of
BASELINE config 4's rank 0.  The JSON `value` is then grad-steps/sec.

Launch: python bench.py --gpus N --steps K --warmup W
  (N>1 via torchrun --nnodes=1 --nproc-per-node N --master-addr 127.0.0.1)
"""

from __future__ import annotations

import argparse
import json
import os
import time

import torch

ACTION_REPEAT = 4  # raw Atari frames per env step (reference env semantics)


def prefill_ring(replay, ring_id: int, n: int, n_actions: int, device,
                 seed: int = 0, chunk: int = 16384) -> None:
    """Fill one replay ring with n synthetic transitions of the benchmark
    shape, generated on-device (GPU replay) or via numpy (CPU replay)."""
    dev = torch.device(device)
    g = torch.Generator(device="cpu").manual_seed(seed)
    cpu_rings = hasattr(replay, "rings")  # CPU ShardedReplay signature
    done = 0
    while done < n:
        c = min(chunk, n - done)
        frames = torch.randint(0, 256, (c, 84, 84), dtype=torch.uint8,
                               generator=g)
        actions = torch.randint(0, n_actions, (c,), generator=g)
        rewards = torch.rand(c, generator=g)
        dones = torch.rand(c, generator=g) < 0.01
        if cpu_rings:
            replay.append_actor_chunk(
                ring_id, frames.numpy(), actions.numpy(),
                rewards.numpy(), dones.numpy())
        else:
            meta = torch.stack([actions.float(), rewards,
                                dones.float()], dim=1)
            replay.append_actor_chunk(ring_id, frames.to(dev),
                                      meta=meta.to(dev))
        done += c


def build_runner(cfg, comm, n_envs, chunk_steps):
    from rainbowiqn_amd.env import n_actions_for
    from rainbowiqn_amd.parallel.apex import ApexRunner

    runner = ApexRunner(cfg, comm, n_envs=n_envs, chunk_steps=chunk_steps)
    # Pre-fill the learner replay to learn_start with synthetic transitions of
    # the benchmark shape so grad steps run from the first timed round (the
    # timed region must contain full learner work; data is synthetic either
    # way).  Reference fill point: learn_start transitions in a 1M ring.
    if comm.is_learner:
        n_act = n_actions_for(cfg.game)
        rings = getattr(runner.replay, "n_rings",
                        getattr(runner.replay, "n_actors", comm.n_actors))
        per_ring = cfg.learn_start // rings + 256
        for a in range(rings):
            prefill_ring(runner.replay, a, per_ring, n_act, cfg.device,
                         seed=a)
    return runner


def emit(comm, args, cfg, elapsed_max, grad_steps, env_steps, extra_cfg):
    n = comm.world_size
    frames_per_sec = env_steps * ACTION_REPEAT / elapsed_max
    grad_per_sec = grad_steps / elapsed_max
    learner_only = env_steps == 0
    if comm.rank != 0:
        return
    out = {
        "metric": ("learner_grad_steps_per_sec (learner-only mode)"
                   if learner_only else
                   "actor_env_frames_per_sec (Ape-X; learner grad-steps/sec"
                   " alongside)"),
        "value": round(grad_per_sec if learner_only else frames_per_sec,
                       2 if learner_only else 1),
        "unit": "grad-steps/s" if learner_only else "frames/s",
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
            "memory_capacity": cfg.memory_capacity,
            "learn_start": cfg.learn_start,
            "N_taus": cfg.num_tau_samples,
            "Np_taus": cfg.num_tau_prime_samples,
            "K_taus": cfg.num_quantile_samples,
            **extra_cfg,
        },
    }
    print(json.dumps(out))


def run_learner_only(args, cfg, comm, on_gpu):
    """Learner rank in isolation (whole GPU): grad-steps/s."""
    from rainbowiqn_amd.env import n_actions_for

    n_act = n_actions_for(cfg.game)
    if on_gpu:
        from rainbowiqn_amd.runtime.gpu_learner import GpuLearner

        learner = GpuLearner(cfg, n_act, n_rings=1, device=cfg.device,
                             use_graph=cfg.hipgraph)
        replay = learner.replay
        prefill_ring(replay, 0, cfg.learn_start + 256, n_act, cfg.device)

        def one_round():
            learner.learn_steps(cfg.learn_per_round)

        def grad_count():
            return learner.grad_steps
    else:
        from rainbowiqn_amd.agent import Agent
        from rainbowiqn_amd.parallel.apex import ShardedReplay

        agent = Agent(cfg, n_act, cfg.device)
        replay = ShardedReplay(cfg, 1)
        prefill_ring(replay, 0, cfg.learn_start + 256, n_act, "cpu")
        state = {"g": 0}

        def one_round():
            for _ in range(cfg.learn_per_round):
                agent.learn(replay)
                state["g"] += 1

        def grad_count():
            return state["g"]

    def sync():
        if on_gpu:
            torch.cuda.synchronize()

    for _ in range(args.warmup):
        one_round()
    sync()
    g0 = grad_count()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        one_round()
    sync()
    elapsed = time.perf_counter() - t0
    emit(comm, args, cfg, elapsed, float(grad_count() - g0), 0.0,
         {"parallelism": "learner-only (1 GPU, no actor work)",
          "learn_per_round": cfg.learn_per_round})


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=200)
    p.add_argument("--warmup", type=int, default=10)
    p.add_argument("--n-envs", type=int, default=1024,
                   help="env streams per actor loop")
    p.add_argument("--nb-actor", type=int, default=1,
                   help="concurrent local actor loops (colocated overlap "
                        "runs each on its own stream; total env streams = "
                        "nb_actor x n_envs)")
    p.add_argument("--chunk-steps", type=int, default=8, help="env steps per round")
    p.add_argument("--learn-per-round", type=int, default=8)
    p.add_argument("--device", type=str, default=None)
    p.add_argument("--memory-capacity", type=int, default=None,
                   help="override replay capacity (default: reference 1e6 on "
                        "GPU, small on the CPU plumbing path)")
    p.add_argument("--learn-start", type=int, default=None,
                   help="override prefill (default: reference 80e3 on GPU)")
    p.add_argument("--learner-only", action="store_true",
                   help="measure the learner rank alone on a whole GPU "
                        "(grad-steps/s; the 7+1 split's rank-0 situation)")
    p.add_argument("--fp8-heads", action="store_true",
                   help="MX-fp8 NoisyLinear heads GEMM (BASELINE config 5)")
    p.add_argument("--no-hipgraph", dest="hipgraph", action="store_false")
    p.add_argument("--no-fused-forward", dest="fused_forward",
                   action="store_false")
    p.add_argument("--no-fused-backward", dest="fused_backward",
                   action="store_false")
    args = p.parse_args()

    from rainbowiqn_amd.config import Config
    from rainbowiqn_amd.parallel.comm import Communicator

    on_gpu = torch.cuda.is_available() and args.device != "cpu"
    if on_gpu:
        torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", 0)))

    # GPU: the reference's Atari replay config (1M ring, 80k fill point --
    # config.py defaults).  CPU: the plumbing-test path scales the replay
    # down so the contract test stays fast; the GPU numbers are the metric.
    mem_cap = args.memory_capacity or (Config.memory_capacity if on_gpu
                                       else 20_000)
    learn_start = args.learn_start or (Config.learn_start if on_gpu else 512)
    cfg = Config(
        game="pong",
        learn_start=learn_start,
        memory_capacity=mem_cap,
        learn_per_round=args.learn_per_round,
        batch_size=32,
        dtype="bf16" if on_gpu else "fp32",
        device="cuda" if on_gpu else "cpu",
        seed=int(os.environ.get("RANK", 0)) + 1,
        nb_actor=args.nb_actor,
        fp8_heads=args.fp8_heads,
        hipgraph=args.hipgraph,
        fused_forward=args.fused_forward,
        fused_backward=args.fused_backward,
    )
    comm = Communicator(backend="auto")
    torch.manual_seed(cfg.seed)

    if args.learner_only:
        run_learner_only(args, cfg, comm, on_gpu)
        comm.close()
        return

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
    emit(comm, args, cfg, elapsed_max, grad_steps, env_steps, {
        "parallelism": (
            "apex-colocated(actor+learner, 1 GPU)" if n == 1
            else f"apex(1 learner + {n-1} actor GPUs, RCCL/xGMI)"
        ),
        "n_envs_per_actor": args.n_envs,
        "chunk_steps_per_round": args.chunk_steps,
        "learn_per_round": args.learn_per_round,
        "frame_accounting": "raw Atari frames = env steps x 4 (action repeat)",
    })
    comm.close()


if __name__ == "__main__":
    main()
