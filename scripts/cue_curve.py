#!/usr/bin/env python3
"""Learning-curve artifact: train the full fused stack on the cue task
(reward a deterministic function of the frame) and emit a reward-vs-steps
curve -- the rebuild's analog of the reference's learning-curve correctness
culture, on real hardware.

Usage (GPU box): python scripts/cue_curve.py [rounds] [out_csv]
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import time

import torch

from rainbowiqn_amd.config import Config
from rainbowiqn_amd.parallel.apex import ApexRunner
from rainbowiqn_amd.parallel.comm import Communicator


def main():
    rounds = int(sys.argv[1]) if len(sys.argv) > 1 else 2000
    out = sys.argv[2] if len(sys.argv) > 2 else "gpurun_out/cue_curve.csv"
    lr = float(sys.argv[3]) if len(sys.argv) > 3 else 1e-3
    cfg = Config(game="cue", device="cuda", dtype="bf16", seed=11,
                 lr=lr, multi_step=1, learn_start=4096,
                 memory_capacity=262144, learn_per_round=4,
                 target_update=8192)
    runner = ApexRunner(cfg, Communicator(backend="auto"), n_envs=256,
                        chunk_steps=8)
    assert runner.fast and runner.gactor.task == 1
    rows = []
    t0 = time.perf_counter()
    for rnd in range(rounds):
        runner.run_round()
        if rnd % 10 == 0:
            torch.cuda.synchronize()
            r = float(runner.gactor.chunk_meta[:, 1].mean())
            loss = runner.losses[-1] if runner.losses else float("nan")
            rows.append((rnd, runner.actor.total_steps, r, loss))
            if rnd % 200 == 0:
                print(f"round {rnd} steps {runner.actor.total_steps} "
                      f"avg_reward {r:+.3f} loss {loss:.4f}", flush=True)
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    os.makedirs(os.path.dirname(out) or ".", exist_ok=True)
    with open(out, "w") as f:
        f.write("round,env_steps,avg_reward,loss\n")
        for row in rows:
            f.write(",".join(str(x) for x in row) + "\n")
    early = sum(r for _, _, r, _ in rows[:5]) / 5
    late = sum(r for _, _, r, _ in rows[-5:]) / 5
    print(f"done: {rounds} rounds in {dt:.1f}s, "
          f"avg reward {early:+.3f} -> {late:+.3f} "
          f"(chance -0.5, optimal +1.0); curve -> {out}")


if __name__ == "__main__":
    main()
