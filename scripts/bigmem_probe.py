#!/usr/bin/env python3
"""HBM3E sizing demonstration: an HBM-resident prioritized replay at a
capacity sized for the 288 GB device (BASELINE config 5 territory), with
live append/sample/priority traffic and invariant checks.

Usage (GPU box): python scripts/bigmem_probe.py [capacity_millions]
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import time

import torch

from rainbowiqn_amd.config import Config
from rainbowiqn_amd.replay.gpu_replay import GpuReplay


def main():
    cap_m = int(sys.argv[1]) if len(sys.argv) > 1 else 30
    cap = cap_m * 1_000_000
    cfg = Config(device="cuda", dtype="bf16", batch_size=32)
    t0 = time.perf_counter()
    rep = GpuReplay(cfg, n_rings=8, capacity=cap, device="cuda",
                    out_dtype=torch.bfloat16)
    torch.cuda.synchronize()
    alloc = torch.cuda.memory_allocated() / 2**30
    print(f"alloc {alloc:.1f} GiB for {cap_m}M transitions "
          f"(tree {rep.tree_size} leaves) in "
          f"{time.perf_counter() - t0:.1f}s", flush=True)
    g = torch.Generator(device="cpu").manual_seed(0)
    f = torch.randint(0, 256, (65536, 84, 84), dtype=torch.uint8,
                      generator=g).cuda()
    m = torch.rand(65536, 3, generator=g).cuda()
    t0 = time.perf_counter()
    appended = 0
    samples = 0
    for it in range(120):
        rep.append_actor_chunk(it % 8, f, meta=m)
        appended += 65536
        if len(rep) > 200_000:
            idx, s, a, r, s2, nt, w = rep.sample()
            rep.update_priorities(idx, torch.rand(32, device="cuda") + 0.05)
            samples += 1
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    assert torch.isfinite(w).all() and torch.isfinite(r).all()
    rep.check_invariants(atol=5e-2)
    print(f"traffic OK: {appended / 1e6:.1f}M appends + {samples} "
          f"sample/update rounds in {dt:.1f}s "
          f"({appended / dt / 1e6:.2f}M transitions/s); "
          f"tree invariants hold at {len(rep) / 1e6:.1f}M resident",
          flush=True)


if __name__ == "__main__":
    main()
