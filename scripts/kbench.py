#!/usr/bin/env python3
"""Microbenchmark of the fused forward kernels at the shapes the Ape-X loop
uses (act: B=E envs x K taus; learn: B=32 x {K, Np}).

Usage (GPU box): python scripts/kbench.py [iters]
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import time

import torch

from rainbowiqn_amd.config import Config
from rainbowiqn_amd.models.model import RainbowIQN
from rainbowiqn_amd.ops import get_extension
from rainbowiqn_amd.runtime.fused_forward import FusedForward, noisy_pack


def timeit(fn, iters):
    for _ in range(10):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6  # us


def main():
    iters = int(sys.argv[1]) if len(sys.argv) > 1 else 200
    ext = get_extension()
    cfg = Config(dtype="bf16", device="cuda")
    A = 6
    net = RainbowIQN(cfg, A).cuda().to(torch.bfloat16).train()
    fw = FusedForward(net, "cuda")
    pack = noisy_pack(net, "cuda")
    rng = torch.tensor([123, 0], dtype=torch.int64, device="cuda")

    print(f"{'kernel':<28} {'shape':<18} {'us':>9}")
    for B, nq, tag in [(1024, 32, "act E=1024"), (256, 32, "act E=256"),
                       (64, 32, "act E=64"),
                       (32, 32, "learn sel"), (32, 8, "learn tgt")]:
        R = B * nq
        x = torch.rand(B, 4, 84, 84, device="cuda").to(torch.bfloat16)
        taus = torch.rand(R, device="cuda")
        feat, h, hh = fw._bufs(B, nq)
        q = torch.empty(R, A, device="cuda")
        th = torch.empty(B, nq, device="cuda")
        acts = torch.zeros(B, dtype=torch.int64, device="cuda")

        us = timeit(lambda: ext.conv_encode(x, fw.w1, fw.b1, fw.w2, fw.b2,
                                            fw.w3p, fw.b3, feat), iters)
        print(f"{'conv_encode':<28} {f'B={B}':<18} {us:9.2f}")
        if B >= 256:  # actor path: conv1 gathers from the u8 frame ring
            ring = torch.randint(0, 255, (4, B, 7056), dtype=torch.uint8,
                                 device="cuda")
            est = torch.zeros(B, 4, dtype=torch.int32, device="cuda")
            est[:, 1] = 100  # steps past history: all frames valid
            us = timeit(lambda: ext.conv_encode_gather(
                ring, est.view(-1), 4, 3, fw.w1, fw.b1, fw.w2, fw.b2,
                fw.w3p, fw.b3, feat), iters)
            print(f"{'conv_encode_gather':<28} {f'E={B}':<18} {us:9.2f}")
        us = timeit(lambda: ext.tau_embed(taus, fw.wiqn, fw.biqn, feat, h,
                                          nq, direct=0), iters)
        print(f"{'tau_embed staged':<28} {f'R={R}':<18} {us:9.2f}")
        us = timeit(lambda: ext.tau_embed(taus, fw.wiqn, fw.biqn, feat, h, nq,
                                          direct=True), iters)
        print(f"{'tau_embed direct':<28} {f'R={R}':<18} {us:9.2f}")
        gf = R * 1024 * 3136 * 2 / 1e9
        for var, name in ([(1, "heads_gemm 256x128"), (2, "heads_gemm 256x256"),
                           (3, "heads_gemm 320x256"), (4, "heads_gemm 256rb")]
                          if R >= 8192 else [(1, "heads_gemm")]):
            us = timeit(lambda: ext.heads_gemm(h, fw.wcat, fw.bcat, hh,
                                               variant=var), iters)
            print(f"{name:<28} {f'R={R}':<18} {us:9.2f}  "
                  f"({gf / us * 1e6 / 1e3:.2f} TF)")
        us = timeit(lambda: ext.heads_finish(hh, fw.wz, fw.bz, A, nq, 1,
                                             torch.zeros(1, device='cuda'),
                                             acts), iters)
        print(f"{'heads_finish act':<28} {f'{tag}':<18} {us:9.2f}")
        us = timeit(lambda: ext.heads_finish(hh, fw.wz, fw.bz, A, nq, 0, q,
                                             acts), iters)
        print(f"{'heads_finish q':<28} {f'R={R}':<18} {us:9.2f}")
        us = timeit(lambda: ext.heads_finish(hh, fw.wz, fw.bz, A, nq, 2,
                                             th.view(-1), acts), iters)
        print(f"{'heads_finish theta':<28} {f'R={R}':<18} {us:9.2f}")
        print()

    # fp8 MX heads path
    fw8 = FusedForward(RainbowIQN(cfg, A).cuda().to(torch.bfloat16).train(),
                       "cuda", fp8_heads=True)
    for B, nq in [(1024, 32), (256, 32), (32, 32)]:
        R = B * nq
        bufs8 = fw8._bufs(B, nq)
        feat8, h8, hh8, hq, hsc = bufs8
        taus8 = torch.rand(R, device="cuda")
        us = timeit(lambda: ext.quant_fp8(h8, hq, hsc), iters)
        print(f"{'quant_fp8':<28} {f'R={R}':<18} {us:9.2f}")
        us = timeit(lambda: ext.tau_embed_fp8(taus8, fw8.wiqn, fw8.biqn,
                                              feat8, hq, hsc, nq), iters)
        print(f"{'tau_embed_fp8 (fused q)':<28} {f'R={R}':<18} {us:9.2f}")
        gf = R * 1024 * 3200 * 2 / 1e9
        for var, name in [(0, "fp8 gemm 128x128"), (1, "fp8 gemm 256x128p"),
                          (2, "fp8 gemm 256x256"), (3, "fp8 gemm 128x256")]:
            us = timeit(lambda: ext.heads_gemm_fp8(hq, hsc, fw8.wq, fw8.wsc,
                                                   fw8.bcat, hh8,
                                                   variant=var), iters)
            print(f"{name:<28} {f'R={R}':<18} {us:9.2f}  "
                  f"({gf / us * 1e6 / 1e3:.2f} TF)")

    us = timeit(lambda: ext.noisy_material4(pack, rng), iters)
    print(f"{'noisy_material4':<28} {'4 layers':<18} {us:9.2f}")
    us = timeit(lambda: fw.pack_weights(), iters)
    print(f"{'pack_fwd_frag':<28} {'':<18} {us:9.2f}")

    # replay append at bench chunk size
    from rainbowiqn_amd.replay.gpu_replay import GpuReplay

    rep = GpuReplay(cfg, n_rings=1, capacity=200_000, device="cuda",
                    out_dtype=torch.bfloat16)
    cf = torch.randint(0, 255, (2048, 84, 84), dtype=torch.uint8, device="cuda")
    cm = torch.rand(2048, 3, device="cuda")
    us = timeit(lambda: rep.append_actor_chunk(0, cf, meta=cm), 50)
    print(f"{'replay_append':<28} {'C=2048':<18} {us:9.2f}")
    us = timeit(lambda: rep.sample(), iters)
    print(f"{'replay_sample':<28} {'B=32':<18} {us:9.2f}")


if __name__ == "__main__":
    main()
