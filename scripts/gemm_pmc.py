#!/usr/bin/env python3
"""Isolated heads_gemm_big loop for rocprofv3 --pmc (stall diagnosis).

Usage on the GPU box (counters need their own run, kernel-trace only):
  cd /tmp && export TMPDIR=/tmp
  rocprofv3 --pmc SQ_VALU_MFMA_BUSY_CYCLES SQ_WAVE_CYCLES SQ_WAIT_ANY \
      -d out -- python /root/repo/scripts/gemm_pmc.py
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from rainbowiqn_amd.config import Config
from rainbowiqn_amd.models.model import RainbowIQN
from rainbowiqn_amd.ops import get_extension
from rainbowiqn_amd.runtime.fused_forward import FusedForward

ext = get_extension()
cfg = Config(dtype="bf16", device="cuda")
net = RainbowIQN(cfg, 6).cuda().to(torch.bfloat16).train()
fw = FusedForward(net, "cuda")
B, nq = 256, 32  # R=8192 -> the 128x128 big-tile kernel
R = B * nq
feat, h, hh = fw._bufs(B, nq)[:3]
h.normal_()
iters = int(sys.argv[1]) if len(sys.argv) > 1 else 50
for _ in range(iters):
    ext.heads_gemm(h, fw.wcat, fw.bcat, hh)
torch.cuda.synchronize()
print("done", iters)
