#!/usr/bin/env python3
"""Minimal kernel loop for PMC counter collection (rocprofv3 --pmc)."""
import torch
from rainbowiqn_amd.config import Config
from rainbowiqn_amd.models.model import RainbowIQN
from rainbowiqn_amd.ops import get_extension
from rainbowiqn_amd.runtime.fused_forward import FusedForward

ext = get_extension()
cfg = Config(dtype="bf16", device="cuda")
net = RainbowIQN(cfg, 6).cuda().to(torch.bfloat16).train()
fw = FusedForward(net, "cuda")
B, nq = 256, 32
R = B * nq
x = torch.rand(B, 4, 84, 84, device="cuda").to(torch.bfloat16)
taus = torch.rand(R, device="cuda")
feat, h, hh = fw._bufs(B, nq)
for _ in range(30):
    ext.conv_encode(x, fw.w1, fw.b1, fw.w2, fw.b2, fw.w3p, fw.b3, feat)
for _ in range(30):
    ext.tau_embed(taus, fw.wiqn, fw.biqn, feat, h, nq)
for _ in range(30):
    ext.heads_gemm(h, fw.wcat, fw.bcat, hh)
torch.cuda.synchronize()
print("done")
