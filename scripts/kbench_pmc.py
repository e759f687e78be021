#!/usr/bin/env python3
"""Minimal kernel loop for PMC counter collection (rocprofv3 --pmc)."""
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

# backward kernels at learn shape
B2, N2 = 32, 8
R2 = B2 * N2
s2 = torch.rand(2 * B2, 4, 84, 84, device="cuda").to(torch.bfloat16)
feat2 = torch.zeros(2 * B2, 3136, dtype=torch.bfloat16, device="cuda")
c1g = torch.zeros(2 * B2, 12800, dtype=torch.bfloat16, device="cuda")
c2g = torch.zeros(2 * B2, 5184, dtype=torch.bfloat16, device="cuda")
dfeat = torch.rand(B2, 3136, device="cuda")
w3tp = torch.zeros(64, 1024, dtype=torch.bfloat16, device="cuda")
w2tp = torch.zeros(32, 1024, dtype=torch.bfloat16, device="cuda")
ext.pack_bwd_weights(fw.w3raw, fw.w2raw, w3tp, w2tp)
dpre3g = torch.zeros(64, B2 * 49, dtype=torch.bfloat16, device="cuda")
dc2g = torch.zeros(64, B2 * 81, dtype=torch.bfloat16, device="cuda")
dc1g = torch.zeros(32, B2 * 400, dtype=torch.bfloat16, device="cuda")
db1s = torch.zeros(32, device="cuda"); db2s = torch.zeros(64, device="cuda")
db3s = torch.zeros(64, device="cuda")
ext.conv_encode(s2, fw.w1, fw.b1, fw.w2, fw.b2, fw.w3p, fw.b3, feat2, c1g, c2g)
for _ in range(30):
    ext.conv_bwd_data(dfeat, feat2[:B2].contiguous(), c1g[:B2].contiguous(),
                      c2g[:B2].contiguous(), w3tp, w2tp, dpre3g, dc2g, dc1g,
                      db1s, db2s, db3s)
torch.cuda.synchronize()
print("done2")
