#!/usr/bin/env python3
"""Round-2 PMC loop: the four actor-shape kernels under diagnosis
(R=32768 = E1024 x K32) -- bf16 heads GEMM, fp8 GEMM variants, tau_embed
direct, fused tau_embed_fp8.  Run under rocprofv3 --pmc (counters need
their own pass; kernel names in the csv separate the kernels).

  cd /tmp && export TMPDIR=/tmp
  rocprofv3 --pmc SQ_WAVE_CYCLES SQ_WAIT_ANY SQ_WAIT_INST_ANY \
      SQ_VALU_MFMA_BUSY_CYCLES SQ_ACTIVE_INST_ANY \
      -d outA -- python /root/repo/scripts/perf2_pmc.py 20
  rocprofv3 --pmc TCC_FETCH_SIZE -d outB -- python ... (second pass)
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from rainbowiqn_amd.config import Config
from rainbowiqn_amd.models.model import RainbowIQN
from rainbowiqn_amd.ops import get_extension
from rainbowiqn_amd.runtime.fused_forward import FusedForward

iters = int(sys.argv[1]) if len(sys.argv) > 1 else 20
ext = get_extension()
cfg = Config(dtype="bf16", device="cuda")
net = RainbowIQN(cfg, 6).cuda().to(torch.bfloat16).train()
fw = FusedForward(net, "cuda")
net8 = RainbowIQN(cfg, 6).cuda().to(torch.bfloat16).train()
fw8 = FusedForward(net8, "cuda", fp8_heads=True)
B, nq = 1024, 32
R = B * nq
feat, h, hh = fw._bufs(B, nq)[:3]
feat.normal_()
h.normal_()
taus = torch.rand(R, device="cuda")
f8, h8, hh8, hq, hsc = fw8._bufs(B, nq)
f8.normal_()
h8.normal_()
ext.quant_fp8(h8, hq, hsc)
torch.cuda.synchronize()

for _ in range(iters):
    ext.heads_gemm(h, fw.wcat, fw.bcat, hh)           # bf16 256x128 NBUF3
for _ in range(iters):
    ext.heads_gemm_fp8(hq, hsc, fw8.wq, fw8.wsc, fw8.bcat, hh8, variant=0)
for _ in range(iters):
    ext.heads_gemm_fp8(hq, hsc, fw8.wq, fw8.wsc, fw8.bcat, hh8, variant=1)
for _ in range(iters):
    ext.tau_embed(taus, fw.wiqn, fw.biqn, feat, h, nq, direct=1)
for _ in range(iters):
    ext.tau_embed_fp8(taus, fw8.wiqn, fw8.biqn, f8, hq, hsc, nq)
# actor-path remainder: conv (both variants), finish, env step
x = torch.rand(B, 4, 84, 84, device="cuda").to(torch.bfloat16)
ring = torch.randint(0, 255, (4, B, 7056), dtype=torch.uint8, device="cuda")
est = torch.zeros(B, 4, dtype=torch.int32, device="cuda")
est[:, 1] = 100
rng = torch.tensor([7, 0], dtype=torch.int64, device="cuda")
acts = torch.zeros(B, dtype=torch.int64, device="cuda")
cf = torch.zeros(B * 8, 84, 84, dtype=torch.uint8, device="cuda")
cm = torch.zeros(B * 8, 3, device="cuda")
for _ in range(iters):
    ext.conv_encode(x, fw.w1, fw.b1, fw.w2, fw.b2, fw.w3p, fw.b3,
                    fw._bufs(B, nq)[0])
for _ in range(iters):
    ext.conv_encode_gather(ring, est.view(-1), 4, 3, fw.w1, fw.b1, fw.w2,
                           fw.b2, fw.w3p, fw.b3, fw._bufs(B, nq)[0])
for _ in range(iters):
    ext.heads_finish(hh, fw.wz, fw.bz, 6, nq, 1,
                     torch.zeros(1, device="cuda"), acts)
for _ in range(iters):
    ext.env_step(acts, est.view(-1), rng, ring.view(4, -1), 4, 0, 1.0,
                 4.0 / 512.0, 108000, cf, cm, 0, 0)
torch.cuda.synchronize()
print("done", iters)
