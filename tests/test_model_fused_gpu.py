"""Parity tests: fused gfx950 forward kernels (ops/csrc/model_fwd.hip via
runtime/fused_forward.FusedForward) against the eager RainbowIQN module.

Reference = the fp32 eager module with IDENTICAL weights/noise; the bf16
eager module brackets the tolerance (fused and eager-bf16 should both sit
within bf16 rounding of the fp32 truth).
"""

import pytest
import torch

from rainbowiqn_amd.config import Config

pytestmark = pytest.mark.gpu


def make_nets(A=6, seed=0):
    from rainbowiqn_amd.models.model import RainbowIQN

    torch.manual_seed(seed)
    cfg = Config(dtype="bf16", device="cuda")
    net32 = RainbowIQN(cfg, A).cuda().train()
    net16 = RainbowIQN(cfg, A).cuda().train()
    net16.load_state_dict(net32.state_dict())
    net16 = net16.to(torch.bfloat16)
    for layer in (net16.fc_h_v, net16.fc_h_a, net16.fc_z_v, net16.fc_z_a):
        layer.enable_fused_noise()
    # identical noise in all three: copy eps from net32, refresh noisy bufs
    for l32, l16 in [(net32.fc_h_v, net16.fc_h_v), (net32.fc_h_a, net16.fc_h_a),
                     (net32.fc_z_v, net16.fc_z_v), (net32.fc_z_a, net16.fc_z_a)]:
        l16.weight_epsilon.copy_(l32.weight_epsilon.to(torch.bfloat16))
        l16.bias_epsilon.copy_(l32.bias_epsilon.to(torch.bfloat16))
        with torch.no_grad():
            l16.weight_noisy.copy_(l16.weight_mu + l16.weight_sigma * l16.weight_epsilon)
            l16.bias_noisy.copy_(l16.bias_mu + l16.bias_sigma * l16.bias_epsilon)
    return cfg, net32, net16


def fused_of(net16):
    from rainbowiqn_amd.runtime.fused_forward import FusedForward

    fw = FusedForward(net16, "cuda")
    # FusedForward re-points weight_noisy to concat storage, preserving values
    return fw


def test_conv_encode_parity():
    _, net32, net16 = make_nets()
    fw = fused_of(net16)
    x = torch.rand(32, 4, 84, 84, device="cuda")
    ref = net32.encode(x)  # fp32 truth
    got = fw.encode(x.to(torch.bfloat16)).float()
    torch.cuda.synchronize()
    err = (got - ref).abs().max().item()
    scale = ref.abs().max().item()
    assert err < 0.03 * max(scale, 1.0), f"conv encode err {err} scale {scale}"
    # bracket: eager bf16 should be comparably far from fp32
    eag16 = net16.encode(x.to(torch.bfloat16)).float()
    err16 = (eag16 - ref).abs().max().item()
    assert err < max(3 * err16, 0.02 * max(scale, 1.0)), (err, err16)


def test_conv_encode_gather_matches_staged_stack():
    """Fused frame-stack gather (conv1 reads the u8 ring directly) produces
    bit-identical features to stack_gather -> conv_encode, including the
    pre-episode zeroing of short episodes."""
    from rainbowiqn_amd.ops import get_extension

    ext = get_extension()
    _, _, net16 = make_nets(seed=5)
    fw = fused_of(net16)
    E, h = 64, 4
    torch.manual_seed(7)
    ring = torch.randint(0, 256, (h, E, 7056), dtype=torch.uint8,
                         device="cuda")
    env_state = torch.zeros(E, 4, dtype=torch.int32, device="cuda")
    # mix of fresh episodes (steps 0..2 -> zeroed history) and running ones
    env_state[:, 1] = torch.tensor([i % 6 for i in range(E)],
                                   dtype=torch.int32, device="cuda")
    for ring_head in range(h):
        stack = torch.empty(E, h, 84, 84, dtype=torch.bfloat16,
                            device="cuda")
        ext.stack_gather(ring.view(h, -1), env_state.view(-1), h, ring_head,
                         E, stack)
        ref = fw.encode(stack).clone()
        feat = torch.empty(E, 3136, dtype=torch.bfloat16, device="cuda")
        ext.conv_encode_gather(ring, env_state.view(-1), h, ring_head,
                               fw.w1, fw.b1, fw.w2, fw.b2, fw.w3p, fw.b3,
                               feat)
        torch.cuda.synchronize()
        assert torch.equal(feat, ref), ring_head


def test_tau_embed_direct_matches_staged():
    """tau_embed DIRECT=1 (accumulator stores, no LDS staging) is bit-exact
    vs the staged variant on h, phi and cosb -- same math, different path."""
    from rainbowiqn_amd.ops import get_extension

    ext = get_extension()
    _, _, net16 = make_nets()
    fw = fused_of(net16)
    for B, nq in [(32, 8), (1024, 8), (255, 32)]:  # 255: ragged tail rows
        R = B * nq
        x = torch.rand(B, 4, 84, 84, device="cuda").to(torch.bfloat16)
        feat = fw.encode(x).clone()
        taus = torch.rand(R, device="cuda")
        h1 = torch.empty(R, 3136, dtype=torch.bfloat16, device="cuda")
        h2 = torch.empty_like(h1)
        p1 = torch.empty_like(h1)
        p2 = torch.empty_like(h1)
        c1 = torch.empty(R, 64, dtype=torch.bfloat16, device="cuda")
        c2 = torch.empty_like(c1)
        ext.tau_embed(taus, fw.wiqn, fw.biqn, feat, h1, nq,
                      phi_out=p1, cosb_out=c1, direct=0)
        ext.tau_embed(taus, fw.wiqn, fw.biqn, feat, h2, nq,
                      phi_out=p2, cosb_out=c2, direct=True)
        torch.cuda.synchronize()
        assert torch.equal(h1, h2), (B, nq)
        assert torch.equal(p1, p2) and torch.equal(c1, c2), (B, nq)


def test_heads_gemm_tau_fused_matches_pair():
    """tau-embed fused into the heads GEMM (actor path) is bit-identical to
    tau_embed -> heads_gemm, including clamped tail rows."""
    from rainbowiqn_amd.ops import get_extension

    ext = get_extension()
    _, _, net16 = make_nets(seed=13)
    fw = fused_of(net16)
    for B, nq in [(1024, 8), (256, 32), (100, 8)]:  # 100*8: ragged tail
        R = B * nq
        x = torch.rand(B, 4, 84, 84, device="cuda").to(torch.bfloat16)
        feat = fw.encode(x).clone()
        taus = torch.rand(R, device="cuda")
        h = torch.empty(R, 3136, dtype=torch.bfloat16, device="cuda")
        hh_ref = torch.empty(R, 1024, dtype=torch.bfloat16, device="cuda")
        hh = torch.empty_like(hh_ref)
        ext.tau_embed(taus, fw.wiqn, fw.biqn, feat, h, nq)
        ext.heads_gemm(h, fw.wcat, fw.bcat, hh_ref)
        ext.heads_gemm_tau(taus, fw.wiqn, fw.biqn, feat, fw.wcat, fw.bcat,
                           hh, nq)
        torch.cuda.synchronize()
        assert torch.equal(hh, hh_ref), (B, nq)


def test_heads_gemm_large_r_parity():
    """R=8192 routes to the 256x128 big-tile kernel: check vs torch matmul
    (fp32 accumulate both sides), including a ragged-R launch."""
    from rainbowiqn_amd.ops import get_extension

    ext = get_extension()
    _, _, net16 = make_nets(seed=9)
    fw = fused_of(net16)
    for R in (8192, 8192 + 128):  # ragged: last 256-row block half full
        h = (torch.randn(R, 3136, device="cuda") * 0.1).to(torch.bfloat16)
        hh = torch.empty(R, 1024, dtype=torch.bfloat16, device="cuda")
        ext.heads_gemm(h, fw.wcat, fw.bcat, hh)
        torch.cuda.synchronize()
        ref = torch.relu(h.float() @ fw.wcat.float().t()
                         + fw.bcat.float())
        err = (hh.float() - ref).abs().max().item()
        scale = ref.abs().max().item()
        assert err < 0.02 * max(scale, 1.0) + 0.05, (R, err, scale)


def test_full_forward_parity():
    A = 6
    _, net32, net16 = make_nets(A)
    fw = fused_of(net16)
    B, nq = 32, 8
    x = torch.rand(B, 4, 84, 84, device="cuda")
    taus = torch.rand(B * nq, 1, device="cuda")
    z_ref, _ = net32(x, nq, taus=taus)  # [B, nq, A] fp32 truth
    z16, _ = net16(x.to(torch.bfloat16), nq, taus=taus)
    out = torch.empty(B, nq, A, device="cuda")
    fw.quantiles(x.to(torch.bfloat16), taus.view(-1), nq, out)
    torch.cuda.synchronize()
    err = (out - z_ref).abs().max().item()
    err16 = (z16.float() - z_ref).abs().max().item()
    scale = z_ref.abs().max().item()
    assert torch.isfinite(out).all()
    assert err < 0.05 * max(scale, 1.0) + 0.02, (err, scale)
    assert err < max(3 * err16 + 0.01, 0.03), (err, err16)


def test_act_and_theta_modes_consistent():
    A = 6
    _, _, net16 = make_nets(A, seed=3)
    fw = fused_of(net16)
    E, K = 64, 32
    x = (torch.rand(E, 4, 84, 84, device="cuda")).to(torch.bfloat16)
    taus = torch.rand(E * K, device="cuda")
    q = torch.empty(E, K, A, device="cuda")
    fw.quantiles(x, taus, K, q)
    actions = torch.zeros(E, dtype=torch.int64, device="cuda")
    fw.act(x, taus, K, actions)
    torch.cuda.synchronize()
    expect = q.mean(dim=1).argmax(dim=1)
    assert (actions == expect).all(), (actions[:8], expect[:8])
    # mode 2 gather equals mode 0 at the same actions
    th = torch.empty(E, K, device="cuda")
    fw.theta(x, taus, K, actions, th)
    torch.cuda.synchronize()
    gathered = q.gather(2, actions.view(E, 1, 1).expand(E, K, 1)).squeeze(2)
    assert torch.allclose(th, gathered, atol=1e-4), (th[0, :4], gathered[0, :4])


def test_fused_matches_eager_bf16_closely():
    """Same weights, same noise, same taus: fused vs eager-bf16 should agree
    to bf16 accumulation noise (both fp32-accumulate)."""
    A = 9
    _, _, net16 = make_nets(A, seed=11)
    fw = fused_of(net16)
    B, nq = 16, 8
    x = torch.rand(B, 4, 84, 84, device="cuda").to(torch.bfloat16)
    taus = torch.rand(B * nq, 1, device="cuda")
    z16, _ = net16(x, nq, taus=taus)
    out = torch.empty(B, nq, A, device="cuda")
    fw.quantiles(x, taus.view(-1), nq, out)
    torch.cuda.synchronize()
    err = (out - z16.float()).abs().max().item()
    scale = z16.float().abs().max().item()
    assert err < 0.04 * max(scale, 1.0) + 0.02, (err, scale)


def test_fp8_heads_parity():
    """MX-fp8 heads GEMM (config 5 path): block-scaled e4m3 quantiles within
    fp8 tolerance of the bf16 eager module; act/theta modes stay consistent."""
    A = 6
    _, net32, net16 = make_nets(A, seed=21)
    from rainbowiqn_amd.runtime.fused_forward import FusedForward

    fw = FusedForward(net16, "cuda", fp8_heads=True)
    B, nq = 32, 8
    x = torch.rand(B, 4, 84, 84, device="cuda").to(torch.bfloat16)
    taus = torch.rand(B * nq, 1, device="cuda")
    z16, _ = net16(x, nq, taus=taus)
    out = torch.empty(B, nq, A, device="cuda")
    fw.quantiles(x, taus.view(-1), nq, out)
    torch.cuda.synchronize()
    err = (out - z16.float()).abs().max().item()
    scale = z16.float().abs().max().item()
    assert torch.isfinite(out).all()
    # e4m3 has ~2 significant digits; the K=3136 reduction averages it out
    assert err < 0.10 * max(scale, 1.0) + 0.05, (err, scale)
    # internal consistency: act (mode 1) == argmax of mean quantiles (mode 0)
    E, K = 64, 32
    xa = torch.rand(E, 4, 84, 84, device="cuda").to(torch.bfloat16)
    ta = torch.rand(E * K, device="cuda")
    q = torch.empty(E, K, A, device="cuda")
    fw.quantiles(xa, ta, K, q)
    actions = torch.zeros(E, dtype=torch.int64, device="cuda")
    fw.act(xa, ta, K, actions)
    torch.cuda.synchronize()
    expect = q.mean(dim=1).argmax(dim=1)
    agree = (actions == expect).float().mean().item()
    assert agree == 1.0, agree


def test_tau_embed_fp8_fused_parity():
    """Fused tau-embed+quant (tau_embed_fp8) must produce bit-identical
    e4m3 payloads and e8m0 scales to the unfused pair (tau_embed DIRECT ->
    quant_rows) -- the fp8 actor fast path vs its reference decomposition."""
    from rainbowiqn_amd.ops import get_extension
    from rainbowiqn_amd.runtime.fused_forward import FusedForward

    ext = get_extension()
    A = 6
    _, _, net16 = make_nets(A, seed=31)
    fw = FusedForward(net16, "cuda", fp8_heads=True)
    for B, nq in [(64, 32), (32, 8), (13, 32)]:
        R = B * nq
        feat = (torch.rand(B, 3136, device="cuda") - 0.3).to(torch.bfloat16)
        taus = torch.rand(R, device="cuda")
        h = torch.empty(R, 3136, dtype=torch.bfloat16, device="cuda")
        hq_ref = torch.zeros(R, 3200, dtype=torch.uint8, device="cuda")
        sc_ref = torch.zeros(R, 100, dtype=torch.uint8, device="cuda")
        hq = torch.zeros(R, 3200, dtype=torch.uint8, device="cuda")
        sc = torch.zeros(R, 100, dtype=torch.uint8, device="cuda")
        ext.tau_embed(taus, fw.wiqn, fw.biqn, feat, h, nq, direct=1)
        ext.quant_fp8(h, hq_ref, sc_ref)
        ext.tau_embed_fp8(taus, fw.wiqn, fw.biqn, feat, hq, sc, nq)
        torch.cuda.synchronize()
        assert torch.equal(sc[:, :98], sc_ref[:, :98]), (B, nq)
        assert torch.equal(hq[:, :3136], hq_ref[:, :3136]), (B, nq)
        # padding stays zero (zero-init + kernel never touches k >= 3136)
        assert int(hq[:, 3136:].abs().sum()) == 0
        assert int(sc[:, 98:].abs().sum()) == 0


def test_fp8_gemm_big_variant_parity():
    """256x128 pipelined fp8 GEMM == 128x128 reference kernel bit-for-bit
    (identical per-element k-order MFMA accumulation)."""
    from rainbowiqn_amd.ops import get_extension
    from rainbowiqn_amd.runtime.fused_forward import FusedForward

    ext = get_extension()
    A = 6
    _, _, net16 = make_nets(A, seed=32)
    fw = FusedForward(net16, "cuda", fp8_heads=True)
    for R in [2048, 1024 + 96]:
        h = (torch.rand(R, 3136, device="cuda") - 0.3).to(torch.bfloat16)
        hq = torch.zeros(R, 3200, dtype=torch.uint8, device="cuda")
        sc = torch.zeros(R, 100, dtype=torch.uint8, device="cuda")
        ext.quant_fp8(h, hq, sc)
        out0 = torch.empty(R, 1024, dtype=torch.bfloat16, device="cuda")
        out1 = torch.empty(R, 1024, dtype=torch.bfloat16, device="cuda")
        ext.heads_gemm_fp8(hq, sc, fw.wq, fw.wsc, fw.bcat, out0, variant=0)
        ext.heads_gemm_fp8(hq, sc, fw.wq, fw.wsc, fw.bcat, out1, variant=1)
        torch.cuda.synchronize()
        assert torch.equal(out0, out1), R
        for v in (2, 3):
            ext.heads_gemm_fp8(hq, sc, fw.wq, fw.wsc, fw.bcat, out1,
                               variant=v)
            torch.cuda.synchronize()
            assert torch.equal(out0, out1), (R, v)


def test_bf16_gemm_256n_variant_parity():
    """256x256 NBUF2 bf16 GEMM == shipped 256x128 NBUF3 kernel bit-for-bit
    (identical per-element k-order accumulation)."""
    from rainbowiqn_amd.ops import get_extension
    from rainbowiqn_amd.runtime.fused_forward import FusedForward

    ext = get_extension()
    _, _, net16 = make_nets(6, seed=33)
    fw = FusedForward(net16, "cuda")
    for R in [8192, 8192 + 160]:
        h = (torch.rand(R, 3136, device="cuda") - 0.3).to(torch.bfloat16)
        out1 = torch.empty(R, 1024, dtype=torch.bfloat16, device="cuda")
        out2 = torch.empty(R, 1024, dtype=torch.bfloat16, device="cuda")
        ext.heads_gemm(h, fw.wcat, fw.bcat, out1, variant=1)
        ext.heads_gemm(h, fw.wcat, fw.bcat, out2, variant=2)
        torch.cuda.synchronize()
        assert torch.equal(out1, out2), R
        ext.heads_gemm(h, fw.wcat, fw.bcat, out2, variant=3)
        torch.cuda.synchronize()
        assert torch.equal(out1, out2), (R, "320x256")
        ext.heads_gemm(h, fw.wcat, fw.bcat, out2, variant=4)
        torch.cuda.synchronize()
        assert torch.equal(out1, out2), (R, "256rb")
