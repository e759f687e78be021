"""hipGraph-captured learner and actor steps (MI355X fast path).

The learner's whole step -- GPU-replay sample, online/target forward (bf16),
fused quantile-Huber loss + gradient, backward, fused Adam, priority
write-back, device-RNG tau draws and NoisyNet noise regeneration -- is
captured once into a hip graph and replayed per grad step: one ~10 us replay
instead of ~10^2 eager launches at 3-4 us each (MI355X_MICROARCH.md
boundary/graph-replay-floor rows).  Everything dynamic lives in device
buffers (philox offsets, Adam step counter, ring heads), so replays are
self-advancing; host-side knobs (beta anneal, target sync, tree rebuild)
touch device buffers between replays.

The actor act step (kernels K1-K9+K15) is captured the same
way: static state buffer in, greedy action buffer out.
"""

from __future__ import annotations

from typing import Optional

import torch

from ..models.model import RainbowIQN
from ..ops import get_extension
from ..ops.flat_params import (FusedAdam, FusedAdamMixed, attach_flat_grads,
                               flatten_module, master_state_dict)
from ..ops.functional import quantile_huber_loss_fused
from ..replay.gpu_replay import GpuReplay


def _noisy_layers(net: RainbowIQN):
    return [net.fc_h_v, net.fc_h_a, net.fc_z_v, net.fc_z_a]


class GpuLearner:
    def __init__(self, cfg, action_space: int, n_rings: int = 1,
                 device: str = "cuda", use_graph: bool = True,
                 replay=None, fork_level: int = 2):
        # fork_level: how much of the learn step rides the side queue.
        #   2 = full (theta/double-Q chain, z-grads, dphi consumers, dw2) --
        #       best when the learner owns the whole GPU (learner-only
        #       mode 1995 -> 2102 grad/s; multi-GPU learner rank);
        #   1 = theta chain + z-grad family only -- the colocated CU-
        #       partitioned mode: the deeper fork's extra cross-queue sync
        #       nodes in the 8-step-unrolled graph cost far more than the
        #       overlap there (5.74 -> 3.4M fr/s measured).
        self.fork_level = fork_level
        self.cfg = cfg
        self.ext = get_extension()
        self.device = torch.device(device)
        self.action_space = action_space
        self.B = cfg.batch_size
        self.N = cfg.num_tau_samples
        self.Np = cfg.num_tau_prime_samples
        self.K = cfg.num_quantile_samples

        # Native-bf16 weights (cfg.dtype == "bf16"): the modules' parameters,
        # buffers and activations are bf16 end to end -- no autocast, no
        # per-forward weight casts -- with an fp32 Adam master kept by the
        # fused mixed-precision optimizer kernel.
        self.bf16 = cfg.dtype == "bf16"
        dt = torch.bfloat16 if self.bf16 else torch.float32
        self.online = RainbowIQN(cfg, action_space).to(self.device).to(dt)
        self.target = RainbowIQN(cfg, action_space).to(self.device).to(dt)
        for net in (self.online, self.target):
            for layer in _noisy_layers(net):
                layer.enable_fused_noise()
        self.online.train()
        self.target.train()  # target uses its own noise draws (reference learn())
        self.flat_online = flatten_module(self.online)
        self.flat_target = flatten_module(self.target)
        self.flat_target.copy_(self.flat_online)
        for p in self.target.parameters():
            p.requires_grad_(False)
        self.flat_grad = attach_flat_grads(self.online)
        if self.bf16:
            self.opt = FusedAdamMixed(self.flat_online, self.flat_grad,
                                      lr=cfg.lr, eps=cfg.adam_eps)
        else:
            self.opt = FusedAdam(self.flat_online, self.flat_grad, lr=cfg.lr,
                                 eps=cfg.adam_eps)

        # replay may be external (sharded mode: a RemoteReplay facade over
        # the actor ranks' shards -- parallel/sharded.py)
        self.replay = replay if replay is not None else GpuReplay(
            cfg, n_rings=n_rings, device=device, out_dtype=dt)
        self.rng_state = torch.tensor(
            [cfg.seed * 7919 + 13, 0], dtype=torch.int64, device=self.device
        )
        # target-net noise gets its OWN philox state so its regeneration can
        # ride the side stream concurrently with the online net's (two
        # streams bumping one offset would race; the reference's target net
        # draws independent noise anyway)
        self.rng_state_tgt = torch.tensor(
            [cfg.seed * 7919 + 7001, 0], dtype=torch.int64,
            device=self.device)
        b = self.B
        # one contiguous buffer -> ONE rand_uniform launch per step
        self._taus_all = torch.empty(b * (self.N + self.Np + self.K), 1,
                                     device=self.device)
        self.taus_online = self._taus_all[: b * self.N]
        self.taus_target = self._taus_all[b * self.N : b * (self.N + self.Np)]
        self.taus_select = self._taus_all[b * (self.N + self.Np):]
        self.loss_static = torch.zeros(1, device=self.device)
        self.use_graph = use_graph
        self.graph: Optional[torch.cuda.CUDAGraph] = None
        self.steps_per_replay = 1
        self.grad_steps = 0
        # fused no-grad forwards (double-Q select + target quantiles): 4
        # hand-written kernels each instead of the eager module stack
        self.fused_online = self.fused_target = None
        self.manual_bwd = False
        if self.bf16 and getattr(cfg, "fused_forward", True):
            from .fused_forward import FusedForward

            fp8 = getattr(cfg, "fp8_heads", False)
            self.fused_online = FusedForward.for_net(self.online, self.device,
                                                     fp8_heads=fp8)
            self.fused_target = FusedForward.for_net(self.target, self.device,
                                                     fp8_heads=fp8)
            self.a_star = torch.zeros(b, dtype=torch.int64, device=self.device)
            self.theta_p_buf = torch.zeros(b, self.Np, device=self.device)
            if getattr(cfg, "fused_backward", True):
                self._init_manual_bwd()

    def _init_manual_bwd(self) -> None:
        """Buffers for the fully hand-written learn step (model_bwd.hip):
        no torch.autograd anywhere in the hot loop."""
        d, bf = self.device, torch.bfloat16
        B, N, R = self.B, self.N, self.B * self.N
        z = lambda *s, dt=bf: torch.zeros(*s, dtype=dt, device=d)
        self.w3tp = z(128 * 512)  # fragment-major transposed conv3
        self.w2tp = z(64 * 512)   # fragment-major transposed conv2
        self.c1g = z(2 * B, 12800)  # [s; s'] batch; backward uses [:B]
        self.c2g = z(2 * B, 5184)
        self.feat_cat = z(2 * B, 3136)
        self.phi = z(R, 3136)
        self.dhh = z(R, 1024)
        self.dphi = z(R, 3136)
        self.dfeat = z(B, 3136, dt=torch.float32)
        self.cosb = z(R, 64)
        self.dpre3g = z(64, B * 49)
        self.dc2g = z(64, B * 81)
        self.dc1g = z(32, B * 400)
        self.dW1s = z(32, 256, dt=torch.float32)
        self.dW2s = z(64, 512, dt=torch.float32)
        self.dW3s = z(64, 576, dt=torch.float32)
        self.db1s = z(32, dt=torch.float32)
        self.db2s = z(64, dt=torch.float32)
        self.db3s = z(64, dt=torch.float32)
        self.theta_buf = z(B, N, dt=torch.float32)
        self.dtheta = z(B, N, dt=torch.float32)
        self.loss_buf = z(1, dt=torch.float32)
        self.per_sample = z(B, dt=torch.float32)
        # side stream for the learn step's independent chains (double-Q +
        # target forward; z/heads weight grads).  The grids are tiny (16-64
        # WGs), so the captured graph replays the two chains CONCURRENTLY
        # on the learner's CU partition instead of serializing ~30 small
        # kernels.  Fork/join edges via wait_stream during capture.
        self._bstream = torch.cuda.Stream(device=self.device)
        # grad views (slices of the flat bf16 grad buffer, written in-place)
        net = self.online
        g = lambda p: p.grad
        zv, za = net.fc_z_v, net.fc_z_a
        hv, ha = net.fc_h_v, net.fc_h_a
        self._z_eps = [zv.weight_epsilon, za.weight_epsilon,
                       zv.bias_epsilon, za.bias_epsilon]
        self._z_grads = [g(zv.weight_mu), g(zv.weight_sigma), g(zv.bias_mu),
                         g(zv.bias_sigma), g(za.weight_mu), g(za.weight_sigma),
                         g(za.bias_mu), g(za.bias_sigma)]
        self._h_wgrads = [g(hv.weight_mu), g(hv.weight_sigma),
                          g(ha.weight_mu), g(ha.weight_sigma)]
        self._h_bgrads = [g(hv.bias_mu), g(hv.bias_sigma),
                          g(ha.bias_mu), g(ha.bias_sigma)]
        self._h_weps = (hv.weight_epsilon, ha.weight_epsilon)
        self._h_beps = (hv.bias_epsilon, ha.bias_epsilon)
        cw = net.convs
        self._conv_scr = [self.dW1s, self.dW2s, self.dW3s,
                          self.db1s, self.db2s, self.db3s]
        self._conv_grads = [g(cw[0].weight).view(32, 256),
                            g(cw[2].weight).view(64, 512),
                            g(cw[4].weight).view(64, 576),
                            g(cw[0].bias), g(cw[2].bias), g(cw[4].bias)]
        self._iqn_wgrad = g(net.iqn_fc.weight)
        self._iqn_bgrad = g(net.iqn_fc.bias)
        self.manual_bwd = True

    def _step_body_manual(self):
        """One grad step, every op a hand-written HIP kernel (K1-K16)."""
        cfg = self.cfg
        ext = self.ext
        b, N, A = self.B, self.N, self.action_space
        fo, ft = self.fused_online, self.fused_target
        ext.rand_uniform(self._taus_all, self.rng_state)
        # target-net noise + packs on the side stream (own philox state),
        # concurrent with the online net's
        main0 = torch.cuda.current_stream()
        self._bstream.wait_stream(main0)
        with torch.cuda.stream(self._bstream):
            self._reset_noise_device(self.target, self.rng_state_tgt)
            ft.pack_weights()
        self._reset_noise_device(self.online)
        fo.pack_weights()
        ext.pack_bwd_weights(fo.w3raw, fo.w2raw, self.w3tp, self.w2tp)
        idx, s, a, r, s2, nt, w = self.replay.sample()
        # ---- online conv over [s; s'] in ONE call (activations saved for
        # the backward; the s' half feeds the double-Q select) -------------
        _, h_g, hh_g = fo._bufs(b, N)[:3]
        ext.conv_encode(self.replay.states_cat, fo.w1, fo.b1, fo.w2, fo.b2,
                        fo.w3p, fo.b3, self.feat_cat, self.c1g, self.c2g)
        feat_g = self.feat_cat[:b]
        from .fused_forward import TAU_DIRECT
        main = torch.cuda.current_stream()
        sb = self._bstream
        # fork: double-Q select + target forward (side) run concurrently
        # with the online theta chain (main); both need only the conv
        # features and the pre-drawn taus
        sb.wait_stream(main)
        with torch.cuda.stream(sb):
            fo.act_from_feat(self.feat_cat[b:], self.taus_select.view(-1),
                             self.K, self.a_star)
            ft.theta(s2, self.taus_target.view(-1), self.Np, self.a_star,
                     self.theta_p_buf)
        ext.tau_embed(self.taus_online.view(-1), fo.wiqn, fo.biqn, feat_g,
                      h_g, N, phi_out=self.phi, cosb_out=self.cosb,
                      direct=TAU_DIRECT)
        ext.heads_gemm(h_g, fo.wcat, fo.bcat, hh_g)
        ext.heads_finish(hh_g, fo.wz, fo.bz, A, N, 2,
                         self.theta_buf.view(-1), a)
        main.wait_stream(sb)  # join: loss needs theta AND theta'
        # ---- fused loss: loss, per-sample priorities, dtheta -------------
        ext.quantile_huber(self.theta_buf, self.theta_p_buf,
                           self.taus_online.view(b, N), r, nt, w,
                           cfg.discount ** cfg.multi_step, cfg.kappa,
                           self.loss_buf, self.per_sample, self.dtheta)
        # ---- backward chain ----------------------------------------------
        dt_flat = self.dtheta.view(-1)
        ext.finish_bwd_dhh(dt_flat, a, hh_g, fo.wz, self.dhh, A, N)
        # fork: the z-layer / heads weight-grad family only needs dhh and
        # saved activations -- it runs beside the dx -> conv backward chain
        sb.wait_stream(main)
        with torch.cuda.stream(sb):
            ext.zgrad(dt_flat, a, hh_g, A, N, self._z_eps, self._z_grads)
            ext.gemm_dwcat(self.dhh, h_g, self._h_weps[0], self._h_weps[1],
                           self._h_wgrads)
            ext.bias_grad_heads(self.dhh, self._h_beps[0], self._h_beps[1],
                                self._h_bgrads)
        self.dfeat.zero_()
        ext.gemm_dx(self.dhh, fo.wcat, feat_g, self.phi, self.dphi,
                    self.dfeat, N)
        if self.fork_level >= 2:
            # dphi consumers join the side queue behind the z/heads grads
            sb.wait_stream(main)
            with torch.cuda.stream(sb):
                ext.tau_dw(self.dphi, self.cosb, self._iqn_wgrad)
                ext.bias_grad_cols(self.dphi, self._iqn_bgrad)
            ext.conv_bwd_data(self.dfeat, feat_g, self.c1g, self.c2g,
                              self.w3tp, self.w2tp, self.dpre3g, self.dc2g,
                              self.dc1g, self.db1s, self.db2s, self.db3s)
            # the three conv dW kernels are mutually independent: split
            # them across the two queues behind conv_bwd_data
            sb.wait_stream(main)
            with torch.cuda.stream(sb):
                ext.conv_bwd_dw(2, self.dc2g, self.c1g, self.dW2s, b, 12)
            ext.conv_bwd_dw(3, self.dpre3g, self.c2g, self.dW3s, b, 8)
            ext.conv_bwd_dw(1, self.dc1g, s, self.dW1s, b, 32)
            main.wait_stream(sb)  # join before finalize + Adam
        else:
            ext.tau_dw(self.dphi, self.cosb, self._iqn_wgrad)
            ext.bias_grad_cols(self.dphi, self._iqn_bgrad)
            ext.conv_bwd_data(self.dfeat, feat_g, self.c1g, self.c2g,
                              self.w3tp, self.w2tp, self.dpre3g, self.dc2g,
                              self.dc1g, self.db1s, self.db2s, self.db3s)
            ext.conv_bwd_dw(3, self.dpre3g, self.c2g, self.dW3s, b, 8)
            ext.conv_bwd_dw(2, self.dc2g, self.c1g, self.dW2s, b, 12)
            ext.conv_bwd_dw(1, self.dc1g, s, self.dW1s, b, 32)
            main.wait_stream(sb)  # join the z-grad family before Adam
        ext.conv_finalize(self._conv_scr, self._conv_grads)
        # ---- optimizer + priorities --------------------------------------
        self.opt.step()
        self.replay.update_priorities(idx, self.per_sample)
        self.loss_static.copy_(self.loss_buf)

    # -- pieces -------------------------------------------------------------
    def _reset_noise_device(self, net: RainbowIQN, rng=None) -> None:
        # batched: all 4 NoisyLinear layers in 2 kernels (K6/K7 fused).
        # Packs are built lazily AFTER FusedForward re-points weight_noisy.
        packs = getattr(self, "_noisy_packs", None)
        if packs is None:
            packs = self._noisy_packs = {}
        key = id(net)
        if key not in packs:
            from .fused_forward import noisy_pack

            packs[key] = noisy_pack(net, self.device)
        self.ext.noisy_material4(packs[key],
                                 rng if rng is not None else self.rng_state)

    def _step_body(self):
        if self.manual_bwd:
            return self._step_body_manual()
        cfg = self.cfg
        b = self.B
        ext = self.ext
        ext.rand_uniform(self._taus_all, self.rng_state)
        self._reset_noise_device(self.online)
        self._reset_noise_device(self.target)
        idx, s, a, r, s2, nt, w = self.replay.sample()
        z, _ = self.online(s, self.N, taus=self.taus_online)
        theta = z.gather(2, a.view(b, 1, 1).expand(b, self.N, 1)).squeeze(2)
        if self.fused_online is not None:
            self.fused_online.pack_weights()
            self.fused_target.pack_weights()
            self.fused_online.act(s2, self.taus_select.view(-1), self.K,
                                  self.a_star)
            self.fused_target.theta(s2, self.taus_target.view(-1), self.Np,
                                    self.a_star, self.theta_p_buf)
            theta_p = self.theta_p_buf
        else:
            with torch.no_grad():
                zsel, _ = self.online(s2, self.K, taus=self.taus_select)
                a_star = zsel.mean(dim=1).argmax(dim=1)
                zp, _ = self.target(s2, self.Np, taus=self.taus_target)
                theta_p = zp.gather(
                    2, a_star.view(b, 1, 1).expand(b, self.Np, 1)
                ).squeeze(2)
        loss, per_sample = quantile_huber_loss_fused(
            theta.float(), theta_p.float(),
            self.taus_online.view(b, self.N).float(), r, nt, w,
            cfg.discount, cfg.multi_step, cfg.kappa,
        )
        self.opt.zero_grad()
        loss.backward()
        self.opt.step()
        self.replay.update_priorities(idx, per_sample)
        self.loss_static.copy_(loss.detach().view(1))

    # -- graph lifecycle -----------------------------------------------------
    def capture(self, steps_per_replay: int = 1) -> bool:
        """Capture `steps_per_replay` unrolled learn steps into one hip graph
        (after replay has data): one graph replay per learn round removes all
        per-step host launch/boundary cost."""
        if not self.use_graph:
            return False
        self.steps_per_replay = steps_per_replay
        torch.cuda.synchronize()
        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            for _ in range(3):  # warmup: MIOpen find, workspace allocs
                self._step_body()
        torch.cuda.current_stream().wait_stream(side)
        torch.cuda.synchronize()
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g):
            for _ in range(steps_per_replay):
                self._step_body()
        self.graph = g
        return True

    # -- public API -----------------------------------------------------------
    def append_actor_chunk(self, ring_id: int, frames, meta=None, **kw) -> None:
        self.replay.append_actor_chunk(ring_id, frames, meta=meta, **kw)

    def learn_steps(self, k: int) -> None:
        if self.graph is None and self.use_graph and len(self.replay) >= self.cfg.learn_start:
            # one replay per learn round: unroll the round's steps in-graph
            self.capture(steps_per_replay=max(1, self.cfg.learn_per_round))
            # the 3 warmup iterations executed real steps; the capture pass
            # only records (kernels do not run during stream capture)
            self.grad_steps += 3
            k = max(0, k - 3)
        if self.graph is not None:
            spr = self.steps_per_replay
            while k >= spr:
                self.graph.replay()
                k -= spr
                self.grad_steps += spr
        for _ in range(k):
            self._step_body()
        self.grad_steps += k

    def last_loss(self) -> float:
        return float(self.loss_static.item())

    def update_target(self) -> None:
        self.flat_target.copy_(self.flat_online)

    @property
    def master_flat(self) -> torch.Tensor:
        return self.opt.master if self.bf16 else self.flat_online

    def anneal_beta(self, beta: float) -> None:
        self.replay.beta.fill_(min(1.0, beta))

    def maintenance(self) -> None:
        """Periodic fp32-drift cleanup of the tree (outside the graph)."""
        self.replay.rebuild_tree()

    # -- checkpoint (reference-parity state_dict: fp32 from the Adam master) ---
    def state_dict(self):
        if self.bf16:
            return master_state_dict(self.online, self.opt.master)
        return self.online.state_dict()

    def save(self, path: str, name: str = "model.pth") -> None:
        import os

        os.makedirs(path, exist_ok=True)
        torch.save(self.state_dict(), os.path.join(path, name))

    def load_state_dict_ref(self, sd) -> None:
        """Load a reference-format fp32 state_dict (checkpoint resume): module
        weights (cast to the compute dtype), Adam fp32 master, and target."""
        from ..utils.checkpoint import remap_reference_state_dict

        sd = remap_reference_state_dict(sd, self.online)
        dt = torch.bfloat16 if self.bf16 else torch.float32
        self.online.load_state_dict({k: v.to(dt) for k, v in sd.items()})
        if self.bf16:
            self.opt.master.copy_(self.flat_online.float())
        self.update_target()


class GraphedActor:
    """hipGraph-captured greedy act step over a batch of E env streams.

    In: static u8 frame-stack buffer [E, h, 84, 84]; out: static action
    buffer [E].  NoisyNet exploration noise and tau draws are regenerated
    in-graph from device philox state.
    """

    def __init__(self, cfg, net: RainbowIQN, n_envs: int, device="cuda",
                 use_graph: bool = True, seed_offset: int = 0):
        self.cfg = cfg
        self.ext = get_extension()
        self.net = net
        self.E = n_envs
        self.K = cfg.num_quantile_samples
        self.device = torch.device(device)
        self.dtype = torch.bfloat16 if cfg.dtype == "bf16" else torch.float32
        self.stack_u8 = torch.zeros(n_envs, cfg.history_length, 84, 84,
                                    dtype=torch.uint8, device=self.device)
        self.stack_f32 = torch.zeros(n_envs, cfg.history_length, 84, 84,
                                     dtype=self.dtype, device=self.device)
        self.taus = torch.empty(n_envs * self.K, 1, device=self.device)
        self.actions = torch.zeros(n_envs, dtype=torch.int64, device=self.device)
        self.rng_state = torch.tensor(
            [cfg.seed * 104729 + 7 + seed_offset, 0], dtype=torch.int64,
            device=self.device,
        )
        self.use_graph = use_graph
        self.graph: Optional[torch.cuda.CUDAGraph] = None

    def _body(self):
        # stack_gather-style preprocess: u8 -> model dtype
        if self.dtype == torch.float32:
            self.ext.preprocess(self.stack_u8, self.stack_f32)
        else:
            self.stack_f32.copy_(self.stack_u8)
            self.stack_f32.div_(255.0)
        self.ext.rand_uniform(self.taus, self.rng_state)
        with torch.no_grad():
            z, _ = self.net(self.stack_f32, self.K, taus=self.taus)
            self.actions.copy_(z.float().mean(dim=1).argmax(dim=1))

    def reset_noise(self) -> None:
        for layer in _noisy_layers(self.net):
            nin = torch.zeros(layer.in_features, device=self.device)
            nout = torch.zeros(layer.out_features, device=self.device)
            self.ext.noisy_reset(nin, nout, layer.weight_epsilon,
                                 layer.bias_epsilon, self.rng_state)

    def capture(self) -> None:
        if not self.use_graph:
            return
        torch.cuda.synchronize()
        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            for _ in range(3):
                self._body()
        torch.cuda.current_stream().wait_stream(side)
        torch.cuda.synchronize()
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g):
            self._body()
        self.graph = g

    def act(self) -> torch.Tensor:
        """Run the captured act step on the current stack_u8; returns actions."""
        if self.graph is None:
            if self.use_graph:
                self.capture()
                self.graph.replay()
            else:
                self._body()
        else:
            self.graph.replay()
        return self.actions
