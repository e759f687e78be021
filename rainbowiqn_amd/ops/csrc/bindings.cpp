// Python bindings for the gfx950 HIP kernels.  Torch headers stay in this
// translation unit; device code lives in the *.hip files behind a C ABI
// (riqn_kernels.h).
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include <hip/hip_runtime_api.h>
#include <vector>

#include "riqn_kernels.h"

namespace {

void* cur_stream() { return (void*)at::cuda::getCurrentCUDAStream().stream(); }

#define CHECK_DEV(t) TORCH_CHECK((t).is_cuda(), #t " must be on GPU")
#define CHECK_CONT(t) TORCH_CHECK((t).is_contiguous(), #t " must be contiguous")
#define CHECK_T(t, ty) \
    CHECK_DEV(t); CHECK_CONT(t); TORCH_CHECK((t).scalar_type() == ty, #t " dtype")

void replay_append(torch::Tensor frames, torch::Tensor actions,
                   torch::Tensor rewards, torch::Tensor terminals,
                   torch::Tensor timesteps, torch::Tensor tree,
                   int64_t tree_size, int64_t capacity,
                   torch::Tensor ring_state, int64_t n_rings, int64_t ring_cap,
                   torch::Tensor max_priority, torch::Tensor chunk_frames,
                   torch::Tensor chunk_meta, int64_t ring_id) {
    CHECK_T(frames, torch::kUInt8);
    CHECK_T(tree, torch::kFloat32);
    CHECK_T(chunk_frames, torch::kUInt8);
    CHECK_T(chunk_meta, torch::kFloat32);
    CHECK_T(ring_state, torch::kInt32);
    int64_t c = chunk_frames.size(0);
    TORCH_CHECK(chunk_meta.size(0) == c && chunk_meta.size(1) == 3);
    TORCH_CHECK(c < ring_cap, "chunk longer than ring");
    riqn_replay_append(
        frames.data_ptr<uint8_t>(), actions.data_ptr<int32_t>(),
        rewards.data_ptr<float>(), terminals.data_ptr<uint8_t>(),
        timesteps.data_ptr<int32_t>(), tree.data_ptr<float>(), tree_size,
        capacity, ring_state.data_ptr<int32_t>(), (int)n_rings, ring_cap,
        max_priority.data_ptr<float>(), chunk_frames.data_ptr<uint8_t>(),
        chunk_meta.data_ptr<float>(), c, (int)ring_id, cur_stream());
}

void replay_sample(torch::Tensor frames, torch::Tensor actions,
                   torch::Tensor rewards, torch::Tensor terminals,
                   torch::Tensor timesteps, torch::Tensor tree,
                   int64_t tree_size, int64_t capacity,
                   torch::Tensor ring_state, int64_t n_rings, int64_t ring_cap,
                   int64_t history, int64_t n_step, double discount,
                   torch::Tensor beta, torch::Tensor rng_state, int64_t batch,
                   torch::Tensor out_idx, torch::Tensor out_states,
                   torch::Tensor out_next_states, torch::Tensor out_actions,
                   torch::Tensor out_returns, torch::Tensor out_nonterminals,
                   torch::Tensor out_weights) {
    CHECK_DEV(out_states); CHECK_CONT(out_states);
    bool out_bf16 = out_states.scalar_type() == torch::kBFloat16;
    TORCH_CHECK(out_bf16 || out_states.scalar_type() == torch::kFloat32);
    TORCH_CHECK(out_next_states.scalar_type() == out_states.scalar_type());
    CHECK_T(out_actions, torch::kInt64);
    CHECK_T(out_idx, torch::kInt32);
    TORCH_CHECK(batch <= 512, "sample batch must be <= 512 (single block)");
    riqn_replay_sample(
        frames.data_ptr<uint8_t>(), actions.data_ptr<int32_t>(),
        rewards.data_ptr<float>(), terminals.data_ptr<uint8_t>(),
        timesteps.data_ptr<int32_t>(), tree.data_ptr<float>(), tree_size,
        capacity, ring_state.data_ptr<int32_t>(), (int)n_rings, ring_cap,
        (int)history, (int)n_step, (float)discount, beta.data_ptr<float>(),
        (uint64_t*)rng_state.data_ptr<int64_t>(), (int)batch,
        out_idx.data_ptr<int32_t>(), (float*)out_states.data_ptr(),
        (float*)out_next_states.data_ptr(), out_actions.data_ptr<int64_t>(),
        out_returns.data_ptr<float>(), out_nonterminals.data_ptr<float>(),
        out_weights.data_ptr<float>(), out_bf16 ? 1 : 0, cur_stream());
}

void replay_update_priorities(torch::Tensor tree, int64_t tree_size,
                              torch::Tensor max_priority, torch::Tensor idxs,
                              torch::Tensor priorities,
                              double priority_exponent) {
    CHECK_T(tree, torch::kFloat32);
    CHECK_T(idxs, torch::kInt32);
    CHECK_T(priorities, torch::kFloat32);
    riqn_replay_update_priorities(
        tree.data_ptr<float>(), tree_size, max_priority.data_ptr<float>(),
        idxs.data_ptr<int32_t>(), priorities.data_ptr<float>(),
        idxs.numel(), (float)priority_exponent, cur_stream());
}

void tree_rebuild(torch::Tensor tree, int64_t tree_size) {
    CHECK_T(tree, torch::kFloat32);
    for (int64_t nodes = tree_size / 2; nodes >= 1; nodes /= 2)
        riqn_tree_level_pass(tree.data_ptr<float>(), nodes, cur_stream());
}

void quantile_huber(torch::Tensor theta, torch::Tensor target_q,
                    torch::Tensor taus, torch::Tensor returns,
                    torch::Tensor nonterm,
                    c10::optional<torch::Tensor> weights, double gamma_n,
                    double kappa, torch::Tensor out_loss,
                    torch::Tensor out_per_sample,
                    torch::Tensor out_grad_theta) {
    CHECK_T(theta, torch::kFloat32);
    CHECK_T(target_q, torch::kFloat32);
    CHECK_T(taus, torch::kFloat32);
    int B = theta.size(0), N = theta.size(1), Np = target_q.size(1);
    TORCH_CHECK(N <= 64 && Np <= 64, "N/N' must be <= 64");
    const float* w = nullptr;
    if (weights.has_value()) {
        CHECK_T(weights.value(), torch::kFloat32);
        w = weights.value().data_ptr<float>();
    }
    riqn_quantile_huber(theta.data_ptr<float>(), target_q.data_ptr<float>(),
                        taus.data_ptr<float>(), returns.data_ptr<float>(),
                        nonterm.data_ptr<float>(), w, (float)gamma_n,
                        (float)kappa, B, N, Np, out_loss.data_ptr<float>(),
                        out_per_sample.data_ptr<float>(),
                        out_grad_theta.data_ptr<float>(), cur_stream());
}

void adam_step(torch::Tensor param, torch::Tensor grad, torch::Tensor exp_avg,
               torch::Tensor exp_avg_sq, double lr, double beta1, double beta2,
               double eps, torch::Tensor step) {
    CHECK_T(param, torch::kFloat32);
    CHECK_T(grad, torch::kFloat32);
    CHECK_T(step, torch::kInt32);
    riqn_adam_step(param.data_ptr<float>(), grad.data_ptr<float>(),
                   exp_avg.data_ptr<float>(), exp_avg_sq.data_ptr<float>(),
                   param.numel(), (float)lr, (float)beta1, (float)beta2,
                   (float)eps, step.data_ptr<int32_t>(), cur_stream());
}

void preprocess(torch::Tensor in, torch::Tensor out) {
    CHECK_T(in, torch::kUInt8);
    CHECK_T(out, torch::kFloat32);
    TORCH_CHECK(in.numel() == out.numel());
    riqn_preprocess(in.data_ptr<uint8_t>(), out.data_ptr<float>(), in.numel(),
                    cur_stream());
}

void rand_uniform(torch::Tensor out, torch::Tensor rng_state) {
    CHECK_T(out, torch::kFloat32);
    riqn_rand_uniform(out.data_ptr<float>(), out.numel(),
                      (uint64_t*)rng_state.data_ptr<int64_t>(), cur_stream());
}

void noisy_reset(torch::Tensor eps_in, torch::Tensor eps_out,
                 torch::Tensor weight_eps, torch::Tensor bias_eps,
                 torch::Tensor rng_state) {
    bool eps_bf16 = weight_eps.scalar_type() == torch::kBFloat16;
    TORCH_CHECK(eps_bf16 || weight_eps.scalar_type() == torch::kFloat32);
    TORCH_CHECK(bias_eps.scalar_type() == weight_eps.scalar_type());
    riqn_noisy_reset(eps_in.data_ptr<float>(), eps_in.numel(),
                     eps_out.data_ptr<float>(), eps_out.numel(),
                     (float*)weight_eps.data_ptr(), (float*)bias_eps.data_ptr(),
                     eps_bf16 ? 1 : 0,
                     (uint64_t*)rng_state.data_ptr<int64_t>(), cur_stream());
}

void noisy_material(torch::Tensor eps_in, torch::Tensor eps_out,
                    torch::Tensor w_mu, torch::Tensor w_sigma,
                    torch::Tensor b_mu, torch::Tensor b_sigma,
                    torch::Tensor weight_eps, torch::Tensor bias_eps,
                    torch::Tensor w_noisy, torch::Tensor b_noisy,
                    torch::Tensor rng_state) {
    bool bf16 = w_mu.scalar_type() == torch::kBFloat16;
    TORCH_CHECK(bf16 || w_mu.scalar_type() == torch::kFloat32);
    TORCH_CHECK(w_noisy.sizes() == w_mu.sizes());
    riqn_noisy_material(eps_in.data_ptr<float>(), eps_in.numel(),
                        eps_out.data_ptr<float>(), eps_out.numel(),
                        w_mu.data_ptr(), w_sigma.data_ptr(), b_mu.data_ptr(),
                        b_sigma.data_ptr(), weight_eps.data_ptr(),
                        bias_eps.data_ptr(), w_noisy.data_ptr(),
                        b_noisy.data_ptr(), bf16 ? 1 : 0,
                        (uint64_t*)rng_state.data_ptr<int64_t>(),
                        cur_stream());
}

void adam_step_mixed(torch::Tensor master, torch::Tensor grad_bf16,
                     torch::Tensor param_bf16, torch::Tensor exp_avg,
                     torch::Tensor exp_avg_sq, double lr, double beta1,
                     double beta2, double eps, torch::Tensor step) {
    CHECK_T(master, torch::kFloat32);
    CHECK_T(grad_bf16, torch::kBFloat16);
    CHECK_T(param_bf16, torch::kBFloat16);
    CHECK_T(step, torch::kInt32);
    riqn_adam_step_mixed(master.data_ptr<float>(),
                         (const uint16_t*)grad_bf16.data_ptr(),
                         (uint16_t*)param_bf16.data_ptr(),
                         exp_avg.data_ptr<float>(),
                         exp_avg_sq.data_ptr<float>(), master.numel(),
                         (float)lr, (float)beta1, (float)beta2, (float)eps,
                         step.data_ptr<int32_t>(), cur_stream());
}

// CU-masked HIP streams (colocated overlap): partition the 256 CUs between
// the actor and learner queues so the actor's chip-filling grids cannot
// starve the learner's small ones (AMD per-queue CU mask; wrapped on the
// Python side with torch.cuda.ExternalStream).
int64_t create_cu_masked_stream(std::vector<int64_t> mask_words) {
    std::vector<uint32_t> m(mask_words.begin(), mask_words.end());
    hipStream_t s = nullptr;
    auto err = hipExtStreamCreateWithCUMask(&s, (uint32_t)m.size(), m.data());
    TORCH_CHECK(err == hipSuccess, "hipExtStreamCreateWithCUMask failed: ",
                hipGetErrorString(err));
    return (int64_t)(uintptr_t)s;
}

void destroy_stream(int64_t ptr) {
    if (ptr) hipStreamDestroy((hipStream_t)(uintptr_t)ptr);
}

void env_step(torch::Tensor actions, torch::Tensor env_state,
              torch::Tensor rng_state, torch::Tensor ring, int64_t history,
              int64_t ring_head_new, double reward_clip, double p_done,
              int64_t max_episode_frames, torch::Tensor chunk_frames,
              torch::Tensor chunk_meta, int64_t chunk_pos, int64_t task) {
    CHECK_T(actions, torch::kInt64);
    CHECK_T(ring, torch::kUInt8);
    int n_envs = actions.size(0);
    riqn_env_step(actions.data_ptr<int64_t>(), env_state.data_ptr<int32_t>(),
                  (uint64_t*)rng_state.data_ptr<int64_t>(),
                  ring.data_ptr<uint8_t>(), (int)history, (int)ring_head_new,
                  n_envs, (float)reward_clip, (float)p_done,
                  (int)max_episode_frames, chunk_frames.data_ptr<uint8_t>(),
                  chunk_meta.data_ptr<float>(), (int)chunk_pos, (int)task,
                  cur_stream());
}

void stack_gather(torch::Tensor ring, torch::Tensor env_state, int64_t history,
                  int64_t ring_head, int64_t n_envs, torch::Tensor out) {
    CHECK_T(ring, torch::kUInt8);
    bool out_bf16 = out.scalar_type() == torch::kBFloat16;
    TORCH_CHECK(out_bf16 || out.scalar_type() == torch::kFloat32);
    riqn_stack_gather(ring.data_ptr<uint8_t>(), env_state.data_ptr<int32_t>(),
                      (int)history, (int)ring_head, (int)n_envs,
                      (float*)out.data_ptr(), out_bf16 ? 1 : 0, cur_stream());
}

// Batched noise regeneration for the 4 NoisyLinear layers of one net.
// `ts` = 4 groups of 10 tensors: w_mu, w_sigma, b_mu, b_sigma, w_eps, b_eps,
// w_noisy, b_noisy, eps_in, eps_out.
void noisy_material4(std::vector<torch::Tensor> ts, torch::Tensor rng_state) {
    TORCH_CHECK(ts.size() == 40, "need 4 layers x 10 tensors");
    NoisyPack4 pack;
    bool bf16 = ts[0].scalar_type() == torch::kBFloat16;
    for (int L = 0; L < 4; ++L) {
        auto* g = &ts[L * 10];
        for (int j = 0; j < 10; ++j) {
            CHECK_DEV(g[j]);
            CHECK_CONT(g[j]);
        }
        TORCH_CHECK(g[8].scalar_type() == torch::kFloat32);  // eps_in
        TORCH_CHECK(g[9].scalar_type() == torch::kFloat32);  // eps_out
        NoisyDesc& d = pack.l[L];
        d.w_mu = g[0].data_ptr();
        d.w_sigma = g[1].data_ptr();
        d.b_mu = g[2].data_ptr();
        d.b_sigma = g[3].data_ptr();
        d.w_eps = g[4].data_ptr();
        d.b_eps = g[5].data_ptr();
        d.w_noisy = g[6].data_ptr();
        d.b_noisy = g[7].data_ptr();
        d.eps_in = g[8].data_ptr<float>();
        d.eps_out = g[9].data_ptr<float>();
        d.n_in = g[0].size(-1);
        d.n_out = g[0].numel() / d.n_in;
        TORCH_CHECK(g[8].numel() >= d.n_in && g[9].numel() >= d.n_out);
    }
    riqn_noisy_material4(pack, bf16 ? 1 : 0,
                         (uint64_t*)rng_state.data_ptr<int64_t>(),
                         cur_stream());
}

// ---- fused RainbowIQN forward (model_fwd.hip) ------------------------------

#define CHECK_BF16(t) CHECK_T(t, torch::kBFloat16)

void pack_fwd_frag(torch::Tensor w1, torch::Tensor w2, torch::Tensor w3,
                   torch::Tensor wiqn, torch::Tensor w1f, torch::Tensor w2f,
                   torch::Tensor w3f, torch::Tensor wiqnf) {
    CHECK_BF16(w1); CHECK_BF16(w2); CHECK_BF16(w3); CHECK_BF16(wiqn);
    CHECK_BF16(w1f); CHECK_BF16(w2f); CHECK_BF16(w3f); CHECK_BF16(wiqnf);
    TORCH_CHECK(w1.numel() == 32 * 256 && w1f.numel() == 16 * 512);
    TORCH_CHECK(w2.numel() == 64 * 512 && w2f.numel() == 64 * 512);
    TORCH_CHECK(w3.numel() == 64 * 576 && w3f.numel() == 128 * 512);
    TORCH_CHECK(wiqn.numel() == 3136 * 64 && wiqnf.numel() == 392 * 512);
    riqn_pack_fwd_frag(w1.data_ptr(), w2.data_ptr(), w3.data_ptr(),
                       wiqn.data_ptr(), w1f.data_ptr(), w2f.data_ptr(),
                       w3f.data_ptr(), wiqnf.data_ptr(), cur_stream());
}

void conv_encode(torch::Tensor x, torch::Tensor w1, torch::Tensor b1,
                 torch::Tensor w2, torch::Tensor b2, torch::Tensor w3p,
                 torch::Tensor b3, torch::Tensor feat,
                 c10::optional<torch::Tensor> c1g = c10::nullopt,
                 c10::optional<torch::Tensor> c2g = c10::nullopt) {
    CHECK_BF16(x);
    CHECK_BF16(w1); CHECK_BF16(w2); CHECK_BF16(w3p);
    CHECK_BF16(b1); CHECK_BF16(b2); CHECK_BF16(b3);
    CHECK_BF16(feat);
    int B = (int)x.size(0);
    TORCH_CHECK(x.numel() == (int64_t)B * 4 * 84 * 84, "history must be 4");
    TORCH_CHECK(w1.numel() == 16 * 512 && w2.numel() == 64 * 512);
    TORCH_CHECK(w3p.numel() == 128 * 512);
    TORCH_CHECK(feat.numel() == (int64_t)B * 3136);
    void *p1 = nullptr, *p2 = nullptr;
    if (c1g.has_value()) {
        TORCH_CHECK(c1g->numel() == (int64_t)B * 12800 &&
                    c2g->numel() == (int64_t)B * 5184);
        p1 = c1g->data_ptr();
        p2 = c2g->data_ptr();
    }
    riqn_conv_encode(x.data_ptr(), w1.data_ptr(), b1.data_ptr(), w2.data_ptr(),
                     b2.data_ptr(), w3p.data_ptr(), b3.data_ptr(),
                     feat.data_ptr(), p1, p2, B, cur_stream());
}

void conv_encode_gather(torch::Tensor ring, torch::Tensor env_state,
                        int64_t history, int64_t ring_head, torch::Tensor w1,
                        torch::Tensor b1, torch::Tensor w2, torch::Tensor b2,
                        torch::Tensor w3p, torch::Tensor b3,
                        torch::Tensor feat) {
    CHECK_T(ring, torch::kUInt8);
    CHECK_T(env_state, torch::kInt32);
    CHECK_BF16(w1); CHECK_BF16(w2); CHECK_BF16(w3p);
    CHECK_BF16(b1); CHECK_BF16(b2); CHECK_BF16(b3);
    CHECK_BF16(feat);
    TORCH_CHECK(history == 4, "fused frame-stack gather assumes history 4");
    int64_t E = env_state.numel() / 4;
    TORCH_CHECK(ring.numel() == history * E * 7056, "ring shape");
    TORCH_CHECK(feat.numel() == E * 3136);
    TORCH_CHECK(ring_head >= 0 && ring_head < history);
    riqn_conv_encode_gather(ring.data_ptr(), env_state.data_ptr(),
                            (int)ring_head, w1.data_ptr(), b1.data_ptr(),
                            w2.data_ptr(), b2.data_ptr(), w3p.data_ptr(),
                            b3.data_ptr(), feat.data_ptr(), (int)E,
                            cur_stream());
}

void heads_gemm_tau(torch::Tensor taus, torch::Tensor wiqnf,
                    torch::Tensor biqn, torch::Tensor feat,
                    torch::Tensor wcat, torch::Tensor bcat,
                    torch::Tensor out, int64_t nq) {
    CHECK_T(taus, torch::kFloat32);
    CHECK_BF16(wiqnf); CHECK_BF16(biqn); CHECK_BF16(feat);
    CHECK_BF16(wcat); CHECK_BF16(bcat); CHECK_BF16(out);
    int R = (int)taus.numel();
    TORCH_CHECK(wiqnf.numel() == 392 * 512 && biqn.numel() == 3136);
    TORCH_CHECK(feat.numel() * nq == (int64_t)R * 3136, "feat rows * nq != R");
    TORCH_CHECK(wcat.numel() == 1024 * 3136);
    TORCH_CHECK(out.numel() == (int64_t)R * 1024);
    riqn_heads_gemm_tau(taus.data_ptr<float>(), wiqnf.data_ptr(),
                        biqn.data_ptr(), feat.data_ptr(), wcat.data_ptr(),
                        bcat.data_ptr(), out.data_ptr(), R, (int)nq,
                        cur_stream());
}

void tau_embed(torch::Tensor taus, torch::Tensor w, torch::Tensor bias,
               torch::Tensor feat, torch::Tensor h, int64_t nq,
               c10::optional<torch::Tensor> phi_out = c10::nullopt,
               c10::optional<torch::Tensor> cosb_out = c10::nullopt,
               int64_t direct = -1) {  // -1 auto, 0 staged, 1 direct
    CHECK_T(taus, torch::kFloat32);
    CHECK_BF16(w); CHECK_BF16(bias); CHECK_BF16(feat); CHECK_BF16(h);
    int R = (int)taus.numel();
    TORCH_CHECK(h.numel() == (int64_t)R * 3136);
    TORCH_CHECK(w.numel() == 392 * 512);
    TORCH_CHECK(feat.numel() * nq == h.numel(), "feat rows * nq != R");
    void* pp = nullptr;
    void* cb = nullptr;
    if (phi_out.has_value()) {
        TORCH_CHECK(phi_out->numel() == h.numel());
        pp = phi_out->data_ptr();
    }
    if (cosb_out.has_value()) {
        TORCH_CHECK(cosb_out->numel() == (int64_t)R * 64);
        cb = cosb_out->data_ptr();
    }
    riqn_tau_embed(taus.data_ptr<float>(), w.data_ptr(), bias.data_ptr(),
                   feat.data_ptr(), h.data_ptr(), pp, cb, R, (int)nq,
                   (int)direct, cur_stream());
}

void heads_gemm(torch::Tensor h, torch::Tensor wcat, torch::Tensor bcat,
                torch::Tensor out, int64_t variant) {
    CHECK_BF16(h); CHECK_BF16(wcat); CHECK_BF16(bcat); CHECK_BF16(out);
    int R = (int)h.size(0);
    TORCH_CHECK(h.size(1) == 3136 && wcat.numel() == 1024 * 3136);
    TORCH_CHECK(out.numel() == (int64_t)R * 1024);
    riqn_heads_gemm(h.data_ptr(), wcat.data_ptr(), bcat.data_ptr(),
                    out.data_ptr(), R, (int)variant, cur_stream());
}

// mode 0: q_out [R,A] f32; mode 1: act_io out [R/nq]; mode 2: q_out [R] f32,
// act_io in [R/nq]
void heads_finish(torch::Tensor hh, torch::Tensor wz, torch::Tensor bz,
                  int64_t A, int64_t nq, int64_t mode, torch::Tensor q_out,
                  torch::Tensor act_io) {
    CHECK_BF16(hh); CHECK_BF16(wz); CHECK_BF16(bz);
    int R = (int)hh.size(0);
    TORCH_CHECK(hh.size(1) == 1024);
    TORCH_CHECK(wz.numel() == (1 + A) * 512 && A <= 18);
    TORCH_CHECK(q_out.scalar_type() == torch::kFloat32);
    TORCH_CHECK(act_io.scalar_type() == torch::kInt64);
    if (mode == 1 || mode == 2) TORCH_CHECK(act_io.numel() == R / nq);
    if (mode == 1) TORCH_CHECK(R % nq == 0);
    riqn_heads_finish(hh.data_ptr(), wz.data_ptr(), bz.data_ptr(), R, (int)A,
                      (int)nq, (int)mode, q_out.data_ptr<float>(),
                      act_io.data_ptr<int64_t>(), cur_stream());
}


// ---- hand-written learn-step backward (model_bwd.hip) ----------------------

void finish_bwd_dhh(torch::Tensor dtheta, torch::Tensor actions,
                    torch::Tensor hh, torch::Tensor wz, torch::Tensor dhh,
                    int64_t A, int64_t nq) {
    CHECK_T(dtheta, torch::kFloat32);
    CHECK_T(actions, torch::kInt64);
    CHECK_BF16(hh); CHECK_BF16(wz); CHECK_BF16(dhh);
    int R = (int)dtheta.numel();
    TORCH_CHECK(hh.numel() == (int64_t)R * 1024 && dhh.numel() == hh.numel());
    riqn_finish_bwd_dhh(dtheta.data_ptr<float>(),
                        actions.data_ptr<int64_t>(), hh.data_ptr(),
                        wz.data_ptr(), dhh.data_ptr(), R, (int)A, (int)nq,
                        cur_stream());
}

void zgrad(torch::Tensor dtheta, torch::Tensor actions, torch::Tensor hh,
           int64_t A, int64_t nq, std::vector<torch::Tensor> eps,
           std::vector<torch::Tensor> grads) {
    TORCH_CHECK(eps.size() == 4 && grads.size() == 8);
    int R = (int)dtheta.numel();
    riqn_zgrad(dtheta.data_ptr<float>(), actions.data_ptr<int64_t>(),
               hh.data_ptr(), R, (int)A, (int)nq, eps[0].data_ptr(),
               eps[1].data_ptr(), eps[2].data_ptr(), eps[3].data_ptr(),
               grads[0].data_ptr(), grads[1].data_ptr(), grads[2].data_ptr(),
               grads[3].data_ptr(), grads[4].data_ptr(), grads[5].data_ptr(),
               grads[6].data_ptr(), grads[7].data_ptr(), cur_stream());
}

void gemm_dx(torch::Tensor dhh, torch::Tensor wcat, torch::Tensor feat,
             torch::Tensor phi, torch::Tensor dphi, torch::Tensor dfeat,
             int64_t nq) {
    CHECK_BF16(dhh); CHECK_BF16(wcat); CHECK_BF16(feat); CHECK_BF16(phi);
    CHECK_BF16(dphi);
    CHECK_T(dfeat, torch::kFloat32);
    int R = (int)dhh.size(0);
    TORCH_CHECK(phi.numel() == (int64_t)R * 3136 && dphi.numel() == phi.numel());
    TORCH_CHECK(dfeat.numel() * nq == (int64_t)R * 3136);
    riqn_gemm_dx(dhh.data_ptr(), wcat.data_ptr(), feat.data_ptr(),
                 phi.data_ptr(), dphi.data_ptr(), dfeat.data_ptr<float>(), R,
                 (int)nq, cur_stream());
}

void gemm_dwcat(torch::Tensor dhh, torch::Tensor h, torch::Tensor weps_v,
                torch::Tensor weps_a, std::vector<torch::Tensor> grads) {
    TORCH_CHECK(grads.size() == 4);
    CHECK_BF16(dhh); CHECK_BF16(h);
    int R = (int)dhh.size(0);
    riqn_gemm_dwcat(dhh.data_ptr(), h.data_ptr(), weps_v.data_ptr(),
                    weps_a.data_ptr(), grads[0].data_ptr(),
                    grads[1].data_ptr(), grads[2].data_ptr(),
                    grads[3].data_ptr(), R, cur_stream());
}

void bias_grad_heads(torch::Tensor dhh, torch::Tensor beps_v,
                     torch::Tensor beps_a, std::vector<torch::Tensor> grads) {
    TORCH_CHECK(grads.size() == 4);
    int R = (int)dhh.size(0);
    riqn_bias_grad_heads(dhh.data_ptr(), R, beps_v.data_ptr(),
                         beps_a.data_ptr(), grads[0].data_ptr(),
                         grads[1].data_ptr(), grads[2].data_ptr(),
                         grads[3].data_ptr(), cur_stream());
}

void bias_grad_cols(torch::Tensor X, torch::Tensor db) {
    CHECK_BF16(X); CHECK_BF16(db);
    riqn_bias_grad_cols(X.data_ptr(), (int)X.size(0), (int)X.size(1),
                        db.data_ptr(), cur_stream());
}

void tau_dw(torch::Tensor dphi, torch::Tensor cosb, torch::Tensor dwiqn) {
    CHECK_BF16(dphi); CHECK_BF16(dwiqn); CHECK_BF16(cosb);
    TORCH_CHECK(dwiqn.numel() == 3136 * 64);
    int R = (int)cosb.size(0);
    TORCH_CHECK(R <= 256 && cosb.size(1) == 64, "tau_dw: R <= 256");
    riqn_tau_dw(dphi.data_ptr(), cosb.data_ptr(), dwiqn.data_ptr(), R,
                cur_stream());
}

void conv_bwd_data(torch::Tensor dfeat, torch::Tensor feat, torch::Tensor c1g,
                   torch::Tensor c2g, torch::Tensor w3tp, torch::Tensor w2tp,
                   torch::Tensor dpre3g, torch::Tensor dc2g,
                   torch::Tensor dc1g, torch::Tensor db1s, torch::Tensor db2s,
                   torch::Tensor db3s) {
    CHECK_T(dfeat, torch::kFloat32);
    int B = (int)dfeat.size(0);
    TORCH_CHECK(dpre3g.numel() == (int64_t)64 * B * 49);
    TORCH_CHECK(dc2g.numel() == (int64_t)64 * B * 81);
    TORCH_CHECK(dc1g.numel() == (int64_t)32 * B * 400);
    riqn_conv_bwd_data(dfeat.data_ptr<float>(), feat.data_ptr(),
                       c1g.data_ptr(), c2g.data_ptr(), w3tp.data_ptr(),
                       w2tp.data_ptr(), dpre3g.data_ptr(), dc2g.data_ptr(),
                       dc1g.data_ptr(), db1s.data_ptr<float>(),
                       db2s.data_ptr<float>(), db3s.data_ptr<float>(), B,
                       cur_stream());
}

void conv_bwd_dw(int64_t layer, torch::Tensor dY, torch::Tensor act,
                 torch::Tensor dWs, int64_t B, int64_t splitk) {
    CHECK_BF16(dY); CHECK_BF16(act);
    CHECK_T(dWs, torch::kFloat32);
    riqn_conv_bwd_dw((int)layer, dY.data_ptr(), act.data_ptr(),
                     dWs.data_ptr<float>(), (int)B, (int)splitk,
                     cur_stream());
}

void conv_finalize(std::vector<torch::Tensor> scr,
                   std::vector<torch::Tensor> grads) {
    TORCH_CHECK(scr.size() == 6 && grads.size() == 6);
    riqn_conv_finalize(scr[0].data_ptr<float>(), scr[1].data_ptr<float>(),
                       scr[2].data_ptr<float>(), scr[3].data_ptr<float>(),
                       scr[4].data_ptr<float>(), scr[5].data_ptr<float>(),
                       grads[0].data_ptr(), grads[1].data_ptr(),
                       grads[2].data_ptr(), grads[3].data_ptr(),
                       grads[4].data_ptr(), grads[5].data_ptr(),
                       cur_stream());
}

void pack_bwd_weights(torch::Tensor w3, torch::Tensor w2, torch::Tensor w3tp,
                      torch::Tensor w2tp) {
    CHECK_BF16(w3); CHECK_BF16(w2); CHECK_BF16(w3tp); CHECK_BF16(w2tp);
    TORCH_CHECK(w3tp.numel() == 64 * 1024 && w2tp.numel() == 32 * 1024);
    riqn_pack_bwd_weights(w3.data_ptr(), w2.data_ptr(), w3tp.data_ptr(),
                          w2tp.data_ptr(), cur_stream());
}


void quant_fp8(torch::Tensor src, torch::Tensor dst, torch::Tensor scales) {
    CHECK_BF16(src);
    CHECK_T(dst, torch::kUInt8);
    CHECK_T(scales, torch::kUInt8);
    int M = (int)src.size(0);
    TORCH_CHECK(src.size(1) == 3136 && dst.numel() == (int64_t)M * 3200 &&
                scales.numel() == (int64_t)M * 100);
    riqn_quant_fp8(src.data_ptr(), dst.data_ptr(), scales.data_ptr(), M,
                   cur_stream());
}

void tau_embed_fp8(torch::Tensor taus, torch::Tensor wiqnf,
                   torch::Tensor bias, torch::Tensor feat, torch::Tensor hq,
                   torch::Tensor hsc, int64_t nq) {
    CHECK_T(taus, torch::kFloat32);
    CHECK_BF16(wiqnf); CHECK_BF16(bias); CHECK_BF16(feat);
    CHECK_T(hq, torch::kUInt8); CHECK_T(hsc, torch::kUInt8);
    int R = (int)taus.numel();
    TORCH_CHECK(hq.size(0) == R && hq.size(1) == 3200);
    TORCH_CHECK(hsc.size(0) == R && hsc.size(1) == 100);
    TORCH_CHECK(R % (int)nq == 0);
    riqn_tau_embed_fp8(taus.data_ptr<float>(), wiqnf.data_ptr(),
                       bias.data_ptr(), feat.data_ptr(), hq.data_ptr(),
                       hsc.data_ptr(), R, (int)nq, cur_stream());
}

void heads_gemm_fp8(torch::Tensor hq, torch::Tensor hsc, torch::Tensor wq,
                    torch::Tensor wsc, torch::Tensor bcat, torch::Tensor out,
                    int64_t variant) {
    CHECK_T(hq, torch::kUInt8);
    CHECK_T(wq, torch::kUInt8);
    CHECK_BF16(bcat); CHECK_BF16(out);
    int R = (int)hq.size(0);
    TORCH_CHECK(hq.size(1) == 3200 && wq.numel() == 1024 * 3200);
    TORCH_CHECK(out.numel() == (int64_t)R * 1024);
    riqn_heads_gemm_fp8(hq.data_ptr(), hsc.data_ptr(), wq.data_ptr(),
                        wsc.data_ptr(), bcat.data_ptr(), out.data_ptr(), R,
                        (int)variant, cur_stream());
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
    m.def("replay_append", &replay_append);
    m.def("replay_sample", &replay_sample);
    m.def("replay_update_priorities", &replay_update_priorities);
    m.def("tree_rebuild", &tree_rebuild);
    m.def("quantile_huber", &quantile_huber);
    m.def("adam_step", &adam_step);
    m.def("preprocess", &preprocess);
    m.def("rand_uniform", &rand_uniform);
    m.def("noisy_reset", &noisy_reset);
    m.def("env_step", &env_step, py::arg("actions"), py::arg("env_state"),
          py::arg("rng_state"), py::arg("ring"), py::arg("history"),
          py::arg("ring_head_new"), py::arg("reward_clip"), py::arg("p_done"),
          py::arg("max_episode_frames"), py::arg("chunk_frames"),
          py::arg("chunk_meta"), py::arg("chunk_pos"), py::arg("task") = 0);
    m.def("create_cu_masked_stream", &create_cu_masked_stream);
    m.def("destroy_stream", &destroy_stream);
    m.def("adam_step_mixed", &adam_step_mixed);
    m.def("noisy_material", &noisy_material);
    m.def("noisy_material4", &noisy_material4);
    m.def("stack_gather", &stack_gather);
    m.def("pack_fwd_frag", &pack_fwd_frag);
    m.def("quant_fp8", &quant_fp8);
    m.def("heads_gemm_fp8", &heads_gemm_fp8, py::arg("hq"), py::arg("hsc"),
          py::arg("wq"), py::arg("wsc"), py::arg("bcat"), py::arg("out"),
          py::arg("variant") = -1);
    m.def("tau_embed_fp8", &tau_embed_fp8);
    m.def("conv_encode", &conv_encode, py::arg("x"), py::arg("w1"), py::arg("b1"), py::arg("w2"), py::arg("b2"), py::arg("w3p"), py::arg("b3"), py::arg("feat"), py::arg("c1g") = c10::nullopt, py::arg("c2g") = c10::nullopt);
    m.def("conv_encode_gather", &conv_encode_gather, py::arg("ring"), py::arg("env_state"), py::arg("history"), py::arg("ring_head"), py::arg("w1"), py::arg("b1"), py::arg("w2"), py::arg("b2"), py::arg("w3p"), py::arg("b3"), py::arg("feat"));
    m.def("heads_gemm_tau", &heads_gemm_tau, py::arg("taus"), py::arg("wiqnf"), py::arg("biqn"), py::arg("feat"), py::arg("wcat"), py::arg("bcat"), py::arg("out"), py::arg("nq"));
    m.def("tau_embed", &tau_embed, py::arg("taus"), py::arg("w"), py::arg("bias"), py::arg("feat"), py::arg("h"), py::arg("nq"), py::arg("phi_out") = c10::nullopt, py::arg("cosb_out") = c10::nullopt, py::arg("direct") = -1);
    m.def("heads_gemm", &heads_gemm, py::arg("h"), py::arg("wcat"),
          py::arg("bcat"), py::arg("out"), py::arg("variant") = -1);
    m.def("heads_finish", &heads_finish);
    m.def("finish_bwd_dhh", &finish_bwd_dhh);
    m.def("zgrad", &zgrad);
    m.def("gemm_dx", &gemm_dx);
    m.def("gemm_dwcat", &gemm_dwcat);
    m.def("bias_grad_heads", &bias_grad_heads);
    m.def("bias_grad_cols", &bias_grad_cols);
    m.def("tau_dw", &tau_dw);
    m.def("conv_bwd_data", &conv_bwd_data);
    m.def("conv_bwd_dw", &conv_bwd_dw);
    m.def("conv_finalize", &conv_finalize);
    m.def("pack_bwd_weights", &pack_bwd_weights);
}
