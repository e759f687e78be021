// C ABI between the torch bindings (bindings.cpp, built against torch headers)
// and the HIP kernels (*.hip, built by hipcc for gfx950 only).  Plain types
// keep torch headers out of device compilation.
#pragma once
#include <cstdint>

extern "C" {

// ---- replay (GPU prioritized sum-tree) ----------------------
// meta/ring_state layout documented in replay.hip.

void riqn_replay_append(
    // replay state
    uint8_t* frames, int32_t* actions, float* rewards, uint8_t* terminals,
    int32_t* timesteps, float* tree, int64_t tree_size, int64_t capacity,
    int32_t* ring_state, int n_rings, int64_t ring_cap, float* max_priority,
    // chunk
    const uint8_t* chunk_frames, const float* chunk_meta, int64_t chunk_len,
    int ring_id, void* stream);

void riqn_replay_sample(
    const uint8_t* frames, const int32_t* actions, const float* rewards,
    const uint8_t* terminals, const int32_t* timesteps, const float* tree,
    int64_t tree_size, int64_t capacity, const int32_t* ring_state, int n_rings,
    int64_t ring_cap, int history, int n_step, float discount,
    const float* beta,       // [1] device scalar (annealed outside the graph)
    uint64_t* rng_state,     // [2] device (seed, offset)
    int batch,
    // outputs
    int32_t* out_idx, float* out_states, float* out_next_states,
    int64_t* out_actions, float* out_returns, float* out_nonterminals,
    float* out_weights, int out_bf16, void* stream);

void riqn_replay_update_priorities(
    float* tree, int64_t tree_size, float* max_priority,
    const int32_t* idxs, const float* priorities, int64_t k,
    float priority_exponent, void* stream);

void riqn_tree_level_pass(float* tree, int64_t level_nodes, void* stream);

// ---- fused IQN quantile-Huber loss (K10+K11+K12 loss part) -----------------
// Computes loss, per-sample priorities, and d(loss)/d(theta) in one kernel.
void riqn_quantile_huber(
    const float* theta,      // [B, N] online quantiles (taken action)
    const float* target_q,   // [B, Np] target-net quantiles Z'(s', a*)
    const float* taus,       // [B, N]
    const float* returns,    // [B]
    const float* nonterm,    // [B]
    const float* weights,    // [B] or nullptr
    float gamma_n, float kappa, int B, int N, int Np,
    float* out_loss,         // [1] (mean over batch of weighted per-sample)
    float* out_per_sample,   // [B]
    float* out_grad_theta,   // [B, N]  d(out_loss)/d(theta)
    void* stream);

// ---- fused Adam over the flat parameter buffer (K13) -----------------------
void riqn_adam_step(
    float* param, const float* grad, float* exp_avg, float* exp_avg_sq,
    int64_t numel, float lr, float beta1, float beta2, float eps,
    int32_t* step,           // [1] device step counter (graph-safe)
    void* stream);

// mixed-precision Adam: fp32 master + bf16 grads -> fp32 master AND a bf16
// parameter copy (the module weights) updated in one pass
void riqn_adam_step_mixed(
    float* master, const uint16_t* grad_bf16, uint16_t* param_bf16,
    float* exp_avg, float* exp_avg_sq, int64_t numel, float lr, float beta1,
    float beta2, float eps, int32_t* step, void* stream);

// ---- frame preprocess: u8 -> f32/255 (K15) ---------------------------------
void riqn_preprocess(const uint8_t* in, float* out, int64_t numel, void* stream);

// ---- device counter RNG fills (K4/K7) --------------------------------------
void riqn_rand_uniform(float* out, int64_t n, uint64_t* rng_state, void* stream);
void riqn_noisy_reset(float* eps_in, int64_t n_in, float* eps_out,
                      int64_t n_out, float* weight_eps, float* bias_eps,
                      int eps_bf16, uint64_t* rng_state, void* stream);
void riqn_noisy_material(float* eps_in, int64_t n_in, float* eps_out,
                         int64_t n_out, const void* w_mu, const void* w_sigma,
                         const void* b_mu, const void* b_sigma,
                         void* weight_eps, void* bias_eps, void* w_noisy,
                         void* b_noisy, int bf16, uint64_t* rng_state,
                         void* stream);

// ---- fused RainbowIQN forward (model_fwd.hip) -------------
// All tensors bf16 unless noted.  Layouts documented in model_fwd.hip.
void riqn_pack_fwd_frag(const void* w1, const void* w2, const void* w3,
                        const void* wiqn, void* w1f, void* w2f, void* w3f,
                        void* wiqnf, void* stream);
void riqn_conv_encode(const void* x, const void* w1, const void* b1,
                      const void* w2, const void* b2, const void* w3p,
                      const void* b3, void* feat, void* c1g, void* c2g, int B,
                      void* stream);
void riqn_conv_encode_gather(const void* ring, const void* env_state,
                             int ring_head, const void* w1, const void* b1,
                             const void* w2, const void* b2, const void* w3p,
                             const void* b3, void* feat, int E, void* stream);
void riqn_tau_embed(const float* taus, const void* w, const void* bias,
                    const void* feat, void* h, void* phi_out, void* cosb_out,
                    int R, int nq, int direct, void* stream);
void riqn_heads_gemm(const void* h, const void* wcat, const void* bcat,
                     void* out, int R, int variant, void* stream);
void riqn_heads_gemm_tau(const float* taus, const void* wiqnf,
                         const void* biqn, const void* feat, const void* wcat,
                         const void* bcat, void* out, int R, int nq,
                         void* stream);
void riqn_heads_finish(const void* hh, const void* wz, const void* bz, int R,
                       int A, int nq, int mode, float* q_out, int64_t* act_io,
                       void* stream);


// ---- hand-written learn-step backward (model_bwd.hip) -------
void riqn_finish_bwd_dhh(const float* dtheta, const int64_t* actions,
                         const void* hh, const void* wz, void* dhh, int R,
                         int A, int nq, void* stream);
void riqn_zgrad(const float* dtheta, const int64_t* actions, const void* hh,
                int R, int A, int nq, const void* weps_v, const void* weps_a,
                const void* beps_v, const void* beps_a, void* dw_mu_v,
                void* dw_sg_v, void* db_mu_v, void* db_sg_v, void* dw_mu_a,
                void* dw_sg_a, void* db_mu_a, void* db_sg_a, void* stream);
void riqn_gemm_dx(const void* dhh, const void* wcat, const void* feat,
                  const void* phi, void* dphi, float* dfeat, int R, int nq,
                  void* stream);
void riqn_gemm_dwcat(const void* dhh, const void* h, const void* weps_v,
                     const void* weps_a, void* dw_mu_v, void* dw_sg_v,
                     void* dw_mu_a, void* dw_sg_a, int R, void* stream);
void riqn_bias_grad_heads(const void* dhh, int R, const void* beps_v,
                          const void* beps_a, void* db_mu_v, void* db_sg_v,
                          void* db_mu_a, void* db_sg_a, void* stream);
void riqn_bias_grad_cols(const void* X, int R, int C, void* db_mu,
                         void* stream);
void riqn_tau_dw(const void* dphi, const void* cosb, void* dwiqn, int R,
                 void* stream);
void riqn_conv_bwd_data(const float* dfeat, const void* feat, const void* c1g,
                        const void* c2g, const void* w3tp, const void* w2tp,
                        void* dpre3g, void* dc2g, void* dc1g, float* db1s,
                        float* db2s, float* db3s, int B, void* stream);
void riqn_conv_bwd_dw(int layer, const void* dY, const void* act, float* dWs,
                      int B, int splitk, void* stream);
void riqn_conv_finalize(float* dW1s, float* dW2s, float* dW3s, float* db1s,
                        float* db2s, float* db3s, void* dw1, void* dw2,
                        void* dw3, void* db1, void* db2, void* db3,
                        void* stream);
void riqn_pack_bwd_weights(const void* w3, const void* w2, void* w3tp,
                           void* w2tp, void* stream);

// ---- fp8 MX heads path (model_fp8.hip; BASELINE config 5) ------------------
void riqn_quant_fp8(const void* src, void* dst, void* scales, int M,
                    void* stream);
void riqn_tau_embed_fp8(const float* taus, const void* wiqnf,
                        const void* bias, const void* feat, void* hq,
                        void* hsc, int R, int nq, void* stream);
void riqn_heads_gemm_fp8(const void* hq, const void* hsc, const void* wq,
                         const void* wsc, const void* bcat, void* out, int R,
                         int variant, void* stream);

// batched 4-layer NoisyLinear regeneration (rng.hip)
struct NoisyDesc {
    const void *w_mu, *w_sigma, *b_mu, *b_sigma;
    void *w_eps, *b_eps, *w_noisy, *b_noisy;
    float *eps_in, *eps_out;
    int64_t n_in, n_out;
};
struct NoisyPack4 { NoisyDesc l[4]; };
void riqn_noisy_material4(NoisyPack4 pack, int bf16, uint64_t* rng_state,
                          void* stream);

// ---- device synthetic env (bench frame source; env.hip) --------------------
void riqn_env_step(const int64_t* actions, int32_t* env_state,
                   uint64_t* rng_state, uint8_t* ring, int history,
                   int ring_head_new, int n_envs, float reward_clip,
                   float p_done, int max_episode_frames, uint8_t* chunk_frames,
                   float* chunk_meta, int chunk_pos, int task, void* stream);
void riqn_stack_gather(const uint8_t* ring, const int32_t* env_state,
                       int history, int ring_head, int n_envs, float* out,
                       int out_bf16, void* stream);

}  // extern "C"
