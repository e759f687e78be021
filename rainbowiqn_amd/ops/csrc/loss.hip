// Fused IQN n-step quantile-Huber loss (SURVEY.md kernels K10 + K11 and the
// loss part of K12).
//
// Reference semantics (rainbowiqn/agent.py learn()):
//   T theta_j = R^(n) + gamma^n * nonterminal * Z'_tau'_j(s', a*)
//   delta_ij  = T theta_j - theta_i
//   rho       = |tau_i - 1{delta<0}| * Huber_kappa(delta) / kappa
//   per-sample loss = sum_i mean_j rho ;  total = mean_b w_b * per-sample
//
// One kernel computes the TD target, the pairwise loss, the per-sample
// priorities AND d(total)/d(theta) -- the autograd backward is then a single
// scale by the incoming gradient.  B blocks x 64 threads; N, Np <= 64.
#include "common.h"
#include "riqn_kernels.h"

namespace {

__global__ void quantile_huber_kernel(
    const float* __restrict__ theta, const float* __restrict__ target_q,
    const float* __restrict__ taus, const float* __restrict__ returns,
    const float* __restrict__ nonterm, const float* __restrict__ weights,
    float gamma_n, float kappa, int B, int N, int Np, float* out_loss,
    float* out_per_sample, float* out_grad_theta) {
    __shared__ float tt[64];  // T theta_j for this sample
    __shared__ float lds[16];
    int b = blockIdx.x;
    int i = threadIdx.x;
    float R = returns[b];
    float nt_g = gamma_n * nonterm[b];
    if (i < Np) tt[i] = R + nt_g * target_q[b * Np + i];
    __syncthreads();

    float lsum = 0.0f, gsum = 0.0f;
    float th = 0.0f, tau = 0.0f;
    if (i < N) {
        th = theta[b * N + i];
        tau = taus[b * N + i];
        float inv_kappa = 1.0f / kappa;
        for (int j = 0; j < Np; ++j) {
            float d = tt[j] - th;
            float ad = fabsf(d);
            float asym = fabsf(tau - (d < 0.0f ? 1.0f : 0.0f));
            float huber, dhuber;  // dhuber = d(huber)/d(d)
            if (ad <= kappa) {
                huber = 0.5f * d * d;
                dhuber = d;
            } else {
                huber = kappa * (ad - 0.5f * kappa);
                dhuber = kappa * (d < 0.0f ? -1.0f : 1.0f);
            }
            lsum += asym * huber * inv_kappa;
            gsum -= asym * dhuber * inv_kappa;  // d(d)/d(theta) = -1
        }
    }
    float w = (weights != nullptr) ? weights[b] : 1.0f;
    float inv_np = 1.0f / (float)Np;
    if (i < N)
        out_grad_theta[b * N + i] = w * gsum * inv_np / (float)B;
    float per = block_reduce_sum(lsum * inv_np, lds);
    if (threadIdx.x == 0) {
        out_per_sample[b] = per;
        atomicAdd(out_loss, w * per / (float)B);
    }
}

}  // namespace

extern "C" void riqn_quantile_huber(
    const float* theta, const float* target_q, const float* taus,
    const float* returns, const float* nonterm, const float* weights,
    float gamma_n, float kappa, int B, int N, int Np, float* out_loss,
    float* out_per_sample, float* out_grad_theta, void* stream) {
    hipStream_t s = (hipStream_t)stream;
    hipMemsetAsync(out_loss, 0, sizeof(float), s);
    int bt = ((max(N, Np) + 63) / 64) * 64;
    hipLaunchKernelGGL(quantile_huber_kernel, dim3(B), dim3(bt), 0, s, theta,
                       target_q, taus, returns, nonterm, weights, gamma_n,
                       kappa, B, N, Np, out_loss, out_per_sample,
                       out_grad_theta);
}
