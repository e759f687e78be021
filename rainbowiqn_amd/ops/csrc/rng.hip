// Device-side counter RNG fills: uniform tau draws and
// factorised NoisyLinear epsilon regeneration, all hipGraph-replayable (the
// philox offset lives in device memory and is bumped by the kernels).
#include "common.h"
#include "riqn_kernels.h"

namespace {

// Philox-offset advance discipline: a kernel that READS rng_state[1] must
// never bump it from one of its own blocks -- with more blocks than resident
// occupancy a late block can observe the bumped value and duplicate the next
// launch's counter stream.  The bump therefore always rides
// a LATER kernel in stream order: the second kernel of a fill pair, or this
// dedicated 1-thread kernel appended by the launcher.
__global__ void rng_bump_kernel(uint64_t* rng_state) { rng_state[1] += 1; }

__global__ void rand_uniform_kernel(float* __restrict__ out, int64_t n,
                                    uint64_t* rng_state) {
    uint64_t seed = rng_state[0], off = rng_state[1];
    int64_t tid = threadIdx.x + (int64_t)blockIdx.x * blockDim.x;
    Philox rng(seed, off, (uint32_t)tid);
    float u[4];
    rng.uniform4(u);
    int64_t base = tid * 4;
    #pragma unroll
    for (int i = 0; i < 4; ++i)
        if (base + i < n) out[base + i] = u[i];
}

// f(x) = sign(x) * sqrt(|x|) of a standard normal
DEV float scale_noise(float z) { return copysignf(sqrtf(fabsf(z)), z); }

__global__ void noisy_eps_vec_kernel(float* __restrict__ eps_in, int64_t n_in,
                                     float* __restrict__ eps_out,
                                     int64_t n_out, uint64_t* rng_state) {
    uint64_t seed = rng_state[0], off = rng_state[1];
    int64_t tid = threadIdx.x + (int64_t)blockIdx.x * blockDim.x;
    Philox rng(seed, off, (uint32_t)(tid + 0x10000));
    float z0, z1;
    rng.normal2(z0, z1);
    int64_t i = tid * 2;
    float* buf;
    // first n_in/2 thread-pairs fill eps_in, rest fill eps_out
    if (i < n_in) {
        if (i < n_in) eps_in[i] = scale_noise(z0);
        if (i + 1 < n_in) eps_in[i + 1] = scale_noise(z1);
    } else {
        int64_t j = i - n_in;
        if (j < n_out) eps_out[j] = scale_noise(z0);
        if (j + 1 < n_out) eps_out[j + 1] = scale_noise(z1);
    }
    (void)buf;
}

template <int OUT_BF16>
__global__ void noisy_eps_outer_kernel(const float* __restrict__ eps_in,
                                       int64_t n_in,
                                       const float* __restrict__ eps_out,
                                       int64_t n_out,
                                       float* __restrict__ weight_eps,
                                       float* __restrict__ bias_eps,
                                       uint64_t* rng_state) {
    // advances the offset CONSUMED by the preceding eps-vec kernel (stream
    // order makes this race-free; see rng_bump_kernel comment)
    if (threadIdx.x == 0 && blockIdx.x == 0) rng_state[1] += 1;
    for (int64_t i = threadIdx.x + (int64_t)blockIdx.x * blockDim.x;
         i < n_in * n_out; i += (int64_t)gridDim.x * blockDim.x) {
        float v = eps_out[i / n_in] * eps_in[i % n_in];
        if (OUT_BF16)
            ((unsigned short*)(void*)weight_eps)[i] = f32_to_bf16(v);
        else
            weight_eps[i] = v;
    }
    int64_t t = threadIdx.x + (int64_t)blockIdx.x * blockDim.x;
    if (t < n_out) {
        if (OUT_BF16)
            ((unsigned short*)(void*)bias_eps)[t] = f32_to_bf16(eps_out[t]);
        else
            bias_eps[t] = eps_out[t];
    }
}

// eps outer product AND the materialized noisy weight w = mu + sigma*eps in
// one pass (K6/K7 fused): the per-forward w/b materialization kernels vanish
// from the learn/act graphs -- noise (and thus w_noisy) changes once per step.
template <int BF16>
__global__ void noisy_material_kernel(
    const float* __restrict__ eps_in, int64_t n_in,
    const float* __restrict__ eps_out, int64_t n_out,
    const void* __restrict__ w_mu, const void* __restrict__ w_sigma,
    const void* __restrict__ b_mu, const void* __restrict__ b_sigma,
    void* __restrict__ weight_eps, void* __restrict__ bias_eps,
    void* __restrict__ w_noisy, void* __restrict__ b_noisy,
    uint64_t* rng_state) {
    if (threadIdx.x == 0 && blockIdx.x == 0) rng_state[1] += 1;
    for (int64_t i = threadIdx.x + (int64_t)blockIdx.x * blockDim.x;
         i < n_in * n_out; i += (int64_t)gridDim.x * blockDim.x) {
        float e = eps_out[i / n_in] * eps_in[i % n_in];
        float mu, sg;
        if (BF16) {
            mu = bf16_to_f32(((const unsigned short*)w_mu)[i]);
            sg = bf16_to_f32(((const unsigned short*)w_sigma)[i]);
            ((unsigned short*)weight_eps)[i] = f32_to_bf16(e);
            ((unsigned short*)w_noisy)[i] = f32_to_bf16(mu + sg * e);
        } else {
            mu = ((const float*)w_mu)[i];
            sg = ((const float*)w_sigma)[i];
            ((float*)weight_eps)[i] = e;
            ((float*)w_noisy)[i] = mu + sg * e;
        }
    }
    int64_t t = threadIdx.x + (int64_t)blockIdx.x * blockDim.x;
    if (t < n_out) {
        float e = eps_out[t];
        if (BF16) {
            float mu = bf16_to_f32(((const unsigned short*)b_mu)[t]);
            float sg = bf16_to_f32(((const unsigned short*)b_sigma)[t]);
            ((unsigned short*)bias_eps)[t] = f32_to_bf16(e);
            ((unsigned short*)b_noisy)[t] = f32_to_bf16(mu + sg * e);
        } else {
            float mu = ((const float*)b_mu)[t];
            float sg = ((const float*)b_sigma)[t];
            ((float*)bias_eps)[t] = e;
            ((float*)b_noisy)[t] = mu + sg * e;
        }
    }
}

// ---------------------------------------------------------------------------
// Batched noise regeneration: ALL FOUR NoisyLinear layers of one net in two
// kernels (eps vectors, then outer-product + materialized weights) instead
// of 16 launches per step.
// ---------------------------------------------------------------------------

__global__ void noisy_eps_vec4_kernel(NoisyPack4 p, uint64_t* rng_state) {
    uint64_t seed = rng_state[0], off = rng_state[1];
    int64_t tid = threadIdx.x + (int64_t)blockIdx.x * blockDim.x;
    Philox rng(seed, off, (uint32_t)(tid + 0x20000));
    float z0, z1;
    rng.normal2(z0, z1);
    int64_t i = tid * 2;
    // linear index space: [l0.in | l0.out | l1.in | l1.out | ...]
    #pragma unroll
    for (int L = 0; L < 4; ++L) {
        int64_t nin = p.l[L].n_in, nout = p.l[L].n_out;
        if (i < nin) {
            p.l[L].eps_in[i] = scale_noise(z0);
            if (i + 1 < nin) p.l[L].eps_in[i + 1] = scale_noise(z1);
            break;
        }
        i -= nin;
        if (i < nout) {
            p.l[L].eps_out[i] = scale_noise(z0);
            if (i + 1 < nout) p.l[L].eps_out[i + 1] = scale_noise(z1);
            break;
        }
        i -= nout;
    }
}

template <int BF16>
__global__ void noisy_material4_kernel(NoisyPack4 p, uint64_t* rng_state) {
    if (threadIdx.x == 0 && blockIdx.x == 0) rng_state[1] += 1;
    int64_t i = threadIdx.x + (int64_t)blockIdx.x * blockDim.x;
    const int64_t stride = (int64_t)gridDim.x * blockDim.x;
    // weight segments: TWO adjacent elements per iteration on the bf16
    // path (n_in is even for every layer, so a pair never straddles a
    // row: one div, one u32 load/store pair instead of two of each)
    for (int L = 0; L < 4; ++L) {
        const NoisyDesc& d = p.l[L];
        const int64_t tot = d.n_in * d.n_out;
        const bool aligned4 =
            ((((uintptr_t)d.w_mu) | ((uintptr_t)d.w_sigma) |
              ((uintptr_t)d.w_eps) | ((uintptr_t)d.w_noisy)) & 3) == 0;
        if (BF16 && (d.n_in & 1) == 0 && aligned4) {
            const int64_t tot2 = tot / 2;
            for (int64_t k2 = i; k2 < tot2; k2 += stride) {
                const int64_t k = k2 * 2;
                const float eo = d.eps_out[k / d.n_in];
                const int64_t col = k % d.n_in;
                const float e0 = eo * d.eps_in[col];
                const float e1 = eo * d.eps_in[col + 1];
                const uint32_t muw = ((const uint32_t*)d.w_mu)[k2];
                const uint32_t sgw = ((const uint32_t*)d.w_sigma)[k2];
                const float mu0 = bf16_to_f32((unsigned short)muw);
                const float mu1 = bf16_to_f32((unsigned short)(muw >> 16));
                const float sg0 = bf16_to_f32((unsigned short)sgw);
                const float sg1 = bf16_to_f32((unsigned short)(sgw >> 16));
                ((uint32_t*)d.w_eps)[k2] =
                    (uint32_t)f32_to_bf16(e0) |
                    ((uint32_t)f32_to_bf16(e1) << 16);
                ((uint32_t*)d.w_noisy)[k2] =
                    (uint32_t)f32_to_bf16(mu0 + sg0 * e0) |
                    ((uint32_t)f32_to_bf16(mu1 + sg1 * e1) << 16);
            }
            continue;
        }
        for (int64_t k = i; k < tot; k += stride) {
            float e = d.eps_out[k / d.n_in] * d.eps_in[k % d.n_in];
            if (BF16) {
                float mu = bf16_to_f32(((const unsigned short*)d.w_mu)[k]);
                float sg = bf16_to_f32(((const unsigned short*)d.w_sigma)[k]);
                ((unsigned short*)d.w_eps)[k] = f32_to_bf16(e);
                ((unsigned short*)d.w_noisy)[k] = f32_to_bf16(mu + sg * e);
            } else {
                float mu = ((const float*)d.w_mu)[k];
                float sg = ((const float*)d.w_sigma)[k];
                ((float*)d.w_eps)[k] = e;
                ((float*)d.w_noisy)[k] = mu + sg * e;
            }
        }
        // bias segment
        for (int64_t k = i; k < d.n_out; k += stride) {
            float e = d.eps_out[k];
            if (BF16) {
                float mu = bf16_to_f32(((const unsigned short*)d.b_mu)[k]);
                float sg = bf16_to_f32(((const unsigned short*)d.b_sigma)[k]);
                ((unsigned short*)d.b_eps)[k] = f32_to_bf16(e);
                ((unsigned short*)d.b_noisy)[k] = f32_to_bf16(mu + sg * e);
            } else {
                float mu = ((const float*)d.b_mu)[k];
                float sg = ((const float*)d.b_sigma)[k];
                ((float*)d.b_eps)[k] = e;
                ((float*)d.b_noisy)[k] = mu + sg * e;
            }
        }
    }
}

}  // namespace

extern "C" {

void riqn_noisy_material4(NoisyPack4 pack, int bf16, uint64_t* rng_state,
                          void* stream) {
    hipStream_t s = (hipStream_t)stream;
    int64_t vec_total = 0;
    for (int L = 0; L < 4; ++L)
        vec_total += pack.l[L].n_in + pack.l[L].n_out;
    int64_t pairs = (vec_total + 1) / 2;
    hipLaunchKernelGGL(noisy_eps_vec4_kernel,
                       dim3((uint32_t)((pairs + 255) / 256)), dim3(256), 0, s,
                       pack, rng_state);
    if (bf16)
        hipLaunchKernelGGL(noisy_material4_kernel<1>, dim3(2048), dim3(256),
                           0, s, pack, rng_state);
    else
        hipLaunchKernelGGL(noisy_material4_kernel<0>, dim3(2048), dim3(256),
                           0, s, pack, rng_state);
}

void riqn_rand_uniform(float* out, int64_t n, uint64_t* rng_state,
                       void* stream) {
    hipStream_t s = (hipStream_t)stream;
    int64_t threads = (n + 3) / 4;
    int blocks = (int)min((threads + 255) / 256, (int64_t)2048);
    hipLaunchKernelGGL(rand_uniform_kernel, dim3(max(blocks, 1)), dim3(256), 0,
                       s, out, n, rng_state);
    hipLaunchKernelGGL(rng_bump_kernel, dim3(1), dim3(1), 0, s, rng_state);
}

void riqn_noisy_reset(float* eps_in, int64_t n_in, float* eps_out,
                      int64_t n_out, float* weight_eps, float* bias_eps,
                      int eps_bf16, uint64_t* rng_state, void* stream) {
    hipStream_t s = (hipStream_t)stream;
    int64_t pairs = (n_in + n_out + 1) / 2;
    int blocks = (int)((pairs + 255) / 256);
    hipLaunchKernelGGL(noisy_eps_vec_kernel, dim3(max(blocks, 1)), dim3(256),
                       0, s, eps_in, n_in, eps_out, n_out, rng_state);
    int64_t total = n_in * n_out;
    int blocks2 = (int)min((total + 255) / 256, (int64_t)2048);
    if (eps_bf16)
        hipLaunchKernelGGL(noisy_eps_outer_kernel<1>, dim3(blocks2), dim3(256),
                           0, s, eps_in, n_in, eps_out, n_out, weight_eps,
                           bias_eps, rng_state);
    else
        hipLaunchKernelGGL(noisy_eps_outer_kernel<0>, dim3(blocks2), dim3(256),
                           0, s, eps_in, n_in, eps_out, n_out, weight_eps,
                           bias_eps, rng_state);
}

void riqn_noisy_material(float* eps_in, int64_t n_in, float* eps_out,
                         int64_t n_out, const void* w_mu, const void* w_sigma,
                         const void* b_mu, const void* b_sigma,
                         void* weight_eps, void* bias_eps, void* w_noisy,
                         void* b_noisy, int bf16, uint64_t* rng_state,
                         void* stream) {
    hipStream_t s = (hipStream_t)stream;
    int64_t pairs = (n_in + n_out + 1) / 2;
    int blocks = (int)((pairs + 255) / 256);
    hipLaunchKernelGGL(noisy_eps_vec_kernel, dim3(max(blocks, 1)), dim3(256),
                       0, s, eps_in, n_in, eps_out, n_out, rng_state);
    int64_t total = n_in * n_out;
    int blocks2 = (int)min((total + 255) / 256, (int64_t)2048);
    if (bf16)
        hipLaunchKernelGGL(noisy_material_kernel<1>, dim3(blocks2), dim3(256),
                           0, s, eps_in, n_in, eps_out, n_out, w_mu, w_sigma,
                           b_mu, b_sigma, weight_eps, bias_eps, w_noisy,
                           b_noisy, rng_state);
    else
        hipLaunchKernelGGL(noisy_material_kernel<0>, dim3(blocks2), dim3(256),
                           0, s, eps_in, n_in, eps_out, n_out, w_mu, w_sigma,
                           b_mu, b_sigma, weight_eps, bias_eps, w_noisy,
                           b_noisy, rng_state);
}

}  // extern "C"
