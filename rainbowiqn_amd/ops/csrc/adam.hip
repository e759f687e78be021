// Fused Adam over the flat parameter buffer (SURVEY.md kernel K13) + the u8
// preprocess kernel (K15).
//
// All of the model's ~6.8M fp32 parameters live in ONE contiguous buffer
// (module params are views -- ops/flat_params.py), so the whole optimizer
// step is a single grid-stride kernel: memory-bound, ~4 tensors x 27 MB
// through HBM3E.  The step counter is a device int so the kernel sequence
// (bump_step -> adam) is hipGraph-replayable with correct bias correction.
#include "common.h"
#include "riqn_kernels.h"

namespace {

__global__ void bump_step_kernel(int32_t* step) {
    if (threadIdx.x == 0 && blockIdx.x == 0) step[0] += 1;
}

__global__ void adam_kernel(float* __restrict__ param,
                            const float* __restrict__ grad,
                            float* __restrict__ m, float* __restrict__ v,
                            int64_t numel, float lr, float beta1, float beta2,
                            float eps, const int32_t* __restrict__ step) {
    int t = step[0];
    float bc1 = 1.0f - __powf(beta1, (float)t);
    float bc2 = 1.0f - __powf(beta2, (float)t);
    float step_size = lr / bc1;
    for (int64_t i = threadIdx.x + (int64_t)blockIdx.x * blockDim.x; i < numel;
         i += (int64_t)gridDim.x * blockDim.x) {
        float g = grad[i];
        float mi = beta1 * m[i] + (1.0f - beta1) * g;
        float vi = beta2 * v[i] + (1.0f - beta2) * g * g;
        m[i] = mi;
        v[i] = vi;
        param[i] -= step_size * mi / (sqrtf(vi / bc2) + eps);
    }
}

__global__ void adam_mixed_kernel(
    float* __restrict__ master, const unsigned short* __restrict__ grad_bf16,
    unsigned short* __restrict__ param_bf16, float* __restrict__ m,
    float* __restrict__ v, int64_t numel, float lr, float beta1, float beta2,
    float eps, const int32_t* __restrict__ step) {
    int t = step[0];
    float bc1 = 1.0f - __powf(beta1, (float)t);
    float bc2 = 1.0f - __powf(beta2, (float)t);
    float step_size = lr / bc1;
    // four params per iteration (float4 state, u16x4 grad/param): the
    // scalar loop was memory-ISSUE bound (8 insts per element; VALU is
    // trivial next to it).  Flat buffers are 16B-aligned allocations.
    typedef __attribute__((ext_vector_type(4))) float f4;
    typedef __attribute__((ext_vector_type(4))) unsigned short u16x4a;
    const int64_t n4 = numel / 4;
    for (int64_t i4 = threadIdx.x + (int64_t)blockIdx.x * blockDim.x;
         i4 < n4; i4 += (int64_t)gridDim.x * blockDim.x) {
        u16x4a gv = ((const u16x4a*)grad_bf16)[i4];
        f4 mv = ((const f4*)m)[i4];
        f4 vv = ((const f4*)v)[i4];
        f4 pv = ((const f4*)master)[i4];
        u16x4a ov;
        #pragma unroll
        for (int jj = 0; jj < 4; ++jj) {
            float g = bf16_to_f32(gv[jj]);
            float mi = beta1 * mv[jj] + (1.0f - beta1) * g;
            float vi = beta2 * vv[jj] + (1.0f - beta2) * g * g;
            mv[jj] = mi;
            vv[jj] = vi;
            float p = pv[jj] - step_size * mi / (sqrtf(vi / bc2) + eps);
            pv[jj] = p;
            ov[jj] = f32_to_bf16(p);
        }
        ((f4*)m)[i4] = mv;
        ((f4*)v)[i4] = vv;
        ((f4*)master)[i4] = pv;
        ((u16x4a*)param_bf16)[i4] = ov;
    }
    // tail
    for (int64_t i = n4 * 4 + threadIdx.x + (int64_t)blockIdx.x * blockDim.x;
         i < numel; i += (int64_t)gridDim.x * blockDim.x) {
        float g = bf16_to_f32(grad_bf16[i]);
        float mi = beta1 * m[i] + (1.0f - beta1) * g;
        float vi = beta2 * v[i] + (1.0f - beta2) * g * g;
        m[i] = mi;
        v[i] = vi;
        float p = master[i] - step_size * mi / (sqrtf(vi / bc2) + eps);
        master[i] = p;
        param_bf16[i] = f32_to_bf16(p);
    }
}

__global__ void preprocess_kernel(const uint8_t* __restrict__ in,
                                  float* __restrict__ out, int64_t numel) {
    const float inv255 = 1.0f / 255.0f;
    int64_t i4 = threadIdx.x + (int64_t)blockIdx.x * blockDim.x;
    int64_t n4 = numel / 4;
    const uchar4* in4 = (const uchar4*)in;
    for (int64_t i = i4; i < n4; i += (int64_t)gridDim.x * blockDim.x) {
        uchar4 u = in4[i];
        float4 f = make_float4(u.x * inv255, u.y * inv255, u.z * inv255,
                               u.w * inv255);
        ((float4*)out)[i] = f;
    }
    // tail
    if (i4 == 0)
        for (int64_t i = n4 * 4; i < numel; ++i) out[i] = in[i] * inv255;
}

}  // namespace

extern "C" {

void riqn_adam_step(float* param, const float* grad, float* exp_avg,
                    float* exp_avg_sq, int64_t numel, float lr, float beta1,
                    float beta2, float eps, int32_t* step, void* stream) {
    hipStream_t s = (hipStream_t)stream;
    hipLaunchKernelGGL(bump_step_kernel, dim3(1), dim3(64), 0, s, step);
    int blocks = (int)min((numel + 255) / 256, (int64_t)2048);
    hipLaunchKernelGGL(adam_kernel, dim3(blocks), dim3(256), 0, s, param, grad,
                       exp_avg, exp_avg_sq, numel, lr, beta1, beta2, eps, step);
}

void riqn_adam_step_mixed(float* master, const uint16_t* grad_bf16,
                          uint16_t* param_bf16, float* exp_avg,
                          float* exp_avg_sq, int64_t numel, float lr,
                          float beta1, float beta2, float eps, int32_t* step,
                          void* stream) {
    hipStream_t s = (hipStream_t)stream;
    hipLaunchKernelGGL(bump_step_kernel, dim3(1), dim3(64), 0, s, step);
    int blocks = (int)min((numel + 255) / 256, (int64_t)2048);
    hipLaunchKernelGGL(adam_mixed_kernel, dim3(blocks), dim3(256), 0, s,
                       master, (const unsigned short*)grad_bf16,
                       (unsigned short*)param_bf16, exp_avg, exp_avg_sq, numel,
                       lr, beta1, beta2, eps, step);
}

void riqn_preprocess(const uint8_t* in, float* out, int64_t numel,
                     void* stream) {
    hipStream_t s = (hipStream_t)stream;
    int blocks = (int)min((numel / 4 + 255) / 256, (int64_t)2048);
    hipLaunchKernelGGL(preprocess_kernel, dim3(max(blocks, 1)), dim3(256), 0, s,
                       in, out, numel);
}

}  // extern "C"
