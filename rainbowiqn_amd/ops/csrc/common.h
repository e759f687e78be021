// Shared device helpers for the CDNA4 (gfx950) kernels.
#pragma once
#include <hip/hip_runtime.h>
#include <cstdint>

#define DEV __device__ __forceinline__

// ---------------------------------------------------------------------------
// Philox4x32-10 counter RNG: deterministic, stateless per (seed, offset, tid)
// -- required inside hipGraph-captured kernels (no host RNG in the graph;
// SURVEY.md K4 note).
// ---------------------------------------------------------------------------
struct Philox {
    uint32_t k0, k1;
    uint32_t c0, c1, c2, c3;
    DEV Philox(uint64_t seed, uint64_t offset, uint32_t tid) {
        k0 = (uint32_t)seed;
        k1 = (uint32_t)(seed >> 32);
        c0 = (uint32_t)offset;
        c1 = (uint32_t)(offset >> 32);
        c2 = tid;
        c3 = 0x9E3779B9u;
    }
    DEV static void mulhilo(uint32_t a, uint32_t b, uint32_t& hi, uint32_t& lo) {
        uint64_t p = (uint64_t)a * b;
        hi = (uint32_t)(p >> 32);
        lo = (uint32_t)p;
    }
    DEV void round_() {
        uint32_t h0, l0, h1, l1;
        mulhilo(0xD2511F53u, c0, h0, l0);
        mulhilo(0xCD9E8D57u, c2, h1, l1);
        uint32_t n0 = h1 ^ c1 ^ k0, n1 = l1, n2 = h0 ^ c3 ^ k1, n3 = l0;
        c0 = n0; c1 = n1; c2 = n2; c3 = n3;
        k0 += 0x9E3779B9u; k1 += 0xBB67AE85u;
    }
    DEV void next4(uint32_t out[4]) {
        #pragma unroll
        for (int i = 0; i < 10; ++i) round_();
        out[0] = c0; out[1] = c1; out[2] = c2; out[3] = c3;
        c3 += 1;  // advance counter for subsequent draws
    }
    // uniform in [0, 1)
    DEV float uniform() {
        uint32_t r[4];
        next4(r);
        return (float)(r[0] >> 8) * (1.0f / 16777216.0f);
    }
    DEV void uniform4(float u[4]) {
        uint32_t r[4];
        next4(r);
        #pragma unroll
        for (int i = 0; i < 4; ++i) u[i] = (float)(r[i] >> 8) * (1.0f / 16777216.0f);
    }
    // standard normal via Box-Muller (pairs)
    DEV void normal2(float& z0, float& z1) {
        uint32_t r[4];
        next4(r);
        float u1 = ((float)(r[0] >> 8) + 0.5f) * (1.0f / 16777216.0f);
        float u2 = (float)(r[1] >> 8) * (1.0f / 16777216.0f);
        float rad = sqrtf(-2.0f * __logf(u1));
        float s, c;
        __sincosf(6.2831853071795864f * u2, &s, &c);
        z0 = rad * c;
        z1 = rad * s;
    }
};

// ---------------------------------------------------------------------------
// Wave/block reductions (wave = 64 lanes on CDNA4)
// ---------------------------------------------------------------------------
DEV float wave_reduce_sum(float v) {
    #pragma unroll
    for (int off = 32; off > 0; off >>= 1) v += __shfl_down(v, off, 64);
    return v;
}
// butterfly: every lane ends with the full sum
DEV float wave_allreduce_sum(float v) {
    #pragma unroll
    for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, 64);
    return v;
}
DEV float wave_reduce_max(float v) {
    #pragma unroll
    for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_down(v, off, 64));
    return v;
}

// block reduce for blockDim.x <= 1024, wave64; result valid on thread 0
template <int MAXWAVES = 16>
DEV float block_reduce_sum(float v, float* lds /* >= MAXWAVES floats */) {
    int lane = threadIdx.x & 63, wave = threadIdx.x >> 6;
    v = wave_reduce_sum(v);
    if (lane == 0) lds[wave] = v;
    __syncthreads();
    int nw = (blockDim.x + 63) >> 6;
    v = (threadIdx.x < nw) ? lds[threadIdx.x] : 0.0f;
    if (wave == 0) v = wave_reduce_sum(v);
    return v;
}
template <int MAXWAVES = 16>
DEV float block_reduce_max(float v, float* lds) {
    int lane = threadIdx.x & 63, wave = threadIdx.x >> 6;
    v = wave_reduce_max(v);
    if (lane == 0) lds[wave] = v;
    __syncthreads();
    int nw = (blockDim.x + 63) >> 6;
    v = (threadIdx.x < nw) ? lds[threadIdx.x] : -3.4e38f;
    if (wave == 0) v = wave_reduce_max(v);
    return v;
}

// f32 -> bf16 (round-to-nearest-even)
DEV unsigned short f32_to_bf16(float f) {
    unsigned int b = __float_as_uint(f);
    b += 0x7FFF + ((b >> 16) & 1);
    return (unsigned short)(b >> 16);
}
DEV float bf16_to_f32(unsigned short h) {
    unsigned int b = ((unsigned int)h) << 16;
    return __uint_as_float(b);
}
struct ushort4_t { unsigned short x, y, z, w; };
DEV ushort4_t f32x4_to_bf16x4(float a, float b, float c, float d) {
    ushort4_t r;
    r.x = f32_to_bf16(a); r.y = f32_to_bf16(b);
    r.z = f32_to_bf16(c); r.w = f32_to_bf16(d);
    return r;
}

// atomicMax for non-negative floats via ordered uint bit pattern
DEV void atomic_max_pos_float(float* addr, float val) {
    atomicMax((unsigned int*)addr, __float_as_uint(val));
}

#define HIP_CHECK(x)                                                          \
    do {                                                                      \
        hipError_t err__ = (x);                                               \
        if (err__ != hipSuccess) {                                            \
            printf("HIP error %s at %s:%d\n", hipGetErrorString(err__),       \
                   __FILE__, __LINE__);                                       \
        }                                                                     \
    } while (0)
