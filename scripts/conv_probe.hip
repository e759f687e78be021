// Standalone stage-wise probe of the fused conv encoder: times each stage
// combination to localize the per-WG latency.  Build (CPU box, cross):
//   hipcc --offload-arch=gfx950 -O3 -std=c++17 scripts/conv_probe.hip \
//         -o scripts/conv_probe
// Run on the GPU box: ./scripts/conv_probe [B]
#include <hip/hip_runtime.h>
#include <cstdio>
#include <cstdlib>
#include "../rainbowiqn_amd/ops/csrc/common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(8))) unsigned short u16x8;
typedef __attribute__((ext_vector_type(4))) unsigned short u16x4;
typedef __attribute__((ext_vector_type(4))) float f32x4;
DEV f32x4 mfma16(bf16x8 a, bf16x8 b, f32x4 c) {
    return __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
}
DEV bf16x8 as_bf16x8(u16x8 v) { return __builtin_bit_cast(bf16x8, v); }
DEV float bfc(const unsigned short u) { return bf16_to_f32(u); }

#define IN_STRIDE 88
#define C1_STRIDE 24
#define C2_STRIDE 12
#define LDS_IN 0
#define LDS_C1 29568
#define LDS_C2 (29568 + 15360)
#define LDS_TOTAL (29568 + 15360 + 6912)

// STAGES bitmask: 1=input stage, 2=conv1, 4=conv2, 8=conv3
template <int STAGES>
__global__ __launch_bounds__(512, 1) void probe_kernel(
    const unsigned short* __restrict__ x, const unsigned short* __restrict__ w1,
    const unsigned short* __restrict__ b1, const unsigned short* __restrict__ w2,
    const unsigned short* __restrict__ b2, const unsigned short* __restrict__ w3p,
    const unsigned short* __restrict__ b3, unsigned short* __restrict__ feat) {
    __shared__ unsigned short lds[LDS_TOTAL];
    const int b = blockIdx.x;
    const int t = threadIdx.x;
    const int lane = t & 63;
    const int wave = t >> 6;
    const int l15 = lane & 15;
    const int l4 = lane >> 4;

    if (STAGES & 1) {
        const unsigned short* src = x + (int64_t)b * (4 * 84 * 84);
        u16x4 tmp[14];
        #pragma unroll
        for (int i = 0; i < 14; ++i) {
            int e = t * 4 + i * 2048;
            tmp[i] = (e < 4 * 84 * 84) ? *(const u16x4*)(src + e)
                                       : u16x4{0, 0, 0, 0};
        }
        #pragma unroll
        for (int i = 0; i < 14; ++i) {
            int e = t * 4 + i * 2048;
            if (e < 4 * 84 * 84) {
                int c = e / 7056, r = (e % 7056) / 84, cc = e % 84;
                *(u16x4*)&lds[LDS_IN + (c * 84 + r) * IN_STRIDE + cc] = tmp[i];
            }
        }
    }
    __syncthreads();

    if (STAGES & 2) {
        bf16x8 b1f[2][8];
        #pragma unroll
        for (int nt = 0; nt < 2; ++nt)
            #pragma unroll
            for (int ks = 0; ks < 8; ++ks)
                b1f[nt][ks] = as_bf16x8(*(const u16x8*)(
                    w1 + ((ks * 2 + nt) << 9) + lane * 8));
        for (int mt = wave; mt < 25; mt += 8) {
            f32x4 acc[2] = {f32x4{0, 0, 0, 0}, f32x4{0, 0, 0, 0}};
            const int m = mt * 16 + l15;
            const int oy = m / 20, ox = m % 20;
            #pragma unroll
            for (int ks = 0; ks < 8; ++ks) {
                const int k0 = ks * 32 + l4 * 8;
                const int c = k0 >> 6, ky = (k0 >> 3) & 7;
                u16x8 av;
                {
                    const int idx =
                        LDS_IN + (c * 84 + oy * 4 + ky) * IN_STRIDE + ox * 4;
                    u16x4 lo = *(const u16x4*)&lds[idx];
                    u16x4 hi = *(const u16x4*)&lds[idx + 4];
                    av = u16x8{lo.x, lo.y, lo.z, lo.w, hi.x, hi.y, hi.z, hi.w};
                }
                bf16x8 a = as_bf16x8(av);
                acc[0] = mfma16(a, b1f[0][ks], acc[0]);
                acc[1] = mfma16(a, b1f[1][ks], acc[1]);
            }
            #pragma unroll
            for (int nt = 0; nt < 2; ++nt) {
                const int n = nt * 16 + l15;
                const float bias = bfc(b1[n]);
                #pragma unroll
                for (int j = 0; j < 4; ++j) {
                    const int mm = mt * 16 + l4 * 4 + j;
                    if (mm < 400) {
                        unsigned short o =
                            f32_to_bf16(fmaxf(acc[nt][j] + bias, 0.0f));
                        lds[LDS_C1 + (n * 20 + mm / 20) * C1_STRIDE + mm % 20] = o;
                    }
                }
            }
        }
    }
    __syncthreads();

    if (STAGES & 4) {
        for (int u = wave; u < 12; u += 8) {
            const int mt = u >> 1, nh = u & 1;
            f32x4 acc[2] = {f32x4{0, 0, 0, 0}, f32x4{0, 0, 0, 0}};
            const int m = mt * 16 + l15;
            const int oy = m / 9, ox = m % 9;
            const bool mv = m < 81;
            bf16x8 bbuf[2][2];
            #pragma unroll
            for (int nt = 0; nt < 2; ++nt)
                bbuf[0][nt] = as_bf16x8(*(const u16x8*)(
                    w2 + ((nh * 2 + nt) << 9) + lane * 8));
            #pragma unroll 2
            for (int ks = 0; ks < 16; ++ks) {
                const int cur = ks & 1;
                if (ks + 1 < 16) {
                    const int k1 = (ks + 1) * 32 + l4 * 8;
                    #pragma unroll
                    for (int nt = 0; nt < 2; ++nt)
                        bbuf[cur ^ 1][nt] = as_bf16x8(*(const u16x8*)(
                            w2 + (((ks + 1) * 4 + nh * 2 + nt) << 9) + lane * 8));
                }
                const int k0 = ks * 32 + l4 * 8;
                const int c = k0 >> 4, t16 = k0 & 15;
                const int ky0 = t16 >> 2;
                u16x8 av;
                {
                    const int base = mv ? LDS_C1 +
                        (c * 20 + oy * 2 + ky0) * C1_STRIDE + ox * 2 : LDS_C1;
                    #pragma unroll
                    for (int j = 0; j < 4; ++j) av[j] = lds[base + j];
                    #pragma unroll
                    for (int j = 0; j < 4; ++j)
                        av[4 + j] = lds[base + C1_STRIDE + j];
                }
                bf16x8 a = as_bf16x8(av);
                acc[0] = mfma16(a, bbuf[cur][0], acc[0]);
                acc[1] = mfma16(a, bbuf[cur][1], acc[1]);
            }
            #pragma unroll
            for (int nt = 0; nt < 2; ++nt) {
                const int n = (nh * 2 + nt) * 16 + l15;
                const float bias = bfc(b2[n]);
                #pragma unroll
                for (int j = 0; j < 4; ++j) {
                    const int mm = mt * 16 + l4 * 4 + j;
                    if (mm < 81) {
                        unsigned short o =
                            f32_to_bf16(fmaxf(acc[nt][j] + bias, 0.0f));
                        lds[LDS_C2 + (n * 9 + mm / 9) * C2_STRIDE + mm % 9] = o;
                    }
                }
            }
        }
    }
    __syncthreads();

    if (STAGES & 8) {
        const int mt = wave >> 1, nh = wave & 1;
        f32x4 acc[2] = {f32x4{0, 0, 0, 0}, f32x4{0, 0, 0, 0}};
        const int m = mt * 16 + l15;
        const int oy = m / 7, ox = m % 7;
        const bool mv = m < 49;
        bf16x8 bbuf[2][2];
        #pragma unroll
        for (int nt = 0; nt < 2; ++nt)
            bbuf[0][nt] = as_bf16x8(*(const u16x8*)(
                w3p + ((nh * 2 + nt) << 9) + lane * 8));
        #pragma unroll 2
        for (int ks = 0; ks < 32; ++ks) {
            const int cur = ks & 1;
            if (ks + 1 < 32) {
                const int k1 = (ks + 1) * 32 + l4 * 8;
                #pragma unroll
                for (int nt = 0; nt < 2; ++nt)
                    bbuf[cur ^ 1][nt] = as_bf16x8(*(const u16x8*)(
                        w3p + (((ks + 1) * 4 + nh * 2 + nt) << 9) + lane * 8));
            }
            const int k0 = ks * 32 + l4 * 8;
            const int c = k0 >> 4, t16 = k0 & 15;
            u16x8 av;
            #pragma unroll
            for (int j = 0; j < 8; ++j) {
                const int tt = t16 + j, ky = tt >> 2, kx = tt & 3;
                const bool ok = mv && ky < 3 && kx < 3;
                const int idx = LDS_C2 + (c * 9 + (ok ? oy + ky : 0)) * C2_STRIDE +
                                (ok ? ox + kx : 0);
                unsigned short v = lds[idx];
                av[j] = ok ? v : (unsigned short)0;
            }
            bf16x8 a = as_bf16x8(av);
            acc[0] = mfma16(a, bbuf[cur][0], acc[0]);
            acc[1] = mfma16(a, bbuf[cur][1], acc[1]);
        }
        unsigned short* out = feat + (int64_t)b * 3136;
        #pragma unroll
        for (int nt = 0; nt < 2; ++nt) {
            const int n = (nh * 2 + nt) * 16 + l15;
            const float bias = bfc(b3[n]);
            #pragma unroll
            for (int j = 0; j < 4; ++j) {
                const int mm = mt * 16 + l4 * 4 + j;
                if (mm < 49)
                    out[n * 49 + mm] =
                        f32_to_bf16(fmaxf(acc[nt][j] + bias, 0.0f));
            }
        }
    } else if (t == 0 && b == 0) {
        feat[0] = lds[0];  // keep LDS alive
    }
}

template <int STAGES>
float run(int B, unsigned short* x, unsigned short* w1, unsigned short* b1,
          unsigned short* w2, unsigned short* b2, unsigned short* w3p,
          unsigned short* b3, unsigned short* feat, int iters) {
    for (int i = 0; i < 20; ++i)
        hipLaunchKernelGGL(probe_kernel<STAGES>, dim3(B), dim3(512), 0, 0, x,
                           w1, b1, w2, b2, w3p, b3, feat);
    hipEvent_t e0, e1;
    hipEventCreate(&e0);
    hipEventCreate(&e1);
    hipEventRecord(e0);
    for (int i = 0; i < iters; ++i)
        hipLaunchKernelGGL(probe_kernel<STAGES>, dim3(B), dim3(512), 0, 0, x,
                           w1, b1, w2, b2, w3p, b3, feat);
    hipEventRecord(e1);
    hipEventSynchronize(e1);
    float ms;
    hipEventElapsedTime(&ms, e0, e1);
    return ms * 1000.0f / iters;
}


// ---------------------------------------------------------------------------
// conv3 cost isolation: V bit0 = real A-gathers, bit1 = real B weight loads,
// bit2 = stage weights in LDS first and read B-frags from LDS
// ---------------------------------------------------------------------------
template <int V>
__global__ __launch_bounds__(512, 1) void c3probe_kernel(
    const unsigned short* __restrict__ w3p,
    unsigned short* __restrict__ feat) {
    __shared__ unsigned short lds[LDS_TOTAL];
    const int b = blockIdx.x;
    const int t = threadIdx.x;
    const int lane = t & 63;
    const int wave = t >> 6;
    const int l15 = lane & 15;
    const int l4 = lane >> 4;
    // fill c2out region with something
    for (int i = t; i < 6912; i += 512) lds[LDS_C2 + i] = (unsigned short)(i);
    unsigned short* wlds = lds;  // reuse input region for staged weights
    if (V & 4) {
        // stage HALF of w3p (the 64 KB this wave-group needs) into LDS
        for (int i = t * 8; i < 32 * 1024; i += 512 * 8)
            *(u16x8*)&wlds[i] = *(const u16x8*)(w3p + i);
    }
    __syncthreads();

    const int mt = wave >> 1, nh = wave & 1;
    f32x4 acc[2] = {f32x4{0, 0, 0, 0}, f32x4{0, 0, 0, 0}};
    const int m = mt * 16 + l15;
    const int oy = m / 7, ox = m % 7;
    const bool mv = m < 49;
    int ga[8];
    unsigned short okm[8];
    {
        const int c0 = l4 >> 1;
        const int t16 = (l4 & 1) * 8;
        #pragma unroll
        for (int j = 0; j < 8; ++j) {
            const int tt = t16 + j, ky = tt >> 2, kx = tt & 3;
            const bool ok = mv && ky < 3 && kx < 3;
            ga[j] = LDS_C2 + (c0 * 9 + (ok ? oy + ky : 0)) * C2_STRIDE +
                    (ok ? ox + kx : 0);
            okm[j] = ok ? 0xFFFFu : 0;
        }
    }
    bf16x8 bbuf[2][2];
    #pragma unroll
    for (int nt = 0; nt < 2; ++nt)
        bbuf[0][nt] = as_bf16x8(*(const u16x8*)(
            w3p + ((nh * 2 + nt) * 16 + l15) * 1024 + l4 * 8));
    #pragma unroll 2
    for (int ks = 0; ks < 32; ++ks) {
        const int cur = ks & 1;
        if ((V & 2) && ks + 1 < 32) {
            const int k1 = (ks + 1) * 32 + l4 * 8;
            #pragma unroll
            for (int nt = 0; nt < 2; ++nt) {
                if (V & 4)
                    bbuf[cur ^ 1][nt] = as_bf16x8(*(const u16x8*)(
                        &wlds[(((nh * 2 + nt) * 16 + l15) & 31) * 1024 + k1]));
                else
                    bbuf[cur ^ 1][nt] = as_bf16x8(*(const u16x8*)(
                        w3p + ((nh * 2 + nt) * 16 + l15) * 1024 + k1));
            }
        }
        u16x8 av;
        if (V & 1) {
            #pragma unroll
            for (int j = 0; j < 8; ++j) av[j] = lds[ga[j]] & okm[j];
            #pragma unroll
            for (int j = 0; j < 8; ++j) ga[j] += 2 * 9 * C2_STRIDE;
        } else {
            #pragma unroll
            for (int j = 0; j < 8; ++j) av[j] = okm[j];
        }
        bf16x8 a = as_bf16x8(av);
        acc[0] = mfma16(a, bbuf[cur][0], acc[0]);
        acc[1] = mfma16(a, bbuf[cur][1], acc[1]);
    }
    unsigned short* outp = feat + (int64_t)b * 3136;
    #pragma unroll
    for (int nt = 0; nt < 2; ++nt) {
        const int n = (nh * 2 + nt) * 16 + l15;
        #pragma unroll
        for (int j = 0; j < 4; ++j) {
            const int mm = mt * 16 + l4 * 4 + j;
            if (mm < 49)
                outp[n * 49 + mm] = f32_to_bf16(fmaxf(acc[nt][j], 0.0f));
        }
    }
}

template <int V>
float runc3(int B, unsigned short* w3p, unsigned short* feat, int iters) {
    for (int i = 0; i < 20; ++i)
        hipLaunchKernelGGL(c3probe_kernel<V>, dim3(B), dim3(512), 0, 0, w3p, feat);
    hipEvent_t e0, e1;
    hipEventCreate(&e0); hipEventCreate(&e1);
    hipEventRecord(e0);
    for (int i = 0; i < iters; ++i)
        hipLaunchKernelGGL(c3probe_kernel<V>, dim3(B), dim3(512), 0, 0, w3p, feat);
    hipEventRecord(e1);
    hipEventSynchronize(e1);
    float ms; hipEventElapsedTime(&ms, e0, e1);
    return ms * 1000.0f / iters;
}

int main(int argc, char** argv) {
    int B = argc > 1 ? atoi(argv[1]) : 256;
    auto alloc = [](size_t n) {
        void* p;
        HIP_CHECK(hipMalloc(&p, n * 2));
        HIP_CHECK(hipMemset(p, 0x3c, n * 2));
        return (unsigned short*)p;
    };
    auto* x = alloc((size_t)B * 4 * 84 * 84);
    auto* w1 = alloc(32 * 256);
    auto* b1 = alloc(32);
    auto* w2 = alloc(64 * 512);
    auto* b2 = alloc(64);
    auto* w3p = alloc(64 * 1024);
    auto* b3 = alloc(64);
    auto* feat = alloc((size_t)B * 3136);
    int it = 300;
    printf("B=%d\n", B);
    printf("stage  in                %7.2f us\n", run<1>(B, x, w1, b1, w2, b2, w3p, b3, feat, it));
    printf("stage  in+c1             %7.2f us\n", run<3>(B, x, w1, b1, w2, b2, w3p, b3, feat, it));
    printf("stage  in+c1+c2          %7.2f us\n", run<7>(B, x, w1, b1, w2, b2, w3p, b3, feat, it));
    printf("stage  full              %7.2f us\n", run<15>(B, x, w1, b1, w2, b2, w3p, b3, feat, it));
    printf("stage  c1 only           %7.2f us\n", run<2>(B, x, w1, b1, w2, b2, w3p, b3, feat, it));
    printf("stage  c2 only           %7.2f us\n", run<4>(B, x, w1, b1, w2, b2, w3p, b3, feat, it));
    printf("stage  c3 only           %7.2f us\n", run<8>(B, x, w1, b1, w2, b2, w3p, b3, feat, it));
    printf("c3 iso: mfma only        %7.2f us\n", runc3<0>(B, w3p, feat, it));
    printf("c3 iso: +gather          %7.2f us\n", runc3<1>(B, w3p, feat, it));
    printf("c3 iso: +wload           %7.2f us\n", runc3<2>(B, w3p, feat, it));
    printf("c3 iso: gather+wload     %7.2f us\n", runc3<3>(B, w3p, feat, it));
    printf("c3 iso: gather+ldsw      %7.2f us\n", runc3<7>(B, w3p, feat, it));
    return 0;
}
