// Fused RainbowIQN forward for CDNA4 (gfx950) -- SURVEY.md K1-K9 as four
// hand-written MFMA kernels instead of ~40 library launches:
//
//   1. conv_encode : Nature-CNN (4->32 8x8s4, 32->64 4x4s2, 64->64 3x3s1,
//      ReLU after each) fused into ONE kernel.  One workgroup per state;
//      conv1 gathers its im2col operand straight from L2 (optionally from
//      the u8 frame ring, fusing the actor's frame-stack assembly) and the
//      two intermediate activation planes stay in LDS (30+14 KB -> 2
//      WGs/CU); each conv layer is an implicit GEMM on
//      v_mfma_f32_16x16x32_bf16 tiles.  Weights ride in from L2 (re-read by
//      all B workgroups).  Replaces MIOpen igemm + NHWC transposes + ReLU.
//   2. tau_embed   : cos(pi*i*tau) basis computed in-register (no [R,64]
//      cosine tensor in HBM), GEMM against iqn_fc.weight via MFMA, bias +
//      ReLU + broadcast-multiply with the conv features fused in the
//      epilogue -> h [R,3136] bf16.
//   3. heads_gemm  : h [R,3136] x Wcat^T [3136,1024] -> [R,1024] where Wcat
//      is fc_h_v/fc_h_a's noisy weights CONCATENATED row-wise (the
//      noisy_material kernel writes them into one buffer, so both dueling
//      branches are one LDS-tiled MFMA GEMM).  Bias + ReLU fused.
//   4. heads_finish: z-layer GEMV (Wz = [fc_z_v; fc_z_a] concat, row 0 from
//      the hv half, rows 1..A from the ha half) + dueling combine
//      Q = V + A - mean_a(A), with three epilogue modes: quantile values
//      out (learn), mean-over-taus argmax -> actions (act, K9 fused), or
//      gather at a given action -> theta (target-net path, K10 feed).
//
// Numerics: bf16 operands, fp32 MFMA accumulation (same contract as the
// hipBLASLt/MIOpen path it replaces); parity tests in
// tests/test_model_fused_gpu.py compare against the eager module.
#include "common.h"
#include "riqn_kernels.h"

namespace {

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(8))) unsigned short u16x8;
typedef __attribute__((ext_vector_type(4))) unsigned short u16x4;
typedef __attribute__((ext_vector_type(4))) float f32x4;

DEV f32x4 mfma16(bf16x8 a, bf16x8 b, f32x4 c) {
    return __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
}
DEV bf16x8 as_bf16x8(u16x8 v) { return __builtin_bit_cast(bf16x8, v); }
DEV float bf(const unsigned short u) { return bf16_to_f32(u); }

// ---------------------------------------------------------------------------
// 1. Fused conv encoder.  One WG (512 threads, 8 waves) per state.
// LDS layout (ushorts):
//   c1out: [32][20][24]  -> 15360 ush (30720 B)
//   c2out: [64][9][12]   ->  6912 ush (13824 B)
// conv1 reads its im2col A-operand straight from global memory: the input
// state is 56 KB (L2-resident; quarter-wave-coalesced 16 B segments), and
// NOT staging it drops LDS 103 -> 44.5 KB, doubling occupancy to 2 WGs/CU
// -- the B=1024 actor launch goes from 4 serial occupancy rounds to 2.
// ---------------------------------------------------------------------------
#define C1_STRIDE 24
#define C2_STRIDE 12
#define LDS_C1 0
#define LDS_C2 15360
#define LDS_TOTAL (15360 + 6912)

// GATHER=1 additionally fuses the actor's frame-stack assembly (K15,
// env.hip stack_gather): conv1 gathers its taps straight from the u8 frame
// RING (per-channel slot pointers, pre-episode frames zeroed via a 0 scale)
// -- the [E,4,84,84] bf16 stack tensor never exists, saving its ~58 MB
// write + re-read per act at E=1024.
template <int GATHER>
__global__ __launch_bounds__(512) void conv_encode_kernel(
    const unsigned short* __restrict__ x,    // [B,4,84,84] bf16 (GATHER=0)
    const uint8_t* __restrict__ ring,        // [h][E][7056] u8 (GATHER=1)
    const int32_t* __restrict__ env_state,   // [E,4] (GATHER=1)
    int ring_head,                           // newest-frame slot (GATHER=1)
    const unsigned short* __restrict__ w1,   // [32,256]  bf16
    const unsigned short* __restrict__ b1,   // [32]      bf16
    const unsigned short* __restrict__ w2,   // [64,512]  bf16
    const unsigned short* __restrict__ b2,   // [64]
    const unsigned short* __restrict__ w3p,  // [64,1024] bf16 (9->16 padded)
    const unsigned short* __restrict__ b3,   // [64]
    unsigned short* __restrict__ feat,       // [B,3136]  bf16
    unsigned short* __restrict__ c1g,        // [B,32*400] or nullptr (bwd)
    unsigned short* __restrict__ c2g) {      // [B,64*81]  or nullptr (bwd)
    __shared__ unsigned short lds[LDS_TOTAL];
    const int b = blockIdx.x;
    const int t = threadIdx.x;
    const int lane = t & 63;
    const int wave = t >> 6;
    const int l15 = lane & 15;
    const int l4 = lane >> 4;  // 0..3

    const unsigned short* src = x + (int64_t)b * (4 * 84 * 84);
    // per-channel ring frame base + scale (WG-uniform; history fixed at 4)
    const uint8_t* rb[4];
    float rs[4];
    if (GATHER) {
        const int steps = env_state[4 * b + 1];  // frames since episode start
        const int E = gridDim.x;
        #pragma unroll
        for (int ch = 0; ch < 4; ++ch) {
            const int back = 3 - ch;
            const int slot = (ring_head - back + 8) & 3;
            rb[ch] = ring + ((int64_t)slot * E + b) * 7056;
            rs[ch] = (steps >= back) ? (1.0f / 255.0f) : 0.0f;
        }
    }

    // ---- conv1: M=400 (25 tiles), N=32 (2 tiles), K=256 (8 steps) --------
    // All 16 B-fragments live in registers; the m-loop runs on L2-cached
    // global A reads + MFMA.
    bf16x8 b1f[2][8];
    #pragma unroll
    for (int nt = 0; nt < 2; ++nt)
        #pragma unroll
        for (int ks = 0; ks < 8; ++ks)
            b1f[nt][ks] = as_bf16x8(
                *(const u16x8*)(w1 + ((ks * 2 + nt) << 9) + lane * 8));

    for (int mt = wave; mt < 25; mt += 8) {
        f32x4 acc[2] = {f32x4{0, 0, 0, 0}, f32x4{0, 0, 0, 0}};
        const int m = mt * 16 + l15;           // output pixel (A-row)
        const int oy = m / 20, ox = m % 20;
        #pragma unroll
        for (int ks = 0; ks < 8; ++ks) {
            const int k0 = ks * 32 + l4 * 8;
            const int c = k0 >> 6, ky = (k0 >> 3) & 7;
            // A: 8 taps = input row segment [c][oy*4+ky][ox*4 .. +7]
            u16x8 av;
            if (GATHER) {
                // c == ks>>1 is wave-uniform, so rb[c]/rs[c] stay scalar
                const uint8_t* p = rb[c] + (oy * 4 + ky) * 84 + ox * 4;
                const uint32_t lo = *(const uint32_t*)p;
                const uint32_t hi = *(const uint32_t*)(p + 4);
                const float sc = rs[c];
                #pragma unroll
                for (int j = 0; j < 4; ++j) {
                    av[j] = f32_to_bf16((float)((lo >> (8 * j)) & 0xFF) * sc);
                    av[4 + j] =
                        f32_to_bf16((float)((hi >> (8 * j)) & 0xFF) * sc);
                }
            } else {
                const int idx = (c * 84 + oy * 4 + ky) * 84 + ox * 4;
                u16x4 lo = *(const u16x4*)(src + idx);
                u16x4 hi = *(const u16x4*)(src + idx + 4);
                av = u16x8{lo.x, lo.y, lo.z, lo.w, hi.x, hi.y, hi.z, hi.w};
            }
            bf16x8 a = as_bf16x8(av);
            acc[0] = mfma16(a, b1f[0][ks], acc[0]);
            acc[1] = mfma16(a, b1f[1][ks], acc[1]);
        }
        // epilogue: D row = mt*16 + l4*4 + j (pixel), col = l15 (channel)
        #pragma unroll
        for (int nt = 0; nt < 2; ++nt) {
            const int n = nt * 16 + l15;
            const float bias = bf(b1[n]);
            #pragma unroll
            for (int j = 0; j < 4; ++j) {
                const int mm = mt * 16 + l4 * 4 + j;
                if (mm < 400) {
                    unsigned short o = f32_to_bf16(fmaxf(acc[nt][j] + bias, 0.0f));
                    lds[LDS_C1 + (n * 20 + mm / 20) * C1_STRIDE + mm % 20] = o;
                    if (c1g != nullptr)
                        c1g[(int64_t)b * 12800 + n * 400 + mm] = o;
                }
            }
        }
    }
    __syncthreads();

    // ---- conv2: M=81 (6 tiles), N=64 (4 tiles), K=512 (16 steps) ---------
    // Work unit = (m-tile, n-quarter): 24 single-fragment units over 8
    // waves (3 balanced rounds; the (m, n-half) split left half the waves
    // idle on its second round).  k-outer loop with a register
    // double-buffer on the B-fragments.
    for (int u = wave; u < 24; u += 8) {
        const int mt = u >> 2, nq2 = u & 3;
        f32x4 acc1 = f32x4{0, 0, 0, 0};
        const int m = mt * 16 + l15;
        const int oy = m / 9, ox = m % 9;
        const bool mv = m < 81;
        // A-taps via ALIGNED dword LDS reads: each lane's 8 taps are two
        // rows of 4 consecutive x positions, and (c*20 + y)*C1_STRIDE +
        // ox*2 is always even (C1_STRIDE = 24), so each row is two aligned
        // u32 reads (ds_read2-able) instead of four scalar u16 reads --
        // the scalar gather's 8 LDS issues per MFMA dominated this loop.
        // Addresses advance +2 channel planes per k-step (stays even).
        int ga0, ga4;
        {
            const int c0 = l4 >> 1;
            const int ky0 = (l4 & 1) * 2;  // t16 = (l4*8)&15 in {0,8}
            ga0 = mv ? LDS_C1 + (c0 * 20 + oy * 2 + ky0) * C1_STRIDE + ox * 2
                     : LDS_C1 + c0 * 20 * C1_STRIDE;
            ga4 = mv ? ga0 + C1_STRIDE : LDS_C1 + c0 * 20 * C1_STRIDE;
        }
        bf16x8 bbuf[2];
        bbuf[0] = as_bf16x8(*(const u16x8*)(w2 + (nq2 << 9) + lane * 8));
        #pragma unroll 2
        for (int ks = 0; ks < 16; ++ks) {
            const int cur = ks & 1;
            if (ks + 1 < 16)
                bbuf[cur ^ 1] = as_bf16x8(*(const u16x8*)(
                    w2 + (((ks + 1) * 4 + nq2) << 9) + lane * 8));
            const uint32_t* p0 = (const uint32_t*)&lds[ga0];
            const uint32_t* p1 = (const uint32_t*)&lds[ga4];
            const uint32_t d0 = p0[0], d1 = p0[1], d2 = p1[0], d3 = p1[1];
            u16x8 av;
            av[0] = (unsigned short)d0; av[1] = (unsigned short)(d0 >> 16);
            av[2] = (unsigned short)d1; av[3] = (unsigned short)(d1 >> 16);
            av[4] = (unsigned short)d2; av[5] = (unsigned short)(d2 >> 16);
            av[6] = (unsigned short)d3; av[7] = (unsigned short)(d3 >> 16);
            ga0 += 2 * 20 * C1_STRIDE;
            ga4 += 2 * 20 * C1_STRIDE;
            acc1 = mfma16(as_bf16x8(av), bbuf[cur], acc1);
        }
        {
            const int n = nq2 * 16 + l15;
            const float bias = bf(b2[n]);
            #pragma unroll
            for (int j = 0; j < 4; ++j) {
                const int mm = mt * 16 + l4 * 4 + j;
                if (mm < 81) {
                    unsigned short o = f32_to_bf16(fmaxf(acc1[j] + bias, 0.0f));
                    lds[LDS_C2 + (n * 9 + mm / 9) * C2_STRIDE + mm % 9] = o;
                    if (c2g != nullptr)
                        c2g[(int64_t)b * 5184 + n * 81 + mm] = o;
                }
            }
        }
    }
    __syncthreads();

    // ---- conv3: M=49 (4 tiles), N=64 (4 tiles), K=1024 padded (32 steps) -
    // 8 units = (m-tile, n-half), exactly one per wave.
    {
        const int mt = wave >> 1, nh = wave & 1;
        f32x4 acc[2] = {f32x4{0, 0, 0, 0}, f32x4{0, 0, 0, 0}};
        const int m = mt * 16 + l15;
        const int oy = m / 7, ox = m % 7;
        const bool mv = m < 49;
        // A-taps via aligned dword superset reads: a lane's 8 taps are two
        // rows of (kx 0..2 + a masked kx=3 slot); each row's 3 ushorts sit
        // inside the two aligned dwords at (base & ~1), selected by the
        // lane-invariant base parity -- 4 u32 LDS reads + a few VALU per
        // MFMA instead of 8 scalar u16 reads (same rewrite as conv2).
        // Invalid rows (ky = 3 at t16 = 8, or m >= 49) read a safe dummy
        // and are fully zero-masked, matching the scalar gather exactly.
        int gb[2];
        unsigned int rm[2];
        bool codd;
        {
            const int c0 = l4 >> 1;
            const int t16 = (l4 & 1) * 8;
            const int ky0 = t16 >> 2;  // 0 or 2
            codd = ((ox & 1) != 0);
            #pragma unroll
            for (int rr2 = 0; rr2 < 2; ++rr2) {
                const int ky = ky0 + rr2;
                const bool ok = mv && ky < 3;
                const int base = LDS_C2 +
                    (c0 * 9 + (ok ? oy + ky : 0)) * C2_STRIDE +
                    (ok ? ox : 0);
                gb[rr2] = base & ~1;
                rm[rr2] = ok ? 0xFFFFFFFFu : 0u;
            }
        }
        bf16x8 bbuf[2][2];
        #pragma unroll
        for (int nt = 0; nt < 2; ++nt)
            bbuf[0][nt] = as_bf16x8(*(const u16x8*)(
                w3p + ((nh * 2 + nt) << 9) + lane * 8));
        #pragma unroll 2
        for (int ks = 0; ks < 32; ++ks) {
            const int cur = ks & 1;
            if (ks + 1 < 32) {
                #pragma unroll
                for (int nt = 0; nt < 2; ++nt)
                    bbuf[cur ^ 1][nt] = as_bf16x8(*(const u16x8*)(
                        w3p + (((ks + 1) * 4 + nh * 2 + nt) << 9) + lane * 8));
            }
            u16x8 av;
            #pragma unroll
            for (int rr2 = 0; rr2 < 2; ++rr2) {
                const uint32_t* p = (const uint32_t*)&lds[gb[rr2]];
                const uint32_t d0 = p[0] & rm[rr2], d1 = p[1] & rm[rr2];
                // kx 0..2 from the superset, kx 3 slot always zero
                const uint32_t w01 =
                    codd ? __builtin_amdgcn_alignbit(d1, d0, 16) : d0;
                const unsigned short w2v = (unsigned short)(
                    codd ? (d1 >> 16) : d1);
                av[rr2 * 4 + 0] = (unsigned short)w01;
                av[rr2 * 4 + 1] = (unsigned short)(w01 >> 16);
                av[rr2 * 4 + 2] = w2v;
                av[rr2 * 4 + 3] = 0;
                gb[rr2] += 2 * 9 * C2_STRIDE;
            }
            bf16x8 a = as_bf16x8(av);
            acc[0] = mfma16(a, bbuf[cur][0], acc[0]);
            acc[1] = mfma16(a, bbuf[cur][1], acc[1]);
        }
        unsigned short* out = feat + (int64_t)b * 3136;
        #pragma unroll
        for (int nt = 0; nt < 2; ++nt) {
            const int n = (nh * 2 + nt) * 16 + l15;
            const float bias = bf(b3[n]);
            #pragma unroll
            for (int j = 0; j < 4; ++j) {
                const int mm = mt * 16 + l4 * 4 + j;
                if (mm < 49)
                    out[n * 49 + mm] = f32_to_bf16(fmaxf(acc[nt][j] + bias, 0.0f));
            }
        }
    }
}

// Fragment-major weight packing: MFMA B-fragments are 8 contiguous k per
// lane at a ROW decided by lane&15 -- loading them from the natural [n][k]
// layout touches 16 cache lines per instruction (the dominant cost of the
// conv/tau k-loops, measured by scripts/conv_probe).  Packing weights as
// [frag][lane][8] makes every fragment load one contiguous 1 KB block.
// frag(W, ks, nt) element (lane, j) = W[nt*16 + (lane&15)][ks*32 + (lane>>4)*8 + j]
DEV unsigned short w3_tap(const unsigned short* w3, int n, int k) {
    int c = k >> 4, t = k & 15, ky = t >> 2, kx = t & 3;
    return (ky < 3 && kx < 3) ? w3[n * 576 + c * 9 + ky * 3 + kx]
                              : (unsigned short)0;
}

__global__ void pack_fwd_frag_kernel(
    const unsigned short* __restrict__ w1,    // [32,256]
    const unsigned short* __restrict__ w2,    // [64,512]
    const unsigned short* __restrict__ w3,    // [64,576]
    const unsigned short* __restrict__ wiqn,  // [3136,64]
    unsigned short* __restrict__ w1f,         // [16][512]  frag = ks*2+nt
    unsigned short* __restrict__ w2f,         // [64][512]  frag = ks*4+nt
    unsigned short* __restrict__ w3f,         // [128][512] frag = ks*4+nt
    unsigned short* __restrict__ wiqnf) {     // [392][512] frag = nt*2+ks
    const int i = threadIdx.x + blockIdx.x * blockDim.x;
    const int lane = (i >> 3) & 63, j = i & 7;
    const int l15 = lane & 15, l4 = (lane >> 4) & 3;
    const int frag = i >> 9;
    if (i < 16 * 512) {  // w1f: ks in [0,8), nt in [0,2)
        const int ks = frag >> 1, nt = frag & 1;
        w1f[i] = w1[(nt * 16 + l15) * 256 + ks * 32 + l4 * 8 + j];
    }
    if (i < 64 * 512) {  // w2f: ks in [0,16), nt in [0,4)
        const int ks = frag >> 2, nt = frag & 3;
        w2f[i] = w2[(nt * 16 + l15) * 512 + ks * 32 + l4 * 8 + j];
    }
    if (i < 128 * 512) {  // w3f: ks in [0,32), nt in [0,4)
        const int ks = frag >> 2, nt = frag & 3;
        w3f[i] = w3_tap(w3, nt * 16 + l15, ks * 32 + l4 * 8 + j);
    }
    if (i < 392 * 512) {  // wiqnf: nt in [0,196), ks in [0,2)
        const int nt = frag >> 1, ks = frag & 1;
        wiqnf[i] = wiqn[(nt * 16 + l15) * 64 + ks * 32 + l4 * 8 + j];
    }
}

// ---------------------------------------------------------------------------
// 2. tau embed: h[r][n] = relu(sum_k cos(pi*k*tau_r) * W[n][k] + b[n]) * feat[r/nq][n]
// Grid (ceil(R/16), NSPLIT=4); 196 n-tiles split across gridDim.y * 4 waves.
// ---------------------------------------------------------------------------
// Output staged through an LDS tile so the [16 rows x 784 cols] block this
// WG owns leaves as coalesced 8-byte row stores instead of strided 2-byte
// scatters; the <=4 feat rows the tile touches are staged the same way.
// Optionally also writes phi (pre-multiply embedding) for the backward pass.
// DIRECT=1 skips the LDS output staging and stores straight from the MFMA
// accumulators (2-byte lanes, 32 B segments per quarter-wave): fewer LDS
// ops + one barrier less, and LDS drops 49 KB -> occupancy is VGPR-bound.
template <int DIRECT>
__global__ __launch_bounds__(256) void tau_embed_kernel(
    const float* __restrict__ taus,         // [R]
    const unsigned short* __restrict__ w,   // [3136,64] bf16 (iqn_fc.weight)
    const unsigned short* __restrict__ bias,// [3136] bf16
    const unsigned short* __restrict__ feat,// [B,3136] bf16
    unsigned short* __restrict__ h,         // [R,3136] bf16
    unsigned short* __restrict__ phi_out,   // [R,3136] bf16 or nullptr
    unsigned short* __restrict__ cosb_out,  // [R,64] bf16 or nullptr (bwd)
    int R, int nq) {
    __shared__ unsigned short tile[DIRECT ? 2 : 2 * 16 * 784];
    __shared__ unsigned short ftile[4 * 784];
    const int t = threadIdx.x;
    const int lane = t & 63;
    const int wave = t >> 6;
    const int l15 = lane & 15;
    const int l4 = lane >> 4;
    const int mt0 = blockIdx.x * 2;        // two m-tiles -> W re-read halved
    const int nt0 = blockIdx.y * 49;       // first n-tile; cols nt0*16..+784

    // stage the feat rows of the states this tile touches (<= 4 for nq >= 8)
    const int s0 = (mt0 * 16) / nq;
    const int span = min((mt0 * 16 + 31) / nq - s0 + 1, 4);
    #pragma unroll 4
    for (int i = t * 4; i < span * 784; i += 256 * 4) {
        int si = i / 784, cc = i % 784;
        int64_t src = (int64_t)min(s0 + si, (R - 1) / nq) * 3136 + nt0 * 16 + cc;
        *(u16x4*)&ftile[si * 784 + cc] = *(const u16x4*)(feat + src);
    }

    // A-fragments for both m-tiles x both k-steps (k = l4*8 + j + 32*ks).
    // cos(k*pi*tau) for the 16 consecutive k of a lane comes from the
    // Chebyshev recurrence cos((k+1)t) = 2cos(t)cos(kt) - cos((k-1)t):
    // 3 accurate cosf per lane instead of 16 (this kernel was ALU-bound).
    bf16x8 afrag[2][2];
    #pragma unroll
    for (int mi = 0; mi < 2; ++mi) {
        const int r_a = (mt0 + mi) * 16 + l15;
        const float tau = (r_a < R) ? taus[r_a] : 0.0f;
        const float th = 3.14159265358979f * tau;
        const int kbase = l4 * 8;
        float cm1 = cosf((float)kbase * th);        // cos(k0 * t)
        float c0 = cosf((float)(kbase + 1) * th);   // cos((k0+1) * t)
        const float two_ct = 2.0f * cosf(th);
        #pragma unroll
        for (int ks = 0; ks < 2; ++ks) {
            u16x8 av;
            #pragma unroll
            for (int j = 0; j < 8; ++j) {
                if (j == 0) {
                    av[j] = f32_to_bf16(cm1);
                } else if (j == 1) {
                    av[j] = f32_to_bf16(c0);
                } else {
                    float c1 = two_ct * c0 - cm1;
                    cm1 = c0;
                    c0 = c1;
                    av[j] = f32_to_bf16(c1);
                }
            }
            if (ks == 0) {
                // jump from k0+7 to k0+32: re-seed the recurrence so the
                // ks=1 fragment starts at cos((k0+32)t), cos((k0+33)t)
                cm1 = cosf((float)(kbase + 32) * th);
                c0 = two_ct * cm1 - cosf((float)(kbase + 31) * th);
            }
            if (cosb_out != nullptr && blockIdx.y == 0 && r_a < R)
                *(u16x8*)(cosb_out + (int64_t)r_a * 64 + ks * 32 + l4 * 8) = av;
            afrag[mi][ks] = as_bf16x8(av);
        }
    }
    __syncthreads();

    // per-tile body; the caller software-pipelines the two W-fragment
    // loads (wv0/wv1) one tile ahead with STATICALLY-named registers -- a
    // dynamically indexed buffer array lowered to ~1300 v_cndmask selects
    // and ran 10x slower (PERF_LOG r02 rejected row)
    auto tile_body = [&](int nt, u16x8 wv0, u16x8 wv1) {
        const int n = nt * 16 + l15;
        f32x4 acc[2] = {f32x4{0, 0, 0, 0}, f32x4{0, 0, 0, 0}};
        acc[0] = mfma16(afrag[0][0], as_bf16x8(wv0), acc[0]);
        acc[1] = mfma16(afrag[1][0], as_bf16x8(wv0), acc[1]);
        acc[0] = mfma16(afrag[0][1], as_bf16x8(wv1), acc[0]);
        acc[1] = mfma16(afrag[1][1], as_bf16x8(wv1), acc[1]);
        const float bb = bf(bias[n]);
        const int col = (nt - nt0) * 16 + l15;
        #pragma unroll
        for (int mi = 0; mi < 2; ++mi) {
            #pragma unroll
            for (int j = 0; j < 4; ++j) {
                const int rr = l4 * 4 + j;       // row within m-tile
                const int r = (mt0 + mi) * 16 + rr;
                unsigned short ph = f32_to_bf16(fmaxf(acc[mi][j] + bb, 0.0f));
                float fv = bf(ftile[min(r / nq - s0, span - 1) * 784 + col]);
                if (DIRECT) {
                    if (r < R)
                        h[(int64_t)r * 3136 + nt * 16 + l15] =
                            f32_to_bf16(bf16_to_f32(ph) * fv);
                } else {
                    tile[(mi * 16 + rr) * 784 + col] =
                        f32_to_bf16(bf16_to_f32(ph) * fv);
                }
                if (phi_out != nullptr && r < R)
                    phi_out[(int64_t)r * 3136 + nt * 16 + l15] = ph;
            }
        }
    };
    auto wload = [&](int nt, int ks) {
        return *(const u16x8*)(w + ((nt * 2 + ks) << 9) + lane * 8);
    };
    {
        const int end = nt0 + 49;
        int nt = nt0 + wave;
        if (nt < end) {
            u16x8 a0 = wload(nt, 0), a1 = wload(nt, 1);
            while (true) {
                const int ntB = nt + 4;
                u16x8 b0, b1;
                if (ntB < end) { b0 = wload(ntB, 0); b1 = wload(ntB, 1); }
                tile_body(nt, a0, a1);
                if (ntB >= end) break;
                const int ntC = ntB + 4;
                if (ntC < end) { a0 = wload(ntC, 0); a1 = wload(ntC, 1); }
                tile_body(ntB, b0, b1);
                if (ntC >= end) break;
                nt = ntC;
            }
        }
    }
    if (!DIRECT) {
        __syncthreads();
        // coalesced row stores: 32 rows x 784 cols
        #pragma unroll 8
        for (int i = t * 4; i < 32 * 784; i += 256 * 4) {
            int rr = i / 784, cc = i % 784;
            int r = mt0 * 16 + rr;
            if (r < R)
                *(u16x4*)(h + (int64_t)r * 3136 + nt0 * 16 + cc) =
                    *(const u16x4*)&tile[rr * 784 + cc];
        }
    }
}

// ---------------------------------------------------------------------------
// 3. heads GEMM: out[R,1024] = relu(h[R,3136] @ Wcat^T + bcat), bf16.
// Block tile 64x64, BK=32 (one MFMA K), 4 waves each on a 32x32 quadrant.
// ---------------------------------------------------------------------------
#define HG_BM 64
#define HG_BN 64
#define HG_BK 64
// BK=64, two LDS buffers, tiles staged by global_load_lds (16 B per lane,
// lane-linear LDS image), XOR-swizzled on the SOURCE address so the
// ds_read_b128 fragment reads don't hit row-stride bank conflicts
// (cdna_hip_programming.md section 5 glds recipe).  The next K-tile's DMA is
// in flight while the current one computes; __syncthreads drains it.
template <int NBUF>  // 2 at large R (occupancy), 3 at small R (deeper pipe)
__global__ __launch_bounds__(256) void heads_gemm_kernel(
    const unsigned short* __restrict__ hin,  // [R,3136]
    const unsigned short* __restrict__ wcat, // [1024,3136]
    const unsigned short* __restrict__ bcat, // [1024]
    unsigned short* __restrict__ out,        // [R,1024]
    int R) {
    __shared__ unsigned short smem[NBUF * 2 * HG_BM * HG_BK];
    const int t = threadIdx.x;
    const int lane = t & 63, wave = t >> 6;
    const int wr = wave >> 1, wc = wave & 1;  // 2x2 wave grid of 32x32
    const int l15 = lane & 15, l4 = lane >> 4;
    const int bm = blockIdx.x * HG_BM, bn = blockIdx.y * HG_BN;

    // glds partition: per buffer, A = 8 KB = 8 instructions; wave w issues
    // A-pieces {2w, 2w+1} and B-pieces {2w, 2w+1}.  Instruction i writes LDS
    // rows 8i..8i+7 (lane l -> row 8i + l/8, 16-B piece l%8); the global
    // column piece is (l%8) ^ (row%8).
    auto stage = [&](int buf, int k0) {
        unsigned short* base = smem + buf * (2 * HG_BM * HG_BK);
        #pragma unroll
        for (int i = 0; i < 2; ++i) {
            const int inst = wave * 2 + i;
            const int row = inst * 8 + (lane >> 3);
            const int piece = (lane & 7) ^ (row & 7);
            {   // A rows (clamped so out-of-range rows read safe memory)
                const int gr = min(bm + row, R - 1);
                auto* gp = (const __attribute__((address_space(1))) uint32_t*)
                    (hin + (int64_t)gr * 3136 + k0 + piece * 8);
                auto* lp = (__attribute__((address_space(3))) uint32_t*)
                    (base + inst * 512);
                __builtin_amdgcn_global_load_lds(gp, lp, 16, 0, 0);
            }
            {   // B rows
                auto* gp = (const __attribute__((address_space(1))) uint32_t*)
                    (wcat + (int64_t)(bn + row) * 3136 + k0 + piece * 8);
                auto* lp = (__attribute__((address_space(3))) uint32_t*)
                    (base + HG_BM * HG_BK + inst * 512);
                __builtin_amdgcn_global_load_lds(gp, lp, 16, 0, 0);
            }
        }
    };

    f32x4 acc[2][2] = {{{0, 0, 0, 0}, {0, 0, 0, 0}},
                       {{0, 0, 0, 0}, {0, 0, 0, 0}}};
    // 3-buffer glds pipeline with raw barriers + counted vmcnt: one tile
    // always in flight across each barrier (__syncthreads would emit a
    // vmcnt(0) and drain it -- cdna_hip_programming.md section 5 glds table)
    stage(0, 0);
    if (NBUF == 3) stage(1, HG_BK);
    for (int kt = 0; kt < 3136 / HG_BK; ++kt) {
        // wait own glds for buf[kt]; with NBUF=3 buf[kt+1]'s 4 stay in
        // flight (except at the tail, where only buf[kt] is outstanding)
        if (NBUF == 3 && kt + 1 < 3136 / HG_BK)
            asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
        else
            asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        __builtin_amdgcn_s_barrier();
        if (kt + NBUF - 1 < 3136 / HG_BK)
            stage((kt + NBUF - 1) % NBUF, (kt + NBUF - 1) * HG_BK);
        const unsigned short* As = smem + (kt % NBUF) * (2 * HG_BM * HG_BK);
        const unsigned short* Bs = As + HG_BM * HG_BK;
        #pragma unroll
        for (int ksub = 0; ksub < 2; ++ksub) {
            #pragma unroll
            for (int fm = 0; fm < 2; ++fm) {
                const int row = wr * 32 + fm * 16 + l15;
                const int piece = (ksub * 4 + l4) ^ (row & 7);
                bf16x8 a = as_bf16x8(*(const u16x8*)&As[row * 64 + piece * 8]);
                #pragma unroll
                for (int fn = 0; fn < 2; ++fn) {
                    const int nrow = wc * 32 + fn * 16 + l15;
                    const int bp = (ksub * 4 + l4) ^ (nrow & 7);
                    bf16x8 b = as_bf16x8(*(const u16x8*)&Bs[nrow * 64 + bp * 8]);
                    acc[fm][fn] = mfma16(a, b, acc[fm][fn]);
                }
            }
        }
    }
    __builtin_amdgcn_s_barrier();  // epilogue reads nothing from LDS; order only
    #pragma unroll
    for (int fm = 0; fm < 2; ++fm) {
        #pragma unroll
        for (int fn = 0; fn < 2; ++fn) {
            const int n = bn + wc * 32 + fn * 16 + l15;
            const float bias = bf(bcat[n]);
            #pragma unroll
            for (int j = 0; j < 4; ++j) {
                const int r = bm + wr * 32 + fm * 16 + l4 * 4 + j;
                if (r < R)
                    out[(int64_t)r * 1024 + n] =
                        f32_to_bf16(fmaxf(acc[fm][fn][j] + bias, 0.0f));
            }
        }
    }
}

// ---------------------------------------------------------------------------
// 4. finish: z-layer + dueling.  hh[r] = [hv(512) | ha(512)];
//   v = Wz[0] . hv + bz[0];  a_i = Wz[1+i] . ha + bz[1+i]
//   q[r][i] = v + a_i - mean(a)
// mode 0: write q [R,A] f32
// mode 1: actions[e] = argmax_i mean_{k<nq} q[e*nq+k][i]   (one WG per env)
// mode 2: theta[r] = q[r][act[r/nq]]                        (f32 [R])
// ---------------------------------------------------------------------------
// Large-R (act path) tile: 256x128 block, 512 threads, each of 8 waves on
// a 64x64 quadrant (4x4 fragments), BK=64 glds pipeline.  Tile growth is
// the lever here: PMC showed a 128x128 version saturating L2 at ~15 TB/s
// (878 MB/dispatch at R=8192, profiles/r01_gemm_pmc_summary.txt); this
// shape moves 28% fewer operand bytes (917 -> 1076 TF).
template <int NBUF>  // 2 or 3 buffers (3 = 144 KB LDS, deeper pipe)
__global__ __launch_bounds__(512) void heads_gemm_big2_kernel(
    const unsigned short* __restrict__ hin,  // [R,3136]
    const unsigned short* __restrict__ wcat, // [1024,3136]
    const unsigned short* __restrict__ bcat, // [1024]
    unsigned short* __restrict__ out,        // [R,1024]
    int R) {
    __shared__ unsigned short smem[NBUF * (256 + 128) * 64];
    const int t = threadIdx.x;
    const int lane = t & 63, wave = t >> 6;   // 8 waves
    const int wr = wave >> 1, wc = wave & 1;  // 4x2 grid of 64x64 quadrants
    const int l15 = lane & 15, l4 = lane >> 4;
    // XCD-aware tile map: the dispatcher assigns WGs to XCDs round-robin
    // on the linear id, so id%8 IS the XCD; give each XCD a contiguous
    // row-block band x all n-blocks so its co-resident WGs share operands
    // through that XCD's L2 (MI355X_MICROARCH.md XCD/L2 locality rule).
    int bxi = blockIdx.x, byi = blockIdx.y;
    if ((gridDim.x & 7) == 0) {
        const int id = blockIdx.x + blockIdx.y * gridDim.x;
        const int xcd = id & 7, s = id >> 3;
        byi = s % gridDim.y;
        bxi = xcd * (gridDim.x >> 3) + s / gridDim.y;
    }
    const int bm = bxi * 256, bn = byi * 128;

    // 48 glds insts per buffer (A 32 + B 16): wave w issues A [4w,4w+4),
    // B [2w,2w+2); each inst = 8 rows x 64-k, XOR source swizzle.
    auto stage = [&](int buf, int k0) {
        unsigned short* base = smem + buf * ((256 + 128) * 64);
        #pragma unroll
        for (int i = 0; i < 4; ++i) {
            const int inst = wave * 4 + i;
            const int row = inst * 8 + (lane >> 3);
            const int piece = (lane & 7) ^ (row & 7);
            const int gr = min(bm + row, R - 1);
            auto* gp = (const __attribute__((address_space(1))) uint32_t*)
                (hin + (int64_t)gr * 3136 + k0 + piece * 8);
            auto* lp = (__attribute__((address_space(3))) uint32_t*)
                (base + inst * 512);
            __builtin_amdgcn_global_load_lds(gp, lp, 16, 0, 0);
        }
        #pragma unroll
        for (int i = 0; i < 2; ++i) {
            const int inst = wave * 2 + i;
            const int row = inst * 8 + (lane >> 3);
            const int piece = (lane & 7) ^ (row & 7);
            auto* gp = (const __attribute__((address_space(1))) uint32_t*)
                (wcat + (int64_t)(bn + row) * 3136 + k0 + piece * 8);
            auto* lp = (__attribute__((address_space(3))) uint32_t*)
                (base + 256 * 64 + inst * 512);
            __builtin_amdgcn_global_load_lds(gp, lp, 16, 0, 0);
        }
    };

    f32x4 acc[4][4];
    #pragma unroll
    for (int fm = 0; fm < 4; ++fm)
        #pragma unroll
        for (int fn = 0; fn < 4; ++fn) acc[fm][fn] = f32x4{0, 0, 0, 0};
    stage(0, 0);
    if (NBUF == 3) stage(1, 64);
    for (int kt = 0; kt < 3136 / 64; ++kt) {
        // wait own glds for buf[kt]; with NBUF=3 the newest stage's 6 stay
        // in flight across the barrier (counted vmcnt, not a full drain)
        if (NBUF == 3 && kt + 1 < 3136 / 64)
            asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
        else
            asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        __builtin_amdgcn_s_barrier();
        if (kt + NBUF - 1 < 3136 / 64)
            stage((kt + NBUF - 1) % NBUF, (kt + NBUF - 1) * 64);
        const unsigned short* As = smem + (kt % NBUF) * ((256 + 128) * 64);
        const unsigned short* Bs = As + 256 * 64;
        #pragma unroll
        for (int ksub = 0; ksub < 2; ++ksub) {
            #pragma unroll
            for (int fm = 0; fm < 4; ++fm) {
                const int row = wr * 64 + fm * 16 + l15;
                const int piece = (ksub * 4 + l4) ^ (row & 7);
                bf16x8 a = as_bf16x8(*(const u16x8*)&As[row * 64 + piece * 8]);
                #pragma unroll
                for (int fn = 0; fn < 4; ++fn) {
                    const int nrow = wc * 64 + fn * 16 + l15;
                    const int bp = (ksub * 4 + l4) ^ (nrow & 7);
                    bf16x8 b = as_bf16x8(*(const u16x8*)&Bs[nrow * 64 + bp * 8]);
                    acc[fm][fn] = mfma16(a, b, acc[fm][fn]);
                }
            }
        }
    }
    __builtin_amdgcn_s_barrier();
    #pragma unroll
    for (int fm = 0; fm < 4; ++fm) {
        #pragma unroll
        for (int fn = 0; fn < 4; ++fn) {
            const int n = bn + wc * 64 + fn * 16 + l15;
            const float bias = bf(bcat[n]);
            #pragma unroll
            for (int j = 0; j < 4; ++j) {
                const int r = bm + wr * 64 + fm * 16 + l4 * 4 + j;
                if (r < R)
                    out[(int64_t)r * 1024 + n] =
                        f32_to_bf16(fmaxf(acc[fm][fn][j] + bias, 0.0f));
            }
        }
    }
}

// 256x256 tile variant (NBUF=2, 128 KB LDS, 1 WG/CU): per 64-k tile a WG
// stages (256+256)x64x2 = 64 KB for 256x256x64x2 = 8.4 MFLOP -- 131 FLOP/B
// vs the 256x128 kernel's 87, a +50% arithmetic-intensity step aimed at the
// measured L2-traffic bound (profiles/r01_gemm_pmc_summary.txt).  Wave tile
// 64x128: acc[4][8].  A/B experiment vs heads_gemm_big2 (RIQN_GEMM=2).
template <int NBUF>
__global__ __launch_bounds__(512) void heads_gemm_256n_kernel(
    const unsigned short* __restrict__ hin,  // [R,3136]
    const unsigned short* __restrict__ wcat, // [1024,3136]
    const unsigned short* __restrict__ bcat, // [1024]
    unsigned short* __restrict__ out,        // [R,1024]
    int R) {
    __shared__ unsigned short smem[NBUF * (256 + 256) * 64];
    const int t = threadIdx.x;
    const int lane = t & 63, wave = t >> 6;   // 8 waves
    const int wr = wave >> 1, wc = wave & 1;  // 4x2 grid of 64x128 tiles
    const int l15 = lane & 15, l4 = lane >> 4;
    int bxi = blockIdx.x, byi = blockIdx.y;
    if ((gridDim.x & 7) == 0) {
        const int id = blockIdx.x + blockIdx.y * gridDim.x;
        const int xcd = id & 7, s = id >> 3;
        byi = s % gridDim.y;
        bxi = xcd * (gridDim.x >> 3) + s / gridDim.y;
    }
    const int bm = bxi * 256, bn = byi * 256;

    // 64 glds insts per buffer (A 32 + B 32): wave w issues 8.
    auto stage = [&](int buf, int k0) {
        unsigned short* base = smem + buf * ((256 + 256) * 64);
        #pragma unroll
        for (int i = 0; i < 4; ++i) {
            const int inst = wave * 4 + i;
            const int row = inst * 8 + (lane >> 3);
            const int piece = (lane & 7) ^ (row & 7);
            const int gr = min(bm + row, R - 1);
            auto* gp = (const __attribute__((address_space(1))) uint32_t*)
                (hin + (int64_t)gr * 3136 + k0 + piece * 8);
            auto* lp = (__attribute__((address_space(3))) uint32_t*)
                (base + inst * 512);
            __builtin_amdgcn_global_load_lds(gp, lp, 16, 0, 0);
        }
        #pragma unroll
        for (int i = 0; i < 4; ++i) {
            const int inst = wave * 4 + i;
            const int row = inst * 8 + (lane >> 3);
            const int piece = (lane & 7) ^ (row & 7);
            auto* gp = (const __attribute__((address_space(1))) uint32_t*)
                (wcat + (int64_t)(bn + row) * 3136 + k0 + piece * 8);
            auto* lp = (__attribute__((address_space(3))) uint32_t*)
                (base + 256 * 64 + inst * 512);
            __builtin_amdgcn_global_load_lds(gp, lp, 16, 0, 0);
        }
    };

    f32x4 acc[4][8];
    #pragma unroll
    for (int fm = 0; fm < 4; ++fm)
        #pragma unroll
        for (int fn = 0; fn < 8; ++fn) acc[fm][fn] = f32x4{0, 0, 0, 0};
    stage(0, 0);
    if (NBUF == 3) stage(1, 64);
    for (int kt = 0; kt < 3136 / 64; ++kt) {
        if (NBUF == 3 && kt + 1 < 3136 / 64)
            asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
        else
            asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        __builtin_amdgcn_s_barrier();
        if (kt + NBUF - 1 < 3136 / 64)
            stage((kt + NBUF - 1) % NBUF, (kt + NBUF - 1) * 64);
        const unsigned short* As = smem + (kt % NBUF) * ((256 + 256) * 64);
        const unsigned short* Bs = As + 256 * 64;
        #pragma unroll
        for (int ksub = 0; ksub < 2; ++ksub) {
            #pragma unroll
            for (int fm = 0; fm < 4; ++fm) {
                const int row = wr * 64 + fm * 16 + l15;
                const int piece = (ksub * 4 + l4) ^ (row & 7);
                bf16x8 a = as_bf16x8(*(const u16x8*)&As[row * 64 + piece * 8]);
                #pragma unroll
                for (int fn = 0; fn < 8; ++fn) {
                    const int nrow = wc * 128 + fn * 16 + l15;
                    const int bp = (ksub * 4 + l4) ^ (nrow & 7);
                    bf16x8 b = as_bf16x8(*(const u16x8*)&Bs[nrow * 64 + bp * 8]);
                    acc[fm][fn] = mfma16(a, b, acc[fm][fn]);
                }
            }
        }
    }
    __builtin_amdgcn_s_barrier();
    #pragma unroll
    for (int fm = 0; fm < 4; ++fm) {
        #pragma unroll
        for (int fn = 0; fn < 8; ++fn) {
            const int n = bn + wc * 128 + fn * 16 + l15;
            const float bias = bf(bcat[n]);
            #pragma unroll
            for (int j = 0; j < 4; ++j) {
                const int r = bm + wr * 64 + fm * 16 + l4 * 4 + j;
                if (r < R)
                    out[(int64_t)r * 1024 + n] =
                        f32_to_bf16(fmaxf(acc[fm][fn][j] + bias, 0.0f));
            }
        }
    }
}

// 256x256 tile with REGISTER-DIRECT B (variant 4): only A rides the
// LDS-DMA fill path (32 KB/kt, half the 256x256 kernel's); each wave reads
// its B fragments straight from L2 into VGPRs (B is L2-resident -- 6.4 MB
// re-read by every m-band of the XCD).  Splits the per-kt traffic across
// the two independent load paths AND halves LDS to 64 KB -> 2 WGs/CU for
// latency hiding.  Same per-element k-order as the other variants
// (bit-exact).
template <int NBUF>
__global__ __launch_bounds__(512) void heads_gemm_rb_kernel(
    const unsigned short* __restrict__ hin,  // [R,3136]
    const unsigned short* __restrict__ wcat, // [1024,3136]
    const unsigned short* __restrict__ bcat, // [1024]
    unsigned short* __restrict__ out,        // [R,1024]
    int R) {
    __shared__ unsigned short smem[NBUF * 256 * 64];
    const int t = threadIdx.x;
    const int lane = t & 63, wave = t >> 6;   // 8 waves
    const int wr = wave >> 1, wc = wave & 1;  // 4x2 grid of 64x128 tiles
    const int l15 = lane & 15, l4 = lane >> 4;
    int bxi = blockIdx.x, byi = blockIdx.y;
    if ((gridDim.x & 7) == 0) {
        const int id = blockIdx.x + blockIdx.y * gridDim.x;
        const int xcd = id & 7, s = id >> 3;
        byi = s % gridDim.y;
        bxi = xcd * (gridDim.x >> 3) + s / gridDim.y;
    }
    const int bm = bxi * 256, bn = byi * 256;

    // A only: 32 glds insts per buffer, wave w issues 4
    auto stage = [&](int buf, int k0) {
        unsigned short* base = smem + buf * (256 * 64);
        #pragma unroll
        for (int i = 0; i < 4; ++i) {
            const int inst = wave * 4 + i;
            const int row = inst * 8 + (lane >> 3);
            const int piece = (lane & 7) ^ (row & 7);
            const int gr = min(bm + row, R - 1);
            auto* gp = (const __attribute__((address_space(1))) uint32_t*)
                (hin + (int64_t)gr * 3136 + k0 + piece * 8);
            auto* lp = (__attribute__((address_space(3))) uint32_t*)
                (base + inst * 512);
            __builtin_amdgcn_global_load_lds(gp, lp, 16, 0, 0);
        }
    };

    f32x4 acc[4][8];
    #pragma unroll
    for (int fm = 0; fm < 4; ++fm)
        #pragma unroll
        for (int fn = 0; fn < 8; ++fn) acc[fm][fn] = f32x4{0, 0, 0, 0};
    stage(0, 0);
    if (NBUF == 3) stage(1, 64);
    for (int kt = 0; kt < 3136 / 64; ++kt) {
        const int k0 = kt * 64;
        if (NBUF == 3 && kt + 1 < 3136 / 64)
            asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
        else
            asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        __builtin_amdgcn_s_barrier();
        if (kt + NBUF - 1 < 3136 / 64)
            stage((kt + NBUF - 1) % NBUF, (kt + NBUF - 1) * 64);
        const unsigned short* As = smem + (kt % NBUF) * (256 * 64);
        #pragma unroll
        for (int ksub = 0; ksub < 2; ++ksub) {
            // B fragments straight from global (fm-invariant, loaded once)
            bf16x8 bfr[8];
            #pragma unroll
            for (int fn = 0; fn < 8; ++fn) {
                const int nrow = wc * 128 + fn * 16 + l15;
                bfr[fn] = as_bf16x8(*(const u16x8*)(
                    wcat + (int64_t)(bn + nrow) * 3136 + k0 +
                    (ksub * 4 + l4) * 8));
            }
            #pragma unroll
            for (int fm = 0; fm < 4; ++fm) {
                const int row = wr * 64 + fm * 16 + l15;
                const int piece = (ksub * 4 + l4) ^ (row & 7);
                bf16x8 a = as_bf16x8(*(const u16x8*)&As[row * 64 + piece * 8]);
                #pragma unroll
                for (int fn = 0; fn < 8; ++fn)
                    acc[fm][fn] = mfma16(a, bfr[fn], acc[fm][fn]);
            }
        }
    }
    __builtin_amdgcn_s_barrier();
    #pragma unroll
    for (int fm = 0; fm < 4; ++fm) {
        #pragma unroll
        for (int fn = 0; fn < 8; ++fn) {
            const int n = bn + wc * 128 + fn * 16 + l15;
            const float bias = bf(bcat[n]);
            #pragma unroll
            for (int j = 0; j < 4; ++j) {
                const int r = bm + wr * 64 + fm * 16 + l4 * 4 + j;
                if (r < R)
                    out[(int64_t)r * 1024 + n] =
                        f32_to_bf16(fmaxf(acc[fm][fn][j] + bias, 0.0f));
            }
        }
    }
}

// 320x256 tile (NBUF=2, 144 KB LDS): 72 KB staged per 64-k tile for
// 320x256x64x2 = 10.5 MFLOP = 142 FLOP/B, +9% arithmetic intensity over
// 256x256 against the fill-rate bound.  Wave tile 80x128 (acc[5][8], 160
// VGPRs -- the 384-row version spilled at 256).  Experimental variant 3.
template <int NBUF>
__global__ __launch_bounds__(512) void heads_gemm_320_kernel(
    const unsigned short* __restrict__ hin,  // [R,3136]
    const unsigned short* __restrict__ wcat, // [1024,3136]
    const unsigned short* __restrict__ bcat, // [1024]
    unsigned short* __restrict__ out,        // [R,1024]
    int R) {
    __shared__ unsigned short smem[NBUF * (320 + 256) * 64];
    const int t = threadIdx.x;
    const int lane = t & 63, wave = t >> 6;   // 8 waves
    const int wr = wave >> 1, wc = wave & 1;  // 4x2 grid of 80x128 tiles
    const int l15 = lane & 15, l4 = lane >> 4;
    int bxi = blockIdx.x, byi = blockIdx.y;
    if ((gridDim.x & 7) == 0) {
        const int id = blockIdx.x + blockIdx.y * gridDim.x;
        const int xcd = id & 7, s = id >> 3;
        byi = s % gridDim.y;
        bxi = xcd * (gridDim.x >> 3) + s / gridDim.y;
    }
    const int bm = bxi * 320, bn = byi * 256;

    // 72 glds insts per buffer (A 40 + B 32): wave w issues 9.
    auto stage = [&](int buf, int k0) {
        unsigned short* base = smem + buf * ((320 + 256) * 64);
        #pragma unroll
        for (int i = 0; i < 5; ++i) {
            const int inst = wave * 5 + i;
            const int row = inst * 8 + (lane >> 3);
            const int piece = (lane & 7) ^ (row & 7);
            const int gr = min(bm + row, R - 1);
            auto* gp = (const __attribute__((address_space(1))) uint32_t*)
                (hin + (int64_t)gr * 3136 + k0 + piece * 8);
            auto* lp = (__attribute__((address_space(3))) uint32_t*)
                (base + inst * 512);
            __builtin_amdgcn_global_load_lds(gp, lp, 16, 0, 0);
        }
        #pragma unroll
        for (int i = 0; i < 4; ++i) {
            const int inst = wave * 4 + i;
            const int row = inst * 8 + (lane >> 3);
            const int piece = (lane & 7) ^ (row & 7);
            auto* gp = (const __attribute__((address_space(1))) uint32_t*)
                (wcat + (int64_t)(bn + row) * 3136 + k0 + piece * 8);
            auto* lp = (__attribute__((address_space(3))) uint32_t*)
                (base + 320 * 64 + inst * 512);
            __builtin_amdgcn_global_load_lds(gp, lp, 16, 0, 0);
        }
    };

    f32x4 acc[5][8];
    #pragma unroll
    for (int fm = 0; fm < 5; ++fm)
        #pragma unroll
        for (int fn = 0; fn < 8; ++fn) acc[fm][fn] = f32x4{0, 0, 0, 0};
    stage(0, 0);
    for (int kt = 0; kt < 3136 / 64; ++kt) {
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        __builtin_amdgcn_s_barrier();
        if (kt + NBUF - 1 < 3136 / 64)
            stage((kt + NBUF - 1) % NBUF, (kt + NBUF - 1) * 64);
        const unsigned short* As = smem + (kt % NBUF) * ((320 + 256) * 64);
        const unsigned short* Bs = As + 320 * 64;
        #pragma unroll
        for (int ksub = 0; ksub < 2; ++ksub) {
            #pragma unroll
            for (int fm = 0; fm < 5; ++fm) {
                const int row = wr * 80 + fm * 16 + l15;
                const int piece = (ksub * 4 + l4) ^ (row & 7);
                bf16x8 a = as_bf16x8(*(const u16x8*)&As[row * 64 + piece * 8]);
                #pragma unroll
                for (int fn = 0; fn < 8; ++fn) {
                    const int nrow = wc * 128 + fn * 16 + l15;
                    const int bp = (ksub * 4 + l4) ^ (nrow & 7);
                    bf16x8 b = as_bf16x8(*(const u16x8*)&Bs[nrow * 64 + bp * 8]);
                    acc[fm][fn] = mfma16(a, b, acc[fm][fn]);
                }
            }
        }
    }
    __builtin_amdgcn_s_barrier();
    #pragma unroll
    for (int fm = 0; fm < 5; ++fm) {
        #pragma unroll
        for (int fn = 0; fn < 8; ++fn) {
            const int n = bn + wc * 128 + fn * 16 + l15;
            const float bias = bf(bcat[n]);
            #pragma unroll
            for (int j = 0; j < 4; ++j) {
                const int r = bm + wr * 80 + fm * 16 + l4 * 4 + j;
                if (r < R)
                    out[(int64_t)r * 1024 + n] =
                        f32_to_bf16(fmaxf(acc[fm][fn][j] + bias, 0.0f));
            }
        }
    }
}

// Actor-path fusion of tau_embed INTO the heads GEMM (K4-K6 as one kernel):
// h[r,k] = relu(cos_r . Wiqn[k,:] + b_iqn[k]) * feat[r/nq, k] is produced
// per 256x64 k-tile straight into the LDS A-image the main MFMA loop reads
// -- h [R,3136] never exists in HBM (saves its 51 MB write + 8x re-read at
// R=8192 and the whole tau_embed dispatch).  The per-wave cos fragments
// don't depend on k, so they are built once (Chebyshev recurrence, 3 cosf)
// and reused for all 49 k-tiles.  bf16 rounding matches tau_embed exactly
// (round after relu, then after the feat multiply), so the output is
// bit-identical to tau_embed + heads_gemm.  The learner keeps the unfused
// pair (it needs h/phi for the backward).
__global__ __launch_bounds__(512) void heads_gemm_tau_kernel(
    const float* __restrict__ taus,          // [R]
    const unsigned short* __restrict__ wiqnf,// [392][512] fragment-major
    const unsigned short* __restrict__ biqn, // [3136] bf16
    const unsigned short* __restrict__ feat, // [R/nq, 3136] bf16
    const unsigned short* __restrict__ wcat, // [1024,3136] bf16
    const unsigned short* __restrict__ bcat, // [1024] bf16
    unsigned short* __restrict__ out,        // [R,1024] bf16
    int R, int nq) {
    __shared__ unsigned short aimg[2][256 * 64];   // computed A (2 x 32 KB)
    __shared__ unsigned short bsm[2 * 128 * 64];   // staged wcat (2 x 16 KB)
    const int t = threadIdx.x;
    const int lane = t & 63, wave = t >> 6;   // 8 waves
    const int wr = wave >> 1, wc = wave & 1;  // main-GEMM quadrant map
    const int l15 = lane & 15, l4 = lane >> 4;
    int bxi = blockIdx.x, byi = blockIdx.y;
    if ((gridDim.x & 7) == 0) {
        const int id = blockIdx.x + blockIdx.y * gridDim.x;
        const int xcd = id & 7, s = id >> 3;
        byi = s % gridDim.y;
        bxi = xcd * (gridDim.x >> 3) + s / gridDim.y;
    }
    const int bm = bxi * 256, bn = byi * 128;

    // cos fragments for this wave's 32 A'-rows (k-invariant; tau_embed's
    // recurrence verbatim so the bf16 basis values match bit-for-bit)
    bf16x8 afrag[2][2];
    #pragma unroll
    for (int mi = 0; mi < 2; ++mi) {
        const int r_a = bm + wave * 32 + mi * 16 + l15;
        const float tau = (r_a < R) ? taus[r_a] : 0.0f;
        const float th = 3.14159265358979f * tau;
        const int kbase = l4 * 8;
        float cm1 = cosf((float)kbase * th);
        float c0 = cosf((float)(kbase + 1) * th);
        const float two_ct = 2.0f * cosf(th);
        #pragma unroll
        for (int ks = 0; ks < 2; ++ks) {
            u16x8 av;
            #pragma unroll
            for (int j = 0; j < 8; ++j) {
                if (j == 0) {
                    av[j] = f32_to_bf16(cm1);
                } else if (j == 1) {
                    av[j] = f32_to_bf16(c0);
                } else {
                    float c1 = two_ct * c0 - cm1;
                    cm1 = c0;
                    c0 = c1;
                    av[j] = f32_to_bf16(c1);
                }
            }
            if (ks == 0) {
                cm1 = cosf((float)(kbase + 32) * th);
                c0 = two_ct * cm1 - cosf((float)(kbase + 31) * th);
            }
            afrag[mi][ks] = as_bf16x8(av);
        }
    }

    // wcat staging: 16 glds insts per buffer, wave w issues [2w, 2w+2)
    auto stageb = [&](int buf, int k0) {
        unsigned short* base = bsm + buf * (128 * 64);
        #pragma unroll
        for (int i = 0; i < 2; ++i) {
            const int inst = wave * 2 + i;
            const int row = inst * 8 + (lane >> 3);
            const int piece = (lane & 7) ^ (row & 7);
            auto* gp = (const __attribute__((address_space(1))) uint32_t*)
                (wcat + (int64_t)(bn + row) * 3136 + k0 + piece * 8);
            auto* lp = (__attribute__((address_space(3))) uint32_t*)
                (base + inst * 512);
            __builtin_amdgcn_global_load_lds(gp, lp, 16, 0, 0);
        }
    };

    // A'-producer for one 256x64 k-tile (wave owns rows [32w, 32w+32));
    // bf16 rounding matches tau_embed bit-for-bit
    const int smax = (R - 1) / nq;
    auto produceA = [&](int kt) {
        unsigned short* dst = aimg[kt & 1];
        #pragma unroll
        for (int ntl = 0; ntl < 4; ++ntl) {
            const int ntg = kt * 4 + ntl;          // global 16-col tile
            const int kcol = ntg * 16 + l15;
            f32x4 acc2[2] = {f32x4{0, 0, 0, 0}, f32x4{0, 0, 0, 0}};
            #pragma unroll
            for (int ks = 0; ks < 2; ++ks) {
                bf16x8 bfr = as_bf16x8(
                    *(const u16x8*)(wiqnf + ((ntg * 2 + ks) << 9) + lane * 8));
                acc2[0] = mfma16(afrag[0][ks], bfr, acc2[0]);
                acc2[1] = mfma16(afrag[1][ks], bfr, acc2[1]);
            }
            const float bb = bf(biqn[kcol]);
            #pragma unroll
            for (int mi = 0; mi < 2; ++mi) {
                #pragma unroll
                for (int j = 0; j < 4; ++j) {
                    const int rl = wave * 32 + mi * 16 + l4 * 4 + j;
                    const unsigned short ph =
                        f32_to_bf16(fmaxf(acc2[mi][j] + bb, 0.0f));
                    const int s = min((bm + rl) / nq, smax);
                    const float fv = bf(feat[(int64_t)s * 3136 + kcol]);
                    const int cl = ntl * 16 + l15;   // col within k-tile
                    dst[rl * 64 + (((cl >> 3) ^ (rl & 7)) << 3) + (cl & 7)] =
                        f32_to_bf16(bf16_to_f32(ph) * fv);
                }
            }
        }
    };

    f32x4 acc[4][4];
    #pragma unroll
    for (int fm = 0; fm < 4; ++fm)
        #pragma unroll
        for (int fn = 0; fn < 4; ++fn) acc[fm][fn] = f32x4{0, 0, 0, 0};
    stageb(0, 0);
    produceA(0);
    // software pipeline, ONE barrier per iteration: produce A(kt+1) and
    // run the main MFMA on A(kt) back-to-back (independent LDS buffers, so
    // the scheduler interleaves the two instruction chains); the barrier
    // publishes A(kt+1) + guards B-buffer reuse for the next iteration.
    for (int kt = 0; kt < 3136 / 64; ++kt) {
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");  // B buf[kt] landed
        __builtin_amdgcn_s_barrier();                     // + A(kt) visible
        if (kt + 1 < 3136 / 64) {
            stageb((kt + 1) & 1, (kt + 1) * 64);
            produceA(kt + 1);
        }
        const unsigned short* As = aimg[kt & 1];
        const unsigned short* Bs = bsm + (kt & 1) * (128 * 64);
        #pragma unroll
        for (int ksub = 0; ksub < 2; ++ksub) {
            #pragma unroll
            for (int fm = 0; fm < 4; ++fm) {
                const int row = wr * 64 + fm * 16 + l15;
                const int piece = (ksub * 4 + l4) ^ (row & 7);
                bf16x8 a =
                    as_bf16x8(*(const u16x8*)&As[row * 64 + piece * 8]);
                #pragma unroll
                for (int fn = 0; fn < 4; ++fn) {
                    const int nrow = wc * 64 + fn * 16 + l15;
                    const int bp = (ksub * 4 + l4) ^ (nrow & 7);
                    bf16x8 b =
                        as_bf16x8(*(const u16x8*)&Bs[nrow * 64 + bp * 8]);
                    acc[fm][fn] = mfma16(a, b, acc[fm][fn]);
                }
            }
        }
    }
    __builtin_amdgcn_s_barrier();
    #pragma unroll
    for (int fm = 0; fm < 4; ++fm) {
        #pragma unroll
        for (int fn = 0; fn < 4; ++fn) {
            const int n = bn + wc * 64 + fn * 16 + l15;
            const float bias = bf(bcat[n]);
            #pragma unroll
            for (int j = 0; j < 4; ++j) {
                const int r = bm + wr * 64 + fm * 16 + l4 * 4 + j;
                if (r < R)
                    out[(int64_t)r * 1024 + n] =
                        f32_to_bf16(fmaxf(acc[fm][fn][j] + bias, 0.0f));
            }
        }
    }
}

// MODE 0: q [R,A] f32 out.  MODE 2: theta[r] = q[r][act[r/nq]] out.
// Two-pass per row: advantage sums stashed in LDS (sv) -- no dynamically
// indexed register array (which spills to scratch).
template <int MODE>
__global__ __launch_bounds__(256) void heads_finish_kernel(
    const unsigned short* __restrict__ hh,  // [R,1024]
    const unsigned short* __restrict__ wz,  // [1+A,512]
    const unsigned short* __restrict__ bz,  // [1+A]
    int R, int A, int nq,
    float* __restrict__ q_out,              // mode 0: [R,A]; mode 2: [R]
    const int64_t* __restrict__ act_in) {   // mode 2: [R/nq]
    __shared__ unsigned short wzs[19 * 512];
    __shared__ float sv[4][20];
    const int t = threadIdx.x;
    const int lane = t & 63, wave = t >> 6;
    #pragma unroll 8
    for (int i = t * 4; i < (1 + A) * 512; i += 256 * 4)
        *(u16x4*)&wzs[i] = *(const u16x4*)(wz + i);
    __syncthreads();

    const int r0 = blockIdx.x * 4;
    {
        const int r = r0 + wave;  // one row per wave
        if (r >= R) return;
        const unsigned short* hrow = hh + (int64_t)r * 1024;
        float hv[8], ha[8];
        #pragma unroll
        for (int j = 0; j < 8; ++j) {
            hv[j] = bf(hrow[lane * 8 + j]);
            ha[j] = bf(hrow[512 + lane * 8 + j]);
        }
        float p = 0.0f;
        #pragma unroll
        for (int j = 0; j < 8; ++j) p += hv[j] * bf(wzs[lane * 8 + j]);
        const float v = wave_allreduce_sum(p) + bf(bz[0]);
        float amean = 0.0f;
        for (int i = 0; i < A; ++i) {
            float q = 0.0f;
            const unsigned short* wrow = &wzs[(1 + i) * 512];
            #pragma unroll
            for (int j = 0; j < 8; ++j) q += ha[j] * bf(wrow[lane * 8 + j]);
            float s = wave_allreduce_sum(q) + bf(bz[1 + i]);
            if (lane == 0) sv[wave][i] = s;
            amean += s;
        }
        amean /= (float)A;
        if (MODE == 0) {
            if (lane < A) q_out[(int64_t)r * A + lane] = v + sv[wave][lane] - amean;
        } else {  // MODE 2
            if (lane == 0) {
                int a = (int)act_in[r / nq];
                q_out[r] = v + sv[wave][a] - amean;
            }
        }
    }
}

// MODE 1 (act, K9): one WG per env.  argmax_a mean_k Q(s, a; tau_k) ==
// argmax_a [ (sum_k ha_k) . Wz_a + nq * bz_a ] -- the V and mean_a terms are
// constant across a, so the mean over taus folds into ONE summed-ha vector
// per env (nq x fewer dot products than the per-row form).
__global__ __launch_bounds__(256) void heads_act_kernel(
    const unsigned short* __restrict__ hh,  // [R,1024], R = E*nq
    const unsigned short* __restrict__ wz,  // [1+A,512]
    const unsigned short* __restrict__ bz,  // [1+A]
    int A, int nq,
    int64_t* __restrict__ actions) {        // [E]
    __shared__ float habar[4][512];         // per-wave partial -> reduced
    __shared__ float qbar[20];
    const int t = threadIdx.x;
    const int lane = t & 63, wave = t >> 6;
    const int64_t base = (int64_t)blockIdx.x * nq * 1024;

    // per-wave partial sums over this env's nq rows (lane owns 8 columns)
    float part[8] = {0, 0, 0, 0, 0, 0, 0, 0};
    for (int k = wave; k < nq; k += 4) {
        const unsigned short* harow = hh + base + (int64_t)k * 1024 + 512;
        #pragma unroll
        for (int j = 0; j < 8; ++j) part[j] += bf(harow[lane * 8 + j]);
    }
    #pragma unroll
    for (int j = 0; j < 8; ++j) habar[wave][lane * 8 + j] = part[j];
    __syncthreads();
    // reduce 4 wave partials (thread t owns 2 columns)
    for (int c = t * 2; c < 512; c += 512) {
        #pragma unroll
        for (int u = 0; u < 2; ++u)
            habar[0][c + u] += habar[1][c + u] + habar[2][c + u] + habar[3][c + u];
    }
    __syncthreads();
    // one dot per action: wave w covers actions w, w+4, ...
    for (int i = wave; i < A; i += 4) {
        float p = 0.0f;
        const unsigned short* wrow = wz + (int64_t)(1 + i) * 512;
        #pragma unroll
        for (int j = 0; j < 8; ++j)
            p += habar[0][lane * 8 + j] * bf(wrow[lane * 8 + j]);
        float s = wave_allreduce_sum(p) + (float)nq * bf(bz[1 + i]);
        if (lane == 0) qbar[i] = s;
    }
    __syncthreads();
    if (t == 0) {
        int best = 0;
        float bv = qbar[0];
        for (int i = 1; i < A; ++i)
            if (qbar[i] > bv) { bv = qbar[i]; best = i; }
        actions[blockIdx.x] = (int64_t)best;
    }
}

}  // namespace

extern "C" {

void riqn_pack_fwd_frag(const void* w1, const void* w2, const void* w3,
                        const void* wiqn, void* w1f, void* w2f, void* w3f,
                        void* wiqnf, void* stream) {
    hipLaunchKernelGGL(pack_fwd_frag_kernel, dim3(392 * 512 / 256), dim3(256),
                       0, (hipStream_t)stream, (const unsigned short*)w1,
                       (const unsigned short*)w2, (const unsigned short*)w3,
                       (const unsigned short*)wiqn, (unsigned short*)w1f,
                       (unsigned short*)w2f, (unsigned short*)w3f,
                       (unsigned short*)wiqnf);
}

void riqn_conv_encode(const void* x, const void* w1, const void* b1,
                      const void* w2, const void* b2, const void* w3p,
                      const void* b3, void* feat, void* c1g, void* c2g, int B,
                      void* stream) {
    hipLaunchKernelGGL(conv_encode_kernel<0>, dim3(B), dim3(512), 0,
                       (hipStream_t)stream, (const unsigned short*)x,
                       (const uint8_t*)nullptr, (const int32_t*)nullptr, 0,
                       (const unsigned short*)w1, (const unsigned short*)b1,
                       (const unsigned short*)w2, (const unsigned short*)b2,
                       (const unsigned short*)w3p, (const unsigned short*)b3,
                       (unsigned short*)feat, (unsigned short*)c1g,
                       (unsigned short*)c2g);
}

void riqn_conv_encode_gather(const void* ring, const void* env_state,
                             int ring_head, const void* w1, const void* b1,
                             const void* w2, const void* b2, const void* w3p,
                             const void* b3, void* feat, int E,
                             void* stream) {
    hipLaunchKernelGGL(conv_encode_kernel<1>, dim3(E), dim3(512), 0,
                       (hipStream_t)stream, (const unsigned short*)nullptr,
                       (const uint8_t*)ring, (const int32_t*)env_state,
                       ring_head,
                       (const unsigned short*)w1, (const unsigned short*)b1,
                       (const unsigned short*)w2, (const unsigned short*)b2,
                       (const unsigned short*)w3p, (const unsigned short*)b3,
                       (unsigned short*)feat, (unsigned short*)nullptr,
                       (unsigned short*)nullptr);
}

void riqn_tau_embed(const float* taus, const void* w, const void* bias,
                    const void* feat, void* h, void* phi_out, void* cosb_out,
                    int R, int nq, int direct, void* stream) {
    // auto (-1): direct stores win at actor-size R (occupancy unbound by
    // LDS, 37 -> 28 us at R=8192); the staged tile's coalesced rows win at
    // learner-size R where the grid is latency-bound (kbench_tau2).
    if (direct < 0) direct = (R >= 4096);
    dim3 grid((R + 31) / 32, 4);
    auto kern = direct ? tau_embed_kernel<1> : tau_embed_kernel<0>;
    hipLaunchKernelGGL(kern, grid, dim3(256), 0,
                       (hipStream_t)stream, taus, (const unsigned short*)w,
                       (const unsigned short*)bias,
                       (const unsigned short*)feat, (unsigned short*)h,
                       (unsigned short*)phi_out, (unsigned short*)cosb_out,
                       R, nq);
}

void riqn_heads_gemm_tau(const float* taus, const void* wiqnf,
                         const void* biqn, const void* feat, const void* wcat,
                         const void* bcat, void* out, int R, int nq,
                         void* stream) {
    dim3 grid((R + 255) / 256, 1024 / 128);
    hipLaunchKernelGGL(heads_gemm_tau_kernel, grid, dim3(512), 0,
                       (hipStream_t)stream, taus,
                       (const unsigned short*)wiqnf,
                       (const unsigned short*)biqn,
                       (const unsigned short*)feat,
                       (const unsigned short*)wcat,
                       (const unsigned short*)bcat, (unsigned short*)out, R,
                       nq);
}

void riqn_heads_gemm(const void* h, const void* wcat, const void* bcat,
                     void* out, int R, int variant, void* stream) {
    // variant: 1 = 256x128 NBUF3 (shipped), 2 = 256x256 NBUF2,
    // 3 = 256x256 NBUF3 (needs 192KB -- rejected at launch; kept for doc),
    // -1 = default (env RIQN_GEMM override, else 1)
    if (variant < 0) {
        static int env_variant = -1;
        if (env_variant < 0) {
            const char* e = getenv("RIQN_GEMM");
            env_variant = e ? atoi(e) : 1;
        }
        variant = env_variant;
    }
    if (variant == 4 && R >= 8192) {
        dim3 gridb((R + 255) / 256, 1024 / 256);
        hipLaunchKernelGGL(heads_gemm_rb_kernel<3>, gridb, dim3(512), 0,
                           (hipStream_t)stream, (const unsigned short*)h,
                           (const unsigned short*)wcat,
                           (const unsigned short*)bcat, (unsigned short*)out,
                           R);
        return;
    }
    if (variant == 3 && R >= 8192) {
        dim3 gridb((R + 319) / 320, 1024 / 256);
        hipLaunchKernelGGL(heads_gemm_320_kernel<2>, gridb, dim3(512), 0,
                           (hipStream_t)stream, (const unsigned short*)h,
                           (const unsigned short*)wcat,
                           (const unsigned short*)bcat, (unsigned short*)out,
                           R);
        return;
    }
    if ((variant == 2 || (variant == 1 && R >= 16384)) && R >= 8192) {
        // 256x256 wins once the grid fills the chip (R/256 x 4 >= 256 WGs):
        // 193 -> 157.9 us (1090 -> 1333 TF) at the actor shape R=32768; at
        // R=8192 its 128-WG grid half-fills the chip and loses (707 TF) --
        // kbench r02, fill-rate-bound analysis in gpurun_out/r02_pmc_sq.txt
        dim3 gridb((R + 255) / 256, 1024 / 256);
        hipLaunchKernelGGL(heads_gemm_256n_kernel<2>, gridb, dim3(512), 0,
                           (hipStream_t)stream, (const unsigned short*)h,
                           (const unsigned short*)wcat,
                           (const unsigned short*)bcat, (unsigned short*)out,
                           R);
        return;
    }
    if (R >= 8192) {
        dim3 gridb((R + 255) / 256, 1024 / 128);
        hipLaunchKernelGGL(heads_gemm_big2_kernel<3>, gridb, dim3(512), 0,
                           (hipStream_t)stream, (const unsigned short*)h,
                           (const unsigned short*)wcat,
                           (const unsigned short*)bcat, (unsigned short*)out,
                           R);
        return;
    }
    dim3 grid((R + HG_BM - 1) / HG_BM, 1024 / HG_BN);
    if (R >= 4096)
        hipLaunchKernelGGL(heads_gemm_kernel<2>, grid, dim3(256), 0,
                           (hipStream_t)stream, (const unsigned short*)h,
                           (const unsigned short*)wcat,
                           (const unsigned short*)bcat, (unsigned short*)out,
                           R);
    else
        hipLaunchKernelGGL(heads_gemm_kernel<3>, grid, dim3(256), 0,
                           (hipStream_t)stream, (const unsigned short*)h,
                           (const unsigned short*)wcat,
                           (const unsigned short*)bcat, (unsigned short*)out,
                           R);
}

void riqn_heads_finish(const void* hh, const void* wz, const void* bz, int R,
                       int A, int nq, int mode, float* q_out, int64_t* act_io,
                       void* stream) {
    hipStream_t s = (hipStream_t)stream;
    if (mode == 1) {
        hipLaunchKernelGGL(heads_act_kernel, dim3(R / nq), dim3(256), 0, s,
                           (const unsigned short*)hh,
                           (const unsigned short*)wz,
                           (const unsigned short*)bz, A, nq, act_io);
    } else if (mode == 2) {
        hipLaunchKernelGGL(heads_finish_kernel<2>, dim3((R + 3) / 4),
                           dim3(256), 0, s, (const unsigned short*)hh,
                           (const unsigned short*)wz,
                           (const unsigned short*)bz, R, A, nq, q_out, act_io);
    } else {
        hipLaunchKernelGGL(heads_finish_kernel<0>, dim3((R + 3) / 4),
                           dim3(256), 0, s, (const unsigned short*)hh,
                           (const unsigned short*)wz,
                           (const unsigned short*)bz, R, A, nq, q_out, act_io);
    }
}

}  // extern "C"
