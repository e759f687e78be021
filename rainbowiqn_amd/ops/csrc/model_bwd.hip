// Hand-written backward chain for the RainbowIQN learn step
// -- gfx950 kernels replacing torch.autograd's eager backward entirely.
//
// Forward (model_fwd.hip) leaves: feat [B,3136], phi [R,3136] (pre-multiply
// embedding, tau_embed's phi_out), h [R,3136] = feat_rep * phi,
// hh [R,1024] = relu(h @ Wcat^T + b), theta [B,N] (z-layer at the replay
// action).  The loss kernel (loss.hip) gives dtheta = dL/dtheta.  This file
// walks the chain back:
//
//   finish_bwd_dhh : dtheta -> dhh [R,1024]      (z-layer + dueling + relu)
//   zgrad          : dtheta, hh -> fc_z_{v,a} mu/sigma/bias grads
//   gemm_dx        : dhh @ Wcat -> dphi_relu [R,3136] (x feat, x relu mask)
//                    + dfeat [B,3136] f32 (atomic nq-group reduction)
//   gemm_dwcat     : dhh^T @ h -> fc_h_{v,a} weight mu/sigma grads
//   bias_grad_heads: column sums of dhh -> fc_h_{v,a} bias mu/sigma grads
//   tau_dw         : dphi_relu^T @ cos-basis -> iqn_fc.weight grad
//   bias_grad_vec  : column sum of dphi_relu -> iqn_fc.bias grad
//   conv_bwd_data  : dfeat -> dpre3/dc2/dc1 (transposed convs, relu masks,
//                    activations from the forward's c1/c2 side outputs)
//   conv_bwd_dw<L> : per-layer weight grads as MFMA GEMMs over (state, px)
//   conv_finalize  : f32 scratch -> bf16 flat-grad views
//
// All GEMM-shaped pieces run on v_mfma_f32_16x16x32_bf16; everything is
// hipGraph-capturable (fixed shapes, no host sync).  Parity:
// tests/test_learn_fused_gpu.py checks grads against torch.autograd.
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
// dhh[r][j]      = dtheta[r] * Wz[0][j]                        * (hv[j] > 0)
// dhh[r][512+j]  = dtheta[r] * (Wz[1+a_r][j] - mean_i Wz[1+i][j]) * (ha[j] > 0)
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256) void finish_bwd_dhh_kernel(
    const float* __restrict__ dtheta,       // [R]
    const int64_t* __restrict__ actions,    // [B], r's action = actions[r/nq]
    const unsigned short* __restrict__ hh,  // [R,1024]
    const unsigned short* __restrict__ wz,  // [1+A,512]
    unsigned short* __restrict__ dhh,       // [R,1024] out
    int R, int A, int nq) {
    __shared__ float wzv[512];
    __shared__ float wzbar[512];
    const int t = threadIdx.x;
    for (int j = t; j < 512; j += 256) {
        wzv[j] = bf(wz[j]);
        float s = 0.0f;
        for (int i = 0; i < A; ++i) s += bf(wz[(1 + i) * 512 + j]);
        wzbar[j] = s / (float)A;
    }
    __syncthreads();
    const int lane = t & 63, wave = t >> 6;
    const int r0 = blockIdx.x * 16;
    for (int rr = wave; rr < 16; rr += 4) {
        const int r = r0 + rr;
        if (r >= R) break;
        const float g = dtheta[r];
        const int a = (int)actions[r / nq];
        const unsigned short* wza = wz + (1 + a) * 512;
        const unsigned short* hrow = hh + (int64_t)r * 1024;
        unsigned short* drow = dhh + (int64_t)r * 1024;
        for (int j = lane * 8; j < 512; j += 64 * 8) {
            #pragma unroll
            for (int u = 0; u < 8; ++u) {
                float mv = bf(hrow[j + u]) > 0.0f ? 1.0f : 0.0f;
                drow[j + u] = f32_to_bf16(g * wzv[j + u] * mv);
                float ma = bf(hrow[512 + j + u]) > 0.0f ? 1.0f : 0.0f;
                drow[512 + j + u] =
                    f32_to_bf16(g * (bf(wza[j + u]) - wzbar[j + u]) * ma);
            }
        }
    }
}

// ---------------------------------------------------------------------------
// z-layer grads.  Row 0 (V): coeff_r = dtheta[r], input hv.
// Row 1+i (A_i):  coeff_r = dtheta[r] * (1{a_r == i} - 1/A), input ha.
// One WG per output row; f32 accumulate over all R rows; writes mu and
// sigma (= mu-grad * eps) bias grads included.
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256) void zgrad_kernel(
    const float* __restrict__ dtheta, const int64_t* __restrict__ actions,
    const unsigned short* __restrict__ hh, int R, int A, int nq,
    const unsigned short* __restrict__ weps_v,  // [1,512]
    const unsigned short* __restrict__ weps_a,  // [A,512]
    const unsigned short* __restrict__ beps_v,  // [1]
    const unsigned short* __restrict__ beps_a,  // [A]
    unsigned short* __restrict__ dw_mu_v, unsigned short* __restrict__ dw_sg_v,
    unsigned short* __restrict__ db_mu_v, unsigned short* __restrict__ db_sg_v,
    unsigned short* __restrict__ dw_mu_a, unsigned short* __restrict__ dw_sg_a,
    unsigned short* __restrict__ db_mu_a, unsigned short* __restrict__ db_sg_a) {
    __shared__ float part[4][128];
    __shared__ float bpart[4];
    const int row = blockIdx.x;  // 0 = V, 1..A = advantage rows
    const int cq = blockIdx.y;   // column quarter [cq*128, +128)
    const int t = threadIdx.x;
    const int lane = t & 63, wave = t >> 6;
    const float invA = 1.0f / (float)A;
    const int off = ((row == 0) ? 0 : 512) + cq * 128;
    // each wave accumulates rows wave::4, lane covers 2 coalesced columns
    float wacc[2] = {0, 0};
    float wb = 0.0f;
    #pragma unroll 4
    for (int r = wave; r < R; r += 4) {
        float c = dtheta[r];
        if (row > 0) {
            int a = (int)actions[r / nq];
            c *= ((a == row - 1) ? 1.0f : 0.0f) - invA;
        }
        const unsigned short* hrow = hh + (int64_t)r * 1024 + off + lane * 2;
        wacc[0] += c * bf(hrow[0]);
        wacc[1] += c * bf(hrow[1]);
        wb += c;
    }
    part[wave][lane * 2] = wacc[0];
    part[wave][lane * 2 + 1] = wacc[1];
    if (lane == 0) bpart[wave] = wb;
    __syncthreads();
    if (t >= 128) return;
    const int j0 = cq * 128 + t;  // column within the 512-wide layer
    float acc1 = part[0][t] + part[1][t] + part[2][t] + part[3][t];
    float bacc = bpart[0] + bpart[1] + bpart[2] + bpart[3];
    if (row == 0) {
        dw_mu_v[j0] = f32_to_bf16(acc1);
        dw_sg_v[j0] = f32_to_bf16(acc1 * bf(weps_v[j0]));
        if (t == 0 && cq == 0) {
            db_mu_v[0] = f32_to_bf16(bacc);
            db_sg_v[0] = f32_to_bf16(bacc * bf(beps_v[0]));
        }
    } else {
        int i = row - 1;
        dw_mu_a[i * 512 + j0] = f32_to_bf16(acc1);
        dw_sg_a[i * 512 + j0] = f32_to_bf16(acc1 * bf(weps_a[i * 512 + j0]));
        if (t == 0 && cq == 0) {
            db_mu_a[i] = f32_to_bf16(bacc);
            db_sg_a[i] = f32_to_bf16(bacc * bf(beps_a[i]));
        }
    }
}

// ---------------------------------------------------------------------------
// dh = dhh @ Wcat  (M=R, N=3136, K=1024), epilogue:
//   dphi_relu[r][n] = bf16( dh * feat[r/nq][n] * (phi[r][n] > 0) )
//   dfeat[r/nq][n] += dh * phi[r][n]          (f32 atomic)
// B-tile (Wcat slice) is staged transposed through LDS.
// ---------------------------------------------------------------------------
// glds-staged (2 buffers, raw barrier + counted waits, XOR source swizzle
// -- the heads-GEMM recipe).  A = dhh rows; B = Wcat k-rows ([k][n] slice is
// row-major contiguous, no transpose needed).
__global__ __launch_bounds__(256) void gemm_dx_kernel(
    const unsigned short* __restrict__ dhh,   // [R,1024]
    const unsigned short* __restrict__ wcat,  // [1024,3136]
    const unsigned short* __restrict__ feat,  // [B,3136]
    const unsigned short* __restrict__ phi,   // [R,3136]
    unsigned short* __restrict__ dphi,        // [R,3136] out
    float* __restrict__ dfeat,                // [B,3136] f32, pre-zeroed
    int R, int nq) {
    __shared__ unsigned short smem[2 * 2 * 64 * 64];  // 2 bufs x (A|B)
    const int t = threadIdx.x;
    const int lane = t & 63, wave = t >> 6;
    const int wr = wave >> 1, wc = wave & 1;
    const int l15 = lane & 15, l4 = lane >> 4;
    const int bm = blockIdx.x * 64, bn = blockIdx.y * 64;

    auto stage = [&](int buf, int k0) {
        unsigned short* base = smem + buf * (2 * 64 * 64);
        #pragma unroll
        for (int i = 0; i < 2; ++i) {
            const int inst = wave * 2 + i;
            const int row = inst * 8 + (lane >> 3);
            const int piece = (lane & 7) ^ (row & 7);
            {   // A: dhh[bm+row][k0 + piece*8]
                const int gr = min(bm + row, R - 1);
                auto* gp = (const __attribute__((address_space(1))) uint32_t*)
                    (dhh + (int64_t)gr * 1024 + k0 + piece * 8);
                auto* lp = (__attribute__((address_space(3))) uint32_t*)
                    (base + inst * 512);
                __builtin_amdgcn_global_load_lds(gp, lp, 16, 0, 0);
            }
            {   // B: Wcat[k0+row][bn + piece*8]  ([k][n] image)
                auto* gp = (const __attribute__((address_space(1))) uint32_t*)
                    (wcat + (int64_t)(k0 + row) * 3136 + bn + piece * 8);
                auto* lp = (__attribute__((address_space(3))) uint32_t*)
                    (base + 64 * 64 + inst * 512);
                __builtin_amdgcn_global_load_lds(gp, lp, 16, 0, 0);
            }
        }
    };

    f32x4 acc[2][2] = {{{0, 0, 0, 0}, {0, 0, 0, 0}},
                       {{0, 0, 0, 0}, {0, 0, 0, 0}}};
    stage(0, 0);
    for (int kt = 0; kt < 1024 / 64; ++kt) {
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        __builtin_amdgcn_s_barrier();
        if (kt + 1 < 1024 / 64) stage((kt + 1) & 1, (kt + 1) * 64);
        const unsigned short* As = smem + (kt & 1) * (2 * 64 * 64);
        const unsigned short* Bs = As + 64 * 64;
        #pragma unroll
        for (int ksub = 0; ksub < 2; ++ksub) {
            #pragma unroll
            for (int fm = 0; fm < 2; ++fm) {
                const int row = wr * 32 + fm * 16 + l15;
                const int piece = (ksub * 4 + l4) ^ (row & 7);
                bf16x8 a = as_bf16x8(*(const u16x8*)&As[row * 64 + piece * 8]);
                #pragma unroll
                for (int fn = 0; fn < 2; ++fn) {
                    const int n_l = wc * 32 + fn * 16 + l15;
                    // B fragment: 8 k-values (rows of the [k][n] image) for
                    // one n; per-element reads honour the source swizzle
                    u16x8 bv;
                    #pragma unroll
                    for (int u = 0; u < 8; ++u) {
                        const int krow = ksub * 32 + l4 * 8 + u;
                        const int np = ((n_l >> 3) ^ (krow & 7)) * 8 + (n_l & 7);
                        bv[u] = Bs[krow * 64 + np];
                    }
                    acc[fm][fn] = mfma16(a, as_bf16x8(bv), acc[fm][fn]);
                }
            }
        }
    }
    __builtin_amdgcn_s_barrier();
    #pragma unroll
    for (int fm = 0; fm < 2; ++fm) {
        #pragma unroll
        for (int fn = 0; fn < 2; ++fn) {
            const int n = bn + wc * 32 + fn * 16 + l15;
            #pragma unroll
            for (int j = 0; j < 4; ++j) {
                const int r = bm + wr * 32 + fm * 16 + l4 * 4 + j;
                if (r < R) {
                    float dh = acc[fm][fn][j];
                    int64_t bi = (int64_t)(r / nq) * 3136 + n;
                    float ph = bf(phi[(int64_t)r * 3136 + n]);
                    float fv = bf(feat[bi]);
                    dphi[(int64_t)r * 3136 + n] =
                        f32_to_bf16(ph > 0.0f ? dh * fv : 0.0f);
                    atomicAdd(&dfeat[bi], dh * ph);
                }
            }
        }
    }
}

// ---------------------------------------------------------------------------
// dWcat[m][n] = sum_k dhh[k][m] * h[k][n]  (M=1024, N=3136, K=R)
// A-tile (dhh slice) staged transposed; epilogue splits rows 0..511 ->
// fc_h_v, 512..1023 -> fc_h_a, writing mu and sigma (x eps) grads.
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256) void gemm_dwcat_kernel(
    const unsigned short* __restrict__ dhh,  // [R,1024]
    const unsigned short* __restrict__ hbuf, // [R,3136]
    const unsigned short* __restrict__ weps_v,  // [512,3136]
    const unsigned short* __restrict__ weps_a,  // [512,3136]
    unsigned short* __restrict__ dw_mu_v, unsigned short* __restrict__ dw_sg_v,
    unsigned short* __restrict__ dw_mu_a, unsigned short* __restrict__ dw_sg_a,
    int R) {
    __shared__ unsigned short As[64 * 32];  // [k(32)][m(64)] transposed image
    __shared__ unsigned short Bs2[32 * 64]; // [k(32)][n(64)] staged h slice
    const int t = threadIdx.x;
    const int lane = t & 63, wave = t >> 6;
    const int wr = wave >> 1, wc = wave & 1;
    const int l15 = lane & 15, l4 = lane >> 4;
    const int bm = blockIdx.x * 64, bn = blockIdx.y * 64;

    f32x4 acc[2][2] = {{{0, 0, 0, 0}, {0, 0, 0, 0}},
                       {{0, 0, 0, 0}, {0, 0, 0, 0}}};
    for (int k0 = 0; k0 < R; k0 += 32) {
        // stage A^T and the B slice, rows coalesced
        {
            int kk = (t * 8) / 64, mm = (t * 8) % 64;
            u16x8 v = {0, 0, 0, 0, 0, 0, 0, 0};
            if (k0 + kk < R)
                v = *(const u16x8*)(dhh + (int64_t)(k0 + kk) * 1024 + bm + mm);
            *(u16x8*)&As[kk * 64 + mm] = v;
            u16x8 w = {0, 0, 0, 0, 0, 0, 0, 0};
            if (k0 + kk < R)
                w = *(const u16x8*)(hbuf + (int64_t)(k0 + kk) * 3136 + bn + mm);
            *(u16x8*)&Bs2[kk * 64 + mm] = w;
        }
        __syncthreads();
        #pragma unroll
        for (int fm = 0; fm < 2; ++fm) {
            const int m_l = wr * 32 + fm * 16 + l15;
            u16x8 av;
            #pragma unroll
            for (int u = 0; u < 8; ++u) av[u] = As[(l4 * 8 + u) * 64 + m_l];
            bf16x8 a = as_bf16x8(av);
            #pragma unroll
            for (int fn = 0; fn < 2; ++fn) {
                const int n_l = wc * 32 + fn * 16 + l15;
                u16x8 bv;
                #pragma unroll
                for (int u = 0; u < 8; ++u)
                    bv[u] = Bs2[(l4 * 8 + u) * 64 + n_l];
                acc[fm][fn] = mfma16(a, as_bf16x8(bv), acc[fm][fn]);
            }
        }
        __syncthreads();
    }
    #pragma unroll
    for (int fm = 0; fm < 2; ++fm) {
        #pragma unroll
        for (int fn = 0; fn < 2; ++fn) {
            const int n = bn + wc * 32 + fn * 16 + l15;
            #pragma unroll
            for (int j = 0; j < 4; ++j) {
                const int m = bm + wr * 32 + fm * 16 + l4 * 4 + j;
                const float g = acc[fm][fn][j];
                if (m < 512) {
                    int64_t idx = (int64_t)m * 3136 + n;
                    dw_mu_v[idx] = f32_to_bf16(g);
                    dw_sg_v[idx] = f32_to_bf16(g * bf(weps_v[idx]));
                } else {
                    int64_t idx = (int64_t)(m - 512) * 3136 + n;
                    dw_mu_a[idx] = f32_to_bf16(g);
                    dw_sg_a[idx] = f32_to_bf16(g * bf(weps_a[idx]));
                }
            }
        }
    }
}

// B-frag trap note: hbuf reads above are 8 strided u16 gathers; adjacent
// lanes (n contiguous) coalesce within a cacheline, and K=R is small.

// ---------------------------------------------------------------------------
// column sums: db[c] = sum_r X[r][c] -> bias mu/sigma grads
// grid = C/256; each thread owns one column.
// ---------------------------------------------------------------------------
__global__ void bias_grad_cols_kernel(
    const unsigned short* __restrict__ X, int R, int C,
    const unsigned short* __restrict__ beps,  // [C] or nullptr
    unsigned short* __restrict__ db_mu,       // [C]
    unsigned short* __restrict__ db_sg) {     // [C] or nullptr
    // WG covers 64 columns; thread (c_local, row-partition) layout so the
    // 4 waves split the serial row loop 4-ways; LDS-combined.
    __shared__ float part[64];
    const int t = threadIdx.x;
    const int cl = t & 63, rp = t >> 6;
    const int c = blockIdx.x * 64 + cl;
    float s = 0.0f;
    if (c < C) {
        for (int r = rp; r < R; r += 4) s += bf(X[(int64_t)r * C + c]);
    }
    if (t < 64) part[t] = 0.0f;
    __syncthreads();
    atomicAdd(&part[cl], s);
    __syncthreads();
    if (t < 64 && c < C) {
        float v = part[t];
        db_mu[c] = f32_to_bf16(v);
        if (db_sg != nullptr) db_sg[c] = f32_to_bf16(v * bf(beps[c]));
    }
}

// split-column variant for the concat dhh buffer: cols 0..511 -> layer V,
// 512..1023 -> layer A
__global__ void bias_grad_heads_kernel(
    const unsigned short* __restrict__ dhh, int R,
    const unsigned short* __restrict__ beps_v,
    const unsigned short* __restrict__ beps_a,
    unsigned short* __restrict__ db_mu_v, unsigned short* __restrict__ db_sg_v,
    unsigned short* __restrict__ db_mu_a, unsigned short* __restrict__ db_sg_a) {
    __shared__ float part[64];
    const int t = threadIdx.x;
    const int cl = t & 63, rp = t >> 6;
    const int c = blockIdx.x * 64 + cl;
    float s = 0.0f;
    for (int r = rp; r < R; r += 4) s += bf(dhh[(int64_t)r * 1024 + c]);
    if (t < 64) part[t] = 0.0f;
    __syncthreads();
    atomicAdd(&part[cl], s);
    __syncthreads();
    if (t >= 64) return;
    s = part[t];
    if (c < 512) {
        db_mu_v[c] = f32_to_bf16(s);
        db_sg_v[c] = f32_to_bf16(s * bf(beps_v[c]));
    } else {
        db_mu_a[c - 512] = f32_to_bf16(s);
        db_sg_a[c - 512] = f32_to_bf16(s * bf(beps_a[c - 512]));
    }
}

// ---------------------------------------------------------------------------
// dWiqn[n][i] = sum_r dphi[r][n] * cos(pi * i * tau_r)
// M=3136 (n), N=64 (i), K=R.  A-frag = 8 strided dphi gathers (lanes
// coalesce across n); B-frag = cos recomputed in-register from taus.
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256) void tau_dw_kernel(
    const unsigned short* __restrict__ dphi,  // [R,3136]
    const unsigned short* __restrict__ cosb,  // [R,64] bf16 (from tau_embed)
    unsigned short* __restrict__ dwiqn,       // [3136,64] bf16 out
    int R) {
    // one m-tile (16 n-rows) per WG; the 4 waves split the K=R reduction
    // and combine through LDS (196 WGs instead of 49)
    __shared__ unsigned short cs[256 * 64];  // staged cos table (R <= 256)
    __shared__ float part[4][16 * 64];       // per-wave D tiles
    const int t = threadIdx.x;
    const int lane = t & 63, wave = t >> 6;
    const int l15 = lane & 15, l4 = lane >> 4;
    const int mt = blockIdx.x;               // 196 m-tiles
    const int n = mt * 16 + l15;
    #pragma unroll 8
    for (int i = t * 4; i < R * 64; i += 256 * 4)
        *(u16x4*)&cs[i] = *(const u16x4*)(cosb + i);
    __syncthreads();

    f32x4 acc[4] = {f32x4{0, 0, 0, 0}, f32x4{0, 0, 0, 0},
                    f32x4{0, 0, 0, 0}, f32x4{0, 0, 0, 0}};
    const int kbeg = wave * (R / 4);
    #pragma unroll 2
    for (int k0 = kbeg; k0 < kbeg + R / 4; k0 += 32) {
        const int kk = k0 + l4 * 8;
        u16x8 av = {0, 0, 0, 0, 0, 0, 0, 0};
        #pragma unroll
        for (int u = 0; u < 8; ++u)
            if (kk + u < R) av[u] = dphi[(int64_t)(kk + u) * 3136 + n];
        bf16x8 a = as_bf16x8(av);
        #pragma unroll
        for (int it = 0; it < 4; ++it) {  // i-tiles: 64 cols
            u16x8 bv = {0, 0, 0, 0, 0, 0, 0, 0};
            #pragma unroll
            for (int u = 0; u < 8; ++u)
                if (kk + u < R) bv[u] = cs[(kk + u) * 64 + it * 16 + l15];
            acc[it] = mfma16(a, as_bf16x8(bv), acc[it]);
        }
    }
    // cross-wave combine: D element (nn = l4*4+j within tile, i = it*16+l15)
    #pragma unroll
    for (int it = 0; it < 4; ++it)
        #pragma unroll
        for (int j = 0; j < 4; ++j)
            part[wave][(l4 * 4 + j) * 64 + it * 16 + l15] = acc[it][j];
    __syncthreads();
    for (int i = t; i < 16 * 64; i += 256) {
        const int nn = mt * 16 + i / 64, col = i % 64;
        if (nn < 3136)
            dwiqn[(int64_t)nn * 64 + col] = f32_to_bf16(
                part[0][i] + part[1][i] + part[2][i] + part[3][i]);
    }
}

// ---------------------------------------------------------------------------
// conv backward data: one WG (512 thr) per state.
//   dpre3 = dfeat * (feat > 0)
//   dc2   = conv3^T(dpre3) * (c2out > 0)       M=81, N=64c, K=64n*16taps
//   dc1   = conv2^T(dc2)   * (c1out > 0)       M=400, N=32c, K=64n*16taps
// Activations c1/c2 come from the forward's side outputs (c1g/c2g).
// Outputs land transposed-by-channel for the dW GEMMs:
//   dpre3g [64][B*49], dc2g [64][B*81], dc1g [32][B*400]
// plus f32 bias-grad atomics.
// ---------------------------------------------------------------------------
#define BW_C1 12800  // 32*400 ush
#define BW_C2 5184   // 64*81
#define BW_P3 3136   // 64*49

// conv backward data, split for occupancy (B=32 states would otherwise fill
// only 32 CUs):
//   conv_bwd_dc2: grid (B, 3), unit = (m-tile, n-half) of conv3^T; also
//     computes dpre3 = dfeat * relu-mask (y==0 writes dpre3g + db3)
//   conv_bwd_dc1: grid (B, 7), m-tiles of conv2^T over dc2g
// Transposed weights come fragment-major (pack_bwd_frag_kernel).
__global__ __launch_bounds__(256) void conv_bwd_dc2_kernel(
    const float* __restrict__ dfeat,         // [B,3136] f32
    const unsigned short* __restrict__ feat, // [B,3136] bf16 (relu mask)
    const unsigned short* __restrict__ c2g,  // [B][64*81] (relu mask)
    const unsigned short* __restrict__ w3tp, // fragment-major [128*512]
    unsigned short* __restrict__ dpre3g,     // [64][B*49]
    unsigned short* __restrict__ dc2g,       // [64][B*81]
    float* __restrict__ db2s, float* __restrict__ db3s,
    int B) {
    __shared__ unsigned short p3s[BW_P3];
    const int b = blockIdx.x;
    const int y = blockIdx.y;
    const int t = threadIdx.x;
    const int lane = t & 63, wave = t >> 6;
    const int l15 = lane & 15, l4 = lane >> 4;

    #pragma unroll 4
    for (int i = t; i < BW_P3; i += 256) {
        float g = dfeat[(int64_t)b * 3136 + i];
        float m = bf(feat[(int64_t)b * 3136 + i]) > 0.0f ? 1.0f : 0.0f;
        p3s[i] = f32_to_bf16(g * m);
    }
    __syncthreads();
    if (y == 0) {  // db3 + dpre3g once
        if (wave < 8) {
            for (int n = wave; n < 64; n += 4) {
                float s = 0.0f;
                for (int px = lane; px < 49; px += 64) s += bf(p3s[n * 49 + px]);
                s = wave_allreduce_sum(s);
                if (lane == 0) atomicAdd(&db3s[n], s);
                for (int px = lane; px < 49; px += 64)
                    dpre3g[(int64_t)n * (B * 49) + b * 49 + px] =
                        p3s[n * 49 + px];
            }
        }
    }

    // units y*4 + wave in [0,12): (mt, nh)
    const int u = y * 4 + wave;
    const int mt = u >> 1, nh = u & 1;
    f32x4 acc[2] = {f32x4{0, 0, 0, 0}, f32x4{0, 0, 0, 0}};
    const int m = mt * 16 + l15;
    const int iy = m / 9, ix = m % 9;
    const bool mv = m < 81;
    int ga[8];
    unsigned short okm[8];
    {
        const int n0 = l4 >> 1;
        const int t16 = (l4 & 1) * 8;
        #pragma unroll
        for (int j = 0; j < 8; ++j) {
            const int tt = t16 + j, ky = tt >> 2, kx = tt & 3;
            const int oy = iy - ky, ox = ix - kx;
            const bool ok = mv && ky < 3 && kx < 3 && oy >= 0 && oy < 7 &&
                            ox >= 0 && ox < 7;
            ga[j] = n0 * 49 + (ok ? oy * 7 + ox : 0);
            okm[j] = ok ? 0xFFFFu : 0;
        }
    }
    bf16x8 bbuf[2][2];
    #pragma unroll
    for (int nt = 0; nt < 2; ++nt)
        bbuf[0][nt] = as_bf16x8(*(const u16x8*)(
            w3tp + ((nh * 2 + nt) << 9) + lane * 8));
    #pragma unroll 2
    for (int ks = 0; ks < 32; ++ks) {
        const int cur = ks & 1;
        if (ks + 1 < 32) {
            #pragma unroll
            for (int nt = 0; nt < 2; ++nt)
                bbuf[cur ^ 1][nt] = as_bf16x8(*(const u16x8*)(
                    w3tp + (((ks + 1) * 4 + nh * 2 + nt) << 9) + lane * 8));
        }
        u16x8 av;
        #pragma unroll
        for (int j = 0; j < 8; ++j) av[j] = p3s[ga[j]] & okm[j];
        #pragma unroll
        for (int j = 0; j < 8; ++j) ga[j] += 2 * 49;
        bf16x8 a = as_bf16x8(av);
        acc[0] = mfma16(a, bbuf[cur][0], acc[0]);
        acc[1] = mfma16(a, bbuf[cur][1], acc[1]);
    }
    #pragma unroll
    for (int nt = 0; nt < 2; ++nt) {
        const int c = (nh * 2 + nt) * 16 + l15;
        float bsum = 0.0f;
        #pragma unroll
        for (int j = 0; j < 4; ++j) {
            const int mm = mt * 16 + l4 * 4 + j;
            if (mm < 81) {
                float mask =
                    bf(c2g[(int64_t)b * BW_C2 + c * 81 + mm]) > 0.0f ? 1.0f
                                                                     : 0.0f;
                float v = acc[nt][j] * mask;
                dc2g[(int64_t)c * (B * 81) + b * 81 + mm] = f32_to_bf16(v);
                bsum += v;
            }
        }
        // db2: sum over pixels; lanes cover distinct (c, mm) pairs
        bsum = bsum;  // reduce over the 4 l4 groups of the same c
        #pragma unroll
        for (int off = 16; off < 64; off <<= 1)
            bsum += __shfl_xor(bsum, off, 64);
        if (l4 == 0) atomicAdd(&db2s[c], bsum);
    }
}

__global__ __launch_bounds__(256) void conv_bwd_dc1_kernel(
    const unsigned short* __restrict__ dc2g, // [64][B*81]
    const unsigned short* __restrict__ c1g,  // [B][32*400] (relu mask)
    const unsigned short* __restrict__ w2tp, // fragment-major [64*512]
    unsigned short* __restrict__ dc1g,       // [32][B*400]
    float* __restrict__ db1s,
    int B) {
    __shared__ unsigned short d2s[BW_C2];
    const int b = blockIdx.x;
    const int y = blockIdx.y;
    const int t = threadIdx.x;
    const int lane = t & 63, wave = t >> 6;
    const int l15 = lane & 15, l4 = lane >> 4;

    // stage this state's dc2 from its channel-major layout
    #pragma unroll 4
    for (int i = t; i < BW_C2; i += 256) {
        int n = i / 81, px = i % 81;
        d2s[i] = dc2g[(int64_t)n * (B * 81) + b * 81 + px];
    }
    __syncthreads();

    const int mt = y * 4 + wave;  // 25 m-tiles over 28 wave-slots
    if (mt >= 25) return;
    f32x4 acc[2] = {f32x4{0, 0, 0, 0}, f32x4{0, 0, 0, 0}};
    const int m = mt * 16 + l15;
    const int iy = m / 20, ix = m % 20;
    const bool mv = m < 400;
    int ga[8];
    unsigned short okm[8];
    {
        const int n0 = l4 >> 1;
        const int t16 = (l4 & 1) * 8;
        #pragma unroll
        for (int j = 0; j < 8; ++j) {
            const int tt = t16 + j, ky = tt >> 2, kx = tt & 3;
            const int ry = iy - ky, rx = ix - kx;
            const int oy = ry >> 1, ox = rx >> 1;
            const bool ok = mv && ry >= 0 && rx >= 0 && (ry & 1) == 0 &&
                            (rx & 1) == 0 && oy < 9 && ox < 9;
            ga[j] = n0 * 81 + (ok ? oy * 9 + ox : 0);
            okm[j] = ok ? 0xFFFFu : 0;
        }
    }
    bf16x8 bbuf[2][2];
    #pragma unroll
    for (int nt = 0; nt < 2; ++nt)
        bbuf[0][nt] = as_bf16x8(
            *(const u16x8*)(w2tp + (nt << 9) + lane * 8));
    #pragma unroll 2
    for (int ks = 0; ks < 32; ++ks) {
        const int cur = ks & 1;
        if (ks + 1 < 32) {
            #pragma unroll
            for (int nt = 0; nt < 2; ++nt)
                bbuf[cur ^ 1][nt] = as_bf16x8(*(const u16x8*)(
                    w2tp + (((ks + 1) * 2 + nt) << 9) + lane * 8));
        }
        u16x8 av;
        #pragma unroll
        for (int j = 0; j < 8; ++j) av[j] = d2s[ga[j]] & okm[j];
        #pragma unroll
        for (int j = 0; j < 8; ++j) ga[j] += 2 * 81;
        bf16x8 a = as_bf16x8(av);
        acc[0] = mfma16(a, bbuf[cur][0], acc[0]);
        acc[1] = mfma16(a, bbuf[cur][1], acc[1]);
    }
    #pragma unroll
    for (int nt = 0; nt < 2; ++nt) {
        const int c = nt * 16 + l15;
        float bsum = 0.0f;
        #pragma unroll
        for (int j = 0; j < 4; ++j) {
            const int mm = mt * 16 + l4 * 4 + j;
            if (mm < 400) {
                float mask =
                    bf(c1g[(int64_t)b * BW_C1 + c * 400 + mm]) > 0.0f ? 1.0f
                                                                      : 0.0f;
                float v = acc[nt][j] * mask;
                dc1g[(int64_t)c * (B * 400) + b * 400 + mm] = f32_to_bf16(v);
                bsum += v;
            }
        }
        #pragma unroll
        for (int off = 16; off < 64; off <<= 1)
            bsum += __shfl_xor(bsum, off, 64);
        if (l4 == 0) atomicAdd(&db1s[c], bsum);
    }
}

// ---------------------------------------------------------------------------
// conv weight grads as GEMMs over K = (state, out-px):
//  L=3: dW3[64][576]: A = dpre3g [64][B*49], B-tap = c2g im2col
//  L=2: dW2[64][512]: A = dc2g   [64][B*81], B-tap = c1g im2col
//  L=1: dW1[32][256]: A = dc1g   [32][B*400], B-tap = x im2col
// Split-K across gridDim.z; f32 atomic scratch; conv_finalize casts.
// ---------------------------------------------------------------------------
template <int L>
__global__ __launch_bounds__(256) void conv_bwd_dw_kernel(
    const unsigned short* __restrict__ dY,  // [M][B*PX]
    const unsigned short* __restrict__ act, // activations (see L)
    float* __restrict__ dWs,                // [M][N] f32 scratch
    int B, int splitk) {
    constexpr int M = (L == 1) ? 32 : 64;
    constexpr int N = (L == 1) ? 256 : (L == 2 ? 512 : 576);
    constexpr int PX = (L == 1) ? 400 : (L == 2 ? 81 : 49);
    constexpr int OW = (L == 1) ? 20 : (L == 2 ? 9 : 7);       // out width
    constexpr int IW = (L == 1) ? 84 : (L == 2 ? 20 : 9);      // in width
    constexpr int CIN = (L == 1) ? 4 : (L == 2 ? 32 : 64);
    constexpr int KW = (L == 1) ? 8 : (L == 2 ? 4 : 3);        // kernel w
    constexpr int STR = (L == 1) ? 4 : (L == 2 ? 2 : 1);
    const int t = threadIdx.x;
    const int lane = t & 63, wave = t >> 6;
    const int l15 = lane & 15, l4 = lane >> 4;
    const int mt = blockIdx.x, ntile = blockIdx.y * 4 + wave;
    const int n = ntile * 16 + l15;  // tap index (c,ky,kx)
    const int K = B * PX;
    const int kchunk = (K + splitk - 1) / splitk;
    const int kbeg = blockIdx.z * kchunk;
    const int kend = min(kbeg + kchunk, K);
    if (ntile * 16 >= N) return;

    // decode tap n -> (c, ky, kx)
    const int c = n / (KW * KW);
    const int rem = n % (KW * KW);
    const int ky = rem / KW, kx = rem % KW;

    f32x4 acc = {0, 0, 0, 0};
    for (int k0 = kbeg; k0 < kend; k0 += 32) {
        const int kk = k0 + l4 * 8;
        // A-frag: dY[mrow][kk..kk+8) contiguous
        u16x8 av = {0, 0, 0, 0, 0, 0, 0, 0};
        const int mrow = mt * 16 + l15;
        if (mrow < M) {
            #pragma unroll
            for (int u = 0; u < 8; ++u)
                if (kk + u < kend) av[u] = dY[(int64_t)mrow * K + kk + u];
        }
        // B-frag: im2col tap for 8 (state,px) values, fixed tap n
        u16x8 bv;
        #pragma unroll
        for (int u = 0; u < 8; ++u) {
            const int k = kk + u;
            const bool ok = k < kend && n < N;
            const int kc = ok ? k : kbeg;
            const int st = kc / PX, px = kc % PX;
            const int oy = px / OW, ox = px % OW;
            const int iy = oy * STR + ky, ix = ox * STR + kx;
            unsigned short v =
                act[((int64_t)st * CIN + min(c, CIN - 1)) * (IW * IW) +
                    iy * IW + ix];
            bv[u] = ok ? v : (unsigned short)0;
        }
        acc = mfma16(as_bf16x8(av), as_bf16x8(bv), acc);
    }
    // D[m][tap]: row = mt*16 + l4*4+j, col = n
    #pragma unroll
    for (int j = 0; j < 4; ++j) {
        const int m = mt * 16 + l4 * 4 + j;
        if (m < M && n < N) atomicAdd(&dWs[(int64_t)m * N + n], acc[j]);
    }
}

// cast f32 scratch -> bf16 grads (weights + biases), then zero the scratch
// for the next step (saves a memset launch)
__global__ void conv_finalize_kernel(
    float* __restrict__ dW1s, float* __restrict__ dW2s,
    float* __restrict__ dW3s, float* __restrict__ db1s,
    float* __restrict__ db2s, float* __restrict__ db3s,
    unsigned short* __restrict__ dw1, unsigned short* __restrict__ dw2,
    unsigned short* __restrict__ dw3, unsigned short* __restrict__ db1,
    unsigned short* __restrict__ db2, unsigned short* __restrict__ db3) {
    int i = threadIdx.x + blockIdx.x * blockDim.x;
    const int n1 = 32 * 256, n2 = 64 * 512, n3 = 64 * 576;
    if (i < n1) { dw1[i] = f32_to_bf16(dW1s[i]); dW1s[i] = 0.0f; }
    if (i < n2) { dw2[i] = f32_to_bf16(dW2s[i]); dW2s[i] = 0.0f; }
    if (i < n3) { dw3[i] = f32_to_bf16(dW3s[i]); dW3s[i] = 0.0f; }
    if (i < 32) { db1[i] = f32_to_bf16(db1s[i]); db1s[i] = 0.0f; }
    if (i < 64) {
        db2[i] = f32_to_bf16(db2s[i]); db2s[i] = 0.0f;
        db3[i] = f32_to_bf16(db3s[i]); db3s[i] = 0.0f;
    }
}

// Transposed conv weights for the data backward, packed FRAGMENT-MAJOR
// ([frag][lane][8], frag = ks*CT + ct) so each B-fragment load is one
// contiguous 1 KB block (see pack_fwd_frag_kernel in model_fwd.hip).
// Transposed element (c-row, k = n*16 + ky*4 + kx):
//   conv3: w3[n][c*9 + ky*3 + kx] (0 on the padded taps)
//   conv2: w2[n][c*16 + ky*4 + kx]
__global__ void pack_bwd_frag_kernel(const unsigned short* __restrict__ w3,
                                     const unsigned short* __restrict__ w2,
                                     unsigned short* __restrict__ w3tf,
                                     unsigned short* __restrict__ w2tf) {
    const int i = threadIdx.x + blockIdx.x * blockDim.x;
    const int lane = (i >> 3) & 63, j = i & 7;
    const int l15 = lane & 15, l4 = (lane >> 4) & 3;
    const int frag = i >> 9;
    if (i < 128 * 512) {  // w3tf: frag = ks*4 + ct, ks in [0,32), ct in [0,4)
        const int ks = frag >> 2, ct = frag & 3;
        const int c = ct * 16 + l15;
        const int k = ks * 32 + l4 * 8 + j;
        const int n = k >> 4, tt = k & 15, ky = tt >> 2, kx = tt & 3;
        w3tf[i] = (ky < 3 && kx < 3)
                      ? w3[n * 576 + c * 9 + ky * 3 + kx]
                      : (unsigned short)0;
    }
    if (i < 64 * 512) {  // w2tf: frag = ks*2 + ct, ks in [0,32), ct in [0,2)
        const int ks = frag >> 1, ct = frag & 1;
        const int c = ct * 16 + l15;
        const int k = ks * 32 + l4 * 8 + j;
        const int n = k >> 4, tt = k & 15;
        w2tf[i] = w2[n * 512 + c * 16 + tt];
    }
}

}  // namespace

extern "C" {

void riqn_finish_bwd_dhh(const float* dtheta, const int64_t* actions,
                         const void* hh, const void* wz, void* dhh, int R,
                         int A, int nq, void* stream) {
    hipLaunchKernelGGL(finish_bwd_dhh_kernel, dim3((R + 15) / 16), dim3(256),
                       0, (hipStream_t)stream, dtheta, actions,
                       (const unsigned short*)hh, (const unsigned short*)wz,
                       (unsigned short*)dhh, R, A, nq);
}

void riqn_zgrad(const float* dtheta, const int64_t* actions, const void* hh,
                int R, int A, int nq, const void* weps_v, const void* weps_a,
                const void* beps_v, const void* beps_a, void* dw_mu_v,
                void* dw_sg_v, void* db_mu_v, void* db_sg_v, void* dw_mu_a,
                void* dw_sg_a, void* db_mu_a, void* db_sg_a, void* stream) {
    hipLaunchKernelGGL(zgrad_kernel, dim3(1 + A, 4), dim3(256), 0,
                       (hipStream_t)stream, dtheta, actions,
                       (const unsigned short*)hh, R, A, nq,
                       (const unsigned short*)weps_v,
                       (const unsigned short*)weps_a,
                       (const unsigned short*)beps_v,
                       (const unsigned short*)beps_a,
                       (unsigned short*)dw_mu_v, (unsigned short*)dw_sg_v,
                       (unsigned short*)db_mu_v, (unsigned short*)db_sg_v,
                       (unsigned short*)dw_mu_a, (unsigned short*)dw_sg_a,
                       (unsigned short*)db_mu_a, (unsigned short*)db_sg_a);
}

void riqn_gemm_dx(const void* dhh, const void* wcat, const void* feat,
                  const void* phi, void* dphi, float* dfeat, int R, int nq,
                  void* stream) {
    dim3 grid((R + 63) / 64, 3136 / 64);
    hipLaunchKernelGGL(gemm_dx_kernel, grid, dim3(256), 0, (hipStream_t)stream,
                       (const unsigned short*)dhh, (const unsigned short*)wcat,
                       (const unsigned short*)feat, (const unsigned short*)phi,
                       (unsigned short*)dphi, dfeat, R, nq);
}

void riqn_gemm_dwcat(const void* dhh, const void* h, const void* weps_v,
                     const void* weps_a, void* dw_mu_v, void* dw_sg_v,
                     void* dw_mu_a, void* dw_sg_a, int R, void* stream) {
    dim3 grid(1024 / 64, 3136 / 64);
    hipLaunchKernelGGL(gemm_dwcat_kernel, grid, dim3(256), 0,
                       (hipStream_t)stream, (const unsigned short*)dhh,
                       (const unsigned short*)h, (const unsigned short*)weps_v,
                       (const unsigned short*)weps_a,
                       (unsigned short*)dw_mu_v, (unsigned short*)dw_sg_v,
                       (unsigned short*)dw_mu_a, (unsigned short*)dw_sg_a, R);
}

void riqn_bias_grad_heads(const void* dhh, int R, const void* beps_v,
                          const void* beps_a, void* db_mu_v, void* db_sg_v,
                          void* db_mu_a, void* db_sg_a, void* stream) {
    hipLaunchKernelGGL(bias_grad_heads_kernel, dim3(16), dim3(256), 0,
                       (hipStream_t)stream, (const unsigned short*)dhh, R,
                       (const unsigned short*)beps_v,
                       (const unsigned short*)beps_a,
                       (unsigned short*)db_mu_v, (unsigned short*)db_sg_v,
                       (unsigned short*)db_mu_a, (unsigned short*)db_sg_a);
}

void riqn_bias_grad_cols(const void* X, int R, int C, void* db_mu,
                         void* stream) {
    hipLaunchKernelGGL(bias_grad_cols_kernel, dim3((C + 63) / 64), dim3(256),
                       0, (hipStream_t)stream, (const unsigned short*)X, R, C,
                       (const unsigned short*)nullptr, (unsigned short*)db_mu,
                       (unsigned short*)nullptr);
}

void riqn_tau_dw(const void* dphi, const void* cosb, void* dwiqn, int R,
                 void* stream) {
    hipLaunchKernelGGL(tau_dw_kernel, dim3(3136 / 16), dim3(256), 0,
                       (hipStream_t)stream, (const unsigned short*)dphi,
                       (const unsigned short*)cosb, (unsigned short*)dwiqn, R);
}

void riqn_conv_bwd_data(const float* dfeat, const void* feat, const void* c1g,
                        const void* c2g, const void* w3tp, const void* w2tp,
                        void* dpre3g, void* dc2g, void* dc1g, float* db1s,
                        float* db2s, float* db3s, int B, void* stream) {
    hipStream_t s = (hipStream_t)stream;
    hipLaunchKernelGGL(conv_bwd_dc2_kernel, dim3(B, 3), dim3(256), 0, s,
                       dfeat, (const unsigned short*)feat,
                       (const unsigned short*)c2g,
                       (const unsigned short*)w3tp, (unsigned short*)dpre3g,
                       (unsigned short*)dc2g, db2s, db3s, B);
    hipLaunchKernelGGL(conv_bwd_dc1_kernel, dim3(B, 7), dim3(256), 0, s,
                       (const unsigned short*)dc2g,
                       (const unsigned short*)c1g,
                       (const unsigned short*)w2tp, (unsigned short*)dc1g,
                       db1s, B);
}

void riqn_conv_bwd_dw(int layer, const void* dY, const void* act, float* dWs,
                      int B, int splitk, void* stream) {
    hipStream_t s = (hipStream_t)stream;
    if (layer == 1) {
        dim3 grid(2, 4, splitk);
        hipLaunchKernelGGL(conv_bwd_dw_kernel<1>, grid, dim3(256), 0, s,
                           (const unsigned short*)dY,
                           (const unsigned short*)act, dWs, B, splitk);
    } else if (layer == 2) {
        dim3 grid(4, 8, splitk);
        hipLaunchKernelGGL(conv_bwd_dw_kernel<2>, grid, dim3(256), 0, s,
                           (const unsigned short*)dY,
                           (const unsigned short*)act, dWs, B, splitk);
    } else {
        dim3 grid(4, 9, splitk);
        hipLaunchKernelGGL(conv_bwd_dw_kernel<3>, grid, dim3(256), 0, s,
                           (const unsigned short*)dY,
                           (const unsigned short*)act, dWs, B, splitk);
    }
}

void riqn_conv_finalize(float* dW1s, float* dW2s, float* dW3s, float* db1s,
                        float* db2s, float* db3s, void* dw1, void* dw2,
                        void* dw3, void* db1, void* db2, void* db3,
                        void* stream) {
    hipLaunchKernelGGL(conv_finalize_kernel, dim3((64 * 576 + 255) / 256),
                       dim3(256), 0, (hipStream_t)stream, dW1s, dW2s, dW3s,
                       db1s, db2s, db3s, (unsigned short*)dw1,
                       (unsigned short*)dw2, (unsigned short*)dw3,
                       (unsigned short*)db1, (unsigned short*)db2,
                       (unsigned short*)db3);
}

void riqn_pack_bwd_weights(const void* w3, const void* w2, void* w3tp,
                           void* w2tp, void* stream) {
    hipLaunchKernelGGL(pack_bwd_frag_kernel, dim3(128 * 512 / 256), dim3(256),
                       0, (hipStream_t)stream, (const unsigned short*)w3,
                       (const unsigned short*)w2, (unsigned short*)w3tp,
                       (unsigned short*)w2tp);
}

}  // extern "C"
