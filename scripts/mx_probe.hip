// Discover v_mfma_scale_f32_16x16x128_f8f6f4 operand layout on gfx950.
// Build: hipcc --offload-arch=gfx950 -O3 scripts/mx_probe.hip -o scripts/mx_probe
#include <hip/hip_runtime.h>
#include <cmath>
#include <cstdio>

typedef __attribute__((ext_vector_type(8))) int i32x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

// A, B: [16][128] fp8 bytes in row-major; scales [16][4] e8m0 per row per
// 32-block; D out [16][16] f32.  Assumed lane map: lane l: row=l&15,
// k-block=(l>>4), holds 32 bytes A[row][kb*32..+32), scale[row][kb].
__global__ void mx_kernel(const unsigned char* A, const unsigned char* B,
                          const unsigned char* sA, const unsigned char* sB,
                          float* D) {
    int l = threadIdx.x;
    int row = l & 15, kb = l >> 4;
    i32x8 a = *(const i32x8*)(A + row * 128 + kb * 32);
    i32x8 b = *(const i32x8*)(B + row * 128 + kb * 32);
    int sa = sA[row * 4 + kb];
    int sb = sB[row * 4 + kb];
    f32x4 c = {0, 0, 0, 0};
    c = __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(a, b, c, 0, 0, 0, sa,
                                                         0, sb);
    // D map assumption: col=l&15, row=(l>>4)*4+reg
    for (int j = 0; j < 4; ++j) D[((l >> 4) * 4 + j) * 16 + (l & 15)] = c[j];
}

static unsigned char f32_to_e4m3(float x) {
    // encode small integers exactly: use bit twiddle via round-trip table
    // e4m3fn: 1-4-3, bias 7, max 448
    if (x == 0) return 0;
    int s = x < 0;
    float ax = fabsf(x);
    int e = (int)floorf(log2f(ax));
    if (e < -6) e = -6;
    float m = ax / exp2f((float)e);
    int mant = (int)roundf((m - 1.0f) * 8.0f);
    if (mant == 8) { mant = 0; e += 1; }
    int eb = e + 7;
    if (eb < 0) return 0;
    return (unsigned char)((s << 7) | (eb << 3) | mant);
}

static float e4m3_to_f32(unsigned char v) {
    int s = v >> 7, eb = (v >> 3) & 15, m = v & 7;
    if (eb == 0 && m == 0) return 0.0f;
    float val;
    if (eb == 0) val = (m / 8.0f) * exp2f(-6.0f);
    else val = (1.0f + m / 8.0f) * exp2f((float)(eb - 7));
    return s ? -val : val;
}

int main() {
    unsigned char hA[16 * 128], hB[16 * 128], hsA[64], hsB[64];
    for (int i = 0; i < 16; ++i)
        for (int k = 0; k < 128; ++k) {
            hA[i * 128 + k] = f32_to_e4m3(0.25f * ((i + 2 * k) % 5 - 2));
            hB[i * 128 + k] = f32_to_e4m3(0.125f * ((3 * i + k) % 7 - 3));
        }
    unsigned char *dA, *dB, *dsA, *dsB; float* dD;
    hipMalloc(&dA, sizeof hA); hipMalloc(&dB, sizeof hB);
    hipMalloc(&dsA, 64); hipMalloc(&dsB, 64); hipMalloc(&dD, 256 * 4);
    hipMemcpy(dA, hA, sizeof hA, hipMemcpyHostToDevice);
    hipMemcpy(dB, hB, sizeof hB, hipMemcpyHostToDevice);
    float hD[256], ref[256];

    // one-hot B probes for each scale byte kb: which (lane, byte) operand
    // positions (reported as my linear k = lane*32 + byte) it multiplies
    for (int kb = 0; kb < 4; ++kb) {
        for (int i = 0; i < 64; ++i) { hsA[i] = 127; hsB[i] = 127; }
        hsA[1 * 4 + kb] = 128;
        hipMemcpy(dsA, hsA, 64, hipMemcpyHostToDevice);
        hipMemcpy(dsB, hsB, 64, hipMemcpyHostToDevice);
        printf("scale byte kb=%d multiplies k:", kb);
        for (int p = 0; p < 8; ++p) {
            unsigned char oneB[16 * 128];
            for (int j = 0; j < 16; ++j)
                for (int k = 0; k < 128; ++k)
                    oneB[j * 128 + k] = (k == j + 16 * p) ? f32_to_e4m3(1.0f) : 0;
            hipMemcpy(dB, oneB, sizeof oneB, hipMemcpyHostToDevice);
            hipLaunchKernelGGL(mx_kernel, dim3(1), dim3(64), 0, 0, dA, dB,
                               dsA, dsB, dD);
            hipMemcpy(hD, dD, sizeof hD, hipMemcpyDeviceToHost);
            for (int j = 0; j < 16; ++j) {
                int k = j + 16 * p;
                float base = e4m3_to_f32(hA[1 * 128 + k]);
                float got = hD[1 * 16 + j];
                if (base != 0 && fabsf(got / base - 2.0f) < 0.01f)
                    printf(" %d", k);
            }
        }
        printf("\n");
    }
    return 0;
}
