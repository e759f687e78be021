// Device-resident synthetic Atari env (the benchmark's frame source).
//
// BASELINE.json prescribes synthetic 84x84x4 frames (no ALE wheel on these
// boxes -- SURVEY.md 3.2); this kernel family keeps the whole Ape-X actor
// round on-GPU so it can be hipGraph-captured end to end:
//
//   env_step : per env stream e -- score the action (reward/done via philox,
//              same functional form as env.py::VectorSyntheticEnv), record
//              the transition (pre-action last frame + action/reward/done)
//              into the outgoing chunk, then synthesize the next frame into
//              the frame-stack ring.
//   stack_gather : ring -> model input [E, h, 84, 84] f32/255, zeroing
//              pre-episode frames (episode bookkeeping via steps-since-reset).
//
// env_state[e] = {t_frames, steps_since_reset, phase, pad} (int32).
#include "common.h"
#include "riqn_kernels.h"

#define FRAME_PX 7056
#define FRAME_W 84
#define FRAME_H 84

namespace {

DEV uint8_t synth_pixel(int x, int y, int t, int cy, int cx, uint8_t noise) {
    uint8_t v = (uint8_t)((x * 3 + y * 2 + t * 5) & 0xFF);
    if (x >= cx - 4 && x < cx + 4 && x >= 0 && y >= cy - 4 && y < cy + 4)
        v = 255;
    return v | noise;
}

// Learnable cue task (task=1; env.py::CueEnv): dim background + bright
// 20x20 block in one of 4 horizontal slots; the matching action pays +1,
// everything else -1.  Reward is a deterministic function of the frame, so
// the full fused stack can DEMONSTRABLY learn (policy accuracy -> 1), which
// the random-reward throughput env cannot show.
#define CUE_SLOTS 4

DEV uint8_t cue_pixel(int x, int y, int t, int cue, uint8_t noise) {
    uint8_t v = (uint8_t)((x * 3 + y * 2 + t * 5) & 0x3F);
    int x0 = cue * (FRAME_W / CUE_SLOTS) + 1;
    if (y >= 32 && y < 52 && x >= x0 && x < x0 + 19) v = 255;
    return v | noise;
}

__global__ void env_bump_kernel(uint64_t* rng_state) { rng_state[1] += 1; }

__global__ void env_step_kernel(
    const int64_t* __restrict__ actions,  // [E]
    int32_t* __restrict__ env_state,      // [E,4]
    uint64_t* rng_state, uint8_t* __restrict__ ring,  // [h][E][84*84]
    int history, int ring_head_new,  // slot to write the NEW frame into
    int n_envs, float reward_clip, float p_done, int max_episode_frames,
    uint8_t* __restrict__ chunk_frames,  // [C][84*84], entry chunk_pos*E+e
    float* __restrict__ chunk_meta,      // [C][3]
    int chunk_pos, int task) {
    int e = blockIdx.x;
    __shared__ int t_s, phase_s, cy_s, cx_s, cue_s;
    __shared__ uint32_t nbase_s;
    if (threadIdx.x == 0) {
        int t = env_state[4 * e + 0];
        int steps = env_state[4 * e + 1];
        int phase = env_state[4 * e + 2];
        // score the action taken from the CURRENT state
        Philox rng(rng_state[0], rng_state[1], (uint32_t)(e * 4 + 1));
        float u01 = rng.uniform();
        float ud = rng.uniform();
        int t_next = t + 4;
        int a = (int)actions[e];
        float reward = 0.0f;
        if (task == 1) {
            // cue task: pay off against the cue VISIBLE in the acted-on
            // state, then redraw it for the next frame
            int cue_old = env_state[4 * e + 3];
            reward = (a == cue_old) ? 1.0f : -1.0f;
            int cue_new = (int)(u01 * CUE_SLOTS) & (CUE_SLOTS - 1);
            env_state[4 * e + 3] = cue_new;
            cue_s = cue_new;
        } else {
            if (u01 < 0.02f) reward = ((a + t_next) % 3 != 0) ? 1.0f : -1.0f;
        }
        if (reward_clip > 0.0f)
            reward = fmaxf(-reward_clip, fminf(reward_clip, reward));
        bool done = (ud < p_done) || (t_next >= max_episode_frames);
        int64_t ci = (int64_t)chunk_pos * n_envs + e;
        chunk_meta[3 * ci + 0] = (float)a;
        chunk_meta[3 * ci + 1] = reward;
        chunk_meta[3 * ci + 2] = done ? 1.0f : 0.0f;
        if (done) {
            uint32_t r4[4];
            rng.next4(r4);
            t_next = 0;
            phase = (int)(r4[0] & 0xFFFF);
            env_state[4 * e + 1] = 0;
        } else {
            env_state[4 * e + 1] = steps + 1;
        }
        env_state[4 * e + 0] = t_next;
        env_state[4 * e + 2] = phase;
        int tt = (t_next + phase) & 0xFFFF;
        t_s = tt;
        phase_s = phase;
        cy_s = (int)(FRAME_H / 2 + (FRAME_H / 2 - 6) * __sinf(tt * 0.11f));
        cx_s = (int)(FRAME_W / 2 + (FRAME_W / 2 - 6) * __cosf(tt * 0.07f));
        Philox nrng(rng_state[0], rng_state[1], (uint32_t)(e * 4 + 2));
        uint32_t r4[4];
        nrng.next4(r4);
        nbase_s = r4[0];
    }
    __syncthreads();
    // record the pre-action frame (the state's most recent frame) into the
    // chunk: ring slot (head_new - 1) holds it
    int prev_slot = (ring_head_new - 1 + history) % history;
    const uint8_t* prev = ring + ((int64_t)prev_slot * n_envs + e) * FRAME_PX;
    uint8_t* cf = chunk_frames + ((int64_t)chunk_pos * n_envs + e) * FRAME_PX;
    for (int i = threadIdx.x; i < FRAME_PX / 4; i += blockDim.x)
        ((uint32_t*)cf)[i] = ((const uint32_t*)prev)[i];
    // synthesize the new frame
    int t = t_s, cy = cy_s, cx = cx_s;
    uint8_t* dst = ring + ((int64_t)ring_head_new * n_envs + e) * FRAME_PX;
    // four pixels per dword store: the per-byte store loop was store-issue
    // bound (FRAME_PX byte insts per WG); FRAME_PX = 7056 = 4 * 1764
    for (int i4 = threadIdx.x; i4 < FRAME_PX / 4; i4 += blockDim.x) {
        uint32_t word = 0;
        #pragma unroll
        for (int jj = 0; jj < 4; ++jj) {
            const int i = i4 * 4 + jj;
            int x = i % FRAME_W, y = i / FRAME_W;
            // cheap per-pixel noise hash (5 bits, matches CPU env's 0..31)
            uint32_t h = (nbase_s ^ (uint32_t)(i * 2654435761u));
            h ^= h >> 15;
            uint8_t noise = (uint8_t)(h & 31);
            uint8_t px = task == 1 ? cue_pixel(x, y, t, cue_s, noise)
                                   : synth_pixel(x, y, t, cy, cx, noise);
            word |= (uint32_t)px << (8 * jj);
        }
        ((uint32_t*)dst)[i4] = word;
    }
    // NOTE: offset bump rides a trailing 1-thread kernel (launcher):
    // bumping from block 0 here would race late blocks' offset reads when
    // the grid exceeds resident occupancy.
}

template <int OUT_BF16>
__global__ void stack_gather_kernel(
    const uint8_t* __restrict__ ring, const int32_t* __restrict__ env_state,
    int history, int ring_head,  // slot of the NEWEST frame
    int n_envs, float* __restrict__ out /* [E][h][84*84] */) {
    int e = blockIdx.x;
    int steps = env_state[4 * e + 1];  // frames since episode start
    const float inv255 = 1.0f / 255.0f;
    for (int j = 0; j < history; ++j) {
        int back = history - 1 - j;
        int64_t off = ((int64_t)e * history + j) * FRAME_PX;
        bool fill = steps >= back;
        int slot = (ring_head - back + 2 * history) % history;
        const uchar4* src =
            (const uchar4*)(ring + ((int64_t)slot * n_envs + e) * FRAME_PX);
        if (OUT_BF16) {
            ushort4_t* d4 = (ushort4_t*)((unsigned short*)(void*)out + off);
            ushort4_t z = {0, 0, 0, 0};
            for (int i = threadIdx.x; i < FRAME_PX / 4; i += blockDim.x) {
                if (fill) {
                    uchar4 u = src[i];
                    d4[i] = f32x4_to_bf16x4(u.x * inv255, u.y * inv255,
                                            u.z * inv255, u.w * inv255);
                } else {
                    d4[i] = z;
                }
            }
        } else {
            float4* d4 = (float4*)(out + off);
            float4 z = make_float4(0.f, 0.f, 0.f, 0.f);
            for (int i = threadIdx.x; i < FRAME_PX / 4; i += blockDim.x) {
                if (fill) {
                    uchar4 u = src[i];
                    d4[i] = make_float4(u.x * inv255, u.y * inv255,
                                        u.z * inv255, u.w * inv255);
                } else {
                    d4[i] = z;
                }
            }
        }
    }
}

}  // namespace

extern "C" {

void riqn_env_step(const int64_t* actions, int32_t* env_state,
                   uint64_t* rng_state, uint8_t* ring, int history,
                   int ring_head_new, int n_envs, float reward_clip,
                   float p_done, int max_episode_frames, uint8_t* chunk_frames,
                   float* chunk_meta, int chunk_pos, int task, void* stream) {
    hipLaunchKernelGGL(env_step_kernel, dim3(n_envs), dim3(256), 0,
                       (hipStream_t)stream, actions, env_state, rng_state,
                       ring, history, ring_head_new, n_envs, reward_clip,
                       p_done, max_episode_frames, chunk_frames, chunk_meta,
                       chunk_pos, task);
    hipLaunchKernelGGL(env_bump_kernel, dim3(1), dim3(1), 0,
                       (hipStream_t)stream, rng_state);
}

void riqn_stack_gather(const uint8_t* ring, const int32_t* env_state,
                       int history, int ring_head, int n_envs, float* out,
                       int out_bf16, void* stream) {
    if (out_bf16)
        hipLaunchKernelGGL(stack_gather_kernel<1>, dim3(n_envs), dim3(256), 0,
                           (hipStream_t)stream, ring, env_state, history,
                           ring_head, n_envs, out);
    else
        hipLaunchKernelGGL(stack_gather_kernel<0>, dim3(n_envs), dim3(256), 0,
                           (hipStream_t)stream, ring, env_state, history,
                           ring_head, n_envs, out);
}

}  // extern "C"
