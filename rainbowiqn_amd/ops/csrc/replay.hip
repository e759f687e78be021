// On-GPU prioritized sum-tree replay for MI355X (SURVEY.md kernel K16).
//
// The reference keeps its replay as a CPU numpy sum-tree fed through Redis
// (rainbowiqn/memory.py); here the whole replay -- uint8
// frame ring, priority sum-tree, stochastic prefix-sum sampling, n-step
// return assembly, frame-stack reconstruction and importance weights -- lives
// in HBM3E and runs as three kernel families so the learner step can be
// hipGraph-captured end to end:
//
//   append  : meta blit + ballot-scan episode timesteps (1 WG) -> frame
//             blit (C blocks) -> leaf stores + level-wise ancestor range
//             recompute (1 WG, handles ring wrap)
//   sample  : S1 stratified tree descent + IS weights (1 block)
//             S2 gather/assemble + fused u8->f32 preprocess (B blocks)
//   update  : priority delta propagation with atomicExch/atomicAdd
//
// Ring layout: capacity = n_rings * ring_cap; ring r (one per Ape-X actor)
// owns slots [r*ring_cap, (r+1)*ring_cap) so per-actor episode streams stay
// contiguous for frame-stack/n-step reconstruction.  ring_state[r] =
// {head, count, episode_t, pad} (int32).  Tree: float32 heap, nodes[1] root,
// leaves at [tree_size, tree_size + capacity).
#include "common.h"
#include "riqn_kernels.h"

#define FRAME_PX 7056  // 84*84

namespace {

DEV void tree_set_leaf(float* tree, int64_t tree_size, int64_t leaf, float p) {
    float old = atomicExch(&tree[tree_size + leaf], p);
    float delta = p - old;
    uint64_t node = (uint64_t)(tree_size + leaf) >> 1;
    while (node >= 1) {
        atomicAdd(&tree[node], delta);
        if (node == 1) break;
        node >>= 1;
    }
}

struct RingView {
    const int32_t* rs;
    int64_t ring_cap;
    DEV int head(int r) const { return rs[4 * r + 0]; }
    DEV int count(int r) const { return rs[4 * r + 1]; }
};

DEV bool slot_valid(int64_t leaf, int64_t capacity, int64_t ring_cap,
                    const int32_t* ring_state, int history, int n_step,
                    const float* tree, int64_t tree_size) {
    if (leaf >= capacity) return false;
    int r = (int)(leaf / ring_cap);
    int local = (int)(leaf % ring_cap);
    int head = ring_state[4 * r + 0];
    int count = ring_state[4 * r + 1];
    int fwd = (int)((head - local + ring_cap) % ring_cap);
    if (fwd <= n_step) return false;
    if (count < ring_cap) {
        if (local >= count || local + n_step >= count) return false;
    } else {
        int bwd = (int)((local - head + ring_cap) % ring_cap);
        if (bwd < history - 1) return false;
    }
    return tree[tree_size + leaf] > 0.0f;
}

// ---------------------------------------------------------------------------
// APPEND
// ---------------------------------------------------------------------------
// Bulk fields in parallel; the sequential episode-timestep chain
// (t <- term ? 0 : t+1) is a 64-wide ballot scan on wave 0: position k's t
// is (k - j - 1) for the last terminal j < k in the chunk, or t0 + k before
// the first terminal.  One block; ~C/64 scan iterations instead of C.
__global__ void append_meta_kernel(
    int32_t* actions, float* rewards, uint8_t* terminals, int32_t* timesteps,
    int32_t* ring_state, int ring_id, int64_t ring_cap,
    const float* chunk_meta, int64_t C) {
    __shared__ int t_final_s;
    const int head = ring_state[4 * ring_id + 0];
    const int count = ring_state[4 * ring_id + 1];
    const int64_t base = (int64_t)ring_id * ring_cap;
    // bulk: actions/rewards/terminals (all 4 waves)
    for (int64_t k = threadIdx.x; k < C; k += blockDim.x) {
        int64_t slot = base + (head + k) % ring_cap;
        actions[slot] = (int32_t)chunk_meta[3 * k + 0];
        rewards[slot] = chunk_meta[3 * k + 1];
        terminals[slot] = chunk_meta[3 * k + 2] > 0.5f ? 1 : 0;
    }
    // timestep scan: wave 0 only, 64 entries per iteration
    if (threadIdx.x < 64) {
        const int l = threadIdx.x;
        int t0 = ring_state[4 * ring_id + 2];  // episode t at chunk start
        for (int64_t g = 0; g < C; g += 64) {
            int64_t k = g + l;
            bool term = (k < C) && (chunk_meta[3 * k + 2] > 0.5f);
            uint64_t mask = __ballot(term);
            uint64_t below = mask & ((l == 0) ? 0ull : (~0ull >> (64 - l)));
            int t;
            if (below) {
                int j = 63 - __clzll(below);
                t = l - j - 1;
            } else {
                t = t0 + l;
            }
            if (k < C) timesteps[base + (head + k) % ring_cap] = t;
            // carry episode-t across groups
            int valid = (int)min((int64_t)64, C - g);
            uint64_t vmask = mask & ((valid == 64) ? ~0ull : ((1ull << valid) - 1));
            if (vmask) {
                int h = 63 - __clzll(vmask);
                t0 = valid - h - 1;
            } else {
                t0 += valid;
            }
        }
        if (l == 0) t_final_s = t0;
    }
    __syncthreads();  // everyone has read head/count; scan done
    if (threadIdx.x == 0) {
        ring_state[4 * ring_id + 0] = (int)((head + C) % ring_cap);
        ring_state[4 * ring_id + 1] = (int)min((int64_t)count + C, ring_cap);
        ring_state[4 * ring_id + 2] = t_final_s;
    }
}

__global__ void append_frames_kernel(
    uint8_t* frames, const int32_t* ring_state, int ring_id, int64_t ring_cap,
    const uint8_t* chunk_frames, int64_t C) {
    // runs AFTER append_meta: old head = head - C (mod ring_cap)
    int head_new = ring_state[4 * ring_id + 0];
    int64_t head_old = (head_new - C % ring_cap + ring_cap) % ring_cap;
    int64_t k = blockIdx.x;  // one block per chunk entry
    int local = (int)((head_old + k) % ring_cap);
    int64_t slot = (int64_t)ring_id * ring_cap + local;
    // 7056 B per frame = 1764 u32; 256 threads stride
    const uint32_t* src = (const uint32_t*)(chunk_frames + k * FRAME_PX);
    uint32_t* dst = (uint32_t*)(frames + slot * FRAME_PX);
    for (int i = threadIdx.x; i < FRAME_PX / 4; i += blockDim.x) dst[i] = src[i];
}

// Appended leaves are a contiguous ring range, so instead of C atomic
// delta-walks to the root (serialized near the top), set the leaves with
// plain stores and recompute the affected ancestor RANGE level by level
// (node = left + right) in one block with barriers.  Handles ring wrap as
// two ranges.  Also removes fp32 drift (no += deltas on this path).
__global__ __launch_bounds__(1024) void append_tree_kernel(
    float* tree, int64_t tree_size, const int32_t* ring_state, int ring_id,
    int64_t ring_cap, const float* max_priority, int64_t C) {
    const int head_new = ring_state[4 * ring_id + 0];
    const int64_t head_old = (head_new - C % ring_cap + ring_cap) % ring_cap;
    const float p = max_priority[0];
    const int64_t base = (int64_t)ring_id * ring_cap;
    const int t = threadIdx.x;
    // two leaf ranges [s0, e0) and [s1, e1) in absolute leaf index
    const int64_t first = min(C, ring_cap - head_old);
    const int64_t s0 = base + head_old, e0 = s0 + first;
    const int64_t s1 = base, e1 = base + (C - first);
    for (int64_t k = t; k < C; k += blockDim.x) {
        int64_t leaf = (k < first) ? (s0 + k) : (s1 + (k - first));
        tree[tree_size + leaf] = p;
    }
    __syncthreads();
    // recompute ancestors: ranges in node space, inclusive
    int64_t lo0 = (tree_size + s0) >> 1, hi0 = (tree_size + e0 - 1) >> 1;
    int64_t lo1 = (tree_size + s1) >> 1, hi1 = (tree_size + e1 - 1) >> 1;
    const bool two = C > first;
    while (hi0 >= 1) {
        for (int64_t n = lo0 + t; n <= hi0; n += blockDim.x)
            tree[n] = tree[2 * n] + tree[2 * n + 1];
        if (two && hi0 > 1)  // second range merges with the first near the top
            for (int64_t n = lo1 + t; n <= hi1 && n < lo0; n += blockDim.x)
                tree[n] = tree[2 * n] + tree[2 * n + 1];
        __syncthreads();
        if (hi0 == 1) break;
        lo0 >>= 1; hi0 >>= 1; lo1 >>= 1; hi1 >>= 1;
    }
}

// ---------------------------------------------------------------------------
// SAMPLE
// ---------------------------------------------------------------------------
__global__ void sample_indices_kernel(
    const float* tree, int64_t tree_size, int64_t capacity,
    const int32_t* ring_state, int n_rings, int64_t ring_cap, int history,
    int n_step, const float* beta_ptr, uint64_t* rng_state, int batch,
    int32_t* out_idx, float* out_weights) {
    __shared__ float lds[32];
    __shared__ int total_count_s;
    int b = threadIdx.x;
    if (b == 0) {
        int tc = 0;
        for (int r = 0; r < n_rings; ++r) tc += ring_state[4 * r + 1];
        total_count_s = tc;
    }
    __syncthreads();
    float total = tree[1];
    float beta = beta_ptr[0];
    int64_t leaf = -1;
    float prob = 0.0f;
    if (b < batch) {
        Philox rng(rng_state[0], rng_state[1], (uint32_t)b);
        for (int attempt = 0; attempt < 64; ++attempt) {
            float u;
            if (attempt < 8)
                u = ((float)b + rng.uniform()) * (total / batch);
            else
                u = rng.uniform() * total;
            u = fminf(u, total * (1.0f - 1e-6f));
            // prefix-sum descent
            int64_t node = 1;
            float v = u;
            while (node < tree_size) {
                float left = tree[2 * node];
                if (v < left) {
                    node = 2 * node;
                } else {
                    v -= left;
                    node = 2 * node + 1;
                }
            }
            int64_t cand = node - tree_size;
            if (slot_valid(cand, capacity, ring_cap, ring_state, history,
                           n_step, tree, tree_size)) {
                leaf = cand;
                break;
            }
        }
        if (leaf < 0) {
            // last resort: uniform over a definitely-valid region of a ring
            for (int rr = 0; rr < n_rings && leaf < 0; ++rr) {
                int r = (b + rr) % n_rings;
                int head = ring_state[4 * r + 0];
                int count = ring_state[4 * r + 1];
                if (count <= history + n_step + 1) continue;
                int local;
                if (count < ring_cap) {
                    int span = max(1, count - n_step - 1);
                    local = (int)(rng.uniform() * span);
                } else {
                    int lo = (int)((head + history - 1) % ring_cap);
                    int span = max(1, (int)(ring_cap - (n_step + history)));
                    local = (int)((lo + (int)(rng.uniform() * span)) % ring_cap);
                }
                leaf = (int64_t)r * ring_cap + local;
            }
            if (leaf < 0) leaf = 0;  // empty replay guard (caller prevents)
        }
        prob = fmaxf(tree[tree_size + leaf], 1e-12f * fmaxf(total, 1.0f)) / total;
        out_idx[b] = (int32_t)leaf;
    }
    // importance weights: w = (N * P)^-beta / max_b w
    float w = (b < batch) ? __powf((float)total_count_s * prob, -beta) : 0.0f;
    __syncthreads();
    float wmax = block_reduce_max(w, lds);
    if (threadIdx.x == 0) lds[0] = wmax;
    __syncthreads();
    if (b < batch) out_weights[b] = w / lds[0];
    // single-block kernel (grid=1): every thread's offset read happened
    // before the block_reduce barriers above, so this bump cannot race a
    // late block's read (contrast rng.hip's multi-block discipline)
    if (threadIdx.x == 0) rng_state[1] += 1;  // advance philox offset
}

__global__ void sample_scalars_kernel(
    const int32_t* actions, const float* rewards, const uint8_t* terminals,
    int64_t ring_cap, int n_step, float discount, const int32_t* in_idx,
    int batch, int64_t* out_actions, float* out_returns,
    float* out_nonterminals) {
    int b = threadIdx.x + blockIdx.x * blockDim.x;
    if (b >= batch) return;
    int64_t leaf = in_idx[b];
    int r = (int)(leaf / ring_cap);
    int local = (int)(leaf % ring_cap);
    int64_t base = (int64_t)r * ring_cap;
    float ret = 0.0f, g = 1.0f;
    int alive = 1;
    for (int s = 0; s < n_step; ++s) {
        int64_t j = base + (local + s) % ring_cap;
        ret += g * rewards[j];
        if (terminals[j]) { alive = 0; break; }
        g *= discount;
    }
    out_actions[b] = (int64_t)actions[leaf];
    out_returns[b] = ret;
    out_nonterminals[b] = (float)alive;
}

// grid (batch, 2*history): block (b, j<h: state frame j; j>=h: next frame)
// OUT_BF16: write __hip_bfloat16-packed ushort instead of f32 (native-bf16
// model path -- no separate cast kernels in the learn graph)
template <int OUT_BF16>
__global__ void sample_assemble_kernel(
    const uint8_t* __restrict__ frames, const uint8_t* __restrict__ terminals,
    const int32_t* __restrict__ timesteps, int64_t ring_cap, int history,
    int n_step, const int32_t* __restrict__ in_idx,
    const float* __restrict__ out_nonterminals, float* __restrict__ out_states,
    float* __restrict__ out_next_states) {
    int b = blockIdx.x;
    int jj = blockIdx.y;
    bool is_next = jj >= history;
    int j = is_next ? jj - history : jj;
    int back = history - 1 - j;
    int64_t leaf = in_idx[b];
    int r = (int)(leaf / ring_cap);
    int local = (int)(leaf % ring_cap);
    int64_t base = (int64_t)r * ring_cap;
    bool fill;
    int src_local;
    int64_t elem_off = ((int64_t)b * history + j) * FRAME_PX;
    void* out = is_next ? (void*)out_next_states : (void*)out_states;
    if (!is_next) {
        fill = timesteps[leaf] >= back;
        src_local = (local - back + (int)ring_cap) % (int)ring_cap;
    } else {
        int local_n = (local + n_step) % (int)ring_cap;
        bool alive = out_nonterminals[b] > 0.5f;
        fill = alive && timesteps[base + local_n] >= back;
        src_local = (local_n - back + (int)ring_cap) % (int)ring_cap;
    }
    const float inv255 = 1.0f / 255.0f;
    const uchar4* src =
        (const uchar4*)(frames + (base + src_local) * (int64_t)FRAME_PX);
    if (OUT_BF16) {
        ushort4_t* d4 = (ushort4_t*)((unsigned short*)out + elem_off);
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
        float4* d4 = (float4*)((float*)out + elem_off);
        float4 z = make_float4(0.f, 0.f, 0.f, 0.f);
        for (int i = threadIdx.x; i < FRAME_PX / 4; i += blockDim.x) {
            if (fill) {
                uchar4 u = src[i];
                d4[i] = make_float4(u.x * inv255, u.y * inv255, u.z * inv255,
                                    u.w * inv255);
            } else {
                d4[i] = z;
            }
        }
    }
}

// ---------------------------------------------------------------------------
// PRIORITY UPDATE
// ---------------------------------------------------------------------------
__global__ void update_priorities_kernel(
    float* tree, int64_t tree_size, float* max_priority, const int32_t* idxs,
    const float* priorities, int64_t k, float omega) {
    int64_t i = threadIdx.x + (int64_t)blockIdx.x * blockDim.x;
    if (i >= k) return;
    float p = priorities[i];
    if (!isfinite(p)) p = 1.0f;
    p = fmaxf(__powf(fabsf(p), omega), 1e-8f);
    tree_set_leaf(tree, tree_size, idxs[i], p);
    atomic_max_pos_float(max_priority, p);
}

__global__ void tree_level_pass_kernel(float* tree, int64_t level_nodes) {
    // parents at [level_nodes, 2*level_nodes)
    for (int64_t i = level_nodes + threadIdx.x + (int64_t)blockIdx.x * blockDim.x;
         i < 2 * level_nodes; i += (int64_t)gridDim.x * blockDim.x)
        tree[i] = tree[2 * i] + tree[2 * i + 1];
}

}  // namespace

// ---------------------------------------------------------------------------
// C ABI wrappers
// ---------------------------------------------------------------------------
extern "C" {

void riqn_replay_append(
    uint8_t* frames, int32_t* actions, float* rewards, uint8_t* terminals,
    int32_t* timesteps, float* tree, int64_t tree_size, int64_t capacity,
    int32_t* ring_state, int n_rings, int64_t ring_cap, float* max_priority,
    const uint8_t* chunk_frames, const float* chunk_meta, int64_t chunk_len,
    int ring_id, void* stream) {
    hipStream_t s = (hipStream_t)stream;
    hipLaunchKernelGGL(append_meta_kernel, dim3(1), dim3(256), 0, s,
                       actions, rewards, terminals, timesteps, ring_state,
                       ring_id, ring_cap, chunk_meta, chunk_len);
    hipLaunchKernelGGL(append_frames_kernel, dim3((uint32_t)chunk_len), dim3(256), 0, s,
                       frames, ring_state, ring_id, ring_cap, chunk_frames,
                       chunk_len);
    hipLaunchKernelGGL(append_tree_kernel, dim3(1), dim3(1024), 0, s,
                       tree, tree_size, ring_state, ring_id, ring_cap,
                       max_priority, chunk_len);
}

void riqn_replay_sample(
    const uint8_t* frames, const int32_t* actions, const float* rewards,
    const uint8_t* terminals, const int32_t* timesteps, const float* tree,
    int64_t tree_size, int64_t capacity, const int32_t* ring_state, int n_rings,
    int64_t ring_cap, int history, int n_step, float discount,
    const float* beta, uint64_t* rng_state, int batch, int32_t* out_idx,
    float* out_states, float* out_next_states, int64_t* out_actions,
    float* out_returns, float* out_nonterminals, float* out_weights,
    int out_bf16, void* stream) {
    hipStream_t s = (hipStream_t)stream;
    int bt = ((batch + 63) / 64) * 64;
    hipLaunchKernelGGL(sample_indices_kernel, dim3(1), dim3(bt), 0, s,
                       (float*)tree, tree_size, capacity, ring_state, n_rings,
                       ring_cap, history, n_step, beta, rng_state, batch,
                       out_idx, out_weights);
    hipLaunchKernelGGL(sample_scalars_kernel, dim3((batch + 63) / 64),
                       dim3(64), 0, s, actions, rewards, terminals, ring_cap,
                       n_step, discount, out_idx, batch, out_actions,
                       out_returns, out_nonterminals);
    if (out_bf16)
        hipLaunchKernelGGL(sample_assemble_kernel<1>,
                           dim3(batch, 2 * history), dim3(256), 0, s, frames,
                           terminals, timesteps, ring_cap, history, n_step,
                           out_idx, out_nonterminals, out_states,
                           out_next_states);
    else
        hipLaunchKernelGGL(sample_assemble_kernel<0>,
                           dim3(batch, 2 * history), dim3(256), 0, s, frames,
                           terminals, timesteps, ring_cap, history, n_step,
                           out_idx, out_nonterminals, out_states,
                           out_next_states);
}

void riqn_replay_update_priorities(
    float* tree, int64_t tree_size, float* max_priority, const int32_t* idxs,
    const float* priorities, int64_t k, float priority_exponent, void* stream) {
    hipStream_t s = (hipStream_t)stream;
    int blocks = (int)((k + 255) / 256);
    hipLaunchKernelGGL(update_priorities_kernel, dim3(blocks), dim3(256), 0, s,
                       tree, tree_size, max_priority, idxs, priorities, k,
                       priority_exponent);
}

void riqn_tree_level_pass(float* tree, int64_t level_nodes, void* stream) {
    hipStream_t s = (hipStream_t)stream;
    int blocks = (int)min((level_nodes + 255) / 256, (int64_t)2048);
    hipLaunchKernelGGL(tree_level_pass_kernel, dim3(blocks), dim3(256), 0, s,
                       tree, level_nodes);
}

}  // extern "C"
