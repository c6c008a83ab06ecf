// K5/K6 — on-GPU prioritized sum-tree.
//
// Flat 1-indexed complete binary tree over p^alpha (root tree[1], leaves at
// [cap, 2cap)), identical layout to the CPU oracle (replay/sumtree_cpu.py).
//
// K5 sample: stratified inverse-CDF descent, one thread per walk; u_b is
// drawn from a counter-based hash RNG keyed (seed, step, b) so DP ranks are
// deterministic per (rank-seed, step) with no host RNG state — graph-safe.
//
// K6 update: single-workgroup kernel; leaf writes, then LEVEL-SYNCHRONOUS
// re-reduction up the touched paths (duplicate parents recompute the same
// sum — benign identical writes, fully deterministic; no atomics,
// SURVEY.md §3.4 concurrency contract).

#include "common.h"

// counter-based RNG: one round of splitmix64-style mixing -> u in [0,1)
DEV_INLINE float u01(unsigned long long seed, unsigned long long step,
                     unsigned int b) {
    unsigned long long z = seed * 0x9E3779B97F4A7C15ull + step * 0xBF58476D1CE4E5B9ull +
                           (unsigned long long)b * 0x94D049BB133111EBull + 0x2545F4914F6CDD1Dull;
    z ^= z >> 30; z *= 0xBF58476D1CE4E5B9ull;
    z ^= z >> 27; z *= 0x94D049BB133111EBull;
    z ^= z >> 31;
    // 24 high bits -> [0,1)
    return (float)(z >> 40) * (1.0f / 16777216.0f);
}

extern "C" __global__ void __launch_bounds__(256)
sumtree_sample_kernel(const float* __restrict__ tree, int cap, int B,
                      unsigned long long seed, unsigned long long step,
                      long* __restrict__ idx_out, float* __restrict__ prob_out) {
    int b = blockIdx.x * blockDim.x + threadIdx.x;
    if (b >= B) return;
    float total = tree[1];
    float seg = total / (float)B;
    float u = ((float)b + u01(seed, step, b)) * seg;
    int pos = 1;
    while (pos < cap) {
        int left = 2 * pos;
        float lmass = tree[left];
        if (u < lmass) {
            pos = left;
        } else {
            u -= lmass;
            pos = left + 1;
            // fp drift guard: if the right child is empty mass, fall back left
            if (tree[pos] <= 0.0f && lmass > 0.0f) { pos = left; u = lmass * 0.5f; }
        }
    }
    idx_out[b] = pos - cap;
    prob_out[b] = tree[pos] / total;
}

extern "C" __global__ void __launch_bounds__(1024)
sumtree_update_kernel(float* __restrict__ tree, int cap,
                      const long* __restrict__ idx, const float* __restrict__ prio,
                      int M, int depth) {
    const int tid = threadIdx.x;
    for (int m = tid; m < M; m += (int)blockDim.x)
        tree[cap + idx[m]] = prio[m];
    __syncthreads();
    for (int level = 1; level <= depth; ++level) {
        for (int m = tid; m < M; m += (int)blockDim.x) {
            int pos = (int)((cap + idx[m]) >> level);
            tree[pos] = tree[2 * pos] + tree[2 * pos + 1];
        }
        __syncthreads();
    }
}

extern "C" void launch_sumtree_sample(const float* tree, int cap, int B,
                                      unsigned long long seed, unsigned long long step,
                                      long* idx_out, float* prob_out, hipStream_t stream) {
    int blocks = (B + 255) / 256;
    sumtree_sample_kernel<<<dim3(blocks), dim3(256), 0, stream>>>(
        tree, cap, B, seed, step, idx_out, prob_out);
}

// Multi-block variant for large update batches: one kernel launch per level
// (the launch boundary IS the level barrier). Within a level, duplicate
// parents get identical recomputed sums and children were finalized by the
// previous launch — benign, deterministic, no atomics.
extern "C" __global__ void __launch_bounds__(256)
sumtree_update_leaves_kernel(float* __restrict__ tree, int cap,
                             const long* __restrict__ idx,
                             const float* __restrict__ prio, int M) {
    int m = blockIdx.x * (int)blockDim.x + threadIdx.x;
    if (m < M) tree[cap + idx[m]] = prio[m];
}

extern "C" __global__ void __launch_bounds__(256)
sumtree_update_level_kernel(float* __restrict__ tree, int cap,
                            const long* __restrict__ idx, int M, int level) {
    int m = blockIdx.x * (int)blockDim.x + threadIdx.x;
    if (m < M) {
        int pos = (int)((cap + idx[m]) >> level);
        tree[pos] = tree[2 * pos] + tree[2 * pos + 1];
    }
}

extern "C" void launch_sumtree_update(float* tree, int cap, const long* idx,
                                      const float* prio, int M, hipStream_t stream) {
    int depth = 0;
    while ((1 << depth) < cap) ++depth;
    if (M <= 4096) {
        // leaves sit at depth `depth`; root is reached after `depth` halvings
        sumtree_update_kernel<<<dim3(1), dim3(1024), 0, stream>>>(
            tree, cap, idx, prio, M, depth);
        return;
    }
    // large batch (bulk ingest, 8-rank rates): one grid-stride launch per
    // level so all CUs participate; the stream orders the levels
    int blocks = (M + 255) / 256;
    sumtree_update_leaves_kernel<<<dim3(blocks), dim3(256), 0, stream>>>(
        tree, cap, idx, prio, M);
    for (int level = 1; level <= depth; ++level)
        sumtree_update_level_kernel<<<dim3(blocks), dim3(256), 0, stream>>>(
            tree, cap, idx, M, level);
}
