// K3: fused n-step TD-target reduction + per-sequence priorities.
// K4: fused multi-tensor Polyak soft target update.
// SURVEY.md §2.3 K3/K4. Memory-bound elementwise/scan kernels: one wave per
// sequence row (K3), grid-stride flat loop over a device pointer table (K4).

#include "common.h"

// --------------------------------------------------------------------- K3
// One wave per sequence b. Each lane owns train-window steps t = lane,
// lane+64, ... For each t: fold n rewards (n <= 16 typical), mix bootstrap,
// delta vs online q; then wave-reduce max|d| and sum|d| -> priority.
// Optional value-rescale: y = h(folded + g^n * mask * h_inv(q_boot)) applied
// HERE so the priority/TD math matches the learner exactly.
// valid (nullable, [B]): count of REAL (non-pad) steps per sequence counted
// from the window start, UNCLAMPED (may exceed T into the n-step tail).
// Steps >= min(valid, T) contribute neither td_abs nor priority mass (the
// learner masks them out of the loss too), and the reward fold treats
// indices >= valid as r=0/done=0 — stored pad bytes can never leak into a
// real step's target.
extern "C" __global__ void __launch_bounds__(64)
td_targets_kernel(const float* __restrict__ rewards,  // [B, t_ext]
                  const float* __restrict__ dones,    // [B, t_ext]
                  const float* __restrict__ q_boot,   // [B, T]
                  const float* __restrict__ q_online, // [B, T]
                  const int* __restrict__ valid,      // [B] or nullptr
                  float* __restrict__ y_out,          // [B, T]
                  float* __restrict__ td_abs_out,     // [B, T]
                  float* __restrict__ prio_out,       // [B]
                  int B, int T, int t_ext, int n_step,
                  float gamma, float eta, int rescale, float rescale_eps) {
    const int b = blockIdx.x;
    if (b >= B) return;
    const int lane = threadIdx.x;
    const float* r_row = rewards + (long)b * t_ext;
    const float* d_row = dones + (long)b * t_ext;
    const float* qb_row = q_boot + (long)b * T;
    const float* qo_row = q_online + (long)b * T;
    float gamma_n = 1.0f;
    for (int k = 0; k < n_step; ++k) gamma_n *= gamma;
    int v_ext = t_ext;  // real extent of the reward/done rows
    int vT = T;
    if (valid) {
        v_ext = valid[b] < 1 ? 1 : valid[b];
        vT = v_ext > T ? T : v_ext;
    }

    float max_d = 0.0f, sum_d = 0.0f;
    for (int t = lane; t < T; t += WAVE) {
        float acc = 0.0f, g = 1.0f, alive = 1.0f;
        for (int k = 0; k < n_step; ++k) {
            int idx = t + k;
            if (idx >= t_ext) { alive = 0.0f; break; }
            if (idx < v_ext) {  // beyond the real extent: r=0, done=0
                acc += g * alive * r_row[idx];
                alive *= 1.0f - d_row[idx];
            }
            g *= gamma;
        }
        float qb = qb_row[t];
        if (rescale) {  // invert the target net's rescaled value
            float a = fabsf(qb), s = qb >= 0.f ? 1.f : -1.f;
            float root = (sqrtf(1.0f + 4.0f * rescale_eps * (a + 1.0f + rescale_eps)) - 1.0f)
                         / (2.0f * rescale_eps);
            qb = s * (root * root - 1.0f);
        }
        float y = acc + gamma_n * alive * qb;
        if (rescale) {
            float a = fabsf(y), s = y >= 0.f ? 1.f : -1.f;
            y = s * (sqrtf(a + 1.0f) - 1.0f) + rescale_eps * y;
        }
        float d = t < vT ? fabsf(y - qo_row[t]) : 0.0f;  // pad steps: no mass
        y_out[(long)b * T + t] = y;
        td_abs_out[(long)b * T + t] = d;
        max_d = fmaxf(max_d, d);
        sum_d += d;
    }
    // wave reduction (64 lanes)
    for (int off = 32; off > 0; off >>= 1) {
        max_d = fmaxf(max_d, __shfl_down(max_d, off));
        sum_d += __shfl_down(sum_d, off);
    }
    if (lane == 0)
        prio_out[b] = eta * max_d + (1.0f - eta) * (sum_d / (float)vT);
}

// --------------------------------------------------------------------- K4
// ptrs: [2*N] interleaved (target_i, online_i) device pointers;
// offs: [N+1] prefix element offsets. One grid-stride loop over the flat
// parameter space; each thread binary-searches its tensor (few tensors, the
// search is ~5 steps and amortized over a 4-wide vector).
template <typename T>
DEV_INLINE void polyak_elem(T* t, const T* o, float tau);

template <>
DEV_INLINE void polyak_elem<float>(float* t, const float* o, float tau) {
    *t = (1.0f - tau) * (*t) + tau * (*o);
}
template <>
DEV_INLINE void polyak_elem<__hip_bfloat16>(__hip_bfloat16* t, const __hip_bfloat16* o, float tau) {
    *t = f2bf((1.0f - tau) * bf2f(*t) + tau * bf2f(*o));
}

template <typename T>
__global__ void __launch_bounds__(256) polyak_multi_kernel(
    const unsigned long long* __restrict__ ptrs,
    const long* __restrict__ offs,
    int n_tensors, long total, float tau) {
    long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
    const long stride = (long)gridDim.x * blockDim.x;
    for (; i < total; i += stride) {
        // binary search for tensor containing flat index i
        int lo = 0, hi = n_tensors;
        while (hi - lo > 1) {
            int mid = (lo + hi) >> 1;
            if (offs[mid] <= i) lo = mid; else hi = mid;
        }
        long local = i - offs[lo];
        T* tgt = reinterpret_cast<T*>(ptrs[2 * lo]);
        const T* onl = reinterpret_cast<const T*>(ptrs[2 * lo + 1]);
        polyak_elem<T>(tgt + local, onl + local, tau);
    }
}

// ------------------------------------------------------------- launchers
extern "C" void launch_td_targets(
    const float* rewards, const float* dones, const float* q_boot,
    const float* q_online, const int* valid, float* y, float* td_abs,
    float* prio, int B, int T, int t_ext, int n_step, float gamma, float eta,
    int rescale, float rescale_eps, hipStream_t stream) {
    td_targets_kernel<<<dim3(B), dim3(WAVE), 0, stream>>>(
        rewards, dones, q_boot, q_online, valid, y, td_abs, prio,
        B, T, t_ext, n_step, gamma, eta, rescale, rescale_eps);
}

extern "C" void launch_polyak_multi(
    const unsigned long long* ptrs, const long* offs, int n_tensors,
    long total, float tau, int is_bf16, hipStream_t stream) {
    int threads = 256;
    long want = (total + threads - 1) / threads;
    int blocks = (int)(want < 2048 ? (want > 0 ? want : 1) : 2048);
    if (is_bf16)
        polyak_multi_kernel<__hip_bfloat16><<<dim3(blocks), dim3(threads), 0, stream>>>(
            ptrs, offs, n_tensors, total, tau);
    else
        polyak_multi_kernel<float><<<dim3(blocks), dim3(threads), 0, stream>>>(
            ptrs, offs, n_tensors, total, tau);
}
