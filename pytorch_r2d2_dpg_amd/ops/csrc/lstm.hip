// K1/K2 — fused LSTM sequence forward/backward.
//
// MI355X-native design (gfx950, CDNA4):
//  * ONE kernel launch per direction for the WHOLE sequence (persistent
//    T-loop inside) — replaces per-timestep launches, the dominant cost of
//    the reference's nn.LSTM path at these sizes.
//  * Each workgroup (4 waves, 256 threads) owns a 16-row batch tile; its
//    recurrent state never leaves the CU: h lives in LDS (XOR-swizzled for
//    conflict-free ds_read_b128 MFMA fragments), c lives in registers laid
//    out exactly as the MFMA accumulator. No inter-workgroup traffic, no
//    grid sync — the recurrence is row-local by construction.
//  * The per-step recurrent GEMM h @ Wh^T runs on MFMA
//    (v_mfma_f32_16x16x32_bf16 for bf16; exact-f32 v_mfma_f32_16x16x4_f32
//    for fp32 — no TF32/xf32 on gfx950). Wh streams from the XCD's L2
//    (<=2 MB, resident) as per-lane 16 B fragment loads.
//  * gates_x = x@Wx^T + b is a separate large library GEMM (rocBLAS) done by
//    the caller; the kernel folds it in by INITIALIZING the MFMA accumulator
//    with gates_x[t] instead of zero (saves a full add pass).
//  * Wave w owns h-columns [w*H/4, (w+1)*H/4) and computes all four gates
//    for those columns, so the i,f,g,o combine and the c/h update are
//    lane-local on the accumulator fragments.
//
// Backward runs the same structure in reverse time: dgates are computed
// lane-locally from stashed activations, staged through LDS (swizzled), and
// the dh recurrence dgates @ Wh is MFMA with K = 4H against a transposed
// weight copy (whT, [H,4H]) so B-fragments are contiguous 16 B loads.
// Weight/bias/input gradients are large library GEMMs done by the caller.

#include "common.h"

typedef __attribute__((ext_vector_type(8))) short short8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

// Hardware transcendentals: v_exp_f32 (2^x) + v_rcp_f32. Plain `1.0f/x`
// emits the IEEE div_scale/div_fmas/div_fixup sequence (~10 ops each, 40
// per wave-step here — measured as the epilogue's dominant cost).
DEV_INLINE float sigm(float x) {
    return __builtin_amdgcn_rcpf(1.0f + __builtin_amdgcn_exp2f(-1.44269504f * x));
}
DEV_INLINE float tanh_e(float x) {
    return 2.0f * __builtin_amdgcn_rcpf(1.0f + __builtin_amdgcn_exp2f(-2.88539008f * x)) - 1.0f;
}

// XOR swizzle: spread 16 rows of a row-major LDS tile over bank slots so the
// 16-lane groups of ds_read_b128 hit distinct 16B slots (<=2-way).
DEV_INLINE unsigned swz(unsigned byte_off, int row) {
    return byte_off ^ ((unsigned)(row & 7) << 4);
}

// Load an MFMA A/B fragment worth of DT from a (possibly swizzled) address.
template <typename DT> struct Frag;
template <> struct Frag<__hip_bfloat16> {
    using T = short8;                       // 8 bf16 = 16 B
    static constexpr int KSTEP = 32;        // mfma_f32_16x16x32_bf16
    static constexpr int KPL = 8;           // k elems per lane
    static DEV_INLINE f32x4 mfma(T a, T b, f32x4 c) {
        return __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
    }
};
template <> struct Frag<float> {
    using T = float;                        // 1 f32
    static constexpr int KSTEP = 4;         // mfma_f32_16x16x4_f32 (exact f32)
    static constexpr int KPL = 1;
    static DEV_INLINE f32x4 mfma(T a, T b, f32x4 c) {
        return __builtin_amdgcn_mfma_f32_16x16x4f32(a, b, c, 0, 0, 0);
    }
};

template <typename DT>
DEV_INLINE void st_lds(char* smem, unsigned byte_off, float v) {
    st_f(reinterpret_cast<DT*>(&smem[byte_off]), v);
}

// HFRAG (= H/64, fragments per gate per wave) is a compile-time template
// parameter so every accumulator index is static — runtime-indexed
// ext_vector arrays land in scratch (guide §5.4 rule 20).

// ---------------------------------------------------------------- forward
template <typename DT, int HFRAG>
__global__ void __launch_bounds__(HFRAG <= 2 ? 1024 : 512)
lstm_fwd_kernel(const DT* __restrict__ gates_x,  // [T, B, 4H] (x@Wx^T + b)
                const DT* __restrict__ h0,       // [B, H]
                const DT* __restrict__ c0,       // [B, H]
                const DT* __restrict__ wh,       // [4H, H] row-major
                DT* __restrict__ h_seq,          // [T, B, H]
                DT* __restrict__ c_final,        // [B, H]
                DT* __restrict__ gates_act,      // [T, B, 4H] post-act i,f,g,o
                float* __restrict__ c_seq,       // [T, B, H]
                int T, int B, int H, int stash) {  // stash=0: inference (no bwd)
    using F = Frag<DT>;
    // TWO h tiles (double buffer): read h(t-1) from buf[t&1], write h(t) to
    // buf[(t+1)&1] -> one barrier per timestep instead of two.
    extern __shared__ __attribute__((aligned(16))) char smem[];

    const int tid = threadIdx.x;
    const int wid = tid >> 6;
    const int lane = tid & 63;
    const int brow0 = blockIdx.x * 16;
    const int HQ = HFRAG * 16;          // h-cols per wave (waves = H / HQ)
    const int col_base = wid * HQ;
    const int H4 = 4 * H;
    const int rowbytes = H * (int)sizeof(DT);

    const int a_row = lane & 15;              // A-fragment row
    const int a_kgrp = (lane >> 4) * F::KPL;  // A/B-fragment k offset
    const int c_col = lane & 15;              // C/D col within fragment
    const int c_row0 = (lane >> 4) * 4;       // C/D first row (4 regs)

    const unsigned tile_bytes = 16 * rowbytes;
    // ---- load h0 tile into LDS buf[0] (swizzled), c0 into C-layout registers
    for (int e = tid; e < 16 * H; e += (int)blockDim.x) {
        int r = e / H, col = e - r * H;
        int grow = brow0 + r;
        float v = (grow < B) ? ld_f(&h0[(long)grow * H + col]) : 0.0f;
        st_lds<DT>(smem, swz(r * rowbytes + col * sizeof(DT), r), v);
    }
    float c_reg[HFRAG][4];
#pragma unroll
    for (int mf = 0; mf < HFRAG; ++mf)
#pragma unroll
        for (int r = 0; r < 4; ++r) {
            int grow = brow0 + c_row0 + r;
            int col = col_base + mf * 16 + c_col;
            c_reg[mf][r] = (grow < B) ? ld_f(&c0[(long)grow * H + col]) : 0.0f;
        }
    __syncthreads();

    // Register-pressure control: fragments are processed in chunks of <=4
    // (CHUNK), each with its own accumulator set whose lifetime ends at that
    // chunk's epilogue — at H=512 (HFRAG=8) a monolithic acc[4][8] spills.
    // The new h values are held in registers and written to LDS only after
    // EVERY chunk has finished reading h(t-1).
    constexpr int CHUNK = (HFRAG > 2 && HFRAG % 2 == 0) ? 2 : HFRAG;
    constexpr int NCHUNK = HFRAG / CHUNK;

    // ---- time loop
    for (int t = 0; t < T; ++t) {
        const DT* gx_t = gates_x + (long)t * B * H4;
        DT* hs_t = h_seq + (long)t * B * H;
        DT* ga_t = gates_act + (long)t * B * H4;
        float* cs_t = c_seq + (long)t * B * H;
        char* wr_buf = smem + ((t + 1) & 1) * tile_bytes;
#pragma unroll
        for (int cc = 0; cc < NCHUNK; ++cc) {
            // accumulators initialized with gates_x (bias + input projection)
            f32x4 acc[4][CHUNK];
#pragma unroll
            for (int g = 0; g < 4; ++g)
#pragma unroll
                for (int mc = 0; mc < CHUNK; ++mc)
#pragma unroll
                    for (int r = 0; r < 4; ++r) {
                        int grow = brow0 + c_row0 + r;
                        int j = g * H + col_base + (cc * CHUNK + mc) * 16 + c_col;
                        acc[g][mc][r] = (grow < B) ? ld_f(&gx_t[(long)grow * H4 + j]) : 0.0f;
                    }
            // K-loop over H: A from LDS (h tile), B from global (wh,
            // L2-resident). Two-phase body: issue ALL of this k-step's B
            // fragment loads back-to-back, then the MFMAs — one L2
            // round-trip covers the batch instead of one per fragment.
            char* rd_buf = smem + (t & 1) * tile_bytes;
            for (int kk = 0; kk < H; kk += F::KSTEP) {
                int k = kk + a_kgrp;
                typename F::T a_frag = *reinterpret_cast<const typename F::T*>(
                    &rd_buf[swz(a_row * rowbytes + k * sizeof(DT), a_row)]);
                typename F::T breg[4][CHUNK];
                // wh is packed fragment-major ([4H/16][H/KSTEP][64][KPL]):
                // every load is one lane-linear, fully-coalesced 16B/lane
                // instruction (unpacked row-major cost 16 scattered lines
                // per instruction -- measured as 86% of kernel time)
                const int kb = kk / F::KSTEP;
#pragma unroll
                for (int g = 0; g < 4; ++g)
#pragma unroll
                    for (int mc = 0; mc < CHUNK; ++mc) {
                        int jb = g * (H >> 4) + (col_base >> 4) + cc * CHUNK + mc;
                        long off = (((long)jb * (H / F::KSTEP) + kb) * 64 + lane) * F::KPL;
                        breg[g][mc] = *reinterpret_cast<const typename F::T*>(&wh[off]);
                    }
#pragma unroll
                for (int g = 0; g < 4; ++g)
#pragma unroll
                    for (int mc = 0; mc < CHUNK; ++mc)
                        acc[g][mc] = F::mfma(a_frag, breg[g][mc], acc[g][mc]);
            }
            // ---- epilogue: activations, c update, LDS write into the OTHER
            // buffer (no hazard: readers use rd_buf) + global stores
#pragma unroll
            for (int mc = 0; mc < CHUNK; ++mc) {
                int mf = cc * CHUNK + mc;
                int col = col_base + mf * 16 + c_col;
#pragma unroll
                for (int r = 0; r < 4; ++r) {
                    int row = c_row0 + r;
                    int grow = brow0 + row;
                    float gi = sigm(acc[0][mc][r]);
                    float gf = sigm(acc[1][mc][r]);
                    float gg = tanh_e(acc[2][mc][r]);
                    float go = sigm(acc[3][mc][r]);
                    float c = gf * c_reg[mf][r] + gi * gg;
                    float hn = go * tanh_e(c);
                    c_reg[mf][r] = c;
                    st_lds<DT>(wr_buf, swz(row * rowbytes + col * sizeof(DT), row), hn);
                    if (grow < B) {
                        st_f(&hs_t[(long)grow * H + col], hn);
                        if (stash) {  // wave-uniform: skipped for burn-in/target unrolls
                            cs_t[(long)grow * H + col] = c;
                            st_f(&ga_t[(long)grow * H4 + 0 * H + col], gi);
                            st_f(&ga_t[(long)grow * H4 + 1 * H + col], gf);
                            st_f(&ga_t[(long)grow * H4 + 2 * H + col], gg);
                            st_f(&ga_t[(long)grow * H4 + 3 * H + col], go);
                        }
                    }
                }
            }
        }
        __syncthreads();  // h(t) visible; h(t-1) buffer free for overwrite
    }
    // ---- final cell state
#pragma unroll
    for (int mf = 0; mf < HFRAG; ++mf)
#pragma unroll
        for (int r = 0; r < 4; ++r) {
            int grow = brow0 + c_row0 + r;
            int col = col_base + mf * 16 + c_col;
            if (grow < B) st_f(&c_final[(long)grow * H + col], c_reg[mf][r]);
        }
}

// ---------------------------------------------------------------- backward
template <typename DT, int HFRAG>
__global__ void __launch_bounds__(HFRAG <= 2 ? 1024 : 512)
lstm_bwd_kernel(const DT* __restrict__ dh_seq,     // [T, B, H]
                const DT* __restrict__ dhT,        // [B, H]
                const DT* __restrict__ dcT,        // [B, H]
                const DT* __restrict__ gates_act,  // [T, B, 4H]
                const float* __restrict__ c_seq,   // [T, B, H]
                const DT* __restrict__ c0,         // [B, H]
                const DT* __restrict__ whT,        // [H, 4H] (wh transposed)
                DT* __restrict__ dgates,           // [T, B, 4H]
                DT* __restrict__ dh0,              // [B, H]
                DT* __restrict__ dc0,              // [B, H]
                int T, int B, int H, int db) {     // db: double-buffered LDS tile
    using F = Frag<DT>;
    extern __shared__ __attribute__((aligned(16))) char smem[];  // dgates tile [16][4H] DT, swizzled

    const int tid = threadIdx.x;
    const int wid = tid >> 6;
    const int lane = tid & 63;
    const int brow0 = blockIdx.x * 16;
    const int HQ = HFRAG * 16;          // h-cols per wave (waves = H / HQ)
    const int col_base = wid * HQ;
    const int H4 = 4 * H;
    const int rowbytes4 = H4 * (int)sizeof(DT);

    const int a_row = lane & 15;
    const int a_kgrp = (lane >> 4) * F::KPL;
    const int c_col = lane & 15;
    const int c_row0 = (lane >> 4) * 4;

    // carries in C-layout registers
    float dh_reg[HFRAG][4];
    float dc_reg[HFRAG][4];
#pragma unroll
    for (int mf = 0; mf < HFRAG; ++mf)
#pragma unroll
        for (int r = 0; r < 4; ++r) {
            int grow = brow0 + c_row0 + r;
            int col = col_base + mf * 16 + c_col;
            bool ok = grow < B;
            dh_reg[mf][r] = ok ? ld_f(&dhT[(long)grow * H + col]) : 0.0f;
            dc_reg[mf][r] = ok ? ld_f(&dcT[(long)grow * H + col]) : 0.0f;
        }

    for (int t = T - 1; t >= 0; --t) {
        const DT* dh_t = dh_seq + (long)t * B * H;
        const DT* ga_t = gates_act + (long)t * B * H4;
        const float* cs_t = c_seq + (long)t * B * H;
        const float* cs_p = (t > 0) ? c_seq + (long)(t - 1) * B * H : nullptr;
        DT* dg_t = dgates + (long)t * B * H4;
        char* buf = smem + (db ? (t & 1) * (unsigned)(16 * rowbytes4) : 0u);
        // ---- lane-local dgates from stashed activations
#pragma unroll
        for (int mf = 0; mf < HFRAG; ++mf) {
            int col = col_base + mf * 16 + c_col;
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                int row = c_row0 + r;
                int grow = brow0 + row;
                float dgi = 0.f, dgf = 0.f, dgg = 0.f, dgo = 0.f;
                if (grow < B) {
                    float dht = dh_reg[mf][r] + ld_f(&dh_t[(long)grow * H + col]);
                    float gi = ld_f(&ga_t[(long)grow * H4 + 0 * H + col]);
                    float gf = ld_f(&ga_t[(long)grow * H4 + 1 * H + col]);
                    float gg = ld_f(&ga_t[(long)grow * H4 + 2 * H + col]);
                    float go = ld_f(&ga_t[(long)grow * H4 + 3 * H + col]);
                    float ct = cs_t[(long)grow * H + col];
                    float cp = (t > 0) ? cs_p[(long)grow * H + col]
                                       : ld_f(&c0[(long)grow * H + col]);
                    float tc = tanh_e(ct);
                    float do_ = dht * tc;
                    float dc = dht * go * (1.0f - tc * tc) + dc_reg[mf][r];
                    dgi = dc * gg * gi * (1.0f - gi);
                    dgf = dc * cp * gf * (1.0f - gf);
                    dgg = dc * gi * (1.0f - gg * gg);
                    dgo = do_ * go * (1.0f - go);
                    dc_reg[mf][r] = dc * gf;
                    st_f(&dg_t[(long)grow * H4 + 0 * H + col], dgi);
                    st_f(&dg_t[(long)grow * H4 + 1 * H + col], dgf);
                    st_f(&dg_t[(long)grow * H4 + 2 * H + col], dgg);
                    st_f(&dg_t[(long)grow * H4 + 3 * H + col], dgo);
                } else {
                    dc_reg[mf][r] = 0.0f;
                }
                // LDS stage for the recurrence MFMA (pad rows write zeros)
                const float vals[4] = {dgi, dgf, dgg, dgo};
                for (int g = 0; g < 4; ++g)
                    st_lds<DT>(buf, swz(row * rowbytes4 + (g * H + col) * sizeof(DT), row),
                               vals[g]);
            }
            // pin load movement at fragment boundaries: without this the
            // fully unrolled loop hoists every global load (7 per element)
            // and spills ~140 VGPRs at HFRAG=8
            if (HFRAG > 4) asm volatile("" ::: "memory");
        }
        __syncthreads();
        // ---- dh_prev = dgates @ Wh : MFMA over K = 4H (chunked, see fwd)
        constexpr int CHUNK = (HFRAG > 2 && HFRAG % 2 == 0) ? 2 : HFRAG;
        constexpr int NCHUNK = HFRAG / CHUNK;
#pragma unroll
        for (int cc = 0; cc < NCHUNK; ++cc) {
            f32x4 acc[CHUNK];
#pragma unroll
            for (int mc = 0; mc < CHUNK; ++mc) acc[mc] = f32x4{0, 0, 0, 0};
            for (int kk = 0; kk < H4; kk += F::KSTEP) {
                int k = kk + a_kgrp;
                typename F::T a_frag = *reinterpret_cast<const typename F::T*>(
                    &buf[swz(a_row * rowbytes4 + k * sizeof(DT), a_row)]);
                typename F::T breg[CHUNK];
                const int kb = kk / F::KSTEP;
#pragma unroll
                for (int mc = 0; mc < CHUNK; ++mc) {
                    int nb = (col_base >> 4) + cc * CHUNK + mc;
                    long off = (((long)nb * (H4 / F::KSTEP) + kb) * 64 + lane) * F::KPL;
                    breg[mc] = *reinterpret_cast<const typename F::T*>(&whT[off]);
                }
#pragma unroll
                for (int mc = 0; mc < CHUNK; ++mc)
                    acc[mc] = F::mfma(a_frag, breg[mc], acc[mc]);
            }
#pragma unroll
            for (int mc = 0; mc < CHUNK; ++mc)
#pragma unroll
                for (int r = 0; r < 4; ++r) dh_reg[cc * CHUNK + mc][r] = acc[mc][r];
        }
        if (!db) __syncthreads();  // single-buffer: tile reused next iteration
    }
    // ---- dh0 / dc0
#pragma unroll
    for (int mf = 0; mf < HFRAG; ++mf)
#pragma unroll
        for (int r = 0; r < 4; ++r) {
            int grow = brow0 + c_row0 + r;
            int col = col_base + mf * 16 + c_col;
            if (grow < B) {
                st_f(&dh0[(long)grow * H + col], dh_reg[mf][r]);
                st_f(&dc0[(long)grow * H + col], dc_reg[mf][r]);
            }
        }
}

// ------------------------------------------------------------- launchers
// Wave-count selection. The kernel is L2-latency-bound on the per-step Wh
// fragment loads, so prefer MANY SMALL waves (HFRAG=2 -> 4 waves/SIMD of
// latency hiding) over few fat ones: waves = H/(16*HFRAG), block <= 1024
// threads. LSTM_HF=1|2|4|8 in the env forces HFRAG for A/B experiments.
#include <cstdlib>
static inline int pick_geom(int H, int* threads, int* hf) {
    if (H % 64) return -1;
    static int forced = [] { const char* e = getenv("LSTM_HF"); return e ? atoi(e) : 0; }();
    const int prefs_forced[1] = {forced};
    const int prefs_auto[4] = {2, 1, 4, 8};
    const int* prefs = forced ? prefs_forced : prefs_auto;
    int np = forced ? 1 : 4;
    for (int i = 0; i < np; ++i) {
        int f = prefs[i];
        int cols = 16 * f;
        if (H % cols) continue;
        int waves = H / cols;
        if (waves < 2 || waves > 16) continue;
        // must fit the kernel's __launch_bounds__ (1024 for HFRAG<=2, else
        // 512) — a launch wider than the compiled bound fails at dispatch
        if (waves * 64 > (f <= 2 ? 1024 : 512)) continue;
        *threads = waves * 64;
        *hf = f;
        return 0;
    }
    return -1;
}

#define DISPATCH_HFRAG(HF_EXPR, CALL)                                          \
    switch (HF_EXPR) {                                                         \
        case 1: { constexpr int HF = 1; CALL; break; }                         \
        case 2: { constexpr int HF = 2; CALL; break; }                         \
        case 3: { constexpr int HF = 3; CALL; break; }                         \
        case 4: { constexpr int HF = 4; CALL; break; }                         \
        case 5: { constexpr int HF = 5; CALL; break; }                         \
        case 6: { constexpr int HF = 6; CALL; break; }                         \
        case 7: { constexpr int HF = 7; CALL; break; }                         \
        case 8: { constexpr int HF = 8; CALL; break; }                         \
        default: return -1;                                                    \
    }

extern "C" int launch_lstm_fwd_bf16(
    const void* gates_x, const void* h0, const void* c0, const void* wh,
    void* h_seq, void* c_final, void* gates_act, float* c_seq,
    int T, int B, int H, int stash, hipStream_t stream) {
    int blocks = (B + 15) / 16, threads, hf;
    if (pick_geom(H, &threads, &hf)) return -1;
    size_t lds = (size_t)2 * 16 * H * sizeof(__hip_bfloat16);
    DISPATCH_HFRAG(hf,
        (lstm_fwd_kernel<__hip_bfloat16, HF><<<dim3(blocks), dim3(threads), lds, stream>>>(
            (const __hip_bfloat16*)gates_x, (const __hip_bfloat16*)h0,
            (const __hip_bfloat16*)c0, (const __hip_bfloat16*)wh,
            (__hip_bfloat16*)h_seq, (__hip_bfloat16*)c_final,
            (__hip_bfloat16*)gates_act, c_seq, T, B, H, stash)));
    LSTM_CHECK_LAUNCH();
    return 0;
}

extern "C" int launch_lstm_fwd_f32(
    const void* gates_x, const void* h0, const void* c0, const void* wh,
    void* h_seq, void* c_final, void* gates_act, float* c_seq,
    int T, int B, int H, int stash, hipStream_t stream) {
    int blocks = (B + 15) / 16, threads, hf;
    if (pick_geom(H, &threads, &hf)) return -1;
    size_t lds = (size_t)2 * 16 * H * sizeof(float);
    DISPATCH_HFRAG(hf,
        (lstm_fwd_kernel<float, HF><<<dim3(blocks), dim3(threads), lds, stream>>>(
            (const float*)gates_x, (const float*)h0, (const float*)c0,
            (const float*)wh, (float*)h_seq, (float*)c_final,
            (float*)gates_act, c_seq, T, B, H, stash)));
    LSTM_CHECK_LAUNCH();
    return 0;
}

extern "C" int launch_lstm_bwd_bf16(
    const void* dh_seq, const void* dhT, const void* dcT,
    const void* gates_act, const float* c_seq, const void* c0, const void* whT,
    void* dgates, void* dh0, void* dc0, int T, int B, int H, hipStream_t stream) {
    int blocks = (B + 15) / 16, threads, hf;
    if (pick_geom(H, &threads, &hf)) return -1;
    size_t tile = (size_t)16 * 4 * H * sizeof(__hip_bfloat16);
    int db = (2 * tile <= 160 * 1024) ? 1 : 0;
    size_t lds = tile * (db ? 2 : 1);
    DISPATCH_HFRAG(hf,
        (lstm_bwd_kernel<__hip_bfloat16, HF><<<dim3(blocks), dim3(threads), lds, stream>>>(
            (const __hip_bfloat16*)dh_seq, (const __hip_bfloat16*)dhT,
            (const __hip_bfloat16*)dcT, (const __hip_bfloat16*)gates_act, c_seq,
            (const __hip_bfloat16*)c0, (const __hip_bfloat16*)whT,
            (__hip_bfloat16*)dgates, (__hip_bfloat16*)dh0, (__hip_bfloat16*)dc0,
            T, B, H, db)));
    LSTM_CHECK_LAUNCH();
    return 0;
}

extern "C" int launch_lstm_bwd_f32(
    const void* dh_seq, const void* dhT, const void* dcT,
    const void* gates_act, const float* c_seq, const void* c0, const void* whT,
    void* dgates, void* dh0, void* dc0, int T, int B, int H, hipStream_t stream) {
    int blocks = (B + 15) / 16, threads, hf;
    if (pick_geom(H, &threads, &hf)) return -1;
    size_t tile = (size_t)16 * 4 * H * sizeof(float);
    int db = (2 * tile <= 160 * 1024) ? 1 : 0;
    size_t lds = tile * (db ? 2 : 1);
    DISPATCH_HFRAG(hf,
        (lstm_bwd_kernel<float, HF><<<dim3(blocks), dim3(threads), lds, stream>>>(
            (const float*)dh_seq, (const float*)dhT, (const float*)dcT,
            (const float*)gates_act, c_seq, (const float*)c0, (const float*)whT,
            (float*)dgates, (float*)dh0, (float*)dc0, T, B, H, db)));
    LSTM_CHECK_LAUNCH();
    return 0;
}
