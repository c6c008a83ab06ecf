// K1-px — column-split fused LSTM forward (bf16).
//
// The row-local kernel (lstm.hip) streams the FULL Wh (2 MB at H=512)
// through one CU every timestep — per-CU L1-fill-bound at ~9 us/step. This
// variant splits each 16-row batch tile's gate columns over CG blocks
// (grid = RT x CG): each block streams only Wh's rows for its gate columns
// (2MB/CG) and owns H/CG of the h state. The new h is exchanged through L2
// once per timestep as tagged 8-byte granules (tag = t+1, value = 2 x bf16;
// "the data IS the flag" — relaxed agent-scope atomics, no fences, no
// barriers across blocks; cdna_hip_programming.md §6 G16 R2), swept into the
// double-buffered LDS h tile at the top of the next step.
//
// Deadlock safety: all RT*CG blocks must be co-resident (they spin on each
// other within a timestep). The launcher only picks this kernel when
// RT*CG <= 224 (<= one block per CU with margin); every spin is bounded and
// bails to an error word the host can check (never hangs the GPU).

#include "common.h"

typedef __attribute__((ext_vector_type(8))) short short8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

#define RLX_AGENT __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT
typedef __attribute__((address_space(1))) unsigned long long gu64;
typedef __attribute__((address_space(1))) unsigned int gu32;

DEV_INLINE float px_sigm(float x) {
    return __builtin_amdgcn_rcpf(1.0f + __builtin_amdgcn_exp2f(-1.44269504f * x));
}
DEV_INLINE float px_tanh(float x) {
    return 2.0f * __builtin_amdgcn_rcpf(1.0f + __builtin_amdgcn_exp2f(-2.88539008f * x)) - 1.0f;
}
DEV_INLINE unsigned px_swz(unsigned byte_off, int row) {
    return byte_off ^ ((unsigned)(row & 7) << 4);
}

// XCD-aware (rt, cg) decode: the dispatcher round-robins blocks over the 8
// XCDs by linear block id, so decoding cg = flat % CG puts every block of
// one column group on the same XCD (CG=8: exactly one Wh slice per XCD
// L2, resident across all T steps) or on CG/8-aligned subsets (CG=4/2) —
// instead of every XCD streaming pieces of ALL slices and thrashing its
// L2 with the gates/stash traffic. Placement-independent for correctness
// (the exchange protocol only sees rt/cg); affinity is a speed lever only.
// xcd_swz=0 restores the identity decode (A/B).
template <int CG>
DEV_INLINE void px_decode_block(int xcd_swz, int& rt, int& cg) {
    if (xcd_swz) {
        int flat = (int)(blockIdx.x + gridDim.x * blockIdx.y);
        cg = flat % CG;
        rt = flat / CG;
    } else {
        rt = blockIdx.x;
        cg = blockIdx.y;
    }
}

// One 16-col fragment per gate per wave; CG is compile-time so the sweep's
// per-thread granule count (2*CG) is a static register array.
template <int CG>
__global__ void __launch_bounds__(CG >= 8 ? 512 : 1024)
lstm_fwd_px_kernel(const __hip_bfloat16* __restrict__ gates_x,  // [T,B,4H]
                   const __hip_bfloat16* __restrict__ h0,       // [B,H]
                   const __hip_bfloat16* __restrict__ c0,       // [B,H]
                   const __hip_bfloat16* __restrict__ wh,       // packed [4H/16][H/32][64][8]
                   __hip_bfloat16* __restrict__ h_seq,          // [T,B,H]
                   __hip_bfloat16* __restrict__ c_final,        // [B,H]
                   __hip_bfloat16* __restrict__ gates_act,      // [T,B,4H]
                   float* __restrict__ c_seq,                   // [T,B,H]
                   unsigned long long* __restrict__ hx,         // [2][RT][16][H/2] zeroed
                   unsigned int* __restrict__ err,
                   int T, int B, int H, int stash, int xcd_swz) {
    extern __shared__ __attribute__((aligned(16))) char smem[];  // 2 x h tile [16][H] bf16

    const int tid = threadIdx.x;
    const int wid = tid >> 6;
    const int lane = tid & 63;
    int rt, cg;                           // row tile / column group
    px_decode_block<CG>(xcd_swz, rt, cg);
    const int brow0 = rt * 16;
    const int HQg = H / CG;               // h-cols this block owns
    const int col_base = cg * HQg + wid * 16;  // one 16-col fragment per wave
    const int H4 = 4 * H;
    const int rowbytes = H * 2;
    const unsigned tile_bytes = 16u * (unsigned)rowbytes;

    const int a_row = lane & 15;
    const int a_kgrp = (lane >> 4) * 8;
    const int c_col = lane & 15;
    const int c_row0 = (lane >> 4) * 4;

    // ---- h0 -> LDS buf[0]; c0 -> registers
    for (int e = tid; e < 16 * H; e += (int)blockDim.x) {
        int r = e / H, col = e - r * H;
        int grow = brow0 + r;
        float v = (grow < B) ? bf2f(h0[(long)grow * H + col]) : 0.0f;
        *reinterpret_cast<__hip_bfloat16*>(&smem[px_swz(r * rowbytes + col * 2, r)]) = f2bf(v);
    }
    float c_reg[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
        int grow = brow0 + c_row0 + r;
        int col = col_base + c_col;
        c_reg[r] = (grow < B) ? bf2f(c0[(long)grow * H + col]) : 0.0f;
    }
    __syncthreads();

    gu64* hx_g = (gu64*)hx;
    const long rt_off = (long)rt * 16 * (H / 2);
    const long slot_sz = (long)gridDim.x * 16 * (H / 2);
    unsigned spin_limit = 20000000u;  // degrades after the first timeout

    for (int t = 0; t < T; ++t) {
        const __hip_bfloat16* gx_t = gates_x + (long)t * B * H4;
        // gates_x accumulator init first: these HBM loads are independent of
        // the exchange, so they overlap the granule sweep below
        f32x4 acc[4];
#pragma unroll
        for (int g = 0; g < 4; ++g)
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                int grow = brow0 + c_row0 + r;
                int j = g * H + col_base + c_col;
                acc[g][r] = (grow < B) ? bf2f(gx_t[(long)grow * H4 + j]) : 0.0f;
            }
        // ---- sweep ALL h(t-1) granules (peers' and own) into LDS buf[t&1]
        // (skip t=0: h0 already local). Batched: each thread issues its 2*CG
        // granule loads unconditionally (one L2 round-trip covers all),
        // then re-polls only the not-yet-tagged ones.
        if (t > 0) {
            char* bld = smem + (t & 1) * tile_bytes;
            const unsigned tag = (unsigned)t;  // epoch t = h(t-1), tags from 1
            gu64* slot = hx_g + ((t - 1) & 1) * slot_sz + rt_off;
            constexpr int K = 2 * CG;          // 8H granules / (4H/CG threads)
            const int bd = (int)blockDim.x;
            const int sh = 31 - __builtin_clz((unsigned)(H >> 1));  // log2(H/2)
            unsigned long long v[K];
#pragma unroll
            for (int i = 0; i < K; ++i)
                v[i] = __hip_atomic_load(&slot[tid + i * bd], RLX_AGENT);
            unsigned pending = 0;
#pragma unroll
            for (int i = 0; i < K; ++i)
                pending |= (unsigned)((unsigned)(v[i] >> 32) != tag) << i;
            unsigned spins = 0;
            while (pending) {
                if (++spins > spin_limit) {
                    // bounded spin (~0.5 s at s_sleep granularity): flag the error and PROCEED with stale
                    // data — control flow stays convergent so no thread
                    // ever skips a barrier (a diverged exit deadlocks the
                    // block and hangs the device). After the FIRST timeout
                    // the bound degrades to ~1 ms so a genuine co-residency
                    // violation finishes the kernel fast (T bails) instead
                    // of spinning 0.5 s per timestep.
                    __hip_atomic_store((gu32*)err, 1u + (unsigned)t, RLX_AGENT);
                    spin_limit = 40000u;
                    break;
                }
                __builtin_amdgcn_s_sleep(1);
#pragma unroll
                for (int i = 0; i < K; ++i)
                    if (pending & (1u << i)) {
                        unsigned long long nv =
                            __hip_atomic_load(&slot[tid + i * bd], RLX_AGENT);
                        if ((unsigned)(nv >> 32) == tag) {
                            v[i] = nv;
                            pending &= ~(1u << i);
                        }
                    }
            }
#pragma unroll
            for (int i = 0; i < K; ++i) {
                int e = tid + i * bd;
                int r = e >> sh, gcol = e & ((H >> 1) - 1);
                *reinterpret_cast<unsigned*>(
                    &bld[px_swz(r * rowbytes + gcol * 4, r)]) = (unsigned)v[i];
            }
        }
        __syncthreads();

        __hip_bfloat16* hs_t = h_seq + (long)t * B * H;
        __hip_bfloat16* ga_t = gates_act + (long)t * B * H4;
        float* cs_t = c_seq + (long)t * B * H;
        char* rd_buf = smem + (t & 1) * tile_bytes;

        for (int kk = 0; kk < H; kk += 32) {
            int k = kk + a_kgrp;
            short8 a_frag = *reinterpret_cast<const short8*>(
                &rd_buf[px_swz(a_row * rowbytes + k * 2, a_row)]);
            short8 breg[4];
            const int kb = kk >> 5;
#pragma unroll
            for (int g = 0; g < 4; ++g) {
                int jb = g * (H >> 4) + (col_base >> 4);
                long off = (((long)jb * (H >> 5) + kb) * 64 + lane) * 8;
                breg[g] = *reinterpret_cast<const short8*>(&wh[off]);
            }
#pragma unroll
            for (int g = 0; g < 4; ++g)
                acc[g] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                    a_frag, breg[g], acc[g], 0, 0, 0);
        }
        // ---- epilogue: own h cols -> LDS next buf + exchange granules
        {
            gu64* slot = hx_g + (t & 1) * slot_sz + rt_off;
            const unsigned tag = (unsigned)(t + 1);
            const int col = col_base + c_col;
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                int row = c_row0 + r;
                int grow = brow0 + row;
                float gi = px_sigm(acc[0][r]);
                float gf = px_sigm(acc[1][r]);
                float gg = px_tanh(acc[2][r]);
                float go = px_sigm(acc[3][r]);
                float c = gf * c_reg[r] + gi * gg;
                float hn = go * px_tanh(c);
                c_reg[r] = c;
                unsigned short hb = f32_to_bf16_bits(hn);
                // pair columns (even, odd) into one tagged granule
                unsigned other = __shfl_xor((unsigned)hb, 1);
                if ((lane & 1) == 0) {
                    unsigned lo = (unsigned)hb | (other << 16);
                    unsigned long long g64 = ((unsigned long long)tag << 32) | lo;
                    __hip_atomic_store(&slot[row * (H / 2) + col / 2], g64,
                                       RLX_AGENT);
                }
                if (grow < B) {
                    hs_t[(long)grow * H + col] = f2bf(hn);
                    if (stash) {
                        cs_t[(long)grow * H + col] = c;
                        ga_t[(long)grow * H4 + 0 * H + col] = f2bf(gi);
                        ga_t[(long)grow * H4 + 1 * H + col] = f2bf(gf);
                        ga_t[(long)grow * H4 + 2 * H + col] = f2bf(gg);
                        ga_t[(long)grow * H4 + 3 * H + col] = f2bf(go);
                    }
                }
            }
        }
        // NO trailing barrier: the next sweep writes the OTHER LDS buffer,
        // and a wave cannot get two steps ahead — its next-next sweep would
        // need granules its own block's slow waves publish only after they
        // pass the (kept) pre-MFMA barrier. The double buffer + tag polls
        // bound the skew to one buffer; removing this barrier saves one
        // block-wide sync per timestep.
    }
#pragma unroll
    for (int r = 0; r < 4; ++r) {
        int grow = brow0 + c_row0 + r;
        int col = col_base + c_col;
        if (grow < B) c_final[(long)grow * H + col] = f2bf(c_reg[r]);
    }
}

// Defined after PxArgs (file order): dispatches the pipelined per-lane
// forward for the solo entry point. Returns 0 on success, -1 unsupported.
static int launch_fwd_pl_solo(const void* gates_x, const void* h0,
                              const void* c0, const void* wh_packed,
                              void* h_seq, void* c_final, void* gates_act,
                              float* c_seq, unsigned long long* hx,
                              unsigned int* err, int T, int B, int H,
                              int stash, int CG, int waves,
                              hipStream_t stream);

// CG SELECTION is budget-independent (the r1 rule, RT*CG(*2) <= 224 —
// measured fastest: budget-driven CG reduction cost the flagship 17%).
// The budget is a SAFETY NET only: capacity is counted in CU-EQUIVALENTS
// (a CU fits several px blocks, LDS- and wave-slot-limited — e.g. 5 fwd
// blocks at 32 KB LDS each, 2 bwd blocks at 64 KB; cu_eq =
// ceil(blocks/slots)), and a launch whose cu_eq exceeds the REMAINING
// budget returns -2 so the caller serializes it behind the in-flight px
// grids instead of risking a co-residency livelock. Success returns the
// grid's CU-equivalents (positive) for the caller to commit.
// R2D2_PX_CG_MAX caps the column-group fan-out (tuning knob: the best CG
// shifted when the XCD swizzle made Wh L2-resident — re-measure per shape).
static inline int px_cg_max() {
    static const int v = []() {
        const char* e = getenv("R2D2_PX_CG_MAX");
        int x = e ? atoi(e) : 8;
        return (x == 2 || x == 4 || x == 8) ? x : 8;
    }();
    return v;
}

static inline int px_fwd_pl() {
    static const int v = []() {
        const char* e = getenv("R2D2_PX_FWD_PL");
        return (e && e[0] && e[0] != '0') ? 1 : 0;
    }();
    return v;
}

static inline int px_xcd_swz() {
    static const int v = []() {
        const char* e = getenv("R2D2_PX_NO_XCD_SWZ");
        return (e && e[0] && e[0] != '0') ? 0 : 1;
    }();
    return v;
}

static inline int px_cu_eq(int blocks, int waves_pb, long lds_bytes) {
    int slots = (int)(160L * 1024 / (lds_bytes > 0 ? lds_bytes : 1));
    int wslots = 32 / (waves_pb > 0 ? waves_pb : 1);
    if (wslots < slots) slots = wslots;
    if (slots > 8) slots = 8;
    if (slots < 1) slots = 1;
    return (blocks + slots - 1) / slots;
}
extern "C" int launch_lstm_fwd_px_bf16(
    const void* gates_x, const void* h0, const void* c0, const void* wh_packed,
    void* h_seq, void* c_final, void* gates_act, float* c_seq,
    unsigned long long* hx, unsigned int* err,
    int T, int B, int H, int stash, int budget, hipStream_t stream) {
    int RT = (B + 15) / 16;
    // largest CG with co-residency margin and >= 2 fragments per gate
    if (H & (H - 1)) return -1;              // sweep indexing needs pow2 H
    if (budget > 224) budget = 224;
    int CG = 0, cu_eq = 0;
    for (int c : {8, 4, 2}) {
        if (c > px_cg_max()) continue;
        if (H % (16 * c)) continue;          // wave owns 16 cols
        int w = (H / c) / 16;                // waves per block
        if (w < 2 || w > 16) continue;
        if (RT * c > 224) continue;          // all blocks must be co-resident
        CG = c;
        cu_eq = px_cu_eq(RT * c, w, 2L * 16 * H * 2);
        break;
    }
    if (CG == 0) return -1;
    if (cu_eq > budget) return -2;           // caller: drain in-flight, retry
    int waves = (H / CG) / 16;
    if (px_fwd_pl()) {
        if (launch_fwd_pl_solo(gates_x, h0, c0, wh_packed, h_seq, c_final,
                               gates_act, c_seq, hx, err, T, B, H, stash,
                               CG, waves, stream) == 0)
            return cu_eq;
        return -1;
    }
    dim3 grid(RT, CG);
    size_t lds = (size_t)2 * 16 * H * 2;
    switch (CG) {
        case 2:
            lstm_fwd_px_kernel<2><<<grid, dim3(waves * 64), lds, stream>>>(
                (const __hip_bfloat16*)gates_x, (const __hip_bfloat16*)h0,
                (const __hip_bfloat16*)c0, (const __hip_bfloat16*)wh_packed,
                (__hip_bfloat16*)h_seq, (__hip_bfloat16*)c_final,
                (__hip_bfloat16*)gates_act, c_seq, hx, err, T, B, H, stash,
                px_xcd_swz());
            break;
        case 4:
            lstm_fwd_px_kernel<4><<<grid, dim3(waves * 64), lds, stream>>>(
                (const __hip_bfloat16*)gates_x, (const __hip_bfloat16*)h0,
                (const __hip_bfloat16*)c0, (const __hip_bfloat16*)wh_packed,
                (__hip_bfloat16*)h_seq, (__hip_bfloat16*)c_final,
                (__hip_bfloat16*)gates_act, c_seq, hx, err, T, B, H, stash,
                px_xcd_swz());
            break;
        case 8:
            lstm_fwd_px_kernel<8><<<grid, dim3(waves * 64), lds, stream>>>(
                (const __hip_bfloat16*)gates_x, (const __hip_bfloat16*)h0,
                (const __hip_bfloat16*)c0, (const __hip_bfloat16*)wh_packed,
                (__hip_bfloat16*)h_seq, (__hip_bfloat16*)c_final,
                (__hip_bfloat16*)gates_act, c_seq, hx, err, T, B, H, stash,
                px_xcd_swz());
            break;
        default:
            return -1;
    }
    LSTM_CHECK_LAUNCH();
    return cu_eq;
}

// ---------------------------------------------------------------- backward
// Column-split backward. Block (rt, cg) owns h-cols [cg*HQg, +HQg): computes
// dgates for its 4*HQg gate columns lane-locally, publishes them as an sc1
// SLAB (the swizzled LDS-tile byte image) + one tagged flag per block per
// step (flag stored only after every wave drained its sc1 stores — G16 R1),
// sweeps peers' slabs into the LDS tile, then runs the dh recurrence MFMA
// with K = 4H against its whT row slice (2MB/CG streamed per step).
// Per-set backward arguments — one set for the solo kernel, two for the
// paired (grid.z=2) kernel. Every set has its OWN slab/flags workspace.
struct BwdPxArgs {
    const __hip_bfloat16* dh_seq;    // [T,B,H]
    const __hip_bfloat16* dhT;       // [B,H]
    const __hip_bfloat16* dcT;       // [B,H]
    const __hip_bfloat16* gates_act; // [T,B,4H]
    const float* c_seq;              // [T,B,H]
    const __hip_bfloat16* c0;        // [B,H]
    const __hip_bfloat16* whT;       // packed [H/16][4H/32][64][8]
    __hip_bfloat16* dgates;          // [T,B,4H]
    __hip_bfloat16* dh0;             // [B,H]
    __hip_bfloat16* dc0;             // [B,H]
    unsigned char* slab;             // [2][RT][16*4H*2] bytes
    unsigned long long* flags;       // [2][RT][CG] zeroed
    int T;
    int xcd_swz;
};

template <int CG>
DEV_INLINE void lstm_bwd_px_body(const BwdPxArgs& a, unsigned int* err,
                                 int B, int H, char* smem) {
    const __hip_bfloat16* __restrict__ dh_seq = a.dh_seq;
    const __hip_bfloat16* __restrict__ dhT = a.dhT;
    const __hip_bfloat16* __restrict__ dcT = a.dcT;
    const __hip_bfloat16* __restrict__ gates_act = a.gates_act;
    const float* __restrict__ c_seq = a.c_seq;
    const __hip_bfloat16* __restrict__ c0 = a.c0;
    const __hip_bfloat16* __restrict__ whT = a.whT;
    __hip_bfloat16* __restrict__ dgates = a.dgates;
    __hip_bfloat16* __restrict__ dh0 = a.dh0;
    __hip_bfloat16* __restrict__ dc0 = a.dc0;
    unsigned char* __restrict__ slab = a.slab;
    unsigned long long* __restrict__ flags = a.flags;
    const int T = a.T;

    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wid = tid >> 6;
    int rt, cg;
    px_decode_block<CG>(a.xcd_swz, rt, cg);
    const int brow0 = rt * 16;
    const int HQg = H / CG;
    const int col_base = cg * HQg + wid * 16;
    const int H4 = 4 * H;
    const int rowbytes4 = H4 * 2;
    const unsigned tile_bytes = 16u * (unsigned)rowbytes4;

    const int a_row = lane & 15;
    const int a_kgrp = (lane >> 4) * 8;
    const int c_col = lane & 15;
    const int c_row0 = (lane >> 4) * 4;

    float dh_reg[4], dc_reg[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
        int grow = brow0 + c_row0 + r;
        int col = col_base + c_col;
        bool ok = grow < B;
        dh_reg[r] = ok ? bf2f(dhT[(long)grow * H + col]) : 0.0f;
        dc_reg[r] = ok ? bf2f(dcT[(long)grow * H + col]) : 0.0f;
    }

    typedef __attribute__((address_space(1))) unsigned long long gu64b;
    const long slab_tile = (long)tile_bytes;
    const long slab_slot = (long)gridDim.x * slab_tile;
    gu64b* flg = (gu64b*)flags;
    const int own_byte_lo = (cg * HQg) * 2;   // within one gate's row segment
    unsigned spin_limit = 20000000u;          // degrades after first timeout

    for (int t = T - 1; t >= 0; --t) {
        const unsigned tag = (unsigned)(T - t);
        // SINGLE LDS buffer (64KB at H=512): a double buffer would force one
        // block/CU and two concurrent bwd kernels would need every CU
        // co-resident with zero margin — any displaced block then spins out
        char* buf = smem;
        const __hip_bfloat16* dh_t = dh_seq + (long)t * B * H;
        const __hip_bfloat16* ga_t = gates_act + (long)t * B * H4;
        const float* cs_t = c_seq + (long)t * B * H;
        const float* cs_p = (t > 0) ? c_seq + (long)(t - 1) * B * H : nullptr;
        __hip_bfloat16* dg_t = dgates + (long)t * B * H4;

        // ---- lane-local dgates for own columns -> LDS own region (the
        // global dgates write is DEFERRED until after the flag publish:
        // the sc1 drain below is vmcnt(0), so any outstanding global
        // stores here would delay every peer's flag wait each timestep)
        {
            const int col = col_base + c_col;
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                int row = c_row0 + r;
                int grow = brow0 + row;
                float dgi = 0.f, dgf = 0.f, dgg = 0.f, dgo = 0.f;
                if (grow < B) {
                    float dht = dh_reg[r] + bf2f(dh_t[(long)grow * H + col]);
                    float gi = bf2f(ga_t[(long)grow * H4 + 0 * H + col]);
                    float gf = bf2f(ga_t[(long)grow * H4 + 1 * H + col]);
                    float gg = bf2f(ga_t[(long)grow * H4 + 2 * H + col]);
                    float go = bf2f(ga_t[(long)grow * H4 + 3 * H + col]);
                    float ct = cs_t[(long)grow * H + col];
                    float cp = (t > 0) ? cs_p[(long)grow * H + col]
                                       : bf2f(c0[(long)grow * H + col]);
                    float tc = px_tanh(ct);
                    float do_ = dht * tc;
                    float dc = dht * go * (1.0f - tc * tc) + dc_reg[r];
                    dgi = dc * gg * gi * (1.0f - gi);
                    dgf = dc * cp * gf * (1.0f - gf);
                    dgg = dc * gi * (1.0f - gg * gg);
                    dgo = do_ * go * (1.0f - go);
                    dc_reg[r] = dc * gf;
                } else {
                    dc_reg[r] = 0.0f;
                }
                const float vals[4] = {dgi, dgf, dgg, dgo};
#pragma unroll
                for (int g = 0; g < 4; ++g)
                    *reinterpret_cast<__hip_bfloat16*>(
                        &buf[px_swz(row * rowbytes4 + (g * H + col) * 2, row)]) =
                        f2bf(vals[g]);
            }
        }
        __syncthreads();

        // ---- publish own slab region (sc1 16B), drain, flag
        {
            unsigned char* slab_rt = slab + (t & 1) * slab_slot + (long)rt * slab_tile;
            auto rsrc = __builtin_amdgcn_make_buffer_rsrc(slab_rt, 0, (int)tile_bytes,
                                                          0x00020000);
            // own bytes: 16 rows x 4 gates x HQg*2 bytes, each range 128B-aligned
            // and XOR-swizzle-closed, so raw byte copy preserves the image
            const int own_bytes = HQg * 2;
            const int chunks = 16 * 4 * (own_bytes / 16);
            for (int ch = tid; ch < chunks; ch += (int)blockDim.x) {
                int per_row = 4 * (own_bytes / 16);
                int row = ch / per_row;
                int rem = ch - row * per_row;
                int g = rem / (own_bytes / 16);
                int o16 = rem - g * (own_bytes / 16);
                unsigned off = (unsigned)(row * rowbytes4 + g * H * 2 + own_byte_lo + o16 * 16);
                f32x4 v = *reinterpret_cast<const f32x4*>(&buf[off]);
                __builtin_amdgcn_raw_buffer_store_b128(v, rsrc, (int)off, 0, 16);
            }
            asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
            __syncthreads();
            if (tid == 0)
                __hip_atomic_store(&flg[((t & 1) * gridDim.x + rt) * CG + cg],
                                   ((unsigned long long)tag << 32) | 1u, RLX_AGENT);
        }

        // ---- deferred global dgates write (own columns, read back from the
        // LDS tile — bitwise the same bf16 values): fire-and-forget stores
        // that overlap the peer-flag wait + slab sweep and have the whole
        // next timestep to retire before the next vmcnt(0) drain
        {
            const int col = col_base + c_col;
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                int row = c_row0 + r;
                int grow = brow0 + row;
                if (grow < B) {
#pragma unroll
                    for (int g = 0; g < 4; ++g)
                        dg_t[(long)grow * H4 + g * H + col] =
                            *reinterpret_cast<const __hip_bfloat16*>(
                                &buf[px_swz(row * rowbytes4 + (g * H + col) * 2, row)]);
                }
            }
        }

        // ---- wait peers' flags, then sweep their slab regions into LDS
        {
            if (tid < CG && tid != cg) {
                gu64b* f = &flg[((t & 1) * gridDim.x + rt) * CG + tid];
                unsigned long long v = __hip_atomic_load(f, RLX_AGENT);
                unsigned spins = 0;
                while ((unsigned)(v >> 32) != tag) {
                    if (++spins > spin_limit) {  // convergent bail (see fwd)
                        __hip_atomic_store((gu32*)err, 0x10000u + (unsigned)t, RLX_AGENT);
                        break;
                    }
                    __builtin_amdgcn_s_sleep(1);
                    v = __hip_atomic_load(f, RLX_AGENT);
                }
                if (spins > spin_limit) spin_limit = 40000u;
            }
            __syncthreads();
            unsigned char* slab_rt = slab + (t & 1) * slab_slot + (long)rt * slab_tile;
            auto rsrc = __builtin_amdgcn_make_buffer_rsrc(slab_rt, 0, (int)tile_bytes,
                                                          0x00020000);
            const int own_bytes = HQg * 2;
            const int per_gate16 = (H * 2 - own_bytes) / 16;  // peer bytes per gate-row
            const int chunks = 16 * 4 * per_gate16;
            for (int ch = tid; ch < chunks; ch += (int)blockDim.x) {
                int per_row = 4 * per_gate16;
                int row = ch / per_row;
                int rem = ch - row * per_row;
                int g = rem / per_gate16;
                int o16 = rem - g * per_gate16;
                int byte_in_gate = o16 * 16 + (o16 * 16 >= own_byte_lo ? own_bytes : 0);
                unsigned off = (unsigned)(row * rowbytes4 + g * H * 2 + byte_in_gate);
                f32x4 v = __builtin_amdgcn_raw_buffer_load_b128(rsrc, (int)off, 0, 16);
                *reinterpret_cast<f32x4*>(&buf[off]) = v;
            }
        }
        __syncthreads();

        // ---- dh recurrence: dh_prev(own cols) = dgates(all 4H) @ Wh.
        // K = 4H with ONE output fragment: a single accumulator would chain
        // 4H/32 serially-dependent MFMAs (latency-bound at the 2 waves/SIMD
        // this kernel runs at) — split K over 4 accumulators for ILP, sum at
        // the end (unlike the fwd, whose 4 gate fragments give ILP for free)
        f32x4 acc4[4] = {};
        const int nb = (col_base >> 4);
        for (int kk = 0; kk < H4; kk += 128) {
#pragma unroll
            for (int u = 0; u < 4; ++u) {
                int kx = kk + u * 32;
                int k = kx + a_kgrp;
                short8 a_frag = *reinterpret_cast<const short8*>(
                    &buf[px_swz(a_row * rowbytes4 + k * 2, a_row)]);
                long off = (((long)nb * (H4 >> 5) + (kx >> 5)) * 64 + lane) * 8;
                short8 b_frag = *reinterpret_cast<const short8*>(&whT[off]);
                acc4[u] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                    a_frag, b_frag, acc4[u], 0, 0, 0);
            }
        }
#pragma unroll
        for (int r = 0; r < 4; ++r)
            dh_reg[r] = acc4[0][r] + acc4[1][r] + acc4[2][r] + acc4[3][r];
        __syncthreads();  // MFMA reads done before next step's LDS writes
    }
#pragma unroll
    for (int r = 0; r < 4; ++r) {
        int grow = brow0 + c_row0 + r;
        int col = col_base + c_col;
        if (grow < B) {
            dh0[(long)grow * H + col] = f2bf(dh_reg[r]);
            dc0[(long)grow * H + col] = f2bf(dc_reg[r]);
        }
    }
}

template <int CG>
__global__ void __launch_bounds__(CG >= 8 ? 512 : 1024)
lstm_bwd_px_kernel(BwdPxArgs a, unsigned int* err, int B, int H) {
    extern __shared__ __attribute__((aligned(16))) char smem[];  // dg tile [16][4H]
    lstm_bwd_px_body<CG>(a, err, B, H, smem);
}

// Paired (grid.z=2) backward: two independent unrolls (same B/H, T may
// differ), each with its own slab/flags workspace — e.g. the critic-loss
// backward and the actor-chain critic backward, which become ready at
// nearly the same point in the step. One uniform grid co-dispatches both
// (RT*CG*2 blocks), avoiding the cross-stream CU contention of two
// concurrent solo bwd grids.
template <int CG>
__global__ void __launch_bounds__(CG >= 8 ? 512 : 1024)
lstm_bwd_px2_kernel(BwdPxArgs a0, BwdPxArgs a1, unsigned int* err, int B, int H) {
    extern __shared__ __attribute__((aligned(16))) char smem[];
    lstm_bwd_px_body<CG>(blockIdx.z ? a1 : a0, err, B, H, smem);
}


// ---- granule-exchange backward body (opt-in via R2D2_PX_BWD_GX=1;
// measured SLOWER than the slab body — see px_bwd_slab() — and kept as the
// documented negative result + a second implementation of the exchange
// contract). dgates are published as tagged 8-byte
// granules (2 adjacent bf16 columns; tag = T-t; "the data IS the flag" —
// relaxed agent-scope atomics, exactly the forward's proven pattern), so
// the per-timestep release chain shrinks from
//   sc1 stores -> s_waitcnt vmcnt(0) -> __syncthreads -> flag ->
//   peer flag poll -> slab sweep -> __syncthreads
// to a direct per-granule poll overlapped with the fire-and-forget global
// dgates stores: 2 barriers per timestep instead of 5 and no vmcnt drain.
// Workspace: a.slab reinterpreted as u64 ex[2][RT][CG][16*4*HQg/2]
// (identical byte size to the slab image; MUST be zero-initialized so
// garbage can never alias a valid tag). a.flags is unused here.
// Double-buffer lifetime argument: identical to the forward's (a producer
// can only reach epoch parity p again after every consumer of that parity
// finished its previous use — docs/KERNELS.md).
template <int CG>
DEV_INLINE void lstm_bwd_px_body_gx(const BwdPxArgs& a, unsigned int* err,
                                    int B, int H, char* smem) {
    const __hip_bfloat16* __restrict__ dh_seq = a.dh_seq;
    const __hip_bfloat16* __restrict__ dhT = a.dhT;
    const __hip_bfloat16* __restrict__ dcT = a.dcT;
    const __hip_bfloat16* __restrict__ gates_act = a.gates_act;
    const float* __restrict__ c_seq = a.c_seq;
    const __hip_bfloat16* __restrict__ c0 = a.c0;
    const __hip_bfloat16* __restrict__ whT = a.whT;
    __hip_bfloat16* __restrict__ dgates = a.dgates;
    __hip_bfloat16* __restrict__ dh0 = a.dh0;
    __hip_bfloat16* __restrict__ dc0 = a.dc0;
    const int T = a.T;

    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wid = tid >> 6;
    int rt, cg;
    px_decode_block<CG>(a.xcd_swz, rt, cg);
    const int brow0 = rt * 16;
    const int HQg = H / CG;
    const int col_base = cg * HQg + wid * 16;
    const int H4 = 4 * H;
    const int rowbytes4 = H4 * 2;

    const int a_row = lane & 15;
    const int a_kgrp = (lane >> 4) * 8;
    const int c_col = lane & 15;
    const int c_row0 = (lane >> 4) * 4;

    float dh_reg[4], dc_reg[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
        int grow = brow0 + c_row0 + r;
        int col = col_base + c_col;
        bool ok = grow < B;
        dh_reg[r] = ok ? bf2f(dhT[(long)grow * H + col]) : 0.0f;
        dc_reg[r] = ok ? bf2f(dcT[(long)grow * H + col]) : 0.0f;
    }

    gu64* ex = (gu64*)a.slab;                 // [2][RT][CG][reg_sz]
    const int hq2 = HQg >> 1;                 // col-pairs per gate segment
    const long reg_sz = 16L * 4 * hq2;        // granules per (rt, cg) region
    const long rt_sz = (long)CG * reg_sz;
    const long epoch_sz = (long)gridDim.x * rt_sz;
    const int bd = (int)blockDim.x;           // reg_sz / bd == 8 always
    unsigned spin_limit = 20000000u;          // degrades after first timeout

    for (int t = T - 1; t >= 0; --t) {
        const unsigned tag = (unsigned)(T - t);
        char* buf = smem;                     // single LDS dg tile (see slab body)
        const __hip_bfloat16* dh_t = dh_seq + (long)t * B * H;
        const __hip_bfloat16* ga_t = gates_act + (long)t * B * H4;
        const float* cs_t = c_seq + (long)t * B * H;
        const float* cs_p = (t > 0) ? c_seq + (long)(t - 1) * B * H : nullptr;
        __hip_bfloat16* dg_t = dgates + (long)t * B * H4;
        gu64* ep = ex + (t & 1) * epoch_sz + (long)rt * rt_sz;

        // ---- own dgates: registers -> LDS own region + fire-and-forget
        // global write + tagged granule publish (no drain, no flag)
        {
            const int col = col_base + c_col;
            const int cp_base = (wid * 16 + c_col) >> 1;
            gu64* own = ep + (long)cg * reg_sz;
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                int row = c_row0 + r;
                int grow = brow0 + row;
                float dgi = 0.f, dgf = 0.f, dgg = 0.f, dgo = 0.f;
                if (grow < B) {
                    float dht = dh_reg[r] + bf2f(dh_t[(long)grow * H + col]);
                    float gi = bf2f(ga_t[(long)grow * H4 + 0 * H + col]);
                    float gf = bf2f(ga_t[(long)grow * H4 + 1 * H + col]);
                    float gg = bf2f(ga_t[(long)grow * H4 + 2 * H + col]);
                    float go = bf2f(ga_t[(long)grow * H4 + 3 * H + col]);
                    float ct = cs_t[(long)grow * H + col];
                    float cp2 = (t > 0) ? cs_p[(long)grow * H + col]
                                        : bf2f(c0[(long)grow * H + col]);
                    float tc = px_tanh(ct);
                    float do_ = dht * tc;
                    float dc = dht * go * (1.0f - tc * tc) + dc_reg[r];
                    dgi = dc * gg * gi * (1.0f - gi);
                    dgf = dc * cp2 * gf * (1.0f - gf);
                    dgg = dc * gi * (1.0f - gg * gg);
                    dgo = do_ * go * (1.0f - go);
                    dc_reg[r] = dc * gf;
                } else {
                    dc_reg[r] = 0.0f;
                }
                const float vals[4] = {dgi, dgf, dgg, dgo};
#pragma unroll
                for (int g = 0; g < 4; ++g) {
                    unsigned short hb = f32_to_bf16_bits(vals[g]);
                    *reinterpret_cast<__hip_bfloat16*>(
                        &buf[px_swz(row * rowbytes4 + (g * H + col) * 2, row)]) =
                        f2bf(vals[g]);
                    if (grow < B)
                        dg_t[(long)grow * H4 + g * H + col] = f2bf(vals[g]);
                    unsigned other = __shfl_xor((unsigned)hb, 1);
                    if ((lane & 1) == 0) {
                        unsigned lo = (unsigned)hb | (other << 16);
                        __hip_atomic_store(
                            &own[(long)row * (4 * hq2) + g * hq2 + cp_base],
                            ((unsigned long long)tag << 32) | lo, RLX_AGENT);
                    }
                }
            }
        }

        // ---- sweep peers' granules into the LDS tile, one peer region per
        // batch (8 granules/thread/peer; batched loads, re-poll stragglers)
        for (int pofs = 1; pofs < CG; ++pofs) {
            int p = cg + pofs;
            if (p >= CG) p -= CG;             // stagger start peer per block
            gu64* src = ep + (long)p * reg_sz;
            unsigned long long v[8];
            unsigned pending = 0;
#pragma unroll
            for (int i = 0; i < 8; ++i) {
                v[i] = __hip_atomic_load(&src[tid + i * bd], RLX_AGENT);
                pending |= (unsigned)((unsigned)(v[i] >> 32) != tag) << i;
            }
            unsigned spins = 0;
            while (pending) {
                if (++spins > spin_limit) {  // convergent bail (see fwd)
                    __hip_atomic_store((gu32*)err, 0x20000u + (unsigned)t,
                                       RLX_AGENT);
                    spin_limit = 40000u;
                    break;
                }
                __builtin_amdgcn_s_sleep(1);
#pragma unroll
                for (int i = 0; i < 8; ++i)
                    if (pending & (1u << i)) {
                        unsigned long long nv =
                            __hip_atomic_load(&src[tid + i * bd], RLX_AGENT);
                        if ((unsigned)(nv >> 32) == tag) {
                            v[i] = nv;
                            pending &= ~(1u << i);
                        }
                    }
            }
#pragma unroll
            for (int i = 0; i < 8; ++i) {
                int e = tid + i * bd;
                int row = e / (4 * hq2);
                int rem = e - row * (4 * hq2);
                int g = rem / hq2;
                int cpx = rem - g * hq2;
                unsigned off = px_swz(
                    (unsigned)(row * rowbytes4 + (g * H + p * HQg + cpx * 2) * 2),
                    row);
                *reinterpret_cast<unsigned*>(&buf[off]) = (unsigned)v[i];
            }
        }
        __syncthreads();

        // ---- dh recurrence (4-accumulator ILP split, as the slab body)
        f32x4 acc4[4] = {};
        const int nb = (col_base >> 4);
        for (int kk = 0; kk < H4; kk += 128) {
#pragma unroll
            for (int u = 0; u < 4; ++u) {
                int kx = kk + u * 32;
                int k = kx + a_kgrp;
                short8 a_frag = *reinterpret_cast<const short8*>(
                    &buf[px_swz(a_row * rowbytes4 + k * 2, a_row)]);
                long off = (((long)nb * (H4 >> 5) + (kx >> 5)) * 64 + lane) * 8;
                short8 b_frag = *reinterpret_cast<const short8*>(&whT[off]);
                acc4[u] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                    a_frag, b_frag, acc4[u], 0, 0, 0);
            }
        }
#pragma unroll
        for (int r = 0; r < 4; ++r)
            dh_reg[r] = acc4[0][r] + acc4[1][r] + acc4[2][r] + acc4[3][r];
        __syncthreads();  // MFMA reads done before next step's LDS writes
    }
#pragma unroll
    for (int r = 0; r < 4; ++r) {
        int grow = brow0 + c_row0 + r;
        int col = col_base + c_col;
        if (grow < B) {
            dh0[(long)grow * H + col] = f2bf(dh_reg[r]);
            dc0[(long)grow * H + col] = f2bf(dc_reg[r]);
        }
    }
}

template <int CG>
__global__ void __launch_bounds__(CG >= 8 ? 512 : 1024)
lstm_bwd_px_kernel_gx(BwdPxArgs a, unsigned int* err, int B, int H) {
    extern __shared__ __attribute__((aligned(16))) char smem[];
    lstm_bwd_px_body_gx<CG>(a, err, B, H, smem);
}

template <int CG>
__global__ void __launch_bounds__(CG >= 8 ? 512 : 1024)
lstm_bwd_px2_kernel_gx(BwdPxArgs a0, BwdPxArgs a1, unsigned int* err, int B, int H) {
    extern __shared__ __attribute__((aligned(16))) char smem[];
    lstm_bwd_px_body_gx<CG>(blockIdx.z ? a1 : a0, err, B, H, smem);
}

// MEASURED on MI355X (flagship, probe triplets): slab 4.95 ms/step vs
// granules 5.18 — the slab's vectorized 16B bulk stores/loads beat the
// per-granule 8B atomic traffic (2x bytes for tags, 7 sequential per-peer
// poll rounds) despite paying the vmcnt drain + flag chain. The slab body
// stays the DEFAULT; R2D2_PX_BWD_GX=1 selects the granule body.
static inline int px_bwd_slab() {
    static const int v = []() {
        const char* e = getenv("R2D2_PX_BWD_GX");
        return (e && e[0] && e[0] != '0') ? 0 : 1;
    }();
    return v;
}

extern "C" int launch_lstm_bwd_px_bf16(
    const void* dh_seq, const void* dhT, const void* dcT, const void* gates_act,
    const float* c_seq, const void* c0, const void* whT_packed,
    void* dgates, void* dh0, void* dc0,
    unsigned char* slab, unsigned long long* flags, unsigned int* err,
    int T, int B, int H, int budget, hipStream_t stream) {
    int RT = (B + 15) / 16;
    if (H & (H - 1)) return -1;
    if ((size_t)16 * 4 * H * 2 > 128 * 1024) return -1;  // keep 2 blocks/CU
    if (budget > 224) budget = 224;
    int CG = 0, cu_eq = 0;
    for (int c : {8, 4, 2}) {
        if (c > px_cg_max()) continue;
        if (H % (16 * c)) continue;
        int w = (H / c) / 16;
        if (w < 2 || w > 16) continue;
        if (H / c < 64) continue;  // slab regions must be >= 128B: the XOR
                                   // swizzle permutes 16B blocks within a
                                   // 128B window, so raw byte-range copies
                                   // are closed only for >=128B regions
        if (RT * c > 224) continue;
        CG = c;
        cu_eq = px_cu_eq(RT * c, w, 16L * 4 * H * 2);
        break;
    }
    if (CG == 0) return -1;
    if (cu_eq > budget) return -2;           // caller: drain in-flight, retry
    int waves = (H / CG) / 16;
    dim3 grid(RT, CG);
    size_t lds = (size_t)16 * 4 * H * 2;
    BwdPxArgs a{(const __hip_bfloat16*)dh_seq, (const __hip_bfloat16*)dhT,
                (const __hip_bfloat16*)dcT, (const __hip_bfloat16*)gates_act,
                c_seq, (const __hip_bfloat16*)c0,
                (const __hip_bfloat16*)whT_packed, (__hip_bfloat16*)dgates,
                (__hip_bfloat16*)dh0, (__hip_bfloat16*)dc0, slab, flags, T,
                px_xcd_swz()};
    if (px_bwd_slab()) {
        switch (CG) {
            case 2: lstm_bwd_px_kernel<2><<<grid, dim3(waves * 64), lds, stream>>>(a, err, B, H); break;
            case 4: lstm_bwd_px_kernel<4><<<grid, dim3(waves * 64), lds, stream>>>(a, err, B, H); break;
            case 8: lstm_bwd_px_kernel<8><<<grid, dim3(waves * 64), lds, stream>>>(a, err, B, H); break;
            default: return -1;
        }
    } else {
        switch (CG) {
            case 2: lstm_bwd_px_kernel_gx<2><<<grid, dim3(waves * 64), lds, stream>>>(a, err, B, H); break;
            case 4: lstm_bwd_px_kernel_gx<4><<<grid, dim3(waves * 64), lds, stream>>>(a, err, B, H); break;
            case 8: lstm_bwd_px_kernel_gx<8><<<grid, dim3(waves * 64), lds, stream>>>(a, err, B, H); break;
            default: return -1;
        }
    }
    LSTM_CHECK_LAUNCH();
    return cu_eq;
}

extern "C" int launch_lstm_bwd_px2_bf16(
    const void* dh_seq0, const void* dhT0, const void* dcT0, const void* ga0,
    const float* cs0, const void* c00, const void* whT0, void* dg0, void* dh00,
    void* dc00, unsigned char* slab0, unsigned long long* flags0, int T0,
    const void* dh_seq1, const void* dhT1, const void* dcT1, const void* ga1,
    const float* cs1, const void* c01, const void* whT1, void* dg1, void* dh01,
    void* dc01, unsigned char* slab1, unsigned long long* flags1, int T1,
    unsigned int* err, int B, int H, int budget, hipStream_t stream) {
    int RT = (B + 15) / 16;
    if (H & (H - 1)) return -1;
    if ((size_t)16 * 4 * H * 2 > 128 * 1024) return -1;
    if (budget > 224) budget = 224;
    int CG = 0, cu_eq = 0;
    for (int c : {8, 4, 2}) {
        if (c > px_cg_max()) continue;
        if (H % (16 * c)) continue;
        int w = (H / c) / 16;
        if (w < 2 || w > 16) continue;
        if (H / c < 64) continue;        // slab swizzle closure (see solo)
        if (RT * c * 2 > 224) continue;  // BOTH halves must co-reside
        CG = c;
        cu_eq = px_cu_eq(RT * c * 2, w, 16L * 4 * H * 2);
        break;
    }
    if (CG == 0) return -1;
    if (cu_eq > budget) return -2;       // caller: drain in-flight, retry
    int waves = (H / CG) / 16;
    dim3 grid(RT, CG, 2);
    size_t lds = (size_t)16 * 4 * H * 2;
    BwdPxArgs a0{(const __hip_bfloat16*)dh_seq0, (const __hip_bfloat16*)dhT0,
                 (const __hip_bfloat16*)dcT0, (const __hip_bfloat16*)ga0, cs0,
                 (const __hip_bfloat16*)c00, (const __hip_bfloat16*)whT0,
                 (__hip_bfloat16*)dg0, (__hip_bfloat16*)dh00,
                 (__hip_bfloat16*)dc00, slab0, flags0, T0, px_xcd_swz()};
    BwdPxArgs a1{(const __hip_bfloat16*)dh_seq1, (const __hip_bfloat16*)dhT1,
                 (const __hip_bfloat16*)dcT1, (const __hip_bfloat16*)ga1, cs1,
                 (const __hip_bfloat16*)c01, (const __hip_bfloat16*)whT1,
                 (__hip_bfloat16*)dg1, (__hip_bfloat16*)dh01,
                 (__hip_bfloat16*)dc01, slab1, flags1, T1, px_xcd_swz()};
    if (px_bwd_slab()) {
        switch (CG) {
            case 2: lstm_bwd_px2_kernel<2><<<grid, dim3(waves * 64), lds, stream>>>(a0, a1, err, B, H); break;
            case 4: lstm_bwd_px2_kernel<4><<<grid, dim3(waves * 64), lds, stream>>>(a0, a1, err, B, H); break;
            case 8: lstm_bwd_px2_kernel<8><<<grid, dim3(waves * 64), lds, stream>>>(a0, a1, err, B, H); break;
            default: return -1;
        }
    } else {
        switch (CG) {
            case 2: lstm_bwd_px2_kernel_gx<2><<<grid, dim3(waves * 64), lds, stream>>>(a0, a1, err, B, H); break;
            case 4: lstm_bwd_px2_kernel_gx<4><<<grid, dim3(waves * 64), lds, stream>>>(a0, a1, err, B, H); break;
            case 8: lstm_bwd_px2_kernel_gx<8><<<grid, dim3(waves * 64), lds, stream>>>(a0, a1, err, B, H); break;
            default: return -1;
        }
    }
    LSTM_CHECK_LAUNCH();
    return cu_eq;
}

// ------------------------------------------------- paired (z=2) forward
// Two INDEPENDENT LSTM unrolls (same B and H; T and stash may differ per
// set — e.g. the actor/critic burn-ins, or the target-actor unroll paired
// with the grad-tracked online critic forward) in one launch: grid.z
// selects the pointer set. A single grid dispatches uniformly, so the two
// halves co-reside by construction — unlike two concurrent px launches on
// different streams, which contend for CUs and run ~3x slower than
// back-to-back.
struct PxArgs {
    const __hip_bfloat16* gates_x;
    const __hip_bfloat16* h0;
    const __hip_bfloat16* c0;
    const __hip_bfloat16* wh;
    __hip_bfloat16* h_seq;
    __hip_bfloat16* c_final;
    __hip_bfloat16* gates_act;  // stash only
    float* c_seq;               // stash only
    unsigned long long* hx;
    int T;
    int stash;
    int xcd_swz;
};

template <int CG>
__global__ void __launch_bounds__(CG >= 8 ? 512 : 1024)
lstm_fwd_px2_kernel(PxArgs a0, PxArgs a1, unsigned int* err, int B, int H) {
    const PxArgs& a = blockIdx.z ? a1 : a0;
    const int T = a.T;
    extern __shared__ __attribute__((aligned(16))) char smem[];

    const int tid = threadIdx.x;
    const int wid = tid >> 6;
    const int lane = tid & 63;
    int rt, cg;
    px_decode_block<CG>(a.xcd_swz, rt, cg);
    const int brow0 = rt * 16;
    const int HQg = H / CG;
    const int col_base = cg * HQg + wid * 16;
    const int H4 = 4 * H;
    const int rowbytes = H * 2;
    const unsigned tile_bytes = 16u * (unsigned)rowbytes;

    const int a_row = lane & 15;
    const int a_kgrp = (lane >> 4) * 8;
    const int c_col = lane & 15;
    const int c_row0 = (lane >> 4) * 4;

    for (int e = tid; e < 16 * H; e += (int)blockDim.x) {
        int r = e / H, col = e - r * H;
        int grow = brow0 + r;
        float v = (grow < B) ? bf2f(a.h0[(long)grow * H + col]) : 0.0f;
        *reinterpret_cast<__hip_bfloat16*>(&smem[px_swz(r * rowbytes + col * 2, r)]) = f2bf(v);
    }
    float c_reg[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
        int grow = brow0 + c_row0 + r;
        int col = col_base + c_col;
        c_reg[r] = (grow < B) ? bf2f(a.c0[(long)grow * H + col]) : 0.0f;
    }
    __syncthreads();

    gu64* hx_g = (gu64*)a.hx;
    const long rt_off = (long)rt * 16 * (H / 2);
    const long slot_sz = (long)gridDim.x * 16 * (H / 2);
    unsigned spin_limit = 20000000u;  // degrades after the first timeout

    for (int t = 0; t < T; ++t) {
        const __hip_bfloat16* gx_t = a.gates_x + (long)t * B * H4;
        f32x4 acc[4];
#pragma unroll
        for (int g = 0; g < 4; ++g)
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                int grow = brow0 + c_row0 + r;
                int j = g * H + col_base + c_col;
                acc[g][r] = (grow < B) ? bf2f(gx_t[(long)grow * H4 + j]) : 0.0f;
            }
        if (t > 0) {
            char* bld = smem + (t & 1) * tile_bytes;
            const unsigned tag = (unsigned)t;
            gu64* slot = hx_g + ((t - 1) & 1) * slot_sz + rt_off;
            constexpr int K = 2 * CG;
            const int bd = (int)blockDim.x;
            const int sh = 31 - __builtin_clz((unsigned)(H >> 1));
            unsigned long long v[K];
#pragma unroll
            for (int i = 0; i < K; ++i)
                v[i] = __hip_atomic_load(&slot[tid + i * bd], RLX_AGENT);
            unsigned pending = 0;
#pragma unroll
            for (int i = 0; i < K; ++i)
                pending |= (unsigned)((unsigned)(v[i] >> 32) != tag) << i;
            unsigned spins = 0;
            while (pending) {
                if (++spins > spin_limit) {  // convergent bail (see single fwd)
                    __hip_atomic_store((gu32*)err, 1u + (unsigned)t, RLX_AGENT);
                    spin_limit = 40000u;
                    break;
                }
                __builtin_amdgcn_s_sleep(1);
#pragma unroll
                for (int i = 0; i < K; ++i)
                    if (pending & (1u << i)) {
                        unsigned long long nv =
                            __hip_atomic_load(&slot[tid + i * bd], RLX_AGENT);
                        if ((unsigned)(nv >> 32) == tag) {
                            v[i] = nv;
                            pending &= ~(1u << i);
                        }
                    }
            }
#pragma unroll
            for (int i = 0; i < K; ++i) {
                int e = tid + i * bd;
                int r = e >> sh, gcol = e & ((H >> 1) - 1);
                *reinterpret_cast<unsigned*>(
                    &bld[px_swz(r * rowbytes + gcol * 4, r)]) = (unsigned)v[i];
            }
        }
        __syncthreads();

        __hip_bfloat16* hs_t = a.h_seq + (long)t * B * H;
        __hip_bfloat16* ga_t = a.stash ? a.gates_act + (long)t * B * H4 : nullptr;
        float* cs_t = a.stash ? a.c_seq + (long)t * B * H : nullptr;
        char* rd_buf = smem + (t & 1) * tile_bytes;
        for (int kk = 0; kk < H; kk += 32) {
            int k = kk + a_kgrp;
            short8 a_frag = *reinterpret_cast<const short8*>(
                &rd_buf[px_swz(a_row * rowbytes + k * 2, a_row)]);
            short8 breg[4];
            const int kb = kk >> 5;
#pragma unroll
            for (int g = 0; g < 4; ++g) {
                int jb = g * (H >> 4) + (col_base >> 4);
                long off = (((long)jb * (H >> 5) + kb) * 64 + lane) * 8;
                breg[g] = *reinterpret_cast<const short8*>(&a.wh[off]);
            }
#pragma unroll
            for (int g = 0; g < 4; ++g)
                acc[g] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                    a_frag, breg[g], acc[g], 0, 0, 0);
        }
        {
            gu64* slot = hx_g + (t & 1) * slot_sz + rt_off;
            const unsigned tag = (unsigned)(t + 1);
            const int col = col_base + c_col;
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                int row = c_row0 + r;
                int grow = brow0 + row;
                float gi = px_sigm(acc[0][r]);
                float gf = px_sigm(acc[1][r]);
                float gg = px_tanh(acc[2][r]);
                float go = px_sigm(acc[3][r]);
                float c = gf * c_reg[r] + gi * gg;
                float hn = go * px_tanh(c);
                c_reg[r] = c;
                unsigned short hb = f32_to_bf16_bits(hn);
                unsigned other = __shfl_xor((unsigned)hb, 1);
                if ((lane & 1) == 0) {
                    unsigned lo = (unsigned)hb | (other << 16);
                    __hip_atomic_store(&slot[row * (H / 2) + col / 2],
                                       ((unsigned long long)tag << 32) | lo, RLX_AGENT);
                }
                if (grow < B) {
                    hs_t[(long)grow * H + col] = f2bf(hn);
                    if (a.stash) {
                        cs_t[(long)grow * H + col] = c;
                        ga_t[(long)grow * H4 + 0 * H + col] = f2bf(gi);
                        ga_t[(long)grow * H4 + 1 * H + col] = f2bf(gf);
                        ga_t[(long)grow * H4 + 2 * H + col] = f2bf(gg);
                        ga_t[(long)grow * H4 + 3 * H + col] = f2bf(go);
                    }
                }
            }
        }
        // no trailing barrier — see the solo forward kernel
    }
#pragma unroll
    for (int r = 0; r < 4; ++r) {
        int grow = brow0 + c_row0 + r;
        int col = col_base + c_col;
        if (grow < B) a.c_final[(long)grow * H + col] = f2bf(c_reg[r]);
    }
}


// ---- pipelined forward body (opt-in via R2D2_PX_FWD_PL=1): the h exchange
// feeds MFMA A-FRAGMENTS directly from per-lane granule polls — no LDS
// staging, no barriers in the T-loop at all. Key observation: lane l of a
// 16x16x32 MFMA needs exactly rows l&15 x h-cols (l>>4)*8..+8 = 4
// CONSECUTIVE granules of the exchange slot, and the granule payload
// (2 adjacent bf16 cols) is already in a_frag order — 4 tag-checked u64
// loads ARE the fragment. The k-loop polls chunk i while chunk i+1's loads
// are in flight (2-deep pipeline), so MFMA starts on the first-arrived
// chunk instead of after the block-wide sweep of the SLOWEST granule, and
// the two per-step barriers + 16 LDS round-trips per thread disappear.
// Per-lane divergent spins are safe precisely because there are no
// barriers; the bail degrades like the other bodies. Costs: waves re-load
// granules redundantly from L2 (4x traffic on a 16 KB/step tile — cheap)
// and t=0 reads h0 straight from global.
template <int CG>
DEV_INLINE void lstm_fwd_px_body_pl(const PxArgs& a, unsigned int* err,
                                    int B, int H) {
    const int T = a.T;
    const int tid = threadIdx.x;
    const int wid = tid >> 6;
    const int lane = tid & 63;
    int rt, cg;
    px_decode_block<CG>(a.xcd_swz, rt, cg);
    const int brow0 = rt * 16;
    const int HQg = H / CG;
    const int col_base = cg * HQg + wid * 16;
    const int H4 = 4 * H;

    const int a_row = lane & 15;
    const int a_kgrp = (lane >> 4) * 8;
    const int c_col = lane & 15;
    const int c_row0 = (lane >> 4) * 4;

    float c_reg[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
        int grow = brow0 + c_row0 + r;
        int col = col_base + c_col;
        c_reg[r] = (grow < B) ? bf2f(a.c0[(long)grow * H + col]) : 0.0f;
    }

    gu64* hx_g = (gu64*)a.hx;
    const long rt_off = (long)rt * 16 * (H / 2);
    const long slot_sz = (long)gridDim.x * 16 * (H / 2);
    unsigned spin_limit = 20000000u;
    const int a_base_row = a_row * (H >> 1);

    for (int t = 0; t < T; ++t) {
        const __hip_bfloat16* gx_t = a.gates_x + (long)t * B * H4;
        f32x4 acc[4];
#pragma unroll
        for (int g = 0; g < 4; ++g)
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                int grow = brow0 + c_row0 + r;
                int j = g * H + col_base + c_col;
                acc[g][r] = (grow < B) ? bf2f(gx_t[(long)grow * H4 + j]) : 0.0f;
            }

        const unsigned tag = (unsigned)t;  // consuming h(t-1)
        gu64* slot = hx_g + ((t - 1) & 1) * slot_sz + rt_off;
        const int hgrow = brow0 + a_row;

        unsigned long long vc[4], vn[4];
        if (t > 0) {
            int base = a_base_row + (a_kgrp >> 1);
#pragma unroll
            for (int i = 0; i < 4; ++i)
                vc[i] = __hip_atomic_load(&slot[base + i], RLX_AGENT);
        }
        for (int kk = 0; kk < H; kk += 32) {
            // issue next chunk loads while finishing this one
            if (t > 0 && kk + 32 < H) {
                int nbase = a_base_row + ((kk + 32 + a_kgrp) >> 1);
#pragma unroll
                for (int i = 0; i < 4; ++i)
                    vn[i] = __hip_atomic_load(&slot[nbase + i], RLX_AGENT);
            }
            short8 a_frag;
            if (t == 0) {
                if (hgrow < B)
                    a_frag = *reinterpret_cast<const short8*>(
                        &a.h0[(long)hgrow * H + kk + a_kgrp]);
                else
                    a_frag = short8{0, 0, 0, 0, 0, 0, 0, 0};
            } else {
                int base = a_base_row + ((kk + a_kgrp) >> 1);
                unsigned pending = 0;
#pragma unroll
                for (int i = 0; i < 4; ++i)
                    pending |= (unsigned)((unsigned)(vc[i] >> 32) != tag) << i;
                unsigned spins = 0;
                while (pending) {
                    if (++spins > spin_limit) {  // per-lane bail, no barriers
                        __hip_atomic_store((gu32*)err, 1u + (unsigned)t,
                                           RLX_AGENT);
                        spin_limit = 40000u;
                        break;
                    }
                    __builtin_amdgcn_s_sleep(1);
#pragma unroll
                    for (int i = 0; i < 4; ++i)
                        if (pending & (1u << i)) {
                            unsigned long long nv =
                                __hip_atomic_load(&slot[base + i], RLX_AGENT);
                            if ((unsigned)(nv >> 32) == tag) {
                                vc[i] = nv;
                                pending &= ~(1u << i);
                            }
                        }
                }
                unsigned u[4];
#pragma unroll
                for (int i = 0; i < 4; ++i) u[i] = (unsigned)vc[i];
                a_frag = *reinterpret_cast<const short8*>(u);
            }
            short8 breg[4];
            const int kb = kk >> 5;
#pragma unroll
            for (int g = 0; g < 4; ++g) {
                int jb = g * (H >> 4) + (col_base >> 4);
                long off = (((long)jb * (H >> 5) + kb) * 64 + lane) * 8;
                breg[g] = *reinterpret_cast<const short8*>(&a.wh[off]);
            }
#pragma unroll
            for (int g = 0; g < 4; ++g)
                acc[g] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                    a_frag, breg[g], acc[g], 0, 0, 0);
#pragma unroll
            for (int i = 0; i < 4; ++i) vc[i] = vn[i];
        }

        // ---- epilogue: activations, cell update, granule publish, stores
        {
            __hip_bfloat16* hs_t = a.h_seq + (long)t * B * H;
            __hip_bfloat16* ga_t = a.stash ? a.gates_act + (long)t * B * H4 : nullptr;
            float* cs_t = a.stash ? a.c_seq + (long)t * B * H : nullptr;
            gu64* oslot = hx_g + (t & 1) * slot_sz + rt_off;
            const unsigned otag = (unsigned)(t + 1);
            const int col = col_base + c_col;
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                int row = c_row0 + r;
                int grow = brow0 + row;
                float gi = px_sigm(acc[0][r]);
                float gf = px_sigm(acc[1][r]);
                float gg = px_tanh(acc[2][r]);
                float go = px_sigm(acc[3][r]);
                float c = gf * c_reg[r] + gi * gg;
                float hn = go * px_tanh(c);
                c_reg[r] = c;
                unsigned short hb = f32_to_bf16_bits(hn);
                unsigned other = __shfl_xor((unsigned)hb, 1);
                if ((lane & 1) == 0) {
                    unsigned lo = (unsigned)hb | (other << 16);
                    __hip_atomic_store(&oslot[row * (H / 2) + col / 2],
                                       ((unsigned long long)otag << 32) | lo,
                                       RLX_AGENT);
                }
                if (grow < B) {
                    hs_t[(long)grow * H + col] = f2bf(hn);
                    if (a.stash) {
                        cs_t[(long)grow * H + col] = c;
                        ga_t[(long)grow * H4 + 0 * H + col] = f2bf(gi);
                        ga_t[(long)grow * H4 + 1 * H + col] = f2bf(gf);
                        ga_t[(long)grow * H4 + 2 * H + col] = f2bf(gg);
                        ga_t[(long)grow * H4 + 3 * H + col] = f2bf(go);
                    }
                }
            }
        }
    }
#pragma unroll
    for (int r = 0; r < 4; ++r) {
        int grow = brow0 + c_row0 + r;
        int col = col_base + c_col;
        if (grow < B) a.c_final[(long)grow * H + col] = f2bf(c_reg[r]);
    }
}

template <int CG>
__global__ void __launch_bounds__(CG >= 8 ? 512 : 1024)
lstm_fwd_px_pl_kernel(PxArgs a, unsigned int* err, int B, int H) {
    lstm_fwd_px_body_pl<CG>(a, err, B, H);
}

template <int CG>
__global__ void __launch_bounds__(CG >= 8 ? 512 : 1024)
lstm_fwd_px2_pl_kernel(PxArgs a0, PxArgs a1, unsigned int* err, int B, int H) {
    lstm_fwd_px_body_pl<CG>(blockIdx.z ? a1 : a0, err, B, H);
}


extern "C" int launch_lstm_fwd_px2_bf16(
    const void* gx0, const void* h00, const void* c00, const void* wh0,
    void* hs0, void* cf0, void* ga0, float* cs0, int stash0, int T0,
    unsigned long long* hx0,
    const void* gx1, const void* h01, const void* c01, const void* wh1,
    void* hs1, void* cf1, void* ga1, float* cs1, int stash1, int T1,
    unsigned long long* hx1,
    unsigned int* err, int B, int H, int budget, hipStream_t stream) {
    int RT = (B + 15) / 16;
    if (H & (H - 1)) return -1;
    if (budget > 224) budget = 224;
    int CG = 0, cu_eq = 0;
    for (int c : {8, 4, 2}) {
        if (c > px_cg_max()) continue;
        if (H % (16 * c)) continue;
        int w = (H / c) / 16;
        if (w < 2 || w > 16) continue;
        if (RT * c * 2 > 224) continue;  // BOTH halves must co-reside
        CG = c;
        cu_eq = px_cu_eq(RT * c * 2, w, 2L * 16 * H * 2);
        break;
    }
    if (CG == 0) return -1;
    if (cu_eq > budget) return -2;       // caller: drain in-flight, retry
    int waves = (H / CG) / 16;
    dim3 grid(RT, CG, 2);
    size_t lds = (size_t)2 * 16 * H * 2;
    PxArgs a0{(const __hip_bfloat16*)gx0, (const __hip_bfloat16*)h00,
              (const __hip_bfloat16*)c00, (const __hip_bfloat16*)wh0,
              (__hip_bfloat16*)hs0, (__hip_bfloat16*)cf0,
              (__hip_bfloat16*)ga0, cs0, hx0, T0, stash0, px_xcd_swz()};
    PxArgs a1{(const __hip_bfloat16*)gx1, (const __hip_bfloat16*)h01,
              (const __hip_bfloat16*)c01, (const __hip_bfloat16*)wh1,
              (__hip_bfloat16*)hs1, (__hip_bfloat16*)cf1,
              (__hip_bfloat16*)ga1, cs1, hx1, T1, stash1, px_xcd_swz()};
    if (px_fwd_pl()) {
        switch (CG) {
            case 2: lstm_fwd_px2_pl_kernel<2><<<grid, dim3(waves * 64), 0, stream>>>(a0, a1, err, B, H); break;
            case 4: lstm_fwd_px2_pl_kernel<4><<<grid, dim3(waves * 64), 0, stream>>>(a0, a1, err, B, H); break;
            case 8: lstm_fwd_px2_pl_kernel<8><<<grid, dim3(waves * 64), 0, stream>>>(a0, a1, err, B, H); break;
            default: return -1;
        }
    } else {
        switch (CG) {
            case 2: lstm_fwd_px2_kernel<2><<<grid, dim3(waves * 64), lds, stream>>>(a0, a1, err, B, H); break;
            case 4: lstm_fwd_px2_kernel<4><<<grid, dim3(waves * 64), lds, stream>>>(a0, a1, err, B, H); break;
            case 8: lstm_fwd_px2_kernel<8><<<grid, dim3(waves * 64), lds, stream>>>(a0, a1, err, B, H); break;
            default: return -1;
        }
    }
    LSTM_CHECK_LAUNCH();
    return cu_eq;
}

// ---- pipelined-forward solo dispatch (see forward declaration above) ------
static int launch_fwd_pl_solo(const void* gates_x, const void* h0,
                              const void* c0, const void* wh_packed,
                              void* h_seq, void* c_final, void* gates_act,
                              float* c_seq, unsigned long long* hx,
                              unsigned int* err, int T, int B, int H,
                              int stash, int CG, int waves,
                              hipStream_t stream) {
    int RT = (B + 15) / 16;
    dim3 grid(RT, CG);
    PxArgs a{(const __hip_bfloat16*)gates_x, (const __hip_bfloat16*)h0,
             (const __hip_bfloat16*)c0, (const __hip_bfloat16*)wh_packed,
             (__hip_bfloat16*)h_seq, (__hip_bfloat16*)c_final,
             (__hip_bfloat16*)gates_act, c_seq, hx, T, stash, px_xcd_swz()};
    switch (CG) {
        case 2: lstm_fwd_px_pl_kernel<2><<<grid, dim3(waves * 64), 0, stream>>>(a, err, B, H); break;
        case 4: lstm_fwd_px_pl_kernel<4><<<grid, dim3(waves * 64), 0, stream>>>(a, err, B, H); break;
        case 8: lstm_fwd_px_pl_kernel<8><<<grid, dim3(waves * 64), 0, stream>>>(a, err, B, H); break;
        default: return -1;
    }
    LSTM_CHECK_LAUNCH();
    return 0;
}
