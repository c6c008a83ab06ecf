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
                   int T, int B, int H, int stash) {
    extern __shared__ __attribute__((aligned(16))) char smem[];  // 2 x h tile [16][H] bf16

    const int tid = threadIdx.x;
    const int wid = tid >> 6;
    const int lane = tid & 63;
    const int rt = blockIdx.x;            // row tile
    const int cg = blockIdx.y;            // column group
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

    for (int t = 0; t < T; ++t) {
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
                if (++spins > 40000000u) {  // bounded spin: bail, flag error
                    __hip_atomic_store((gu32*)err, 1u + (unsigned)t, RLX_AGENT);
                    goto bail;
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

        const __hip_bfloat16* gx_t = gates_x + (long)t * B * H4;
        __hip_bfloat16* hs_t = h_seq + (long)t * B * H;
        __hip_bfloat16* ga_t = gates_act + (long)t * B * H4;
        float* cs_t = c_seq + (long)t * B * H;
        char* rd_buf = smem + (t & 1) * tile_bytes;

        f32x4 acc[4];
#pragma unroll
        for (int g = 0; g < 4; ++g)
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                int grow = brow0 + c_row0 + r;
                int j = g * H + col_base + c_col;
                acc[g][r] = (grow < B) ? bf2f(gx_t[(long)grow * H4 + j]) : 0.0f;
            }
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
        __syncthreads();  // LDS wr_buf complete before next sweep reads buf
    }
bail:
#pragma unroll
    for (int r = 0; r < 4; ++r) {
        int grow = brow0 + c_row0 + r;
        int col = col_base + c_col;
        if (grow < B) c_final[(long)grow * H + col] = f2bf(c_reg[r]);
    }
}

extern "C" int launch_lstm_fwd_px_bf16(
    const void* gates_x, const void* h0, const void* c0, const void* wh_packed,
    void* h_seq, void* c_final, void* gates_act, float* c_seq,
    unsigned long long* hx, unsigned int* err,
    int T, int B, int H, int stash, hipStream_t stream) {
    int RT = (B + 15) / 16;
    // largest CG with co-residency margin and >= 2 fragments per gate
    if (H & (H - 1)) return -1;              // sweep indexing needs pow2 H
    int CG = 0;
    for (int c : {8, 4, 2}) {
        if (H % (16 * c)) continue;          // wave owns 16 cols
        int w = (H / c) / 16;                // waves per block
        if (w < 2 || w > 16) continue;
        if (RT * c > 224) continue;          // all blocks must be co-resident
        CG = c;
        break;
    }
    if (CG == 0) return -1;
    int waves = (H / CG) / 16;
    dim3 grid(RT, CG);
    size_t lds = (size_t)2 * 16 * H * 2;
    switch (CG) {
        case 2:
            lstm_fwd_px_kernel<2><<<grid, dim3(waves * 64), lds, stream>>>(
                (const __hip_bfloat16*)gates_x, (const __hip_bfloat16*)h0,
                (const __hip_bfloat16*)c0, (const __hip_bfloat16*)wh_packed,
                (__hip_bfloat16*)h_seq, (__hip_bfloat16*)c_final,
                (__hip_bfloat16*)gates_act, c_seq, hx, err, T, B, H, stash);
            break;
        case 4:
            lstm_fwd_px_kernel<4><<<grid, dim3(waves * 64), lds, stream>>>(
                (const __hip_bfloat16*)gates_x, (const __hip_bfloat16*)h0,
                (const __hip_bfloat16*)c0, (const __hip_bfloat16*)wh_packed,
                (__hip_bfloat16*)h_seq, (__hip_bfloat16*)c_final,
                (__hip_bfloat16*)gates_act, c_seq, hx, err, T, B, H, stash);
            break;
        case 8:
            lstm_fwd_px_kernel<8><<<grid, dim3(waves * 64), lds, stream>>>(
                (const __hip_bfloat16*)gates_x, (const __hip_bfloat16*)h0,
                (const __hip_bfloat16*)c0, (const __hip_bfloat16*)wh_packed,
                (__hip_bfloat16*)h_seq, (__hip_bfloat16*)c_final,
                (__hip_bfloat16*)gates_act, c_seq, hx, err, T, B, H, stash);
            break;
        default:
            return -1;
    }
    return 0;
}
