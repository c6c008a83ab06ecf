// K7 — replay ingest/pack: fp8 (OCP e4m3) observation storage with fused
// batch-gather + decode at sample time.
//
// The GPU replay shard stores observations as fp8 e4m3 (gfx950 OCP format,
// NOT the MI300X fnuz variant) to halve/quarter shard bytes; the sample-time
// gather reads B scattered slot rows and decodes straight into the batch
// tensor, so sampled data never round-trips through host memory.

#include "common.h"

#include <hip/hip_fp8.h>

// ---- encode: f32 -> e4m3 bytes (elementwise over one or more slot rows)
extern "C" __global__ void __launch_bounds__(256)
fp8_encode_kernel(const float* __restrict__ src, unsigned char* __restrict__ dst,
                  long n, float scale) {
    long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
    const long stride = (long)gridDim.x * blockDim.x;
    for (; i < n; i += stride) {
        __hip_fp8_e4m3 v(src[i] * scale);
        dst[i] = v.__x;
    }
}

// ---- fused gather + decode: store rows [N, row] u8 -> out [B, row] f32/bf16
template <typename OUT>
__global__ void __launch_bounds__(256)
fp8_gather_decode_kernel(const unsigned char* __restrict__ store,
                         const long* __restrict__ idx,
                         OUT* __restrict__ out,
                         long row_elems, int B, float inv_scale) {
    const int b = blockIdx.y;
    if (b >= B) return;
    const unsigned char* src = store + (long)idx[b] * row_elems;
    OUT* dst = out + (long)b * row_elems;
    // vectorized: 4 bytes per lane per iteration
    long i4 = ((long)blockIdx.x * blockDim.x + threadIdx.x) * 4;
    const long stride4 = (long)gridDim.x * blockDim.x * 4;
    for (; i4 + 3 < row_elems; i4 += stride4) {
        uint32_t packed = *reinterpret_cast<const uint32_t*>(src + i4);
#pragma unroll
        for (int j = 0; j < 4; ++j) {
            __hip_fp8_e4m3 v; v.__x = (packed >> (8 * j)) & 0xff;
            st_f(&dst[i4 + j], (float)v * inv_scale);
        }
    }
    // tail (row_elems % 4 stragglers): unconditionally thread (0,0) of this
    // row's blocks — gating it on the thread's own exhausted i4 position
    // skipped the tail whenever the grid stride jumped past row_elems
    if (blockIdx.x == 0 && threadIdx.x == 0) {
        for (long i = row_elems & ~3L; i < row_elems; ++i) {
            __hip_fp8_e4m3 v; v.__x = src[i];
            st_f(&dst[i], (float)v * inv_scale);
        }
    }
}

extern "C" void launch_fp8_encode(const float* src, unsigned char* dst, long n,
                                  float scale, hipStream_t stream) {
    long want = (n + 255) / 256;
    int blocks = (int)(want < 2048 ? (want > 0 ? want : 1) : 2048);
    fp8_encode_kernel<<<dim3(blocks), dim3(256), 0, stream>>>(src, dst, n, scale);
}

extern "C" void launch_fp8_gather_decode(const unsigned char* store, const long* idx,
                                         void* out, int out_is_bf16, long row_elems,
                                         int B, float inv_scale, hipStream_t stream) {
    long want = (row_elems / 4 + 255) / 256;
    int bx = (int)(want < 32 ? (want > 0 ? want : 1) : 32);
    dim3 grid(bx, B);
    if (out_is_bf16)
        fp8_gather_decode_kernel<__hip_bfloat16><<<grid, dim3(256), 0, stream>>>(
            store, idx, (__hip_bfloat16*)out, row_elems, B, inv_scale);
    else
        fp8_gather_decode_kernel<float><<<grid, dim3(256), 0, stream>>>(
            store, idx, (float*)out, row_elems, B, inv_scale);
}
