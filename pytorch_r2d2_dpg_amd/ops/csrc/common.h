// Common helpers for the CDNA4 (gfx950) HIP kernels.
// Hand-written for MI355X: wave64, MFMA, LDS; no CUDA compatibility paths.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define DEV_INLINE __device__ __forceinline__

// Host-side post-launch check: a dispatch rejected by the runtime (bad
// geometry vs launch bounds, LDS overflow, ...) is otherwise SILENT and
// leaves the output garbage. Returns -3 so callers can fail loudly or
// fall back.
#define LSTM_CHECK_LAUNCH() \
    do { if (hipGetLastError() != hipSuccess) return -3; } while (0)

constexpr int WAVE = 64;

DEV_INLINE float sigmoidf_(float x) { return 1.0f / (1.0f + __expf(-x)); }

// bf16 <-> f32 (scalar). __hip_bfloat16 has conversions but keep it explicit.
DEV_INLINE float bf2f(__hip_bfloat16 v) { return __bfloat162float(v); }
DEV_INLINE __hip_bfloat16 f2bf(float v) { return __float2bfloat16(v); }

// round-to-nearest-even f32 -> bf16 bit pattern (for packed stores)
DEV_INLINE unsigned short f32_to_bf16_bits(float f) {
    union { float f; unsigned int u; } x{f};
    unsigned int lsb = (x.u >> 16) & 1u;
    unsigned int r = x.u + 0x7fffu + lsb;
    return (unsigned short)(r >> 16);
}

DEV_INLINE float bf16_bits_to_f32(unsigned short b) {
    union { unsigned int u; float f; } x{((unsigned int)b) << 16};
    return x.f;
}

// dtype-generic scalar load/store as f32
DEV_INLINE float ld_f(const float* p) { return *p; }
DEV_INLINE float ld_f(const __hip_bfloat16* p) { return bf2f(*p); }
DEV_INLINE void st_f(float* p, float v) { *p = v; }
DEV_INLINE void st_f(__hip_bfloat16* p, float v) { *p = f2bf(v); }

#define HIP_CHECK(expr)                                                        \
    do {                                                                       \
        hipError_t _e = (expr);                                                \
        if (_e != hipSuccess) {                                                \
            printf("HIP error %s at %s:%d\n", hipGetErrorString(_e), __FILE__, \
                   __LINE__);                                                  \
        }                                                                      \
    } while (0)
