// Shared helpers for the MI355X (gfx950 / CDNA4) gateway kernels.
// Compile: hipcc --offload-arch=gfx950 -O3 -std=c++17 -shared -fPIC
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <stdint.h>
#include <stdio.h>

#define WAVE 64  // CDNA wavefront width (hard-coded per guide: warpSize==64 on gfx950)

#define HIP_CHECK(expr)                                                              \
    do {                                                                             \
        hipError_t _e = (expr);                                                      \
        if (_e != hipSuccess) {                                                      \
            fprintf(stderr, "HIP error %s at %s:%d\n", hipGetErrorString(_e),        \
                    __FILE__, __LINE__);                                             \
            return (int)_e;                                                          \
        }                                                                            \
    } while (0)

typedef __hip_bfloat16 bf16_t;

// 8 x bf16 stored as shorts — the vectorized load unit (guide G13).
typedef __attribute__((ext_vector_type(8))) short short8;
typedef __attribute__((ext_vector_type(4))) short short4v;
typedef __attribute__((ext_vector_type(4))) float float4v;

__device__ __forceinline__ float bf16_to_f(short s) {
    union { float f; uint32_t u; } cvt;
    cvt.u = ((uint32_t)(uint16_t)s) << 16;
    return cvt.f;
}

__device__ __forceinline__ short f_to_bf16(float f) {
    union { float f; uint32_t u; } cvt;
    cvt.f = f;
    // round-to-nearest-even
    uint32_t lsb = (cvt.u >> 16) & 1;
    cvt.u += 0x7fff + lsb;
    return (short)(cvt.u >> 16);
}

__device__ __forceinline__ float gelu_tanh(float x) {
    // matches torch.nn.functional.gelu(approximate="tanh")
    const float c = 0.7978845608028654f;  // sqrt(2/pi)
    float x3 = x * x * x;
    return 0.5f * x * (1.0f + tanhf(c * (x + 0.044715f * x3)));
}

__device__ __forceinline__ float sigmoidf(float x) { return 1.0f / (1.0f + expf(-x)); }

static inline int ceil_div(int64_t a, int64_t b) { return (int)((a + b - 1) / b); }
