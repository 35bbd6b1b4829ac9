// Common helpers for the gfx950 (MI355X / CDNA4) kernels.
// Wavefront = 64 lanes; LDS 160 KiB/CU; bf16 loads vectorized as uint4
// (8 x bf16 = 16 B/lane) per the CDNA4 performance rules.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <cstdint>

#define WAVE_SIZE 64

using bf16 = __hip_bfloat16;

// fp8 KV cache (OCP e4m3fn — gfx950's native format, NOT MI300X fnuz).
// Quantization: per (token, head) row scale = amax/448; dequant scales
// factor out of the attention dot products so the kernels multiply once
// per key instead of per element.
#include <hip/hip_fp8.h>
#define FP8_MAX 448.0f
typedef __attribute__((__vector_size__(2 * sizeof(float)))) float f32x2_t;

__device__ __forceinline__ unsigned char f2fp8(float v) {
    return __hip_fp8_e4m3(v).__x;
}
// 4 packed e4m3 bytes -> 4 floats (hardware cvt_pk)
__device__ __forceinline__ void fp8x4_to_f32(uint32_t q, float& a, float& b,
                                             float& c, float& d) {
    const f32x2_t lo = __builtin_amdgcn_cvt_pk_f32_fp8(q, false);
    const f32x2_t hi = __builtin_amdgcn_cvt_pk_f32_fp8(q, true);
    a = lo[0];
    b = lo[1];
    c = hi[0];
    d = hi[1];
}

__device__ __forceinline__ float bf2f(bf16 v) { return __bfloat162float(v); }
__device__ __forceinline__ bf16 f2bf(float v) { return __float2bfloat16(v); }

// raw-bits converters for packed uint <-> bf16 pairs
__device__ __forceinline__ float bfbits2f(uint16_t bits) {
    union { uint32_t u; float f; } c;
    c.u = static_cast<uint32_t>(bits) << 16;
    return c.f;
}
__device__ __forceinline__ uint16_t f2bfbits(float v) {
    // round-to-nearest-even, matching __float2bfloat16
    union { float f; uint32_t u; } c;
    c.f = v;
    uint32_t lsb = (c.u >> 16) & 1u;
    uint32_t rounded = c.u + 0x7FFFu + lsb;
    return static_cast<uint16_t>(rounded >> 16);
}

// unpack a uint32 holding 2 bf16
__device__ __forceinline__ void unpack2(uint32_t p, float& lo, float& hi) {
    lo = bfbits2f(static_cast<uint16_t>(p & 0xFFFFu));
    hi = bfbits2f(static_cast<uint16_t>(p >> 16));
}
__device__ __forceinline__ uint32_t pack2(float lo, float hi) {
    return static_cast<uint32_t>(f2bfbits(lo)) |
           (static_cast<uint32_t>(f2bfbits(hi)) << 16);
}
// truncating pack (round-to-zero): 2 ops instead of ~10 — for paths where
// the input already carries quantization noise (fp8 dequant staging)
__device__ __forceinline__ uint32_t pack2_trunc(float lo, float hi) {
    union { float f; uint32_t u; } a, b;
    a.f = lo;
    b.f = hi;
    return (a.u >> 16) | (b.u & 0xFFFF0000u);
}

// wave-wide reductions (64 lanes)
__device__ __forceinline__ float wave_reduce_sum(float v) {
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, WAVE_SIZE);
    return v;
}
__device__ __forceinline__ float wave_reduce_max(float v) {
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, WAVE_SIZE));
    return v;
}

// reduction across a 16-lane group (used by MFMA 16x16 fragment rows)
__device__ __forceinline__ float group16_reduce_sum(float v) {
#pragma unroll
    for (int off = 8; off > 0; off >>= 1) v += __shfl_xor(v, off, WAVE_SIZE);
    return v;
}
__device__ __forceinline__ float group16_reduce_max(float v) {
#pragma unroll
    for (int off = 8; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, WAVE_SIZE));
    return v;
}

#define HIP_CHECK_LAST()                                                        \
    do {                                                                        \
        hipError_t err__ = hipGetLastError();                                   \
        if (err__ != hipSuccess) return err__;                                  \
    } while (0)

static inline int ceil_div_i(int a, int b) { return (a + b - 1) / b; }
