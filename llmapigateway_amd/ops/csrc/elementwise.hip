// Memory-bound elementwise / normalization kernels for gfx950.
//
// Per the CDNA4 rules (Guideline 13): bf16 traffic is vectorized as uint4
// (8 bf16 = 16 B per lane), fp32 accumulation, grid-stride where the shape
// is unbounded. These replace what the reference gateway delegated to its
// remote providers' model forward (SURVEY.md §2b rows RMSNorm/RoPE/…).

#include "common.h"

// ---------------------------------------------------------------------------
// RMSNorm: y = x * rsqrt(mean(x^2) + eps) * w        (one block per row)
// rmsnorm_residual fuses r = x + residual; residual <- r; y = rmsnorm(r):
// one HBM pass over x and residual instead of three.
// ---------------------------------------------------------------------------

template <bool FUSED_RESIDUAL>
__global__ void rmsnorm_kernel(
    bf16* __restrict__ out,            // [T, H]
    const bf16* __restrict__ x,        // [T, H]
    bf16* __restrict__ residual,       // [T, H] (FUSED only; updated in place)
    const bf16* __restrict__ weight,   // [H]
    float eps,
    int H) {
    const int row = blockIdx.x;
    const int tid = threadIdx.x;
    const int nthreads = blockDim.x;
    const uint4* xrow = reinterpret_cast<const uint4*>(x + (size_t)row * H);
    uint4* rrow = FUSED_RESIDUAL
                      ? reinterpret_cast<uint4*>(residual + (size_t)row * H)
                      : nullptr;
    uint4* orow = reinterpret_cast<uint4*>(out + (size_t)row * H);
    const uint4* wv = reinterpret_cast<const uint4*>(weight);
    const int nvec = H / 8;  // H is a multiple of 8 for all supported models

    // cache up to 4 uint4 per thread in registers (H <= 32*8*nthreads)
    float vals[32];
    int held = 0;
    float ssq = 0.f;
    for (int i = tid; i < nvec; i += nthreads) {
        uint4 vx = xrow[i];
        float f[8];
        unpack2(vx.x, f[0], f[1]);
        unpack2(vx.y, f[2], f[3]);
        unpack2(vx.z, f[4], f[5]);
        unpack2(vx.w, f[6], f[7]);
        if (FUSED_RESIDUAL) {
            uint4 vr = rrow[i];
            float g[8];
            unpack2(vr.x, g[0], g[1]);
            unpack2(vr.y, g[2], g[3]);
            unpack2(vr.z, g[4], g[5]);
            unpack2(vr.w, g[6], g[7]);
#pragma unroll
            for (int j = 0; j < 8; ++j) {
                // round the residual sum to bf16 (the value stored back) and
                // accumulate the norm on the rounded value — matches the
                // fp32-reference semantics of ops/reference.py
                f[j] = bfbits2f(f2bfbits(f[j] + g[j]));
            }
            uint4 vw;
            vw.x = pack2(f[0], f[1]);
            vw.y = pack2(f[2], f[3]);
            vw.z = pack2(f[4], f[5]);
            vw.w = pack2(f[6], f[7]);
            rrow[i] = vw;
        }
#pragma unroll
        for (int j = 0; j < 8; ++j) ssq += f[j] * f[j];
        if (held + 8 <= 32) {
#pragma unroll
            for (int j = 0; j < 8; ++j) vals[held + j] = f[j];
        }
        held += 8;
    }

    // block reduction of sum of squares
    __shared__ float red[16];
    float wsum = wave_reduce_sum(ssq);
    const int wid = tid / WAVE_SIZE;
    const int nwaves = nthreads / WAVE_SIZE;
    if ((tid & (WAVE_SIZE - 1)) == 0) red[wid] = wsum;
    __syncthreads();
    float total = 0.f;
#pragma unroll
    for (int w = 0; w < 16; ++w)
        if (w < nwaves) total += red[w];
    const float inv = rsqrtf(total / (float)H + eps);

    // write pass: reuse register-cached values when they fit
    held = 0;
    for (int i = tid; i < nvec; i += nthreads) {
        float f[8];
        if (held + 8 <= 32) {
#pragma unroll
            for (int j = 0; j < 8; ++j) f[j] = vals[held + j];
        } else {
            uint4 vx = FUSED_RESIDUAL ? rrow[i] : xrow[i];
            unpack2(vx.x, f[0], f[1]);
            unpack2(vx.y, f[2], f[3]);
            unpack2(vx.z, f[4], f[5]);
            unpack2(vx.w, f[6], f[7]);
        }
        uint4 vw = wv[i];
        float w8[8];
        unpack2(vw.x, w8[0], w8[1]);
        unpack2(vw.y, w8[2], w8[3]);
        unpack2(vw.z, w8[4], w8[5]);
        unpack2(vw.w, w8[6], w8[7]);
        uint4 vo;
        vo.x = pack2(f[0] * inv * w8[0], f[1] * inv * w8[1]);
        vo.y = pack2(f[2] * inv * w8[2], f[3] * inv * w8[3]);
        vo.z = pack2(f[4] * inv * w8[4], f[5] * inv * w8[5]);
        vo.w = pack2(f[6] * inv * w8[6], f[7] * inv * w8[7]);
        orow[i] = vo;
        held += 8;
    }
}

// Wave-per-row variant for the decode regime: T <= ~1024 rows means the
// block-per-row form leaves the chip latency-bound (measured ~5.2 us flat
// for T <= 256 vs ~0.3 us of L2 traffic — tools/rmsnorm_micro.py). One
// 64-lane wave owns a whole row: the block barrier and the LDS reduce
// round trip disappear (wave_reduce only), 4 independent rows share a
// block. VPL = row uint4-chunks per lane (H / 512), a template constant so
// the register row cache fully unrolls — with a runtime bound the compiler
// demoted vals[] to 272 B/lane of scratch and the kernel measured SLOWER
// than the block form (8.2 vs 5.2 us); compile-time it is pure VGPRs.
template <bool FUSED_RESIDUAL, int VPL>
__global__ void rmsnorm_wave_kernel(
    bf16* __restrict__ out, const bf16* __restrict__ x,
    bf16* __restrict__ residual, const bf16* __restrict__ weight, float eps,
    int T, int H) {
    const int row = blockIdx.x * 4 + (threadIdx.x >> 6);
    if (row >= T) return;
    const int lane = threadIdx.x & (WAVE_SIZE - 1);
    const uint4* xrow = reinterpret_cast<const uint4*>(x + (size_t)row * H);
    uint4* rrow = FUSED_RESIDUAL
                      ? reinterpret_cast<uint4*>(residual + (size_t)row * H)
                      : nullptr;
    uint4* orow = reinterpret_cast<uint4*>(out + (size_t)row * H);
    const uint4* wv = reinterpret_cast<const uint4*>(weight);

    float vals[VPL * 8];
    float ssq = 0.f;
#pragma unroll
    for (int t = 0; t < VPL; ++t) {
        const int i = lane + t * WAVE_SIZE;
        uint4 vx = xrow[i];
        float f[8];
        unpack2(vx.x, f[0], f[1]);
        unpack2(vx.y, f[2], f[3]);
        unpack2(vx.z, f[4], f[5]);
        unpack2(vx.w, f[6], f[7]);
        if (FUSED_RESIDUAL) {
            uint4 vr = rrow[i];
            float g[8];
            unpack2(vr.x, g[0], g[1]);
            unpack2(vr.y, g[2], g[3]);
            unpack2(vr.z, g[4], g[5]);
            unpack2(vr.w, g[6], g[7]);
#pragma unroll
            for (int j = 0; j < 8; ++j)
                f[j] = bfbits2f(f2bfbits(f[j] + g[j]));
            uint4 vw;
            vw.x = pack2(f[0], f[1]);
            vw.y = pack2(f[2], f[3]);
            vw.z = pack2(f[4], f[5]);
            vw.w = pack2(f[6], f[7]);
            rrow[i] = vw;
        }
#pragma unroll
        for (int j = 0; j < 8; ++j) {
            ssq += f[j] * f[j];
            vals[t * 8 + j] = f[j];
        }
    }
    const float inv = rsqrtf(wave_reduce_sum(ssq) / (float)H + eps);
#pragma unroll
    for (int t = 0; t < VPL; ++t) {
        const int i = lane + t * WAVE_SIZE;
        uint4 vw = wv[i];
        float w8[8];
        unpack2(vw.x, w8[0], w8[1]);
        unpack2(vw.y, w8[2], w8[3]);
        unpack2(vw.z, w8[4], w8[5]);
        unpack2(vw.w, w8[6], w8[7]);
        uint4 vo;
        vo.x = pack2(vals[t * 8] * inv * w8[0], vals[t * 8 + 1] * inv * w8[1]);
        vo.y = pack2(vals[t * 8 + 2] * inv * w8[2], vals[t * 8 + 3] * inv * w8[3]);
        vo.z = pack2(vals[t * 8 + 4] * inv * w8[4], vals[t * 8 + 5] * inv * w8[5]);
        vo.w = pack2(vals[t * 8 + 6] * inv * w8[6], vals[t * 8 + 7] * inv * w8[7]);
        orow[i] = vo;
    }
}

template <bool FUSED>
static hipError_t launch_rms_wave(bf16* out, const bf16* x, bf16* residual,
                                  const bf16* w, float eps, int T, int H,
                                  hipStream_t stream) {
    const dim3 grid((T + 3) / 4);
#define RMS_W(V)                                                               \
    rmsnorm_wave_kernel<FUSED, V>                                              \
        <<<grid, 256, 0, stream>>>(out, x, residual, w, eps, T, H)
    switch (H / 512) {
        case 1: RMS_W(1); break;
        case 2: RMS_W(2); break;
        case 3: RMS_W(3); break;
        case 4: RMS_W(4); break;
        case 5: RMS_W(5); break;
        case 6: RMS_W(6); break;
        case 7: RMS_W(7); break;
        default: RMS_W(8); break;
    }
#undef RMS_W
    HIP_CHECK_LAST();
    return hipSuccess;
}

static inline bool rms_wave_fits(int T, int H) {
    return T <= 1024 && H <= 4096 && H % 512 == 0;
}

extern "C" hipError_t launch_rmsnorm(
    void* out, const void* x, const void* weight, float eps, int T, int H,
    hipStream_t stream) {
    if (H % 8 != 0) return hipErrorInvalidValue;
    if (rms_wave_fits(T, H))
        return launch_rms_wave<false>((bf16*)out, (const bf16*)x, nullptr,
                                      (const bf16*)weight, eps, T, H, stream);
    {
        rmsnorm_kernel<false><<<T, 256, 0, stream>>>(
            (bf16*)out, (const bf16*)x, nullptr, (const bf16*)weight, eps, H);
    }
    HIP_CHECK_LAST();
    return hipSuccess;
}

extern "C" hipError_t launch_rmsnorm_residual(
    void* out, const void* x, void* residual, const void* weight, float eps,
    int T, int H, hipStream_t stream) {
    if (H % 8 != 0) return hipErrorInvalidValue;
    if (rms_wave_fits(T, H))
        return launch_rms_wave<true>((bf16*)out, (const bf16*)x,
                                     (bf16*)residual, (const bf16*)weight,
                                     eps, T, H, stream);
    {
        rmsnorm_kernel<true><<<T, 256, 0, stream>>>(
            (bf16*)out, (const bf16*)x, (bf16*)residual, (const bf16*)weight,
            eps, H);
    }
    HIP_CHECK_LAST();
    return hipSuccess;
}

// ---------------------------------------------------------------------------
// RoPE (llama rotate-half), in place on q [T,Hq,D] and k [T,Hkv,D].
// Rows of q/k may be strided in T (they are views into the fused qkv
// projection). cos_sin: [max_pos, D] fp32 = [cos(half) | sin(half)].
// One block per token; each wave rotates heads; lane d < D/2 handles the
// (d, d+half) pair of one head per iteration.
// ---------------------------------------------------------------------------

__global__ void rope_kernel(
    bf16* __restrict__ q,
    bf16* __restrict__ k,
    const int64_t* __restrict__ positions,  // [T]
    const float* __restrict__ cos_sin,      // [max_pos, D]
    int64_t q_stride,
    int64_t k_stride,
    int Hq,
    int Hkv,
    int D) {
    const int t = blockIdx.x;
    const int half = D / 2;
    const int64_t pos = positions[t];
    const float* cs = cos_sin + pos * D;

    const int lanes_per_head = half;         // one lane per rotation pair
    const int pairs = (Hq + Hkv) * half;     // total rotation pairs this token
    for (int p = threadIdx.x; p < pairs; p += blockDim.x) {
        const int head = p / lanes_per_head;
        const int d = p % lanes_per_head;
        bf16* base;
        if (head < Hq) {
            base = q + (size_t)t * q_stride + (size_t)head * D;
        } else {
            base = k + (size_t)t * k_stride + (size_t)(head - Hq) * D;
        }
        const float c = cs[d];
        const float s = cs[half + d];
        const float x1 = bf2f(base[d]);
        const float x2 = bf2f(base[half + d]);
        base[d] = f2bf(x1 * c - x2 * s);
        base[half + d] = f2bf(x2 * c + x1 * s);
    }
}

extern "C" hipError_t launch_rope(
    void* q, void* k, const int64_t* positions, const float* cos_sin, int T,
    int64_t q_stride, int64_t k_stride, int Hq, int Hkv, int D,
    hipStream_t stream) {
    const int threads = 256;
    rope_kernel<<<T, threads, 0, stream>>>(
        (bf16*)q, (bf16*)k, positions, cos_sin, q_stride, k_stride, Hq, Hkv, D);
    HIP_CHECK_LAST();
    return hipSuccess;
}

// ---------------------------------------------------------------------------
// SwiGLU: out[t, i] = silu(x[t, i]) * x[t, I + i]  (x contiguous [T, 2I])
// ---------------------------------------------------------------------------

__global__ void swiglu_kernel(
    bf16* __restrict__ out, const bf16* __restrict__ x, int I) {
    const int row = blockIdx.x;
    const uint4* grow = reinterpret_cast<const uint4*>(x + (size_t)row * 2 * I);
    const uint4* urow = reinterpret_cast<const uint4*>(x + (size_t)row * 2 * I + I);
    uint4* orow = reinterpret_cast<uint4*>(out + (size_t)row * I);
    const int nvec = I / 8;
    for (int i = threadIdx.x; i < nvec; i += blockDim.x) {
        uint4 vg = grow[i];
        uint4 vu = urow[i];
        float g[8], u[8];
        unpack2(vg.x, g[0], g[1]);
        unpack2(vg.y, g[2], g[3]);
        unpack2(vg.z, g[4], g[5]);
        unpack2(vg.w, g[6], g[7]);
        unpack2(vu.x, u[0], u[1]);
        unpack2(vu.y, u[2], u[3]);
        unpack2(vu.z, u[4], u[5]);
        unpack2(vu.w, u[6], u[7]);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
            const float s = g[j] / (1.f + __expf(-g[j]));
            g[j] = s * u[j];
        }
        uint4 vo;
        vo.x = pack2(g[0], g[1]);
        vo.y = pack2(g[2], g[3]);
        vo.z = pack2(g[4], g[5]);
        vo.w = pack2(g[6], g[7]);
        orow[i] = vo;
    }
}

extern "C" hipError_t launch_swiglu(
    void* out, const void* x, int T, int I, hipStream_t stream) {
    if (I % 8 != 0) return hipErrorInvalidValue;
    swiglu_kernel<<<T, 256, 0, stream>>>((bf16*)out, (const bf16*)x, I);
    HIP_CHECK_LAST();
    return hipSuccess;
}

// ---------------------------------------------------------------------------
// Fused RoPE + KV-cache scatter: rotate q/k in place, then write the roped
// k and v into the paged cache. One launch instead of two — at decode
// batch sizes these kernels are dispatch-ramp-bound (~5 us floor each at
// 256 one-per-CU workgroups), so a fused launch saves ~5 us per layer.
// Same-workgroup global k writes are drained by __syncthreads() before the
// cache-copy phase reads them back (same CU, own L1 — coherent).
// ---------------------------------------------------------------------------


// ---------------------------------------------------------------------------
// fp8 KV write helper: quantize one (token, head) row of D elements to
// e4m3 with a per-row scale (amax/448). One 64-lane wave per head row:
// lane covers D/64 elements, wave-reduced amax, packed byte stores.
// ---------------------------------------------------------------------------
__device__ __forceinline__ void fp8_write_row(
    const bf16* __restrict__ src,      // D contiguous bf16
    unsigned char* __restrict__ dst,   // D bytes
    float* __restrict__ scale_out,     // 1 float
    int D, int lane) {
    const int per = D / WAVE_SIZE;     // 2 at D=128
    float vals[4];                     // per <= 4 supported (D <= 256)
    float amax = 0.f;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
        if (j < per) {
            vals[j] = bf2f(src[lane * per + j]);
            amax = fmaxf(amax, fabsf(vals[j]));
        }
    }
    amax = wave_reduce_max(amax);
    const float sc = amax > 0.f ? amax / FP8_MAX : 1.f;
    if (lane == 0) *scale_out = sc;
    // exact division (not reciprocal-multiply): bitwise-matches the torch
    // reference quantizer, so tests compare exactly
    if (per == 2) {
        unsigned short pk = (unsigned short)f2fp8(vals[0] / sc) |
                            ((unsigned short)f2fp8(vals[1] / sc) << 8);
        reinterpret_cast<unsigned short*>(dst)[lane] = pk;
    } else {
        for (int j = 0; j < per; ++j) dst[lane * per + j] = f2fp8(vals[j] / sc);
    }
}

template <bool FP8>
__global__ void rope_kv_kernel(
    bf16* __restrict__ q,
    bf16* __restrict__ k,
    const bf16* __restrict__ v,
    void* __restrict__ k_cache,        // bf16 or fp8 bytes
    void* __restrict__ v_cache,
    float* __restrict__ k_scale,       // [NB, Hkv, BS] (FP8)
    float* __restrict__ v_scale,
    const int64_t* __restrict__ positions,     // [T]
    const float* __restrict__ cos_sin,         // [max_pos, D]
    const int64_t* __restrict__ slot_mapping,  // [T]
    int64_t q_stride,
    int64_t k_stride,
    int64_t v_stride,
    int Hq,
    int Hkv,
    int D,
    int block_size) {
    const int t = blockIdx.x;
    const int half = D / 2;
    const int64_t pos = positions[t];
    const float* cs = cos_sin + pos * D;

    const int pairs = (Hq + Hkv) * half;
    for (int p = threadIdx.x; p < pairs; p += blockDim.x) {
        const int head = p / half;
        const int d = p % half;
        bf16* base = (head < Hq)
                         ? q + (size_t)t * q_stride + (size_t)head * D
                         : k + (size_t)t * k_stride + (size_t)(head - Hq) * D;
        const float c = cs[d];
        const float s = cs[half + d];
        const float x1 = bf2f(base[d]);
        const float x2 = bf2f(base[half + d]);
        base[d] = f2bf(x1 * c - x2 * s);
        base[half + d] = f2bf(x2 * c + x1 * s);
    }

    const int64_t slot = slot_mapping[t];
    if (slot < 0) return;  // uniform per block: every thread sees the same t
    __syncthreads();       // k row writes above must land before the copy

    const int64_t block = slot / block_size;
    const int64_t off = slot % block_size;
    if (FP8) {
        // one wave per head row (k then v): blockDim 256 = 4 waves
        const int lane = threadIdx.x & (WAVE_SIZE - 1);
        const int wave = threadIdx.x >> 6;
        for (int hh = wave; hh < 2 * Hkv; hh += blockDim.x / WAVE_SIZE) {
            const int h = hh % Hkv;
            const bool is_k = hh < Hkv;
            const size_t row = ((size_t)block * Hkv + h) * block_size + off;
            fp8_write_row(
                (is_k ? k + (size_t)t * k_stride : v + (size_t)t * v_stride) +
                    (size_t)h * D,
                (unsigned char*)(is_k ? k_cache : v_cache) + row * D,
                (is_k ? k_scale : v_scale) + row, D, lane);
        }
        return;
    }
    const int nvec = (Hkv * D) / 8;
    const uint4* ksrc = reinterpret_cast<const uint4*>(k + (size_t)t * k_stride);
    const uint4* vsrc = reinterpret_cast<const uint4*>(v + (size_t)t * v_stride);
    for (int i = threadIdx.x; i < nvec; i += blockDim.x) {
        const int h = (i * 8) / D;
        const int d = (i * 8) % D;
        const size_t dst = (((size_t)block * Hkv + h) * block_size + off) * D + d;
        reinterpret_cast<uint4*>((bf16*)k_cache + dst)[0] = ksrc[i];
        reinterpret_cast<uint4*>((bf16*)v_cache + dst)[0] = vsrc[i];
    }
}

extern "C" hipError_t launch_rope_kv(
    void* q, void* k, const void* v, void* k_cache, void* v_cache,
    float* k_scale, float* v_scale,
    const int64_t* positions, const float* cos_sin,
    const int64_t* slot_mapping, int T, int64_t q_stride, int64_t k_stride,
    int64_t v_stride, int Hq, int Hkv, int D, int block_size,
    hipStream_t stream) {
    if (D % 8 != 0) return hipErrorInvalidValue;
    if (k_scale != nullptr) {
        if (D % WAVE_SIZE != 0 || D > 256) return hipErrorInvalidValue;
        rope_kv_kernel<true><<<T, 256, 0, stream>>>(
            (bf16*)q, (bf16*)k, (const bf16*)v, k_cache, v_cache, k_scale,
            v_scale, positions, cos_sin, slot_mapping, q_stride, k_stride,
            v_stride, Hq, Hkv, D, block_size);
    } else {
        rope_kv_kernel<false><<<T, 256, 0, stream>>>(
            (bf16*)q, (bf16*)k, (const bf16*)v, k_cache, v_cache, nullptr,
            nullptr, positions, cos_sin, slot_mapping, q_stride, k_stride,
            v_stride, Hq, Hkv, D, block_size);
    }
    HIP_CHECK_LAST();
    return hipSuccess;
}

// ---------------------------------------------------------------------------
// KV-cache scatter: write k/v rows [T, Hkv, D] (strided in T) into the paged
// cache [num_blocks, Hkv, block_size, D] at slot_mapping[t].
// One block per token; vectorized 16 B per lane.
// ---------------------------------------------------------------------------

template <bool FP8>
__global__ void kv_cache_write_kernel(
    const bf16* __restrict__ k,
    const bf16* __restrict__ v,
    void* __restrict__ k_cache,
    void* __restrict__ v_cache,
    float* __restrict__ k_scale,
    float* __restrict__ v_scale,
    const int64_t* __restrict__ slot_mapping,  // [T]
    int64_t k_stride,
    int64_t v_stride,
    int Hkv,
    int block_size,
    int D) {
    const int t = blockIdx.x;
    const int64_t slot = slot_mapping[t];
    if (slot < 0) return;
    const int64_t block = slot / block_size;
    const int64_t off = slot % block_size;
    if (FP8) {
        const int lane = threadIdx.x & (WAVE_SIZE - 1);
        const int wave = threadIdx.x >> 6;
        for (int hh = wave; hh < 2 * Hkv; hh += blockDim.x / WAVE_SIZE) {
            const int h = hh % Hkv;
            const bool is_k = hh < Hkv;
            const size_t row = ((size_t)block * Hkv + h) * block_size + off;
            fp8_write_row(
                (is_k ? k + (size_t)t * k_stride : v + (size_t)t * v_stride) +
                    (size_t)h * D,
                (unsigned char*)(is_k ? k_cache : v_cache) + row * D,
                (is_k ? k_scale : v_scale) + row, D, lane);
        }
        return;
    }
    const int nvec = (Hkv * D) / 8;  // uint4 elements per token

    const uint4* ksrc = reinterpret_cast<const uint4*>(k + (size_t)t * k_stride);
    const uint4* vsrc = reinterpret_cast<const uint4*>(v + (size_t)t * v_stride);
    for (int i = threadIdx.x; i < nvec; i += blockDim.x) {
        const int h = (i * 8) / D;
        const int d = (i * 8) % D;
        const size_t dst =
            (((size_t)block * Hkv + h) * block_size + off) * D + d;
        reinterpret_cast<uint4*>((bf16*)k_cache + dst)[0] = ksrc[i];
        reinterpret_cast<uint4*>((bf16*)v_cache + dst)[0] = vsrc[i];
    }
}

extern "C" hipError_t launch_kv_cache_write(
    const void* k, const void* v, void* k_cache, void* v_cache,
    float* k_scale, float* v_scale,
    const int64_t* slot_mapping, int T, int64_t k_stride, int64_t v_stride,
    int Hkv, int block_size, int D, hipStream_t stream) {
    if (D % 8 != 0) return hipErrorInvalidValue;
    if (k_scale != nullptr) {
        if (D % WAVE_SIZE != 0 || D > 256) return hipErrorInvalidValue;
        kv_cache_write_kernel<true><<<T, 256, 0, stream>>>(
            (const bf16*)k, (const bf16*)v, k_cache, v_cache, k_scale,
            v_scale, slot_mapping, k_stride, v_stride, Hkv, block_size, D);
    } else {
        kv_cache_write_kernel<false><<<T, 256, 0, stream>>>(
            (const bf16*)k, (const bf16*)v, k_cache, v_cache, nullptr,
            nullptr, slot_mapping, k_stride, v_stride, Hkv, block_size, D);
    }
    HIP_CHECK_LAST();
    return hipSuccess;
}
