// Skinny-M streaming GEMM for gfx950: Y[M,N] = X[M,K] · W[N,K]^T, bf16.
//
// The decode-step projections (M = batch ≤ 256, W up to 0.5 GB/layer) are
// weight-stream-bound: ideal time is W-bytes / HBM-rate. rocBLAS/hipBLASLt
// run these 3-4x off that bound at skinny M (measured via TunableOp
// cold-cache sweep: down-proj 4096x256x14336 at 66 us vs the ~15 us
// stream bound). This kernel streams W exactly ONCE:
//
// - grid = (N/64 n-stripes) x (split-K slices). One workgroup owns ALL M
//   rows of a 64-wide N stripe for its K-slice, so W is never re-read and
//   the X re-read factor is only N/64 (X lives in the per-XCD L2s).
// - up to 8 waves; wave w owns M rows [w*16*MF, ..): MF m-fragments x
//   NF=4 n-fragments of v_mfma_f32_16x16x32_bf16 per 32-deep k-step. All
//   fragments are 16 B dwordx4 loads straight from global (B/W coalesced
//   by 16 rows x 64 B, or fully contiguous with the opt-in k-major
//   swizzle; A/X from L2).
// - software pipeline up to depth 4, held in shape by sched_barriers
//   (see profiles/r01_gemm_skinny_probe.md for the measured history).
//   Production dispatch is the latency regime (M <= 16, ops/__init__.py);
//   the tuned library keeps M >= 64 (its LDS macro-tiles re-read X less).
// - split-K writes private fp32 slabs (each workgroup fully writes its
//   [M x 64] stripe of slab blockIdx.y — no zero-fill, no atomics, no
//   cross-workgroup visibility hazard); a second tiny kernel reduces the
//   slabs to bf16. Both launches are hipGraph-capturable (static scratch).
//
// Replaces F.linear on the decode hot path (models/llama.py) for
// M <= GS_MAX_M and N <= 28672; prefill, lm_head and big-M stay on the
// TunableOp-tuned library GEMMs.

#include "common.h"

#define GS_MAX_M 256
#define GS_WAVES 4
#define GS_NF 4                    // 16-col n-fragments per stripe
#define GS_NT (GS_NF * 16)         // stripe width = 64 columns

typedef __attribute__((__vector_size__(8 * sizeof(short)))) short bf16x8;
typedef __attribute__((__vector_size__(4 * sizeof(float)))) float f32x4;

__device__ __forceinline__ f32x4 gs_mfma(bf16x8 a, bf16x8 b, f32x4 c) {
    return __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
}

// 16-B load through an explicit global (addrspace 1) pointer: loop-carried
// generic pointers defeat the compiler's addrspace inference and demote
// these to flat_load (which drains lgkmcnt as well — serializing the
// software pipeline into load -> full drain -> mfma).
__device__ __forceinline__ bf16x8 gs_gload(const bf16* p) {
    return *(const __attribute__((address_space(1))) bf16x8*)(unsigned long long)p;
}

// NOTE: nt loads on W measured 20-40% SLOWER here: each 128 B line spans
// two k-steps of a row, and nt drops the line before the second 64 B hit,
// doubling the W fetch traffic.

// scheduling fence: keep the software-pipeline shape — without it the
// scheduler compresses the depth-4 pipeline back to serial loads (84
// VGPRs observed) and the kernel runs latency-bound
#define GS_FENCE __builtin_amdgcn_sched_barrier(0)

// blockDim = nwaves*64 with nwaves = ceil(M / (16*MF)) (<= 8): MF=2 with 8
// waves covers M=256 while the register-feasible depth-4 pipeline applies.
// SWZ: W is pre-swizzled k-major ([K/32, N, 32]) so every B-fragment load
// is part of a contiguous 4 KB (64-row x 64 B) stream per k-step —
// the [N, K] row-major layout reads 16 rows x 64 B per instruction and
// caps at ~3.5 TB/s effective.
template <int MF, bool SPLITK, bool SWZ>
__launch_bounds__(8 * WAVE_SIZE)
__global__ void gemm_skinny_kernel(
    bf16* __restrict__ y,        // [M, N] (!SPLITK)
    float* __restrict__ yw,      // [nsk, M, N] fp32 slabs (SPLITK)
    const bf16* __restrict__ x,  // [M, K]
    const bf16* __restrict__ w,  // [N, K] or swizzled [K/32, N, 32]
    int M,
    int N,
    int K,
    int nsk) {
    const int n0 = blockIdx.x * GS_NT;
    const int kslice = ((K / 32 + nsk - 1) / nsk) * 32;
    const int k0 = (SPLITK ? blockIdx.y : 0) * kslice;
    const int k1 = min(K, k0 + kslice);

    const int lane = threadIdx.x & (WAVE_SIZE - 1);
    const int wave = threadIdx.x >> 6;
    const int lrow = lane & 15;  // A row / B col within a fragment
    const int lk = lane >> 4;    // k-group of 8 bf16

    const int m_base = wave * (MF * 16);

    f32x4 acc[MF][GS_NF];
#pragma unroll
    for (int f = 0; f < MF; ++f)
#pragma unroll
        for (int n = 0; n < GS_NF; ++n) acc[f][n] = (f32x4){0.f, 0.f, 0.f, 0.f};

    const bf16* ap[MF];
#pragma unroll
    for (int f = 0; f < MF; ++f) {
        const int r = m_base + f * 16 + lrow;
        ap[f] = x + (size_t)(r < M ? r : 0) * K + k0 + lk * 8;
    }
    // B pointers: row-major [N, K] strides K per n-row and 32 per k-step;
    // swizzled [K/32, N, 32] strides 32 per n-row and 32*N per k-step
    const size_t bstep = SWZ ? (size_t)32 * N : 32;  // elems per k-step
    const bf16* bp[GS_NF];
#pragma unroll
    for (int n = 0; n < GS_NF; ++n)
        bp[n] = SWZ
                    ? w + (size_t)(k0 / 32) * N * 32 +
                          (size_t)(n0 + n * 16 + lrow) * 32 + lk * 8
                    : w + (size_t)(n0 + n * 16 + lrow) * K + k0 + lk * 8;

    const int nsteps = (k1 - k0) / 32;  // K and kslice are 32-aligned
    if (nsteps > 0 && m_base < M) {
        // Explicit buffer sets, all indices compile-time (a [d][..] array
        // indexed by a runtime phase demotes to scratch; depth 4 at MF=4
        // spilled to 280 VGPRs = 1 wave/SIMD). Without pipelining, ~70%
        // of wave cycles park on the ~900-cycle HBM latency of the W
        // stream (measured SQ_WAIT_ANY).
        bf16x8 a0[MF], b0[GS_NF], a1[MF], b1[GS_NF];
        bf16x8 a2[MF], b2[GS_NF], a3[MF], b3[GS_NF];
#define GS_LOAD(AV, BV, S)                                                     \
    do {                                                                       \
        _Pragma("unroll") for (int f = 0; f < MF; ++f) AV[f] =                 \
            gs_gload(ap[f] + (S) * 32);                                        \
        _Pragma("unroll") for (int n = 0; n < GS_NF; ++n) BV[n] =              \
            gs_gload(bp[n] + (S) * (SWZ ? bstep : 32));                        \
    } while (0)
#define GS_MFMA_ALL(AV, BV)                                                    \
    do {                                                                       \
        _Pragma("unroll") for (int n = 0; n < GS_NF; ++n)                      \
            _Pragma("unroll") for (int f = 0; f < MF; ++f) acc[f][n] =         \
                gs_mfma(AV[f], BV[n], acc[f][n]);                              \
    } while (0)
#define GS_BUMP(STEPS)                                                         \
    do {                                                                       \
        _Pragma("unroll") for (int f = 0; f < MF; ++f) ap[f] += (STEPS) * 32;  \
        _Pragma("unroll") for (int n = 0; n < GS_NF; ++n)                      \
            bp[n] += (STEPS) * (SWZ ? bstep : 32);                             \
    } while (0)
        if constexpr (MF <= 2) {
            // depth-4 pipeline: fits registers at MF<=2 and fully covers
            // HBM latency (small-M decode: latency matters most there)
            if (nsteps < 3) {
                for (int s = 0; s < nsteps; ++s) {
                    GS_LOAD(a0, b0, 0);
                    GS_MFMA_ALL(a0, b0);
                    GS_BUMP(1);
                }
            } else {
                GS_LOAD(a0, b0, 0);
                GS_LOAD(a1, b1, 1);
                GS_LOAD(a2, b2, 2);
                int s = 0;
                for (; s + 7 <= nsteps; s += 4) {  // prefetches reach s+6
                    GS_LOAD(a3, b3, 3);
                    GS_FENCE;
                    GS_MFMA_ALL(a0, b0);
                    GS_FENCE;
                    GS_LOAD(a0, b0, 4);
                    GS_FENCE;
                    GS_MFMA_ALL(a1, b1);
                    GS_FENCE;
                    GS_LOAD(a1, b1, 5);
                    GS_FENCE;
                    GS_MFMA_ALL(a2, b2);
                    GS_FENCE;
                    GS_LOAD(a2, b2, 6);
                    GS_FENCE;
                    GS_MFMA_ALL(a3, b3);
                    GS_BUMP(4);
                }
                // tail: r in [3,6]; a0/a1/a2 hold steps s, s+1, s+2
                const int r = nsteps - s;
                if (r >= 4) GS_LOAD(a3, b3, 3);
                GS_MFMA_ALL(a0, b0);
                if (r >= 5) GS_LOAD(a0, b0, 4);
                GS_MFMA_ALL(a1, b1);
                if (r >= 6) GS_LOAD(a1, b1, 5);
                GS_MFMA_ALL(a2, b2);
                if (r >= 4) GS_MFMA_ALL(a3, b3);
                if (r >= 5) GS_MFMA_ALL(a0, b0);
                if (r >= 6) GS_MFMA_ALL(a1, b1);
            }
        } else {
            // depth-2 ping/pong: MF 3-4 spills at deeper pipelines
            GS_LOAD(a0, b0, 0);
            int s = 0;
            for (; s + 2 <= nsteps - 1; s += 2) {
                GS_LOAD(a1, b1, 1);
                GS_FENCE;
                GS_MFMA_ALL(a0, b0);
                GS_FENCE;
                GS_LOAD(a0, b0, 2);
                GS_FENCE;
                GS_MFMA_ALL(a1, b1);
                GS_BUMP(2);
            }
            if (nsteps - s == 2) {
                GS_LOAD(a1, b1, 1);
                GS_MFMA_ALL(a0, b0);
                GS_MFMA_ALL(a1, b1);
            } else {
                GS_MFMA_ALL(a0, b0);
            }
        }
#undef GS_LOAD
#undef GS_MFMA_ALL
#undef GS_BUMP
    }

    // ---- epilogue: C fragment row = lk*4 + r, col = lrow ----
    if (SPLITK) {
        float* slab = yw + (size_t)blockIdx.y * M * N;
#pragma unroll
        for (int f = 0; f < MF; ++f)
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                const int row = m_base + f * 16 + lk * 4 + r;
                if (row >= M) continue;
#pragma unroll
                for (int n = 0; n < GS_NF; ++n)
                    slab[(size_t)row * N + n0 + n * 16 + lrow] = acc[f][n][r];
            }
    } else {
#pragma unroll
        for (int f = 0; f < MF; ++f)
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                const int row = m_base + f * 16 + lk * 4 + r;
                if (row >= M) continue;
#pragma unroll
                for (int n = 0; n < GS_NF; ++n)
                    y[(size_t)row * N + n0 + n * 16 + lrow] = f2bf(acc[f][n][r]);
            }
    }
}

// sum the split-K fp32 slabs into bf16 y; float4-vectorized grid-stride
__global__ void gemm_skinny_reduce_kernel(
    bf16* __restrict__ y, const float* __restrict__ yw, int64_t mn, int nsk) {
    const int64_t i4 = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) * 4;
    if (i4 >= mn) return;
    float4 s = *reinterpret_cast<const float4*>(yw + i4);
    for (int k = 1; k < nsk; ++k) {
        const float4 t = *reinterpret_cast<const float4*>(yw + (int64_t)k * mn + i4);
        s.x += t.x; s.y += t.y; s.z += t.z; s.w += t.w;
    }
    uint2 packed;
    packed.x = pack2(s.x, s.y);
    packed.y = pack2(s.z, s.w);
    *reinterpret_cast<uint2*>(y + i4) = packed;
}

// standalone split-K slab reduction (shared with gemm_m256.hip)
extern "C" hipError_t launch_gemm_reduce(
    void* y, const float* yw, int64_t mn, int nsk, hipStream_t stream) {
    const int64_t thr = mn / 4;  // mn % 4 == 0 (N % 64 == 0)
    const int tpb = 256;
    gemm_skinny_reduce_kernel<<<dim3((thr + tpb - 1) / tpb), dim3(tpb), 0,
                                stream>>>((bf16*)y, yw, mn, nsk);
    HIP_CHECK_LAST();
    return hipSuccess;
}

extern "C" hipError_t launch_gemm_skinny(
    void* y, float* workspace, const void* x, const void* w, int M, int N,
    int K, int nsk, int swz, hipStream_t stream) {
    if (M <= 0 || M > GS_MAX_M) return hipErrorInvalidValue;
    if ((N % GS_NT) != 0 || (K % 32) != 0) return hipErrorInvalidValue;
    if (nsk < 1) return hipErrorInvalidValue;
    if (nsk > 1 && workspace == nullptr) return hipErrorInvalidValue;
    const int tiles = N / GS_NT;
    // smallest MF whose wave count fits 8 waves: deeper pipelines only fit
    // registers at MF<=2, so prefer more waves over more rows per wave
    int mf = 1;
    while (ceil_div_i(M, 16 * mf) > 8) mf *= 2;
    const int nwaves = ceil_div_i(M, 16 * mf);
    dim3 grid(tiles, nsk);
    dim3 block(nwaves * WAVE_SIZE);
#define GS_LAUNCH1(MFV, SPLIT, SWZV)                                           \
    gemm_skinny_kernel<MFV, SPLIT, SWZV><<<grid, block, 0, stream>>>(          \
        (bf16*)y, workspace, (const bf16*)x, (const bf16*)w, M, N, K, nsk)
#define GS_LAUNCH(MFV, SPLIT)                                                  \
    do {                                                                       \
        if (swz) GS_LAUNCH1(MFV, SPLIT, true);                                 \
        else GS_LAUNCH1(MFV, SPLIT, false);                                    \
    } while (0)
    if (nsk == 1) {
        switch (mf) {
            case 1: GS_LAUNCH(1, false); break;
            case 2: GS_LAUNCH(2, false); break;
            case 3: GS_LAUNCH(3, false); break;
            default: GS_LAUNCH(4, false); break;
        }
    } else {
        switch (mf) {
            case 1: GS_LAUNCH(1, true); break;
            case 2: GS_LAUNCH(2, true); break;
            case 3: GS_LAUNCH(3, true); break;
            default: GS_LAUNCH(4, true); break;
        }
        const int64_t mn = (int64_t)M * N;  // M*N % 4 == 0 (N % 64 == 0)
        const int64_t thr = mn / 4;
        const int tpb = 256;
        gemm_skinny_reduce_kernel<<<dim3((thr + tpb - 1) / tpb), dim3(tpb), 0,
                                    stream>>>((bf16*)y, workspace, mn, nsk);
    }
#undef GS_LAUNCH
#undef GS_LAUNCH1
    HIP_CHECK_LAST();
    return hipSuccess;
}
