// Macro-tile decode-projection GEMM for gfx950: Y[M,N] = X[M,K] · W[N,K]^T,
// bf16, M <= 256 (the continuous-batching decode regime).
//
// Round-1's single-pass streaming kernel (gemm_skinny.hip) lost to the
// tuned library at M >= 64 because its NT=64 stripes re-read X N/64 times
// through fragment-shaped (16-row x 64 B) loads — the per-CU load path,
// not HBM, was the wall (profiles/r01_gemm_skinny_probe.md). This kernel
// is the macro-tile restructure:
//
// - One workgroup owns ALL M rows of a BN-wide column stripe for its
//   K-slice: W is streamed exactly once and X travels the load path once
//   per block in full 128-B lines.
// - BOTH operands are staged into LDS with `global_load_lds` (async DMA,
//   zero staging VGPRs). X tiles land lane-linear with the bank-conflict
//   XOR applied to the per-lane SOURCE address (the swizzle and the
//   ds_read XOR are the same involution); W is pre-swizzled host-side
//   into fragment-major [K/32][N/16][64 lanes][8] order, so its staging
//   is a straight contiguous copy and its ds_read_b128 is conflict-free
//   with no transpose anywhere.
// - 3-deep LDS buffer ring with counted `s_waitcnt vmcnt(N)` and raw
//   s_barriers: up to 2 tiles of DMA stay in flight across each barrier
//   (a plain __syncthreads() would drain the DMA queue to vmcnt(0)).
// - split-K over grid.y writes private fp32 slabs (no atomics; every
//   block fully writes its stripe, empty slices write zeros) reduced by
//   gemm_skinny.hip's reduce kernel; both launches hipGraph-capturable.
//
// Per-wave geometry: wave w owns output rows [32w, 32w+32) — MF=2
// m-fragments x NF n-fragments of v_mfma_f32_16x16x32_bf16 per 32-deep
// k-step, BK=64 (2 k-steps per staged tile).
//
// Replaces the library GEMM on the decode hot path for the projection
// shapes (models/llama.py builds the fragment-major twins); prefill and
// lm_head remain on the TunableOp-tuned library.

#include "common.h"

#define GM_BK 64  // k-depth of one staged tile (2 MFMA k-steps)

typedef __attribute__((__vector_size__(8 * sizeof(short)))) short bf16x8;
typedef __attribute__((__vector_size__(4 * sizeof(float)))) float f32x4;

__device__ __forceinline__ f32x4 gm_mfma(bf16x8 a, bf16x8 b, f32x4 c) {
    return __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
}

// async 16-B-per-lane global->LDS copy. The LDS destination is
// wave-uniform base + lane*16 (hardware rule); the global source is
// per-lane. `ldsoff` must be wave-uniform — computed from wave id, which
// is uniform in fact but not provably so to the compiler (T20), hence the
// readfirstlane.
__device__ __forceinline__ void gm_glds16(const void* g, char* smem, int ldsoff) {
    ldsoff = __builtin_amdgcn_readfirstlane(ldsoff);
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) void*)g,
        (__attribute__((address_space(3))) void*)(smem + ldsoff), 16, 0, 0);
}

template <int N>
__device__ __forceinline__ void gm_vmwait() {
    asm volatile("s_waitcnt vmcnt(%0)" ::"n"(N) : "memory");
}

// compiler-only memory fence: keeps LDS reads/DMA issues from being
// scheduled across a raw s_barrier (the builtin is convergent but not a
// compiler memory barrier)
__device__ __forceinline__ void gm_cfence() { asm volatile("" ::: "memory"); }

__device__ __forceinline__ float gm_silu(float g) {
    return g / (1.f + __expf(-g));
}

// counted wait for "ahead" tiles of GPW glds each still in flight
// (compile-time unrolled: vmcnt immediates must be constants)
template <int GPW, int NBUF, int A = 0>
__device__ __forceinline__ void gm_wait_ahead(int ahead) {
    if constexpr (A < NBUF) {
        if (ahead == A) gm_vmwait<A * GPW>();
        else gm_wait_ahead<GPW, NBUF, A + 1>(ahead);
    }
}

// MW = waves per block (BM = 32*MW rows), NF = 16-col n-fragments (BN =
// 16*NF), BK = staged k-depth per tile (32 or 64), NBUF = LDS ring depth
// (NBUF-1 tiles of DMA in flight across the barriers — the knob that
// covers HBM latency on the W stream).
template <int MW, int NF, int BK, int NBUF, bool SPLITK, bool SWIGLU = false>
__launch_bounds__(MW * WAVE_SIZE)
__global__ void gemm_m256_kernel(
    bf16* __restrict__ y,        // [M, N] (!SPLITK; [M, N/2] when SWIGLU)
    float* __restrict__ yw,      // [nsk, M, N] fp32 slabs (SPLITK)
    const bf16* __restrict__ x,  // [M, K]
    const bf16* __restrict__ w,  // fragment-major [K/32][N/16][64][8]
    int M, int N, int K, int nsk) {
    static_assert(!(SWIGLU && SPLITK), "fused swiglu epilogue needs nsk=1");
    static_assert(!SWIGLU || NF % 2 == 0, "fused swiglu pairs fragments");
    constexpr int BM = MW * 32;
    constexpr int BN = NF * 16;
    constexpr int KS = BK / 32;          // MFMA k-steps per tile
    constexpr int CPR = BK / 8;          // 16-B chunks per X row
    constexpr int XB = BM * BK * 2;      // X tile bytes (row stride BK*2)
    constexpr int WB = BK * BN * 2;      // W tile bytes (KS*NF 1-KiB frags)
    constexpr int BUFB = XB + WB;
    constexpr int XG_W = BK / 16;        // X glds per wave
    constexpr int WU = KS * NF;          // W 1-KiB units per tile
    // W units round-robin over waves; when MW does not divide WU the low
    // waves carry one extra unit and the counted vmcnt waits use each
    // wave's own glds count (the wait is a per-wave counter).
    constexpr int WG_HI = (WU + MW - 1) / MW;  // units on waves < WU%MW (or all)
    constexpr int WG_LO = WU / MW;
    constexpr int GPW_HI = XG_W + WG_HI;
    constexpr int GPW_LO = XG_W + WG_LO;
    static_assert(NBUF * BUFB <= 160 * 1024, "LDS ring exceeds 160 KiB");

    __shared__ __attribute__((aligned(16))) char smem[NBUF * BUFB];

    const int lane = threadIdx.x & (WAVE_SIZE - 1);
    const int wave = threadIdx.x >> 6;
    const int n0 = blockIdx.x * BN;
    const int n16 = N / 16;

    const int ktiles = K / BK;
    const int kt_per = SPLITK ? (ktiles + nsk - 1) / nsk : ktiles;
    const int kt0 = SPLITK ? blockIdx.y * kt_per : 0;
    const int ntiles = min(ktiles - kt0, kt_per) > 0 ? min(ktiles - kt0, kt_per) : 0;

    // ---- staging addresses (loop-invariant parts) ----
    // X: glds i covers LDS rows [32*wave + i*(512/BK), +512/BK); lane l ->
    // row base + l/CPR, 16-B chunk l%CPR, source chunk (l%CPR) ^ (row &
    // (CPR-1)) (the read XOR's inverse — same involution; row&(CPR-1) is
    // invariant under the unit stride). Rows are BK*2-aligned in global
    // (K % 64 == 0), so the permutation stays within one/two cache lines.
    // Per-glds row clamp keeps tail blocks (M < BM) in bounds; clamped
    // rows are skipped by the epilogue.
    const int xrow = wave * 32 + (lane / CPR);  // + (512/BK)*i per glds
    const int xchunk = (lane % CPR) ^ (xrow & (CPR - 1));
    const bf16* xsrc[XG_W];
#pragma unroll
    for (int i = 0; i < XG_W; ++i) {
        const int r = xrow + i * (512 / BK);
        xsrc[i] = x + (size_t)(r < M ? r : 0) * K + xchunk * 8;
    }
    const int xdst0 = wave * 32 * (BK * 2);  // wave-uniform LDS base

    // W: fragment fi = wave + j*MW (j < WG_HI, fi < WU), global fragment
    // index (KS*kt + fi/NF)*n16 + n0/16 + fi%NF; source is 1 KiB
    // contiguous.
    int wfi[WG_HI];
#pragma unroll
    for (int j = 0; j < WG_HI; ++j) wfi[j] = wave + j * MW;
    const bool w_extra = wave < (WU % MW == 0 ? MW : WU % MW);

    f32x4 acc[2][NF];
#pragma unroll
    for (int f = 0; f < 2; ++f)
#pragma unroll
        for (int n = 0; n < NF; ++n) acc[f][n] = (f32x4){0.f, 0.f, 0.f, 0.f};

#define GM_STAGE(T)                                                            \
    do {                                                                       \
        const int kt__ = kt0 + (T);                                            \
        char* buf__ = smem + ((T) % NBUF) * BUFB;                              \
        _Pragma("unroll") for (int i = 0; i < XG_W; ++i)                       \
            gm_glds16(xsrc[i] + (size_t)(kt__)*BK, buf__,                      \
                      xdst0 + i * 1024);                                       \
        _Pragma("unroll") for (int j = 0; j < WG_HI; ++j) {                    \
            if (wfi[j] < WU) {                                                 \
                const size_t gfi__ = (size_t)(KS * kt__ + wfi[j] / NF) * n16 + \
                                     n0 / 16 + wfi[j] % NF;                    \
                gm_glds16(w + gfi__ * 512 + lane * 8, buf__,                   \
                          XB + wfi[j] * 1024);                                 \
            }                                                                  \
        }                                                                      \
    } while (0)

    // ds_read addresses (byte offsets within a buffer)
    const int arow0 = wave * 32 + (lane & 15);          // f=0 row, f=1 adds 16
    const int alk = lane >> 4;                          // k-chunk 0..3

    if (ntiles > 0) {
#pragma unroll
        for (int p = 0; p < NBUF - 1; ++p)
            if (p < ntiles) GM_STAGE(p);
        for (int t = 0; t < ntiles; ++t) {
            if (t + NBUF - 1 < ntiles) GM_STAGE(t + NBUF - 1);
            // wait for tile t's DMA: allow the glds of the tiles beyond t
            // to stay in flight across the barrier (per-wave counts)
            const int ahead = min(ntiles - 1 - t, NBUF - 1);
            if (w_extra) gm_wait_ahead<GPW_HI, NBUF>(ahead);
            else gm_wait_ahead<GPW_LO, NBUF>(ahead);
            __builtin_amdgcn_s_barrier();

            const char* buf = smem + (t % NBUF) * BUFB;
#pragma unroll
            for (int ks = 0; ks < KS; ++ks) {
                bf16x8 a[2], b[NF];
#pragma unroll
                for (int f = 0; f < 2; ++f) {
                    const int row = arow0 + f * 16;
                    const int chunk = (ks * 4 + alk) ^ (row & (CPR - 1));
                    a[f] = *(const __attribute__((address_space(3))) bf16x8*)(
                        (const __attribute__((address_space(3))) char*)buf +
                        row * (BK * 2) + chunk * 16);
                }
#pragma unroll
                for (int n = 0; n < NF; ++n)
                    b[n] = *(const __attribute__((address_space(3))) bf16x8*)(
                        (const __attribute__((address_space(3))) char*)buf +
                        XB + (ks * NF + n) * 1024 + lane * 16);
#pragma unroll
                for (int n = 0; n < NF; ++n)
#pragma unroll
                    for (int f = 0; f < 2; ++f)
                        acc[f][n] = gm_mfma(a[f], b[n], acc[f][n]);
            }
            // all waves done reading buf t%NBUF before iter t+1 refills it
            gm_cfence();
            __builtin_amdgcn_s_barrier();
            gm_cfence();
        }
    }
#undef GM_STAGE

    // ---- epilogue: C fragment row = 4*(lane>>4) + r, col = lane&15 ----
    const int m_base = wave * 32;
    if (SPLITK) {
        float* slab = yw + (size_t)blockIdx.y * M * N;
#pragma unroll
        for (int f = 0; f < 2; ++f)
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                const int row = m_base + f * 16 + (lane >> 4) * 4 + r;
                if (row >= M) continue;
#pragma unroll
                for (int n = 0; n < NF; ++n)
                    slab[(size_t)row * N + n0 + n * 16 + (lane & 15)] =
                        acc[f][n][r];
            }
    } else if constexpr (SWIGLU) {
        // block-16 interleaved gate/up twin (ops.interleave_gate_up):
        // even fragment n holds the GATE columns, fragment n+1 the UP
        // columns of the SAME 16 output columns — the pair lands in the
        // same lane, so silu(g)*u needs no cross-lane traffic and the
        // [M, N] intermediate + separate swiglu kernel disappear.
#pragma unroll
        for (int f = 0; f < 2; ++f)
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                const int row = m_base + f * 16 + (lane >> 4) * 4 + r;
                if (row >= M) continue;
#pragma unroll
                for (int n = 0; n < NF; n += 2)
                    y[(size_t)row * (N >> 1) + (n0 >> 1) + (n >> 1) * 16 +
                      (lane & 15)] =
                        f2bf(gm_silu(acc[f][n][r]) * acc[f][n + 1][r]);
            }
    } else {
#pragma unroll
        for (int f = 0; f < 2; ++f)
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                const int row = m_base + f * 16 + (lane >> 4) * 4 + r;
                if (row >= M) continue;
#pragma unroll
                for (int n = 0; n < NF; ++n)
                    y[(size_t)row * N + n0 + n * 16 + (lane & 15)] =
                        f2bf(acc[f][n][r]);
            }
    }
}

// ---------------------------------------------------------------------------
// Register-staged variant (T14: global->reg loads issued one tile ahead,
// ds_write AFTER the barrier, plain k-loop, 2 LDS buffers).
//
// Why it exists: the all-glds variant above is capped by the LDS-DMA
// landing cadence (~25-40 GB/s per CU; MI355X_MICROARCH.md "ldsdma-fill"),
// which binds exactly when one block must stream a lot of bytes — the
// M=256 big-N shapes measured ~33 GB/s/CU (profiles/r02_gemm_m256_probe).
// Ordinary vector loads ride the full per-CU load path (~122 GB/s), so
// staging through registers + ds_write lifts the ceiling ~3x there. The
// ds_write applies the same X bank-conflict XOR at write time (no source
// permutation needed); W's fragment-major twin writes lane-linear.
// Both ds_write_b128 patterns are conflict-free in the 8-lane
// contiguous store groups.
// ---------------------------------------------------------------------------

// NTW: stream W through non-temporal loads — the weight stream has zero
// reuse, so keeping it out of L2 leaves the cache to X and the consumers.
template <int MW, int NF, bool SPLITK, bool SWIGLU = false, bool NTW = false>
__launch_bounds__(MW * WAVE_SIZE)
__global__ void gemm_m256r_kernel(
    bf16* __restrict__ y, float* __restrict__ yw,
    const bf16* __restrict__ x,  // [M, K]
    const bf16* __restrict__ w,  // fragment-major [K/32][N/16][64][8]
    int M, int N, int K, int nsk) {
    static_assert(!(SWIGLU && SPLITK), "fused swiglu epilogue needs nsk=1");
    static_assert(!SWIGLU || NF % 2 == 0, "fused swiglu pairs fragments");
    constexpr int BM = MW * 32;
    constexpr int BN = NF * 16;
    constexpr int XB = BM * GM_BK * 2;
    constexpr int WB = GM_BK * BN * 2;
    constexpr int BUFB = XB + WB;
    constexpr int XG_W = 4;                       // 1-KiB x units per wave
    constexpr int WG_W = (2 * NF + MW - 1) / MW;  // 1-KiB w units per wave
    static_assert((2 * NF) % MW == 0, "W frag split uneven across waves");

    __shared__ __attribute__((aligned(16))) char smem[2 * BUFB];

    const int lane = threadIdx.x & (WAVE_SIZE - 1);
    const int wave = threadIdx.x >> 6;
    const int n0 = blockIdx.x * BN;
    const int n16 = N / 16;

    const int ktiles = K / GM_BK;
    const int kt_per = SPLITK ? (ktiles + nsk - 1) / nsk : ktiles;
    const int kt0 = SPLITK ? blockIdx.y * kt_per : 0;
    const int ntiles = min(ktiles - kt0, kt_per) > 0 ? min(ktiles - kt0, kt_per) : 0;

    // X staging: unit i = rows [32w+8i, +8); lane -> row base+l/8, chunk
    // l%8. Loads are linear (full 128-B lines); the XOR goes on the
    // ds_write address. Per-unit row clamp for M < BM tails.
    const int xrow = wave * 32 + (lane >> 3);
    const bf16* xsrc[XG_W];
#pragma unroll
    for (int i = 0; i < XG_W; ++i) {
        const int r = xrow + i * 8;
        xsrc[i] = x + (size_t)(r < M ? r : 0) * K + (lane & 7) * 8;
    }
    // ds_write byte offset (swizzled) per unit: row*128 + (c ^ row&7)*16
    const int xwoff = (xrow & (BM - 1)) * 128 + (((lane & 7) ^ (xrow & 7)) << 4);

    int wfi[WG_W];
#pragma unroll
    for (int j = 0; j < WG_W; ++j) wfi[j] = wave + j * MW;

    f32x4 acc[2][NF];
#pragma unroll
    for (int f = 0; f < 2; ++f)
#pragma unroll
        for (int n = 0; n < NF; ++n) acc[f][n] = (f32x4){0.f, 0.f, 0.f, 0.f};

    bf16x8 xr[XG_W], wr[WG_W];  // one in-flight staging tile

#define GMR_LOAD(T)                                                            \
    do {                                                                       \
        const int kt__ = kt0 + (T);                                            \
        _Pragma("unroll") for (int i = 0; i < XG_W; ++i) xr[i] =               \
            *(const __attribute__((address_space(1))) bf16x8*)(                \
                xsrc[i] + (size_t)(kt__)*GM_BK);                               \
        _Pragma("unroll") for (int j = 0; j < WG_W; ++j) {                     \
            const size_t gfi__ = (size_t)(2 * kt__ + wfi[j] / NF) * n16 +      \
                                 n0 / 16 + wfi[j] % NF;                        \
            const __attribute__((address_space(1))) bf16x8* wp__ =             \
                (const __attribute__((address_space(1))) bf16x8*)(             \
                    w + gfi__ * 512 + lane * 8);                               \
            wr[j] = NTW ? __builtin_nontemporal_load(wp__) : *wp__;            \
        }                                                                      \
    } while (0)
#define GMR_WRITE(T)                                                           \
    do {                                                                       \
        char* buf__ = smem + ((T)&1) * BUFB;                                   \
        _Pragma("unroll") for (int i = 0; i < XG_W; ++i)                       \
            *(__attribute__((address_space(3))) bf16x8*)(                      \
                (__attribute__((address_space(3))) char*)buf__ + xwoff +       \
                i * 8 * 128) = xr[i];                                          \
        _Pragma("unroll") for (int j = 0; j < WG_W; ++j)                       \
            *(__attribute__((address_space(3))) bf16x8*)(                      \
                (__attribute__((address_space(3))) char*)buf__ + XB +          \
                wfi[j] * 1024 + lane * 16) = wr[j];                            \
    } while (0)

    const int arow0 = wave * 32 + (lane & 15);
    const int alk = lane >> 4;

    if (ntiles > 0) {
        GMR_LOAD(0);
        GMR_WRITE(0);
        if (ntiles > 1) GMR_LOAD(1);  // in flight during compute(0)
        gm_cfence();
        __builtin_amdgcn_s_barrier();
        gm_cfence();
        for (int t = 0; t < ntiles; ++t) {
            if (t + 1 < ntiles) {
                // regs hold tile t+1 (issued last iter): write after the
                // barrier, then immediately re-issue for tile t+2 so the
                // loads overlap compute(t)
                GMR_WRITE(t + 1);
                if (t + 2 < ntiles) GMR_LOAD(t + 2);
            }
            const char* buf = smem + (t & 1) * BUFB;
#pragma unroll
            for (int ks = 0; ks < 2; ++ks) {
                bf16x8 a[2], b[NF];
#pragma unroll
                for (int f = 0; f < 2; ++f) {
                    const int row = arow0 + f * 16;
                    const int chunk = (ks * 4 + alk) ^ (row & 7);
                    a[f] = *(const __attribute__((address_space(3))) bf16x8*)(
                        (const __attribute__((address_space(3))) char*)buf +
                        (row & (BM - 1)) * 128 + chunk * 16);
                }
#pragma unroll
                for (int n = 0; n < NF; ++n)
                    b[n] = *(const __attribute__((address_space(3))) bf16x8*)(
                        (const __attribute__((address_space(3))) char*)buf +
                        XB + (ks * NF + n) * 1024 + lane * 16);
#pragma unroll
                for (int n = 0; n < NF; ++n)
#pragma unroll
                    for (int f = 0; f < 2; ++f)
                        acc[f][n] = gm_mfma(a[f], b[n], acc[f][n]);
            }
            gm_cfence();
            __builtin_amdgcn_s_barrier();
            gm_cfence();
        }
    }
#undef GMR_LOAD
#undef GMR_WRITE

    const int m_base = wave * 32;
    if (SPLITK) {
        float* slab = yw + (size_t)blockIdx.y * M * N;
#pragma unroll
        for (int f = 0; f < 2; ++f)
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                const int row = m_base + f * 16 + (lane >> 4) * 4 + r;
                if (row >= M) continue;
#pragma unroll
                for (int n = 0; n < NF; ++n)
                    slab[(size_t)row * N + n0 + n * 16 + (lane & 15)] =
                        acc[f][n][r];
            }
    } else if constexpr (SWIGLU) {
        // see gemm_m256_kernel's SWIGLU epilogue (block-16 interleaved twin)
#pragma unroll
        for (int f = 0; f < 2; ++f)
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                const int row = m_base + f * 16 + (lane >> 4) * 4 + r;
                if (row >= M) continue;
#pragma unroll
                for (int n = 0; n < NF; n += 2)
                    y[(size_t)row * (N >> 1) + (n0 >> 1) + (n >> 1) * 16 +
                      (lane & 15)] =
                        f2bf(gm_silu(acc[f][n][r]) * acc[f][n + 1][r]);
            }
    } else {
#pragma unroll
        for (int f = 0; f < 2; ++f)
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                const int row = m_base + f * 16 + (lane >> 4) * 4 + r;
                if (row >= M) continue;
#pragma unroll
                for (int n = 0; n < NF; ++n)
                    y[(size_t)row * N + n0 + n * 16 + (lane & 15)] =
                        f2bf(acc[f][n][r]);
            }
    }
}

// ---------------------------------------------------------------------------
// Producer/consumer variant (v2): wave-specialized for the W-stream-heavy
// shapes where both ring variants above hit the lockstep-straggler wall
// (profiles/r02_gemm_m256_sweep.md).
//
// Why specialization: the hardware has ONE in-order vmcnt counter per
// wave, so any single wave that both stages X (short-latency, barrier-
// synced) and prefetches W (HBM-latency) has its W pipeline drained by
// every X wait. Splitting the roles gives each wave a private counter:
//
// - waves 6,7 (loaders): stage the shared X tile ([256][64] bf16, XOR-
//   swizzled) into a 3-slot LDS ring via ordinary loads + ds_write_b128
//   (T14: issue one tile ahead, write after the barrier). Their waits
//   touch only X loads.
// - waves 0-5 (consumers): each owns a 16-column strip (block BN = 96).
//   W fragments stream straight to VGPRs from the fragment-major twin in
//   an 8-deep register ring (compile-time indices via 8x-unrolled k-loop)
//   — ~8 KiB per wave in HBM flight, never drained: plain register loads
//   survive __syncthreads (no LDS-DMA in this kernel), and each wave's
//   counted waits see only its own stream.
// - one __syncthreads per 64-deep k-tile; per k-step a consumer does
//   16 ds_read_b128 (conflict-free, XOR layout) + 16 MFMA into a 16x1
//   fragment accumulator (64 VGPRs).
// ---------------------------------------------------------------------------

#define PC_WAVES 8
#define PC_CONS 6               // consumer waves -> BN = 96 columns
#define PC_BN (PC_CONS * 16)
#define PC_BK 64                // k-depth per staged X tile (2 k-steps)
#define PC_NSLOT 3              // X LDS ring slots
#define PC_WD 8                 // W register-ring depth in 32-deep k-steps

// ABL: 0 = normal, 1 = loader-only (consumers only hit the barriers),
// 2 = consumer-only (loaders only hit the barriers) — the ablation pair
// that isolates which role is the critical path (guide §5 mistake 8).
template <bool SPLITK, int ABL = 0>
__launch_bounds__(PC_WAVES * WAVE_SIZE)
__global__ void gemm_m256pc_kernel(
    bf16* __restrict__ y,        // [M, N] (!SPLITK)
    float* __restrict__ yw,      // [nsk, M, N] fp32 slabs (SPLITK)
    const bf16* __restrict__ x,  // [M, K]
    const bf16* __restrict__ w,  // fragment-major [K/32][N/16][64][8]
    int M, int N, int K, int nsk) {
    constexpr int XB = 256 * PC_BK * 2;  // one X slot: 32 KiB
    __shared__ __attribute__((aligned(16))) char smem[PC_NSLOT * XB];

    const int lane = threadIdx.x & (WAVE_SIZE - 1);
    const int wave = threadIdx.x >> 6;
    const int n0 = blockIdx.x * PC_BN;
    const int n16 = N / 16;

    const int ktiles = K / PC_BK;
    const int kt_per = SPLITK ? (ktiles + nsk - 1) / nsk : ktiles;
    const int kt0 = SPLITK ? blockIdx.y * kt_per : 0;
    const int ntiles = min(ktiles - kt0, kt_per) > 0 ? min(ktiles - kt0, kt_per) : 0;
    const int nk = ntiles * 2;  // 32-deep k-steps in this slice

    if (wave >= PC_CONS) {
        // ---- loader: rows [128*(wave-6), +128), 16 KiB per tile ----
        // unit i covers 8 rows x 64 cols; lane -> row base + l/8, chunk
        // l%8; the XOR swizzle is applied at the ds_write address.
        const int lw = wave - PC_CONS;  // 0 or 1
        const int xrow = lw * 128 + (lane >> 3);  // + 8*i per unit
        const int xchunk = lane & 7;
        const bf16* xsrc[16];
#pragma unroll
        for (int i = 0; i < 16; ++i) {
            const int r = xrow + i * 8;
            xsrc[i] = x + (size_t)(r < M ? r : 0) * K + xchunk * 8;
        }
        const int xwoff = (xrow)*128 + ((xchunk ^ (xrow & 7)) << 4);
        // two register sets -> two 16-KiB tiles of X in HBM/L2 flight per
        // loader (single-set T14 measured loader-bound at 71 us/block)
        bf16x8 xa[16], xb[16];
#define PC_LOAD(SET, T)                                                        \
    do {                                                                       \
        const size_t kof__ = (size_t)(kt0 + (T)) * PC_BK;                      \
        _Pragma("unroll") for (int i = 0; i < 16; ++i) SET[i] =                \
            *(const __attribute__((address_space(1))) bf16x8*)(xsrc[i] +       \
                                                               kof__);         \
    } while (0)
#define PC_WRITE(SET, T)                                                       \
    do {                                                                       \
        char* buf__ = smem + ((T) % PC_NSLOT) * XB;                            \
        _Pragma("unroll") for (int i = 0; i < 16; ++i)                         \
            *(__attribute__((address_space(3))) bf16x8*)(                      \
                (__attribute__((address_space(3))) char*)buf__ + xwoff +       \
                i * 8 * 128) = SET[i];                                         \
    } while (0)
        if (ntiles > 0) {
            if (ABL != 2) {
                PC_LOAD(xa, 0);
                PC_WRITE(xa, 0);
                if (ntiles > 1) PC_LOAD(xa, 1);
                if (ntiles > 2) PC_LOAD(xb, 2);
            }
            __syncthreads();
            // alternating sets keep indices static: iter t writes the
            // tile loaded two iterations ago and reloads that set for t+3
            for (int t = 0; t < ntiles; ++t) {
                if (ABL != 2 && t + 1 < ntiles) {
                    if (t & 1) {
                        PC_WRITE(xb, t + 1);
                        if (t + 3 < ntiles) PC_LOAD(xb, t + 3);
                    } else {
                        PC_WRITE(xa, t + 1);
                        if (t + 3 < ntiles) PC_LOAD(xa, t + 3);
                    }
                }
                __syncthreads();
            }
        }
#undef PC_LOAD
#undef PC_WRITE
        return;  // loaders take no part in the epilogue
    }

    // ---- consumer wave: columns [n0 + wave*16, +16) ----
    // W fragment stream: one 1-KiB lane-linear fragment per 32-deep
    // k-step, clamped to the last valid fragment for tail/padding reads
    // (results discarded).
    const int nfrag = min(n0 / 16 + wave, n16 - 1);
    const bf16* wbase = w + (size_t)lane * 8;
    auto wsrc = [&](int ks) {  // ks = k-step index within this slice
        const int g = kt0 * 2 + min(ks, nk > 0 ? nk - 1 : 0);
        return (const __attribute__((address_space(1))) bf16x8*)(
            wbase + ((size_t)g * n16 + nfrag) * 512);
    };

    f32x4 acc[16];
#pragma unroll
    for (int m = 0; m < 16; ++m) acc[m] = (f32x4){0.f, 0.f, 0.f, 0.f};

    const int arow_l = lane & 15;  // A-frag row within an m-tile
    const int alk = lane >> 4;

    if (ntiles > 0 && ABL == 1) {
        __syncthreads();
        for (int t = 0; t < ntiles; ++t) __syncthreads();
    } else if (ntiles > 0) {
        bf16x8 wring[PC_WD];
#pragma unroll
        for (int d = 0; d < PC_WD; ++d) wring[d] = *wsrc(d);
        __syncthreads();  // X slot 0 staged

        // consume k-steps in pairs (one X tile); ring indices static via
        // the %PC_WD structure of the 4-tile unrolled main loop
        int t = 0;
        for (; t + 4 <= ntiles && 2 * (t + 4) <= nk; t += 4) {
#pragma unroll
            for (int tt = 0; tt < 4; ++tt) {
                const char* buf = smem + ((t + tt) % PC_NSLOT) * XB;
#pragma unroll
                for (int ks = 0; ks < 2; ++ks) {
                    const int kabs = 2 * (t + tt) + ks;
                    const int ridx = kabs & (PC_WD - 1);
                    const bf16x8 bfrag = wring[ridx];
#pragma unroll
                    for (int m = 0; m < 16; ++m) {
                        const int row = m * 16 + arow_l;
                        const int chunk = (ks * 4 + alk) ^ (row & 7);
                        const bf16x8 afrag =
                            *(const __attribute__((address_space(3))) bf16x8*)(
                                (const __attribute__((address_space(3))) char*)
                                    buf +
                                row * 128 + chunk * 16);
                        acc[m] = gm_mfma(afrag, bfrag, acc[m]);
                    }
                    wring[ridx] = *wsrc(kabs + PC_WD);  // refill the slot
                }
                __syncthreads();
            }
        }
        // tail tiles (ring indices still static: kabs & 7)
        for (; t < ntiles; ++t) {
            const char* buf = smem + (t % PC_NSLOT) * XB;
#pragma unroll
            for (int ks = 0; ks < 2; ++ks) {
                const int kabs = 2 * t + ks;
                bf16x8 bfrag;
#pragma unroll
                for (int d = 0; d < PC_WD; ++d)
                    if (d == (kabs & (PC_WD - 1))) bfrag = wring[d];
#pragma unroll
                for (int m = 0; m < 16; ++m) {
                    const int row = m * 16 + arow_l;
                    const int chunk = (ks * 4 + alk) ^ (row & 7);
                    const bf16x8 afrag =
                        *(const __attribute__((address_space(3))) bf16x8*)(
                            (const __attribute__((address_space(3))) char*)buf +
                            row * 128 + chunk * 16);
                    acc[m] = gm_mfma(afrag, bfrag, acc[m]);
                }
            }
            __syncthreads();
        }
    }

    // ---- epilogue ----
    const int col = n0 + wave * 16 + (lane & 15);
    if (col >= N) return;
    if (SPLITK) {
        float* slab = yw + (size_t)blockIdx.y * M * N;
#pragma unroll
        for (int m = 0; m < 16; ++m)
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                const int row = m * 16 + (lane >> 4) * 4 + r;
                if (row < M) slab[(size_t)row * N + col] = acc[m][r];
            }
    } else {
#pragma unroll
        for (int m = 0; m < 16; ++m)
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                const int row = m * 16 + (lane >> 4) * 4 + r;
                if (row < M) y[(size_t)row * N + col] = f2bf(acc[m][r]);
            }
    }
}

extern "C" hipError_t launch_gemm_reduce(void*, const float*, int64_t, int,
                                         hipStream_t);  // gemm_skinny.hip

// nf: 4 (BN=64) or 8 (BN=128). M <= 256; N % (16*nf) == 0; K % 64 == 0.
// variant: 0 = glds-staged (DMA pipeline, depth per `pipe`),
//          1 = register-staged T14 (2 buffers, loads one tile ahead),
//          2 = producer/consumer wave-specialized (BN=96; nf ignored).
// pipe (variant 0 only): 0=(BK64,NBUF3) 1=(BK64,NBUF4,nf4)
//          2=(BK32,NBUF4,nf8) 3=(BK32,NBUF6,nf8) — deeper rings keep more
//          HBM latency covered on the W stream; 4=(BK64,NBUF2,nf4) and
//          5=(BK32,NBUF3,nf4) fit 2 blocks/CU so a second block fills the
//          barrier-lockstep wait gaps.
extern "C" hipError_t launch_gemm_m256(
    void* y, float* workspace, const void* x, const void* w, int M, int N,
    int K, int nsk, int nf, int variant, int pipe, int swiglu,
    hipStream_t stream) {
    if (M <= 0 || M > 256) return hipErrorInvalidValue;
    if (nsk < 1 || (nsk > 1 && workspace == nullptr)) return hipErrorInvalidValue;
    if ((K % GM_BK) != 0) return hipErrorInvalidValue;
    // fused swiglu epilogue: y is [M, N/2], single-pass only (the fp32
    // split-K slabs hold pre-activation partials — no fusion point)
    if (swiglu && (nsk != 1 || variant == 2)) return hipErrorInvalidValue;
    if (variant == 2) {
        if ((N % 16) != 0) return hipErrorInvalidValue;
        dim3 grid((N + PC_BN - 1) / PC_BN, nsk);
        dim3 block(PC_WAVES * WAVE_SIZE);
#define PC_L(SPLIT)                                                                do {                                                                               if (pipe == 1)                                                                     gemm_m256pc_kernel<SPLIT, 1><<<grid, block, 0, stream>>>(                          (bf16*)y, workspace, (const bf16*)x, (const bf16*)w, M, N, K,                  nsk);                                                                  else if (pipe == 2)                                                                gemm_m256pc_kernel<SPLIT, 2><<<grid, block, 0, stream>>>(                          (bf16*)y, workspace, (const bf16*)x, (const bf16*)w, M, N, K,                  nsk);                                                                  else                                                                               gemm_m256pc_kernel<SPLIT, 0><<<grid, block, 0, stream>>>(                          (bf16*)y, workspace, (const bf16*)x, (const bf16*)w, M, N, K,                  nsk);                                                              } while (0)
        if (nsk > 1) PC_L(true);
        else PC_L(false);
#undef PC_L
        HIP_CHECK_LAST();
        if (nsk > 1)
            return launch_gemm_reduce(y, workspace, (int64_t)M * N, nsk, stream);
        return hipSuccess;
    }
    if (nf != 4 && nf != 8) return hipErrorInvalidValue;
    if ((N % (16 * nf)) != 0) return hipErrorInvalidValue;
    if (variant == 0) {
        if ((pipe == 1 || pipe == 4 || pipe == 5) && nf != 4)
            return hipErrorInvalidValue;
        if ((pipe == 2 || pipe == 3) && nf != 8) return hipErrorInvalidValue;
    }
    int mw = 1;
    while (mw * 32 < M) mw *= 2;  // 1,2,4,8
    const int tiles = N / (16 * nf);
    dim3 grid(tiles, nsk);
    dim3 block(mw * WAVE_SIZE);
#define GM_L2(MWV, NFV, SPLIT, SW)                                             \
    do {                                                                       \
        if (variant == 1 && pipe == 1)                                         \
            gemm_m256r_kernel<MWV, NFV, SPLIT, SW, true><<<grid, block, 0,     \
                                                           stream>>>(         \
                (bf16*)y, workspace, (const bf16*)x, (const bf16*)w, M, N, K,  \
                nsk);                                                          \
        else if (variant == 1)                                                 \
            gemm_m256r_kernel<MWV, NFV, SPLIT, SW><<<grid, block, 0,           \
                                                     stream>>>(               \
                (bf16*)y, workspace, (const bf16*)x, (const bf16*)w, M, N, K,  \
                nsk);                                                          \
        else if (pipe == 1 && NFV == 4)                                        \
            gemm_m256_kernel<MWV, 4, 64, 4, SPLIT, SW><<<grid, block, 0,       \
                                                         stream>>>(           \
                (bf16*)y, workspace, (const bf16*)x, (const bf16*)w, M, N, K,  \
                nsk);                                                          \
        else if (pipe == 2 && NFV == 8)                                        \
            gemm_m256_kernel<MWV, 8, 32, 4, SPLIT, SW><<<grid, block, 0,       \
                                                         stream>>>(           \
                (bf16*)y, workspace, (const bf16*)x, (const bf16*)w, M, N, K,  \
                nsk);                                                          \
        else if (pipe == 3 && NFV == 8)                                        \
            gemm_m256_kernel<MWV, 8, 32, 6, SPLIT, SW><<<grid, block, 0,       \
                                                         stream>>>(           \
                (bf16*)y, workspace, (const bf16*)x, (const bf16*)w, M, N, K,  \
                nsk);                                                          \
        else if (pipe == 4 && NFV == 4)                                        \
            gemm_m256_kernel<MWV, 4, 64, 2, SPLIT, SW><<<grid, block, 0,       \
                                                         stream>>>(           \
                (bf16*)y, workspace, (const bf16*)x, (const bf16*)w, M, N, K,  \
                nsk);                                                          \
        else if (pipe == 5 && NFV == 4)                                        \
            gemm_m256_kernel<MWV, 4, 32, 3, SPLIT, SW><<<grid, block, 0,       \
                                                         stream>>>(           \
                (bf16*)y, workspace, (const bf16*)x, (const bf16*)w, M, N, K,  \
                nsk);                                                          \
        else                                                                   \
            gemm_m256_kernel<MWV, NFV, 64, 3, SPLIT, SW><<<grid, block, 0,     \
                                                           stream>>>(         \
                (bf16*)y, workspace, (const bf16*)x, (const bf16*)w, M, N, K,  \
                nsk);                                                          \
    } while (0)
#define GM_L1(MWV, NFV)                                                        \
    do {                                                                       \
        if (swiglu) GM_L2(MWV, NFV, false, true);                              \
        else if (nsk > 1) GM_L2(MWV, NFV, true, false);                        \
        else GM_L2(MWV, NFV, false, false);                                    \
    } while (0)
#define GM_L0(MWV)                                                             \
    do {                                                                       \
        if (nf == 4) GM_L1(MWV, 4);                                            \
        else GM_L1(MWV, 8);                                                    \
    } while (0)
    switch (mw) {
        case 1: GM_L0(1); break;
        case 2: GM_L0(2); break;
        case 4: GM_L0(4); break;
        default: GM_L0(8); break;
    }
#undef GM_L0
#undef GM_L1
#undef GM_L2
    HIP_CHECK_LAST();
    if (nsk > 1) {
        const int64_t mn = (int64_t)M * N;  // N % 64 == 0 -> mn % 4 == 0
        return launch_gemm_reduce(y, workspace, mn, nsk, stream);
    }
    return hipSuccess;
}
