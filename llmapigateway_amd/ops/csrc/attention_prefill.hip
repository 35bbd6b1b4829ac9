// Flash-style causal prefill attention for gfx950 (MFMA bf16, LDS-tiled).
//
// Produces the prompt phase (TTFT) the reference delegated upstream
// (SURVEY.md §2b "prefill attention kernel"). Varlen batch with GQA.
//
// Geometry: one workgroup per (64-row q tile, q head); 4 waves, each wave
// owns 16 q rows. K/V tiles of 64 keys staged in LDS (V transposed at
// staging so the PV B-fragment reads are contiguous 16 B ds_read_b128).
// MFMA: v_mfma_f32_16x16x32_bf16 throughout —
//   A fragment: lane l holds A[row = l&15][k = (l>>4)*8 + j], j=0..7
//   B fragment: lane l holds B[k = (l>>4)*8 + j][col = l&15]
//   C fragment: lane l holds C[row = (l>>4)*4 + r][col = l&15], r=0..3
// (cdna_hip_programming.md §3; transpose-detecting numerics tests in
// tests/test_ops_gpu.py guard the mapping).
//
// Online softmax per 4-row group (flash rescaling with -1e30 sentinel so
// fully-masked rows stay NaN-free).

#include "common.h"

#define PF_D 128
#define QTILE 64         // q rows per workgroup
#define KVTILE 64        // keys per LDS tile
#define PF_WAVES 4
#define NEG_INF (-1e30f)

typedef __attribute__((__vector_size__(8 * sizeof(short)))) short bf16x8;
typedef __attribute__((__vector_size__(4 * sizeof(float)))) float f32x4;

__device__ __forceinline__ f32x4 mfma16(bf16x8 a, bf16x8 b, f32x4 c) {
    return __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
}

// XOR column swizzles (16-B chunk index ^ row) applied identically at
// write and read: the row-major tiles otherwise put every fragment
// read's 16-lane group on one bank slot — k_lds rows are 256 B (one full
// bank row -> up to 16-way conflict, the Guideline-4 D=128 case) and
// vt/p rows 128 B (4-way). Measured before the fix: 1.27M
// SQ_LDS_BANK_CONFLICT cycles per dispatch (profiles/r02_pmc_bench_kernels.csv).
__device__ __forceinline__ int swz16(int chunk, int row) { return chunk ^ (row & 15); }
__device__ __forceinline__ int swz8(int chunk, int row) { return chunk ^ (row & 7); }

// FP8C: the PAGED-CACHE side (phase 0) holds e4m3 bytes with per-row
// scales; fresh K/V stay bf16. Dequant happens at LDS staging, so the
// compute phases are unchanged.
template <bool FP8C>
__launch_bounds__(PF_WAVES* WAVE_SIZE)
__global__ void attention_prefill_kernel(
    bf16* __restrict__ out,             // [T, Hq, D]
    const bf16* __restrict__ q,         // [T, Hq, D] (rows strided)
    const bf16* __restrict__ k,         // [T, Hkv, D]
    const bf16* __restrict__ v,         // [T, Hkv, D]
    const int* __restrict__ cu_seqlens, // [B+1]
    const int* __restrict__ tile_seq,   // [ntiles] seq index of each tile
    const int* __restrict__ tile_off,   // [ntiles] first q row (local) of tile
    float scale,
    int Hq,
    int Hkv,
    int64_t q_stride,
    int64_t k_stride,
    int64_t v_stride,
    // cached-context phase (prefix caching / chunked prefill): per-seq
    // prior KV already resident in the paged cache; every fresh q row
    // attends to ALL of it (no mask — cached positions precede the tile)
    const void* __restrict__ k_cache,        // [NB, Hkv, BS, D] or null
    const void* __restrict__ v_cache,
    const float* __restrict__ k_scale,       // [NB, Hkv, BS] (FP8C)
    const float* __restrict__ v_scale,
    const int* __restrict__ block_tables,    // [B, max_blocks] or null
    const int* __restrict__ cached_lens,     // [B] or null
    int block_size,
    int max_blocks) {
    const int tile = blockIdx.x;
    const int head = blockIdx.y;
    const int kvh = head / (Hq / Hkv);
    const int seq = tile_seq[tile];
    const int tile0 = tile_off[tile];
    const int seq_start = cu_seqlens[seq];
    const int seq_len = cu_seqlens[seq + 1] - seq_start;
    const int cached = (cached_lens != nullptr) ? cached_lens[seq] : 0;

    const int tid = threadIdx.x;
    const int lane = tid & (WAVE_SIZE - 1);
    const int wave = tid >> 6;
    const int lrow = lane & 15;        // A-fragment row / C col
    const int lk = lane >> 4;          // k-group within fragment

    __shared__ bf16 k_lds[KVTILE][PF_D];
    __shared__ bf16 vt_lds[PF_D][KVTILE];
    __shared__ bf16 p_lds[PF_WAVES][16][KVTILE];

    // ---- load this wave's Q fragments (A operand) straight from global ----
    bf16x8 q_frag[4];
    const int my_qrow = tile0 + wave * 16 + lrow;     // local row in seq
    {
        const bool rvalid = my_qrow < seq_len;
        const size_t base = rvalid
            ? ((size_t)(seq_start + my_qrow) * q_stride + (size_t)head * PF_D)
            : ((size_t)seq_start * q_stride + (size_t)head * PF_D);
#pragma unroll
        for (int kc = 0; kc < 4; ++kc) {
            const uint4 raw = *reinterpret_cast<const uint4*>(
                q + base + kc * 32 + lk * 8);
            q_frag[kc] = *reinterpret_cast<const bf16x8*>(&raw);
        }
    }

    // softmax state per lane: 4 rows (r -> row (lane>>4)*4 + r)
    float m_st[4], l_st[4];
    f32x4 accO[8];  // 8 dim-subtiles x 4 rows
#pragma unroll
    for (int r = 0; r < 4; ++r) {
        m_st[r] = NEG_INF;
        l_st[r] = 0.f;
    }
#pragma unroll
    for (int d = 0; d < 8; ++d) accO[d] = (f32x4){0.f, 0.f, 0.f, 0.f};

    // ---- phase 0: cached context from the paged KV cache (unmasked) ----
    const int n_cached_tiles = (cached + KVTILE - 1) / KVTILE;
    const int* bt = (block_tables != nullptr)
                        ? block_tables + (size_t)seq * max_blocks
                        : nullptr;
    for (int jt = 0; jt < n_cached_tiles; ++jt) {
        const int key0 = jt * KVTILE;
        const int keys_here = min(KVTILE, cached - key0);

        __syncthreads();
        for (int i = tid; i < KVTILE * (PF_D / 8); i += PF_WAVES * WAVE_SIZE) {
            const int kk = i / (PF_D / 8);
            const int d0 = (i % (PF_D / 8)) * 8;
            uint4 kraw = {0, 0, 0, 0}, vraw = {0, 0, 0, 0};
            if (kk < keys_here) {
                const int pos = key0 + kk;
                const size_t row =
                    ((size_t)bt[pos / block_size] * Hkv + kvh) * block_size +
                    pos % block_size;
                if (FP8C) {
                    // dequantize 8 e4m3 bytes (+ row scale) to bf16
                    const uint2 kq = *reinterpret_cast<const uint2*>(
                        (const unsigned char*)k_cache + row * PF_D + d0);
                    const uint2 vq = *reinterpret_cast<const uint2*>(
                        (const unsigned char*)v_cache + row * PF_D + d0);
                    const float ksc = k_scale[row], vsc = v_scale[row];
                    float kf[8], vf[8];
                    fp8x4_to_f32(kq.x, kf[0], kf[1], kf[2], kf[3]);
                    fp8x4_to_f32(kq.y, kf[4], kf[5], kf[6], kf[7]);
                    fp8x4_to_f32(vq.x, vf[0], vf[1], vf[2], vf[3]);
                    fp8x4_to_f32(vq.y, vf[4], vf[5], vf[6], vf[7]);
                    uint32_t* kp = reinterpret_cast<uint32_t*>(&kraw);
                    uint32_t* vp = reinterpret_cast<uint32_t*>(&vraw);
#pragma unroll
                    for (int j = 0; j < 4; ++j) {
                        kp[j] = pack2(kf[2 * j] * ksc, kf[2 * j + 1] * ksc);
                        vp[j] = pack2(vf[2 * j] * vsc, vf[2 * j + 1] * vsc);
                    }
                } else {
                    const size_t base = row * PF_D;
                    kraw = *reinterpret_cast<const uint4*>(
                        (const bf16*)k_cache + base + d0);
                    vraw = *reinterpret_cast<const uint4*>(
                        (const bf16*)v_cache + base + d0);
                }
            }
            *reinterpret_cast<uint4*>(&k_lds[kk][swz16(d0 / 8, kk) * 8]) = kraw;
            const bf16* v8 = reinterpret_cast<const bf16*>(&vraw);
#pragma unroll
            for (int j = 0; j < 8; ++j)
                vt_lds[d0 + j][swz8(kk >> 3, d0 + j) * 8 + (kk & 7)] = v8[j];
        }
        __syncthreads();

        f32x4 s[4];
#pragma unroll
        for (int sub = 0; sub < 4; ++sub) {
            s[sub] = (f32x4){0.f, 0.f, 0.f, 0.f};
#pragma unroll
            for (int kc = 0; kc < 4; ++kc) {
                const bf16x8 bfrag = *reinterpret_cast<const bf16x8*>(
                    &k_lds[sub * 16 + lrow][swz16(kc * 4 + lk, sub * 16 + lrow) * 8]);
                s[sub] = mfma16(q_frag[kc], bfrag, s[sub]);
            }
        }

        float p_val[4][4];
        float row_max[4], row_sum[4];
#pragma unroll
        for (int r = 0; r < 4; ++r) {
            const int qrow = tile0 + wave * 16 + lk * 4 + r;
            float mx = NEG_INF;
#pragma unroll
            for (int sub = 0; sub < 4; ++sub) {
                const int key = key0 + sub * 16 + lrow;
                float val = s[sub][r] * scale;
                // only key-validity masking: all cached keys precede
                // every fresh q row, so no causal term
                if (key >= cached || qrow >= seq_len) val = NEG_INF;
                p_val[sub][r] = val;
                mx = fmaxf(mx, val);
            }
            row_max[r] = group16_reduce_max(mx);
        }
#pragma unroll
        for (int r = 0; r < 4; ++r) {
            const float m_new = fmaxf(m_st[r], row_max[r]);
            const float factor = (m_st[r] <= NEG_INF) ? 0.f : __expf(m_st[r] - m_new);
            float sum = 0.f;
#pragma unroll
            for (int sub = 0; sub < 4; ++sub) {
                const float pv = (m_new <= NEG_INF) ? 0.f : __expf(p_val[sub][r] - m_new);
                p_val[sub][r] = pv;
                sum += pv;
            }
            row_sum[r] = group16_reduce_sum(sum);
            l_st[r] = l_st[r] * factor + row_sum[r];
            m_st[r] = m_new;
#pragma unroll
            for (int d = 0; d < 8; ++d) accO[d][r] *= factor;
        }

#pragma unroll
        for (int sub = 0; sub < 4; ++sub)
#pragma unroll
            for (int r = 0; r < 4; ++r)
                p_lds[wave][lk * 4 + r][swz8(sub * 2 + (lrow >> 3), lk * 4 + r) * 8 +
                                        (lrow & 7)] = f2bf(p_val[sub][r]);

#pragma unroll
        for (int dsub = 0; dsub < 8; ++dsub) {
#pragma unroll
            for (int kc = 0; kc < 2; ++kc) {
                const bf16x8 pa = *reinterpret_cast<const bf16x8*>(
                    &p_lds[wave][lrow][swz8(kc * 4 + lk, lrow) * 8]);
                const bf16x8 vb = *reinterpret_cast<const bf16x8*>(
                    &vt_lds[dsub * 16 + lrow][swz8(kc * 4 + lk, dsub * 16 + lrow) * 8]);
                accO[dsub] = mfma16(pa, vb, accO[dsub]);
            }
        }
    }

    // ---- fresh KV (causal within the new tokens) ----
    const int kv_limit = min(seq_len, tile0 + QTILE);  // causal: keys < limit
    const int n_kv_tiles = (kv_limit + KVTILE - 1) / KVTILE;

    for (int jt = 0; jt < n_kv_tiles; ++jt) {
        const int key0 = jt * KVTILE;
        const int keys_here = min(KVTILE, kv_limit - key0);

        // ---- stage K tile and V^T tile (256 threads cooperative) ----
        __syncthreads();
        for (int i = tid; i < KVTILE * (PF_D / 8); i += PF_WAVES * WAVE_SIZE) {
            const int kk = i / (PF_D / 8);
            const int d0 = (i % (PF_D / 8)) * 8;
            uint4 kraw = {0, 0, 0, 0}, vraw = {0, 0, 0, 0};
            if (kk < keys_here) {
                const size_t t = (size_t)(seq_start + key0 + kk);
                kraw = *reinterpret_cast<const uint4*>(
                    k + t * k_stride + (size_t)kvh * PF_D + d0);
                vraw = *reinterpret_cast<const uint4*>(
                    v + t * v_stride + (size_t)kvh * PF_D + d0);
            }
            *reinterpret_cast<uint4*>(&k_lds[kk][swz16(d0 / 8, kk) * 8]) = kraw;
            const bf16* v8 = reinterpret_cast<const bf16*>(&vraw);
#pragma unroll
            for (int j = 0; j < 8; ++j)
                vt_lds[d0 + j][swz8(kk >> 3, d0 + j) * 8 + (kk & 7)] = v8[j];
        }
        __syncthreads();

        // ---- S = scale * Q K^T  (4 key-subtiles of 16) ----
        f32x4 s[4];
#pragma unroll
        for (int sub = 0; sub < 4; ++sub) {
            s[sub] = (f32x4){0.f, 0.f, 0.f, 0.f};
#pragma unroll
            for (int kc = 0; kc < 4; ++kc) {
                const bf16x8 bfrag = *reinterpret_cast<const bf16x8*>(
                    &k_lds[sub * 16 + lrow][swz16(kc * 4 + lk, sub * 16 + lrow) * 8]);
                s[sub] = mfma16(q_frag[kc], bfrag, s[sub]);
            }
        }

        // ---- mask + online softmax ----
        float p_val[4][4];
        float row_max[4], row_sum[4];
#pragma unroll
        for (int r = 0; r < 4; ++r) {
            const int qrow = tile0 + wave * 16 + lk * 4 + r;  // this lane's C row
            float mx = NEG_INF;
#pragma unroll
            for (int sub = 0; sub < 4; ++sub) {
                const int key = key0 + sub * 16 + lrow;       // this lane's C col
                float val = s[sub][r] * scale;
                if (key > qrow || key >= seq_len || qrow >= seq_len) val = NEG_INF;
                p_val[sub][r] = val;
                mx = fmaxf(mx, val);
            }
            row_max[r] = group16_reduce_max(mx);
        }
#pragma unroll
        for (int r = 0; r < 4; ++r) {
            const float m_new = fmaxf(m_st[r], row_max[r]);
            const float factor = (m_st[r] <= NEG_INF) ? 0.f : __expf(m_st[r] - m_new);
            float sum = 0.f;
#pragma unroll
            for (int sub = 0; sub < 4; ++sub) {
                const float pv = (m_new <= NEG_INF) ? 0.f : __expf(p_val[sub][r] - m_new);
                p_val[sub][r] = pv;
                sum += pv;
            }
            row_sum[r] = group16_reduce_sum(sum);
            l_st[r] = l_st[r] * factor + row_sum[r];
            m_st[r] = m_new;
#pragma unroll
            for (int d = 0; d < 8; ++d) accO[d][r] *= factor;
        }

        // ---- P -> LDS (bf16), per-wave private tile ----
#pragma unroll
        for (int sub = 0; sub < 4; ++sub)
#pragma unroll
            for (int r = 0; r < 4; ++r)
                p_lds[wave][lk * 4 + r][swz8(sub * 2 + (lrow >> 3), lk * 4 + r) * 8 +
                                        (lrow & 7)] = f2bf(p_val[sub][r]);
        // wave-synchronous LDS: the same wave reads it next, no barrier

        // ---- O += P V  (8 dim-subtiles, 2 key-chunks of 32) ----
#pragma unroll
        for (int dsub = 0; dsub < 8; ++dsub) {
#pragma unroll
            for (int kc = 0; kc < 2; ++kc) {
                const bf16x8 pa = *reinterpret_cast<const bf16x8*>(
                    &p_lds[wave][lrow][swz8(kc * 4 + lk, lrow) * 8]);
                const bf16x8 vb = *reinterpret_cast<const bf16x8*>(
                    &vt_lds[dsub * 16 + lrow][swz8(kc * 4 + lk, dsub * 16 + lrow) * 8]);
                accO[dsub] = mfma16(pa, vb, accO[dsub]);
            }
        }
    }

    // ---- epilogue: out[row][d] = acc / l ----
#pragma unroll
    for (int r = 0; r < 4; ++r) {
        const int qrow = tile0 + wave * 16 + lk * 4 + r;
        if (qrow >= seq_len || l_st[r] <= 0.f) continue;
        const float inv_l = 1.f / l_st[r];
        bf16* orow =
            out + (size_t)(seq_start + qrow) * Hq * PF_D + (size_t)head * PF_D;
#pragma unroll
        for (int dsub = 0; dsub < 8; ++dsub)
            orow[dsub * 16 + lrow] = f2bf(accO[dsub][r] * inv_l);
    }
}

// accO[dsub][r] uses the PV C-fragment layout row = lk*4 + r, col = lrow,
// identical to the S layout above: both are mfma_f32_16x16x32 C fragments,
// and the C/D lane map on gfx950 is dtype- and operand-independent
// (cdna_hip_programming.md §3), so no relayout is needed between S and O.

extern "C" hipError_t launch_attention_prefill(
    void* out, const void* q, const void* k, const void* v,
    const int* cu_seqlens, const int* tile_seq, const int* tile_off,
    int ntiles, float scale, int Hq, int Hkv, int D, int64_t q_stride,
    int64_t k_stride, int64_t v_stride, const void* k_cache,
    const void* v_cache, const float* k_scale, const float* v_scale,
    const int* block_tables, const int* cached_lens,
    int block_size, int max_blocks, hipStream_t stream) {
    if (D != PF_D) return hipErrorNotSupported;
    dim3 grid(ntiles, Hq);
    dim3 block(PF_WAVES * WAVE_SIZE);
    if (k_scale != nullptr)
        attention_prefill_kernel<true><<<grid, block, 0, stream>>>(
            (bf16*)out, (const bf16*)q, (const bf16*)k, (const bf16*)v,
            cu_seqlens, tile_seq, tile_off, scale, Hq, Hkv, q_stride, k_stride,
            v_stride, k_cache, v_cache, k_scale, v_scale, block_tables,
            cached_lens, block_size, max_blocks);
    else
        attention_prefill_kernel<false><<<grid, block, 0, stream>>>(
            (bf16*)out, (const bf16*)q, (const bf16*)k, (const bf16*)v,
            cu_seqlens, tile_seq, tile_off, scale, Hq, Hkv, q_stride, k_stride,
            v_stride, k_cache, v_cache, nullptr, nullptr, block_tables,
            cached_lens, block_size, max_blocks);
    HIP_CHECK_LAST();
    return hipSuccess;
}
