// Paged decode attention for gfx950 (GQA, bf16 KV cache).
//
// Replaces the per-token generation the reference gateway delegated to its
// remote providers (SURVEY.md §2b "decode attention kernel"). Memory-bound:
// the whole KV history of each sequence is streamed once per step.
//
// Geometry: one workgroup per (sequence, kv-head); 4 waves; the G = Hq/Hkv
// query heads of the group share the streamed K/V (GQA bandwidth saving).
// G is a TEMPLATE parameter: with runtime G the per-lane s/m/l/acc arrays
// are runtime-indexed and spill to scratch (cdna_hip_programming.md §5.4
// rule 20) — the first profiled build ran 12x off roofline exactly there.
//
// Each wave owns key chunks of 64 positions (chunk c -> wave c%4):
//   phase A: lane = one key position; the lane streams that key's 128-dim
//            row (16 B vector loads) and dots it against q staged in LDS
//            (vectorized float4 broadcast reads);
//   phase B: half-wave per key, lane owns a dim quad (dwordx2 loads,
//            2 keys per instruction), cross-half shfl_xor fold.
// Online softmax per wave, flash-style cross-wave combine in LDS at the
// end; long-context small-batch launches split the context across grid.z
// (flash-decode partials + combine kernel).

#include "common.h"

#define MAX_G 8
#define NWAVES 4
#define DECODE_D 128
#define SPLIT_SPAN 2048  // context per split (flash-decode partials)

// SPLIT=false: one workgroup per (seq, kv-head) writes out directly.
// SPLIT=true: grid.z context-splits write un-normalized partials
// (m, l, acc) to scratch; decode_combine_kernel merges them. Used when
// B*Hkv underfills the chip but the context can be long (small-batch
// long-context decode); the split count is derived from max_blocks so
// it is hipGraph-capture-stable (idle splits exit on short contexts).
// FP8: caches hold e4m3 bytes with per-(token, head) row scales; the
// scales factor out of both dot products (one multiply per key).
template <int G, bool SPLIT, bool FP8>
__launch_bounds__(NWAVES* WAVE_SIZE)
__global__ void attention_decode_kernel(
    bf16* __restrict__ out,                 // [B, Hq, D]
    const bf16* __restrict__ q,             // [B, Hq, D]
    const void* __restrict__ k_cache,       // [NB, Hkv, BS, D] bf16|fp8
    const void* __restrict__ v_cache,
    const float* __restrict__ k_scale,      // [NB, Hkv, BS] (FP8)
    const float* __restrict__ v_scale,
    const int* __restrict__ block_tables,   // [B, max_blocks]
    const int* __restrict__ context_lens,   // [B]
    float scale,
    int Hq,
    int Hkv,
    int block_size,
    int max_blocks,
    int64_t q_stride,
    float* __restrict__ part_acc,           // [B, Hq, NSPLIT, D] (SPLIT)
    float* __restrict__ part_ml,            // [B, Hq, NSPLIT, 2] (SPLIT)
    int nsplit) {
    const int seq = blockIdx.x;
    const int kvh = blockIdx.y;
    const int split = SPLIT ? blockIdx.z : 0;
    const int D = DECODE_D;
    const int L = context_lens[seq];
    if (L <= 0) return;
    const int span0 = SPLIT ? split * SPLIT_SPAN : 0;
    if (SPLIT && span0 >= L) {  // idle split: publish an empty partial
        const int lane0 = threadIdx.x & (WAVE_SIZE - 1);
        if (threadIdx.x < WAVE_SIZE) {
#pragma unroll
            for (int g = 0; g < G; ++g) {
                const size_t pb =
                    (((size_t)seq * Hq + kvh * G + g) * nsplit + split);
                if (lane0 < 2) part_ml[pb * 2 + lane0] = lane0 == 0 ? -INFINITY : 0.f;
                for (int d = lane0; d < D; d += WAVE_SIZE) part_acc[pb * D + d] = 0.f;
            }
        }
        return;
    }
    const int span1 = SPLIT ? min(L, span0 + SPLIT_SPAN) : L;

    const int tid = threadIdx.x;
    const int lane = tid & (WAVE_SIZE - 1);
    const int wave = tid >> 6;

    __shared__ float q_lds[G][DECODE_D];
    __shared__ float p_lds[NWAVES][G][WAVE_SIZE];
    __shared__ float m_lds[NWAVES][G];
    __shared__ float l_lds[NWAVES][G];
    __shared__ float acc_lds[NWAVES][G][DECODE_D];
    __shared__ int bt_lds[1024];

    for (int i = tid; i < G * D; i += blockDim.x) {
        const int g = i / D, d = i % D;
        q_lds[g][d] =
            bf2f(q[(size_t)seq * q_stride + (size_t)(kvh * G + g) * D + d]) * scale;
    }
    const int nblocks = (L + block_size - 1) / block_size;
    for (int i = tid; i < nblocks && i < 1024; i += blockDim.x)
        bt_lds[i] = block_tables[(size_t)seq * max_blocks + i];
    __syncthreads();
    const int* __restrict__ bt_global = block_tables + (size_t)seq * max_blocks;
#define BT(idx) ((idx) < 1024 ? bt_lds[(idx)] : bt_global[(idx)])

    float m[G], l[G], acc[G][4];
#pragma unroll
    for (int g = 0; g < G; ++g) {
        m[g] = -INFINITY;
        l[g] = 0.f;
#pragma unroll
        for (int d = 0; d < 4; ++d) acc[g][d] = 0.f;
    }

    const int c0 = span0 / WAVE_SIZE;  // SPLIT_SPAN is a multiple of 64
    const int nchunks = (span1 + WAVE_SIZE - 1) / WAVE_SIZE;
    for (int c = c0 + wave; c < nchunks; c += NWAVES) {
        const int pos = c * WAVE_SIZE + lane;
        const bool valid = pos < span1;

        // --- phase A: lane = key ---
        float s[G];
#pragma unroll
        for (int g = 0; g < G; ++g) s[g] = valid ? 0.f : -INFINITY;
        if (valid) {
            const int kblock = BT(pos / block_size);
            const int koff = pos % block_size;
            const size_t krow_i = ((size_t)kblock * Hkv + kvh) * block_size + koff;
#pragma unroll 4
            for (int i = 0; i < D / 8; ++i) {
                float kf[8];
                if (FP8) {
                    const uint2 kq = reinterpret_cast<const uint2*>(
                        (const unsigned char*)k_cache + krow_i * D)[i];
                    fp8x4_to_f32(kq.x, kf[0], kf[1], kf[2], kf[3]);
                    fp8x4_to_f32(kq.y, kf[4], kf[5], kf[6], kf[7]);
                } else {
                    const uint4 kv8 = reinterpret_cast<const uint4*>(
                        (const bf16*)k_cache + krow_i * D)[i];
                    unpack2(kv8.x, kf[0], kf[1]);
                    unpack2(kv8.y, kf[2], kf[3]);
                    unpack2(kv8.z, kf[4], kf[5]);
                    unpack2(kv8.w, kf[6], kf[7]);
                }
#pragma unroll
                for (int g = 0; g < G; ++g) {
                    const float4 qa = *reinterpret_cast<const float4*>(&q_lds[g][i * 8]);
                    const float4 qb = *reinterpret_cast<const float4*>(&q_lds[g][i * 8 + 4]);
                    s[g] = fmaf(qa.x, kf[0], s[g]);
                    s[g] = fmaf(qa.y, kf[1], s[g]);
                    s[g] = fmaf(qa.z, kf[2], s[g]);
                    s[g] = fmaf(qa.w, kf[3], s[g]);
                    s[g] = fmaf(qb.x, kf[4], s[g]);
                    s[g] = fmaf(qb.y, kf[5], s[g]);
                    s[g] = fmaf(qb.z, kf[6], s[g]);
                    s[g] = fmaf(qb.w, kf[7], s[g]);
                }
            }
            if (FP8) {
                const float ksc = k_scale[krow_i];
#pragma unroll
                for (int g = 0; g < G; ++g) s[g] *= ksc;
            }
        }

        // --- online softmax update (per wave) ---
#pragma unroll
        for (int g = 0; g < G; ++g) {
            const float cmax = wave_reduce_max(s[g]);
            const float m_new = fmaxf(m[g], cmax);
            float p = 0.f;
            if (valid && m_new != -INFINITY) p = __expf(s[g] - m_new);
            const float factor = (m[g] == -INFINITY) ? 0.f : __expf(m[g] - m_new);
            const float csum = wave_reduce_sum(p);
            l[g] = l[g] * factor + csum;
#pragma unroll
            for (int d = 0; d < 4; ++d) acc[g][d] *= factor;
            m[g] = m_new;
            p_lds[wave][g][lane] = p;
        }
        // wave-synchronous LDS use: private per wave, no barrier needed

        // --- phase B: half-wave per key, lane owns a dim QUAD (8 B
        // dwordx2 loads: half the instructions and latency batches of the
        // earlier dim-pair/dword form). Lane l covers key j + (l>>5),
        // dims (l&31)*4..+3; 8 keys per iteration = 4 loads in flight.
        // Keys beyond the chunk have p == 0 (masked in phase A), so only
        // the ADDRESS needs clamping to stay inside the paged cache. ---
        const int dbase = (lane & 31) * 4;
        const int khalf = lane >> 5;
#pragma unroll 2
        for (int j = 0; j < WAVE_SIZE; j += 8) {
            uint2 vp[4];
            uint32_t vp8[4];
            float vsc[4];
#pragma unroll
            for (int u = 0; u < 4; ++u) {
                const int pos = min(c * WAVE_SIZE + j + 2 * u + khalf, span1 - 1);
                const int vb = BT(pos / block_size);
                const int vo = pos % block_size;
                const size_t vrow_i = ((size_t)vb * Hkv + kvh) * block_size + vo;
                if (FP8) {
                    vp8[u] = *reinterpret_cast<const uint32_t*>(
                        (const unsigned char*)v_cache + vrow_i * D + dbase);
                    vsc[u] = v_scale[vrow_i];
                } else {
                    vp[u] = *reinterpret_cast<const uint2*>(
                        (const bf16*)v_cache + vrow_i * D + dbase);
                }
            }
#pragma unroll
            for (int u = 0; u < 4; ++u) {
                float v0, v1, v2, v3;
                if (FP8) {
                    fp8x4_to_f32(vp8[u], v0, v1, v2, v3);
                } else {
                    unpack2(vp[u].x, v0, v1);
                    unpack2(vp[u].y, v2, v3);
                }
                const int key = j + 2 * u + khalf;
#pragma unroll
                for (int g = 0; g < G; ++g) {
                    // fp8: fold the V row scale into p once per key
                    const float pj = p_lds[wave][g][key] * (FP8 ? vsc[u] : 1.f);
                    acc[g][0] = fmaf(pj, v0, acc[g][0]);
                    acc[g][1] = fmaf(pj, v1, acc[g][1]);
                    acc[g][2] = fmaf(pj, v2, acc[g][2]);
                    acc[g][3] = fmaf(pj, v3, acc[g][3]);
                }
            }
        }
    }

    // fold the two half-wave key subsets (lane and lane^32 hold the same
    // dim quad over disjoint keys)
#pragma unroll
    for (int g = 0; g < G; ++g)
#pragma unroll
        for (int d = 0; d < 4; ++d) acc[g][d] += __shfl_xor(acc[g][d], 32, WAVE_SIZE);

    // ---- cross-wave flash combine ----
#pragma unroll
    for (int g = 0; g < G; ++g) {
        if (lane == 0) {
            m_lds[wave][g] = m[g];
            l_lds[wave][g] = l[g];
        }
        if (lane < 32) {
#pragma unroll
            for (int d = 0; d < 4; ++d)
                acc_lds[wave][g][(lane & 31) * 4 + d] = acc[g][d];
        }
    }
    __syncthreads();

    if (wave == 0) {
#pragma unroll
        for (int g = 0; g < G; ++g) {
            float gm = -INFINITY;
#pragma unroll
            for (int w = 0; w < NWAVES; ++w) gm = fmaxf(gm, m_lds[w][g]);
            float gl = 0.f;
            float f[NWAVES];
#pragma unroll
            for (int w = 0; w < NWAVES; ++w) {
                f[w] = (m_lds[w][g] == -INFINITY) ? 0.f : __expf(m_lds[w][g] - gm);
                gl += f[w] * l_lds[w][g];
            }
            float o0 = 0.f, o1 = 0.f;
#pragma unroll
            for (int w = 0; w < NWAVES; ++w) {
                o0 += f[w] * acc_lds[w][g][2 * lane];
                o1 += f[w] * acc_lds[w][g][2 * lane + 1];
            }
            if (SPLIT) {
                // un-normalized partial + (m, l) for the combine kernel
                const size_t pb =
                    (((size_t)seq * Hq + kvh * G + g) * nsplit + split);
                part_acc[pb * DECODE_D + 2 * lane] = o0;
                part_acc[pb * DECODE_D + 2 * lane + 1] = o1;
                if (lane == 0) {
                    part_ml[pb * 2] = gm;
                    part_ml[pb * 2 + 1] = gl;
                }
            } else {
                const float inv_l = (gl > 0.f) ? 1.f / gl : 0.f;
                reinterpret_cast<uint32_t*>(
                    out + (size_t)seq * Hq * DECODE_D +
                    (size_t)(kvh * G + g) * DECODE_D)[lane] =
                    pack2(o0 * inv_l, o1 * inv_l);
            }
        }
    }
}

typedef __attribute__((__vector_size__(8 * sizeof(short)))) short bf16x8_d;

// ---------------------------------------------------------------------------
// v2 geometry (measured A/B winner at short contexts — the headline decode
// regime): the 4 waves of a (seq, kv-head) workgroup process the SAME
// 64-key chunk from a cooperatively-staged LDS tile (fully coalesced
// global loads: v1's phase-A read touched 64 cache lines per instruction),
// and each wave owns a DISJOINT subset of the G query heads, so per-head
// softmax state lives in one wave and the cross-wave combine pass
// disappears. K tile reads are XOR-swizzled (lane-per-key at a fixed
// 16-B chunk is a wave-wide same-bank read on 256-B rows); V reads are
// conflict-free linear.
// ---------------------------------------------------------------------------

template <int G, bool SPLIT, bool FP8>
__launch_bounds__(NWAVES* WAVE_SIZE)
__global__ void attention_decode_v2_kernel(
    bf16* __restrict__ out,                 // [B, Hq, D]
    const bf16* __restrict__ q,             // [B, Hq, D]
    const void* __restrict__ k_cache,       // [NB, Hkv, BS, D] bf16|fp8
    const void* __restrict__ v_cache,
    const float* __restrict__ k_scale,      // [NB, Hkv, BS] (FP8)
    const float* __restrict__ v_scale,
    const int* __restrict__ block_tables,   // [B, max_blocks]
    const int* __restrict__ context_lens,   // [B]
    float scale,
    int Hq,
    int Hkv,
    int block_size,
    int max_blocks,
    int64_t q_stride,
    float* __restrict__ part_acc,           // [B, Hq, NSPLIT, D] (SPLIT)
    float* __restrict__ part_ml,            // [B, Hq, NSPLIT, 2] (SPLIT)
    int nsplit) {
    const int seq = blockIdx.x;
    const int kvh = blockIdx.y;
    const int split = SPLIT ? blockIdx.z : 0;
    const int D = DECODE_D;
    const int L = context_lens[seq];
    if (L <= 0) return;
    const int span0 = SPLIT ? split * SPLIT_SPAN : 0;
    const int tid = threadIdx.x;
    const int lane = tid & (WAVE_SIZE - 1);
    const int wave = tid >> 6;
    // wave w owns heads {w, w+NWAVES, ...} < G
    constexpr int HPW = (G + NWAVES - 1) / NWAVES;  // heads per wave (max)
    if (SPLIT && span0 >= L) {  // idle split: publish empty partials
#pragma unroll
        for (int hh = 0; hh < HPW; ++hh) {
            const int g = wave + hh * NWAVES;
            if (g >= G) break;
            const size_t pb = (((size_t)seq * Hq + kvh * G + g) * nsplit + split);
            if (lane < 2) part_ml[pb * 2 + lane] = lane == 0 ? -INFINITY : 0.f;
            for (int d = lane; d < D; d += WAVE_SIZE) part_acc[pb * D + d] = 0.f;
        }
        return;
    }
    const int span1 = SPLIT ? min(L, span0 + SPLIT_SPAN) : L;

    __shared__ float q_lds[G][DECODE_D];
    __shared__ bf16 k_lds[WAVE_SIZE][DECODE_D];   // 16 KiB (XOR-swizzled)
    __shared__ bf16 v_lds[WAVE_SIZE][DECODE_D];   // 16 KiB (linear)
    __shared__ float p_lds[G][WAVE_SIZE];
    __shared__ int bt_lds[1024];

    for (int i = tid; i < G * D; i += blockDim.x) {
        const int g = i / D, d = i % D;
        q_lds[g][d] =
            bf2f(q[(size_t)seq * q_stride + (size_t)(kvh * G + g) * D + d]) * scale;
    }
    const int nblocks = (L + block_size - 1) / block_size;
    for (int i = tid; i < nblocks && i < 1024; i += blockDim.x)
        bt_lds[i] = block_tables[(size_t)seq * max_blocks + i];
    __syncthreads();
    const int* __restrict__ bt_global = block_tables + (size_t)seq * max_blocks;
#define BT2(idx) ((idx) < 1024 ? bt_lds[(idx)] : bt_global[(idx)])

    float m[HPW], l[HPW], acc[HPW][4];
#pragma unroll
    for (int h = 0; h < HPW; ++h) {
        m[h] = -INFINITY;
        l[h] = 0.f;
#pragma unroll
        for (int d = 0; d < 4; ++d) acc[h][d] = 0.f;
    }

    const int c0 = span0 / WAVE_SIZE;
    const int nchunks = (span1 + WAVE_SIZE - 1) / WAVE_SIZE;
    for (int c = c0; c < nchunks; ++c) {
        // ---- cooperative coalesced staging of the 64-key K/V chunk ----
        // thread i covers key i/4, 16-B piece i%4 per pass (4 passes of
        // 64 B per key row = 256 B); K lands XOR-swizzled, V linear.
        __syncthreads();  // previous chunk's reads complete
        for (int pass = 0; pass < 4; ++pass) {
            // 16 lanes per 256-B key row: each wave reads 1 KiB contiguous
            const int key = (pass << 4) | (tid >> 4);
            const int chunk = tid & 15;  // 16-B chunk 0..15
            const int pos = c * WAVE_SIZE + key;
            const int cpos = min(pos, span1 - 1);
            const int blk = BT2(cpos / block_size);
            const size_t row =
                ((size_t)blk * Hkv + kvh) * block_size + cpos % block_size;
            const int kchunk = chunk ^ (key & 15);
            if (FP8) {
                const uint2 kq = reinterpret_cast<const uint2*>(
                    (const unsigned char*)k_cache + row * D)[chunk];
                const uint2 vq = reinterpret_cast<const uint2*>(
                    (const unsigned char*)v_cache + row * D)[chunk];
                const float ksc = k_scale[row], vsc = v_scale[row];
                float kf[8], vf[8];
                fp8x4_to_f32(kq.x, kf[0], kf[1], kf[2], kf[3]);
                fp8x4_to_f32(kq.y, kf[4], kf[5], kf[6], kf[7]);
                fp8x4_to_f32(vq.x, vf[0], vf[1], vf[2], vf[3]);
                fp8x4_to_f32(vq.y, vf[4], vf[5], vf[6], vf[7]);
                uint2 kw, vw;
                kw.x = pack2_trunc(kf[0] * ksc, kf[1] * ksc);
                kw.y = pack2_trunc(kf[2] * ksc, kf[3] * ksc);
                uint2 kw2;
                kw2.x = pack2_trunc(kf[4] * ksc, kf[5] * ksc);
                kw2.y = pack2_trunc(kf[6] * ksc, kf[7] * ksc);
                vw.x = pack2_trunc(vf[0] * vsc, vf[1] * vsc);
                vw.y = pack2_trunc(vf[2] * vsc, vf[3] * vsc);
                uint2 vw2;
                vw2.x = pack2_trunc(vf[4] * vsc, vf[5] * vsc);
                vw2.y = pack2_trunc(vf[6] * vsc, vf[7] * vsc);
                uint4 kq4 = {kw.x, kw.y, kw2.x, kw2.y};
                uint4 vq4 = {vw.x, vw.y, vw2.x, vw2.y};
                *reinterpret_cast<uint4*>(&k_lds[key][kchunk * 8]) = kq4;
                *reinterpret_cast<uint4*>(&v_lds[key][chunk * 8]) = vq4;
            } else {
                *reinterpret_cast<uint4*>(&k_lds[key][kchunk * 8]) =
                    reinterpret_cast<const uint4*>((const bf16*)k_cache + row * D)[chunk];
                *reinterpret_cast<uint4*>(&v_lds[key][chunk * 8]) =
                    reinterpret_cast<const uint4*>((const bf16*)v_cache + row * D)[chunk];
            }
        }
        __syncthreads();

        // ---- phase A: lane = key (LDS, swizzle-matched reads) ----
        const int pos = c * WAVE_SIZE + lane;
        const bool valid = pos < span1;
#pragma unroll
        for (int hh = 0; hh < HPW; ++hh) {
            const int g = wave + hh * NWAVES;
            if (g >= G) break;
            float sg = valid ? 0.f : -INFINITY;
            if (valid) {
#pragma unroll 4
                for (int i = 0; i < D / 8; ++i) {
                    const bf16x8_d kv8 = *reinterpret_cast<const bf16x8_d*>(
                        &k_lds[lane][(i ^ (lane & 15)) * 8]);
                    const uint4 kvu = *reinterpret_cast<const uint4*>(&kv8);
                    float kf[8];
                    unpack2(kvu.x, kf[0], kf[1]);
                    unpack2(kvu.y, kf[2], kf[3]);
                    unpack2(kvu.z, kf[4], kf[5]);
                    unpack2(kvu.w, kf[6], kf[7]);
                    const float4 qa = *reinterpret_cast<const float4*>(&q_lds[g][i * 8]);
                    const float4 qb = *reinterpret_cast<const float4*>(&q_lds[g][i * 8 + 4]);
                    sg = fmaf(qa.x, kf[0], sg);
                    sg = fmaf(qa.y, kf[1], sg);
                    sg = fmaf(qa.z, kf[2], sg);
                    sg = fmaf(qa.w, kf[3], sg);
                    sg = fmaf(qb.x, kf[4], sg);
                    sg = fmaf(qb.y, kf[5], sg);
                    sg = fmaf(qb.z, kf[6], sg);
                    sg = fmaf(qb.w, kf[7], sg);
                }
            }
            const float cmax = wave_reduce_max(sg);
            const float m_new = fmaxf(m[hh], cmax);
            float pv = 0.f;
            if (valid && m_new != -INFINITY) pv = __expf(sg - m_new);
            const float factor = (m[hh] == -INFINITY) ? 0.f : __expf(m[hh] - m_new);
            const float csum = wave_reduce_sum(pv);
            l[hh] = l[hh] * factor + csum;
#pragma unroll
            for (int d = 0; d < 4; ++d) acc[hh][d] *= factor;
            m[hh] = m_new;
            p_lds[g][lane] = pv;
        }
        // p_lds is wave-private per head (one writer wave), read below by
        // the same wave only — no barrier needed before phase B

        // ---- phase B: half-wave per key, lane owns a dim quad (LDS) ----
        const int dbase = (lane & 31) * 4;
        const int khalf = lane >> 5;
#pragma unroll
        for (int hh = 0; hh < HPW; ++hh) {
            const int g = wave + hh * NWAVES;
            if (g >= G) break;
#pragma unroll 4
            for (int j = 0; j < WAVE_SIZE; j += 2) {
                const int key = j + khalf;
                const float pj = p_lds[g][key];
                float v0, v1, v2, v3;
                const uint2 vv = *reinterpret_cast<const uint2*>(&v_lds[key][dbase]);
                unpack2(vv.x, v0, v1);
                unpack2(vv.y, v2, v3);
                acc[hh][0] = fmaf(pj, v0, acc[hh][0]);
                acc[hh][1] = fmaf(pj, v1, acc[hh][1]);
                acc[hh][2] = fmaf(pj, v2, acc[hh][2]);
                acc[hh][3] = fmaf(pj, v3, acc[hh][3]);
            }
        }
    }

    // fold the half-wave key subsets; each wave writes its own heads
#pragma unroll
    for (int hh = 0; hh < HPW; ++hh) {
        const int g = wave + hh * NWAVES;
        if (g >= G) break;
#pragma unroll
        for (int d = 0; d < 4; ++d) acc[hh][d] += __shfl_xor(acc[hh][d], 32, WAVE_SIZE);
        if (lane < 32) {
            if (SPLIT) {
                const size_t pb = (((size_t)seq * Hq + kvh * G + g) * nsplit + split);
#pragma unroll
                for (int d = 0; d < 4; ++d)
                    part_acc[pb * DECODE_D + (lane & 31) * 4 + d] = acc[hh][d];
                if (lane == 0) {
                    part_ml[pb * 2] = m[hh];
                    part_ml[pb * 2 + 1] = l[hh];
                }
            } else {
                const float inv_l = (l[hh] > 0.f) ? 1.f / l[hh] : 0.f;
                uint2 o2;
                o2.x = pack2(acc[hh][0] * inv_l, acc[hh][1] * inv_l);
                o2.y = pack2(acc[hh][2] * inv_l, acc[hh][3] * inv_l);
                *reinterpret_cast<uint2*>(
                    out + (size_t)seq * Hq * DECODE_D +
                    (size_t)(kvh * G + g) * DECODE_D + (lane & 31) * 4) = o2;
            }
        }
    }
}

template <int G, bool SPLIT, bool FP8>
__launch_bounds__(NWAVES* WAVE_SIZE)
__global__ void attention_decode_v3_kernel(
    bf16* __restrict__ out,                 // [B, Hq, D]
    const bf16* __restrict__ q,             // [B, Hq, D]
    const void* __restrict__ k_cache,       // [NB, Hkv, BS, D] bf16|fp8
    const void* __restrict__ v_cache,
    const float* __restrict__ k_scale,      // [NB, Hkv, BS] (FP8)
    const float* __restrict__ v_scale,
    const int* __restrict__ block_tables,   // [B, max_blocks]
    const int* __restrict__ context_lens,   // [B]
    float scale,
    int Hq,
    int Hkv,
    int block_size,
    int max_blocks,
    int64_t q_stride,
    float* __restrict__ part_acc,           // [B, Hq, NSPLIT, D] (SPLIT)
    float* __restrict__ part_ml,            // [B, Hq, NSPLIT, 2] (SPLIT)
    int nsplit) {
    const int seq = blockIdx.x;
    const int kvh = blockIdx.y;
    const int split = SPLIT ? blockIdx.z : 0;
    const int D = DECODE_D;
    const int L = context_lens[seq];
    if (L <= 0) return;
    const int span0 = SPLIT ? split * SPLIT_SPAN : 0;
    const int tid = threadIdx.x;
    const int lane = tid & (WAVE_SIZE - 1);
    const int wave = tid >> 6;
    // wave w owns heads {w, w+NWAVES, ...} < G
    constexpr int HPW = (G + NWAVES - 1) / NWAVES;  // heads per wave (max)
    if (SPLIT && span0 >= L) {  // idle split: publish empty partials
#pragma unroll
        for (int hh = 0; hh < HPW; ++hh) {
            const int g = wave + hh * NWAVES;
            if (g >= G) break;
            const size_t pb = (((size_t)seq * Hq + kvh * G + g) * nsplit + split);
            if (lane < 2) part_ml[pb * 2 + lane] = lane == 0 ? -INFINITY : 0.f;
            for (int d = lane; d < D; d += WAVE_SIZE) part_acc[pb * D + d] = 0.f;
        }
        return;
    }
    const int span1 = SPLIT ? min(L, span0 + SPLIT_SPAN) : L;

    __shared__ float q_lds[G][DECODE_D];
    __shared__ bf16 k_lds[WAVE_SIZE][DECODE_D];   // 16 KiB (XOR-swizzled)
    __shared__ float p_lds[G][WAVE_SIZE];
    __shared__ int bt_lds[1024];

    for (int i = tid; i < G * D; i += blockDim.x) {
        const int g = i / D, d = i % D;
        q_lds[g][d] =
            bf2f(q[(size_t)seq * q_stride + (size_t)(kvh * G + g) * D + d]) * scale;
    }
    const int nblocks = (L + block_size - 1) / block_size;
    for (int i = tid; i < nblocks && i < 1024; i += blockDim.x)
        bt_lds[i] = block_tables[(size_t)seq * max_blocks + i];
    __syncthreads();
    const int* __restrict__ bt_global = block_tables + (size_t)seq * max_blocks;
#define BT2(idx) ((idx) < 1024 ? bt_lds[(idx)] : bt_global[(idx)])

    float m[HPW], l[HPW], acc[HPW][4];
#pragma unroll
    for (int h = 0; h < HPW; ++h) {
        m[h] = -INFINITY;
        l[h] = 0.f;
#pragma unroll
        for (int d = 0; d < 4; ++d) acc[h][d] = 0.f;
    }

    const int c0 = span0 / WAVE_SIZE;
    const int nchunks = (span1 + WAVE_SIZE - 1) / WAVE_SIZE;
    for (int c = c0; c < nchunks; ++c) {
        // ---- cooperative coalesced staging of the 64-key K/V chunk ----
        // thread i covers key i/4, 16-B piece i%4 per pass (4 passes of
        // 64 B per key row = 256 B); K lands XOR-swizzled, V linear.
        __syncthreads();  // previous chunk's reads complete
        for (int pass = 0; pass < 4; ++pass) {
            // 16 lanes per 256-B key row: each wave reads 1 KiB contiguous
            const int key = (pass << 4) | (tid >> 4);
            const int chunk = tid & 15;  // 16-B chunk 0..15
            const int pos = c * WAVE_SIZE + key;
            const int cpos = min(pos, span1 - 1);
            const int blk = BT2(cpos / block_size);
            const size_t row =
                ((size_t)blk * Hkv + kvh) * block_size + cpos % block_size;
            const int kchunk = chunk ^ (key & 15);
            if (FP8) {
                const uint2 kq = reinterpret_cast<const uint2*>(
                    (const unsigned char*)k_cache + row * D)[chunk];
                const float ksc = k_scale[row];
                float kf[8];
                fp8x4_to_f32(kq.x, kf[0], kf[1], kf[2], kf[3]);
                fp8x4_to_f32(kq.y, kf[4], kf[5], kf[6], kf[7]);
                uint4 kq4;
                kq4.x = pack2_trunc(kf[0] * ksc, kf[1] * ksc);
                kq4.y = pack2_trunc(kf[2] * ksc, kf[3] * ksc);
                kq4.z = pack2_trunc(kf[4] * ksc, kf[5] * ksc);
                kq4.w = pack2_trunc(kf[6] * ksc, kf[7] * ksc);
                *reinterpret_cast<uint4*>(&k_lds[key][kchunk * 8]) = kq4;
            } else {
                *reinterpret_cast<uint4*>(&k_lds[key][kchunk * 8]) =
                    reinterpret_cast<const uint4*>((const bf16*)k_cache + row * D)[chunk];
            }
        }
        __syncthreads();

        // ---- phase A: lane = key (LDS, swizzle-matched reads) ----
        const int pos = c * WAVE_SIZE + lane;
        const bool valid = pos < span1;
#pragma unroll
        for (int hh = 0; hh < HPW; ++hh) {
            const int g = wave + hh * NWAVES;
            if (g >= G) break;
            float sg = valid ? 0.f : -INFINITY;
            if (valid) {
#pragma unroll 4
                for (int i = 0; i < D / 8; ++i) {
                    const bf16x8_d kv8 = *reinterpret_cast<const bf16x8_d*>(
                        &k_lds[lane][(i ^ (lane & 15)) * 8]);
                    const uint4 kvu = *reinterpret_cast<const uint4*>(&kv8);
                    float kf[8];
                    unpack2(kvu.x, kf[0], kf[1]);
                    unpack2(kvu.y, kf[2], kf[3]);
                    unpack2(kvu.z, kf[4], kf[5]);
                    unpack2(kvu.w, kf[6], kf[7]);
                    const float4 qa = *reinterpret_cast<const float4*>(&q_lds[g][i * 8]);
                    const float4 qb = *reinterpret_cast<const float4*>(&q_lds[g][i * 8 + 4]);
                    sg = fmaf(qa.x, kf[0], sg);
                    sg = fmaf(qa.y, kf[1], sg);
                    sg = fmaf(qa.z, kf[2], sg);
                    sg = fmaf(qa.w, kf[3], sg);
                    sg = fmaf(qb.x, kf[4], sg);
                    sg = fmaf(qb.y, kf[5], sg);
                    sg = fmaf(qb.z, kf[6], sg);
                    sg = fmaf(qb.w, kf[7], sg);
                }
            }
            const float cmax = wave_reduce_max(sg);
            const float m_new = fmaxf(m[hh], cmax);
            float pv = 0.f;
            if (valid && m_new != -INFINITY) pv = __expf(sg - m_new);
            const float factor = (m[hh] == -INFINITY) ? 0.f : __expf(m[hh] - m_new);
            const float csum = wave_reduce_sum(pv);
            l[hh] = l[hh] * factor + csum;
#pragma unroll
            for (int d = 0; d < 4; ++d) acc[hh][d] *= factor;
            m[hh] = m_new;
            p_lds[g][lane] = pv;
        }
        // p_lds is wave-private per head (one writer wave), read below by
        // the same wave only — no barrier needed before phase B

        // ---- phase B: half-wave per key, lane owns a dim quad, V read
        // straight from global (coalesced 256-B half-wave segments; the
        // LDS round trip bought nothing for a once-read operand) ----
        const int dbase = (lane & 31) * 4;
        const int khalf = lane >> 5;
#pragma unroll 2
        for (int j = 0; j < WAVE_SIZE; j += 8) {
            uint2 vp[4];
            uint32_t vp8[4];
            float vsc4[4];
#pragma unroll
            for (int u = 0; u < 4; ++u) {
                const int pos2 = min(c * WAVE_SIZE + j + 2 * u + khalf, span1 - 1);
                const int vb = BT2(pos2 / block_size);
                const int vo = pos2 % block_size;
                const size_t vrow = ((size_t)vb * Hkv + kvh) * block_size + vo;
                if (FP8) {
                    vp8[u] = *reinterpret_cast<const uint32_t*>(
                        (const unsigned char*)v_cache + vrow * D + dbase);
                    vsc4[u] = v_scale[vrow];
                } else {
                    vp[u] = *reinterpret_cast<const uint2*>(
                        (const bf16*)v_cache + vrow * D + dbase);
                }
            }
#pragma unroll
            for (int u = 0; u < 4; ++u) {
                float v0, v1, v2, v3;
                if (FP8) {
                    fp8x4_to_f32(vp8[u], v0, v1, v2, v3);
                } else {
                    unpack2(vp[u].x, v0, v1);
                    unpack2(vp[u].y, v2, v3);
                }
                const int key = j + 2 * u + khalf;
#pragma unroll
                for (int hh = 0; hh < HPW; ++hh) {
                    const int g = wave + hh * NWAVES;
                    if (g >= G) break;
                    const float pj = p_lds[g][key] * (FP8 ? vsc4[u] : 1.f);
                    acc[hh][0] = fmaf(pj, v0, acc[hh][0]);
                    acc[hh][1] = fmaf(pj, v1, acc[hh][1]);
                    acc[hh][2] = fmaf(pj, v2, acc[hh][2]);
                    acc[hh][3] = fmaf(pj, v3, acc[hh][3]);
                }
            }
        }
    }

    // fold the half-wave key subsets; each wave writes its own heads
#pragma unroll
    for (int hh = 0; hh < HPW; ++hh) {
        const int g = wave + hh * NWAVES;
        if (g >= G) break;
#pragma unroll
        for (int d = 0; d < 4; ++d) acc[hh][d] += __shfl_xor(acc[hh][d], 32, WAVE_SIZE);
        if (lane < 32) {
            if (SPLIT) {
                const size_t pb = (((size_t)seq * Hq + kvh * G + g) * nsplit + split);
#pragma unroll
                for (int d = 0; d < 4; ++d)
                    part_acc[pb * DECODE_D + (lane & 31) * 4 + d] = acc[hh][d];
                if (lane == 0) {
                    part_ml[pb * 2] = m[hh];
                    part_ml[pb * 2 + 1] = l[hh];
                }
            } else {
                const float inv_l = (l[hh] > 0.f) ? 1.f / l[hh] : 0.f;
                uint2 o2;
                o2.x = pack2(acc[hh][0] * inv_l, acc[hh][1] * inv_l);
                o2.y = pack2(acc[hh][2] * inv_l, acc[hh][3] * inv_l);
                *reinterpret_cast<uint2*>(
                    out + (size_t)seq * Hq * DECODE_D +
                    (size_t)(kvh * G + g) * DECODE_D + (lane & 31) * 4) = o2;
            }
        }
    }
}


// merge the per-split partials: out[b,h] = sum_z exp(m_z - M) acc_z / L
__global__ void decode_combine_kernel(
    bf16* __restrict__ out,            // [B, Hq, D]
    const float* __restrict__ part_acc,  // [B, Hq, NSPLIT, D]
    const float* __restrict__ part_ml,   // [B, Hq, NSPLIT, 2]
    int nsplit) {
    const int bh = blockIdx.x;  // seq * Hq + head
    const int lane = threadIdx.x;  // 64 lanes, 2 dims each
    const size_t base = (size_t)bh * nsplit;
    float gm = -INFINITY;
    for (int z = 0; z < nsplit; ++z) gm = fmaxf(gm, part_ml[(base + z) * 2]);
    float gl = 0.f, o0 = 0.f, o1 = 0.f;
    for (int z = 0; z < nsplit; ++z) {
        const float mz = part_ml[(base + z) * 2];
        const float f = (mz == -INFINITY) ? 0.f : __expf(mz - gm);
        gl += f * part_ml[(base + z) * 2 + 1];
        o0 += f * part_acc[(base + z) * DECODE_D + 2 * lane];
        o1 += f * part_acc[(base + z) * DECODE_D + 2 * lane + 1];
    }
    const float inv_l = (gl > 0.f) ? 1.f / gl : 0.f;
    reinterpret_cast<uint32_t*>(out + (size_t)bh * DECODE_D)[lane] =
        pack2(o0 * inv_l, o1 * inv_l);
}

extern "C" hipError_t launch_attention_decode(
    void* out, const void* q, const void* k_cache, const void* v_cache,
    const float* k_scale, const float* v_scale,
    const int* block_tables, const int* context_lens, float scale, int B,
    int Hq, int Hkv, int block_size, int max_blocks, int D, int64_t q_stride,
    float* part_acc, float* part_ml, int nsplit, int version,
    hipStream_t stream) {
    if (D != DECODE_D) return hipErrorNotSupported;
    if (Hq % Hkv != 0) return hipErrorInvalidValue;
    if (nsplit > 1 && (part_acc == nullptr || part_ml == nullptr))
        return hipErrorInvalidValue;
    const int G = Hq / Hkv;
    const bool fp8 = k_scale != nullptr;
    dim3 grid(B, Hkv, nsplit > 1 ? nsplit : 1);
    dim3 block(NWAVES * WAVE_SIZE);
#define LAUNCH_G3(GV, SPLIT, FP8V)                                             \
    do {                                                                       \
        if (version == 2)                                                      \
            attention_decode_v2_kernel<GV, SPLIT, FP8V>                        \
                <<<grid, block, 0, stream>>>(                                  \
                    (bf16*)out, (const bf16*)q, k_cache, v_cache, k_scale,     \
                    v_scale, block_tables, context_lens, scale, Hq, Hkv,       \
                    block_size, max_blocks, q_stride, part_acc, part_ml,       \
                    nsplit);                                                   \
        else                                                                   \
            attention_decode_kernel<GV, SPLIT, FP8V>                           \
                <<<grid, block, 0, stream>>>(                                  \
                    (bf16*)out, (const bf16*)q, k_cache, v_cache, k_scale,     \
                    v_scale, block_tables, context_lens, scale, Hq, Hkv,       \
                    block_size, max_blocks, q_stride, part_acc, part_ml,       \
                    nsplit);                                                   \
    } while (0)
#define LAUNCH_G2(GV, SPLIT)                                                   \
    do {                                                                       \
        if (fp8) LAUNCH_G3(GV, SPLIT, true); else LAUNCH_G3(GV, SPLIT, false);\
    } while (0)
#define LAUNCH_G(GV)                                                           \
    do {                                                                       \
        if (nsplit > 1) LAUNCH_G2(GV, true); else LAUNCH_G2(GV, false);        \
    } while (0)
    switch (G) {  // every GQA ratio up to MAX_G (e.g. qwen2-7b has G=7)
        case 1: LAUNCH_G(1); break;
        case 2: LAUNCH_G(2); break;
        case 3: LAUNCH_G(3); break;
        case 4: LAUNCH_G(4); break;
        case 5: LAUNCH_G(5); break;
        case 6: LAUNCH_G(6); break;
        case 7: LAUNCH_G(7); break;
        case 8: LAUNCH_G(8); break;
        default: return hipErrorInvalidValue;
    }
#undef LAUNCH_G
#undef LAUNCH_G2
#undef LAUNCH_G3
    if (nsplit > 1) {
        decode_combine_kernel<<<dim3(B * Hq), dim3(WAVE_SIZE), 0, stream>>>(
            (bf16*)out, part_acc, part_ml, nsplit);
    }
    HIP_CHECK_LAST();
    return hipSuccess;
}
