// Token sampling for gfx950: greedy argmax / Gumbel-max temperature sampling
// over [B, V] fp32 logits (V up to ~128k). Honors the OpenAI temperature
// param routed through the gateway payload (reference chat.py:116-119).
//
// Two-stage split-V reduction: a single workgroup per sequence leaves 3/4
// of the chip idle at B<=64 and runs latency-bound (measured 160 us for
// 64x128k); stage 1 spreads (B x SPLITS) workgroups over the vocab, stage 2
// combines the per-split winners. Gumbel-max: argmax(logits/T +
// -log(-log(u))) samples softmax(logits/T); noise u comes from the caller
// so CPU and GPU paths share one RNG.

#include "common.h"

#define SAMPLE_SPLITS 16

__device__ __forceinline__ void better(
    float v, int i, float& bv, int& bi) {
    if (v > bv || (v == bv && i < bi)) {
        bv = v;
        bi = i;
    }
}

__launch_bounds__(256)
__global__ void sample_partial_kernel(
    float* __restrict__ part_val,         // [B, SPLITS]
    int* __restrict__ part_idx,           // [B, SPLITS]
    const float* __restrict__ logits,     // [B, V]
    const float* __restrict__ temperature,// [B]
    const float* __restrict__ noise,      // [B, V] or nullptr
    int V) {
    const int b = blockIdx.x;
    const int split = blockIdx.y;
    const int tid = threadIdx.x;
    const float T = temperature[b];
    const bool greedy = (T <= 0.f) || (noise == nullptr);
    const float invT = greedy ? 1.f : 1.f / fmaxf(T, 1e-6f);
    const float* row = logits + (size_t)b * V;
    const float* nrow = noise ? noise + (size_t)b * V : nullptr;

    const int chunk = (V + SAMPLE_SPLITS - 1) / SAMPLE_SPLITS;
    const int lo = split * chunk;
    const int hi = min(V, lo + chunk);

    float best = -INFINITY;
    int best_idx = lo;
    for (int i = lo + tid; i < hi; i += blockDim.x) {
        float val = row[i];
        if (!greedy) {
            const float u = fmaxf(nrow[i], 1e-20f);
            const float g = -logf(fmaxf(-logf(u), 1e-20f));
            val = val * invT + g;
        }
        better(val, i, best, best_idx);
    }
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) {
        const float ov = __shfl_xor(best, off, WAVE_SIZE);
        const int oi = __shfl_xor(best_idx, off, WAVE_SIZE);
        better(ov, oi, best, best_idx);
    }
    __shared__ float sv[4];
    __shared__ int si[4];
    const int wave = tid >> 6;
    if ((tid & 63) == 0) {
        sv[wave] = best;
        si[wave] = best_idx;
    }
    __syncthreads();
    if (tid == 0) {
#pragma unroll
        for (int w = 1; w < 4; ++w) better(sv[w], si[w], best, best_idx);
        part_val[b * SAMPLE_SPLITS + split] = best;
        part_idx[b * SAMPLE_SPLITS + split] = best_idx;
    }
}

__launch_bounds__(64)
__global__ void sample_combine_kernel(
    int64_t* __restrict__ out,
    const float* __restrict__ part_val,
    const int* __restrict__ part_idx) {
    const int b = blockIdx.x;
    const int lane = threadIdx.x;
    float v = lane < SAMPLE_SPLITS ? part_val[b * SAMPLE_SPLITS + lane] : -INFINITY;
    int i = lane < SAMPLE_SPLITS ? part_idx[b * SAMPLE_SPLITS + lane] : 0x7FFFFFFF;
#pragma unroll
    for (int off = 8; off > 0; off >>= 1) {
        const float ov = __shfl_xor(v, off, WAVE_SIZE);
        const int oi = __shfl_xor(i, off, WAVE_SIZE);
        better(ov, oi, v, i);
    }
    if (lane == 0) out[b] = i;
}

extern "C" hipError_t launch_sample(
    int64_t* out, const float* logits, const float* temperature,
    const float* noise, float* part_val, int* part_idx, int B, int V,
    hipStream_t stream) {
    dim3 grid1(B, SAMPLE_SPLITS);
    sample_partial_kernel<<<grid1, 256, 0, stream>>>(
        part_val, part_idx, logits, temperature, noise, V);
    HIP_CHECK_LAST();
    sample_combine_kernel<<<B, 64, 0, stream>>>(out, part_val, part_idx);
    HIP_CHECK_LAST();
    return hipSuccess;
}
