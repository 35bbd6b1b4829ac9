// Token sampling for gfx950: greedy argmax / Gumbel-max temperature sampling
// over [B, V] fp32 logits (V up to ~128k). Honors the OpenAI temperature
// param routed through the gateway payload (reference chat.py:116-119).
//
// One workgroup (4 waves) per sequence; lanes stride the vocab tracking
// (best value, index); LDS tree combine. Gumbel-max: argmax(logits/T +
// -log(-log(u))) == a sample from softmax(logits/T) — noise u supplied by
// the caller so CPU and GPU paths are comparable under one RNG.

#include "common.h"

__launch_bounds__(256)
__global__ void sample_kernel(
    int64_t* __restrict__ out,            // [B]
    const float* __restrict__ logits,     // [B, V]
    const float* __restrict__ temperature,// [B]
    const float* __restrict__ noise,      // [B, V] or nullptr
    int V) {
    const int b = blockIdx.x;
    const int tid = threadIdx.x;
    const float T = temperature[b];
    const bool greedy = (T <= 0.f) || (noise == nullptr);
    const float invT = greedy ? 1.f : 1.f / fmaxf(T, 1e-6f);
    const float* row = logits + (size_t)b * V;
    const float* nrow = noise ? noise + (size_t)b * V : nullptr;

    float best = -INFINITY;
    int best_idx = 0;
    for (int i = tid; i < V; i += blockDim.x) {
        float val = row[i];
        if (!greedy) {
            const float u = fmaxf(nrow[i], 1e-20f);
            const float g = -logf(fmaxf(-logf(u), 1e-20f));
            val = val * invT + g;
        }
        // strict > keeps the lowest index on ties (argmax parity with torch)
        if (val > best || (val == best && i < best_idx)) {
            best = val;
            best_idx = i;
        }
    }

    // wave reduce (value, index)
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) {
        const float ov = __shfl_xor(best, off, WAVE_SIZE);
        const int oi = __shfl_xor(best_idx, off, WAVE_SIZE);
        if (ov > best || (ov == best && oi < best_idx)) {
            best = ov;
            best_idx = oi;
        }
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
        float bv = sv[0];
        int bi = si[0];
#pragma unroll
        for (int w = 1; w < 4; ++w) {
            if (sv[w] > bv || (sv[w] == bv && si[w] < bi)) {
                bv = sv[w];
                bi = si[w];
            }
        }
        out[b] = bi;
    }
}

extern "C" hipError_t launch_sample(
    int64_t* out, const float* logits, const float* temperature,
    const float* noise, int B, int V, hipStream_t stream) {
    sample_kernel<<<B, 256, 0, stream>>>(out, logits, temperature, noise, V);
    HIP_CHECK_LAST();
    return hipSuccess;
}
