// Torch extension binding for the gfx950 kernels (built in-tree by
// `python setup.py build_ext --inplace` with PYTORCH_ROCM_ARCH=gfx950).
// The kernels themselves live in the .hip files (torch-free, raw pointers);
// this file validates tensors, extracts strides and dispatches on the
// current HIP stream.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include <vector>

#include "kv_manager.cpp"

extern "C" {
hipError_t launch_rmsnorm(void*, const void*, const void*, float, int, int, hipStream_t);
hipError_t launch_rmsnorm_residual(void*, const void*, void*, const void*, float, int, int, hipStream_t);
hipError_t launch_rope(void*, void*, const int64_t*, const float*, int, int64_t, int64_t, int, int, int, hipStream_t);
hipError_t launch_swiglu(void*, const void*, int, int, hipStream_t);
hipError_t launch_kv_cache_write(const void*, const void*, void*, void*, float*, float*, const int64_t*, int, int64_t, int64_t, int, int, int, hipStream_t);
hipError_t launch_rope_kv(void*, void*, const void*, void*, void*, float*, float*, const int64_t*, const float*, const int64_t*, int, int64_t, int64_t, int64_t, int, int, int, int, hipStream_t);
hipError_t launch_attention_decode(void*, const void*, const void*, const void*, const float*, const float*, const int*, const int*, float, int, int, int, int, int, int, int64_t, float*, float*, int, int, hipStream_t);
hipError_t launch_attention_prefill(void*, const void*, const void*, const void*, const int*, const int*, const int*, int, float, int, int, int, int64_t, int64_t, int64_t, const void*, const void*, const float*, const float*, const int*, const int*, int, int, hipStream_t);
hipError_t launch_sample(int64_t*, const float*, const float*, const float*, float*, int*, int, int, hipStream_t);
hipError_t launch_topk_topp_filter(float*, const float*, const int*, int, int, hipStream_t);
hipError_t launch_gemm_skinny(void*, float*, const void*, const void*, int, int, int, int, int, hipStream_t);
hipError_t launch_gemm_m256(void*, float*, const void*, const void*, int, int, int, int, int, int, int, int, hipStream_t);
}

namespace {

#define CHECK_HIP(call)                                                        \
    do {                                                                       \
        hipError_t err = (call);                                               \
        TORCH_CHECK(err == hipSuccess, "HIP kernel failed: ",                  \
                    hipGetErrorString(err));                                   \
    } while (0)

hipStream_t current_stream() {
    return at::hip::getCurrentHIPStream().stream();
}

void check_bf16(const torch::Tensor& t, const char* name) {
    TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
    TORCH_CHECK(t.scalar_type() == torch::kBFloat16, name, " must be bf16");
}

// fp8 KV caches are stored as uint8 (e4m3 bits) with fp32 row scales
void check_cache(const torch::Tensor& t, const char* name) {
    TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
    TORCH_CHECK(t.scalar_type() == torch::kBFloat16 ||
                t.scalar_type() == torch::kUInt8,
                name, " must be bf16 or uint8(e4m3)");
}

float* scale_ptr(const c10::optional<torch::Tensor>& s) {
    if (!s.has_value() || !s->defined()) return nullptr;
    TORCH_CHECK(s->scalar_type() == torch::kFloat32 && s->is_contiguous());
    return s->data_ptr<float>();
}

void rmsnorm(torch::Tensor out, torch::Tensor x, torch::Tensor weight, double eps) {
    check_bf16(x, "x");
    check_bf16(out, "out");
    TORCH_CHECK(x.is_contiguous() && out.is_contiguous() && weight.is_contiguous());
    const int H = x.size(-1);
    const int T = x.numel() / H;
    CHECK_HIP(launch_rmsnorm(out.data_ptr(), x.data_ptr(), weight.data_ptr(),
                             (float)eps, T, H, current_stream()));
}

void rmsnorm_residual(torch::Tensor out, torch::Tensor x, torch::Tensor residual,
                      torch::Tensor weight, double eps) {
    check_bf16(x, "x");
    TORCH_CHECK(x.is_contiguous() && out.is_contiguous() && residual.is_contiguous());
    const int H = x.size(-1);
    const int T = x.numel() / H;
    CHECK_HIP(launch_rmsnorm_residual(out.data_ptr(), x.data_ptr(),
                                      residual.data_ptr(), weight.data_ptr(),
                                      (float)eps, T, H, current_stream()));
}

void rope_inplace(torch::Tensor q, torch::Tensor k, torch::Tensor positions,
                  torch::Tensor cos_sin) {
    check_bf16(q, "q");
    check_bf16(k, "k");
    TORCH_CHECK(q.dim() == 3 && k.dim() == 3, "q/k must be [T, H, D]");
    TORCH_CHECK(q.stride(2) == 1 && k.stride(2) == 1, "head_dim must be contiguous");
    TORCH_CHECK(q.stride(1) == q.size(2) && k.stride(1) == k.size(2),
                "heads must be contiguous within a row");
    TORCH_CHECK(positions.scalar_type() == torch::kInt64);
    TORCH_CHECK(cos_sin.scalar_type() == torch::kFloat32 && cos_sin.is_contiguous());
    const int T = q.size(0);
    CHECK_HIP(launch_rope(q.data_ptr(), k.data_ptr(),
                          positions.data_ptr<int64_t>(),
                          cos_sin.data_ptr<float>(), T, q.stride(0), k.stride(0),
                          q.size(1), k.size(1), q.size(2), current_stream()));
}

void swiglu(torch::Tensor out, torch::Tensor x) {
    check_bf16(x, "x");
    TORCH_CHECK(x.is_contiguous() && out.is_contiguous());
    const int I = out.size(-1);
    TORCH_CHECK(x.size(-1) == 2 * I, "x last dim must be 2*I");
    const int T = x.numel() / (2 * I);
    CHECK_HIP(launch_swiglu(out.data_ptr(), x.data_ptr(), T, I, current_stream()));
}

void kv_cache_write(torch::Tensor k, torch::Tensor v, torch::Tensor k_cache,
                    torch::Tensor v_cache, torch::Tensor slot_mapping,
                    c10::optional<torch::Tensor> k_scale,
                    c10::optional<torch::Tensor> v_scale) {
    check_bf16(k, "k");
    check_cache(k_cache, "k_cache");
    TORCH_CHECK(k.dim() == 3 && k_cache.dim() == 4);
    TORCH_CHECK(k.stride(2) == 1 && k.stride(1) == k.size(2));
    TORCH_CHECK(v.stride(2) == 1 && v.stride(1) == v.size(2));
    TORCH_CHECK(k_cache.is_contiguous() && v_cache.is_contiguous());
    TORCH_CHECK(slot_mapping.scalar_type() == torch::kInt64);
    const int T = k.size(0);
    const int Hkv = k_cache.size(1);
    const int block_size = k_cache.size(2);
    const int D = k_cache.size(3);
    TORCH_CHECK((k_cache.scalar_type() == torch::kUInt8) ==
                (k_scale.has_value() && k_scale->defined()),
                "fp8 cache needs scales; bf16 cache must not have them");
    CHECK_HIP(launch_kv_cache_write(
        k.data_ptr(), v.data_ptr(), k_cache.data_ptr(), v_cache.data_ptr(),
        scale_ptr(k_scale), scale_ptr(v_scale),
        slot_mapping.data_ptr<int64_t>(), T, k.stride(0), v.stride(0), Hkv,
        block_size, D, current_stream()));
}

void rope_kv_write(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                   torch::Tensor k_cache, torch::Tensor v_cache,
                   torch::Tensor positions, torch::Tensor cos_sin,
                   torch::Tensor slot_mapping,
                   c10::optional<torch::Tensor> k_scale,
                   c10::optional<torch::Tensor> v_scale) {
    check_bf16(q, "q");
    check_bf16(k, "k");
    check_cache(k_cache, "k_cache");
    TORCH_CHECK(q.dim() == 3 && k.dim() == 3 && v.dim() == 3 && k_cache.dim() == 4);
    TORCH_CHECK(q.stride(2) == 1 && k.stride(2) == 1 && v.stride(2) == 1);
    TORCH_CHECK(q.stride(1) == q.size(2) && k.stride(1) == k.size(2) &&
                v.stride(1) == v.size(2));
    TORCH_CHECK(k_cache.is_contiguous() && v_cache.is_contiguous());
    TORCH_CHECK(positions.scalar_type() == torch::kInt64);
    TORCH_CHECK(slot_mapping.scalar_type() == torch::kInt64);
    TORCH_CHECK(cos_sin.scalar_type() == torch::kFloat32 && cos_sin.is_contiguous());
    const int T = q.size(0);
    CHECK_HIP(launch_rope_kv(
        q.data_ptr(), k.data_ptr(), v.data_ptr(), k_cache.data_ptr(),
        v_cache.data_ptr(), scale_ptr(k_scale), scale_ptr(v_scale),
        positions.data_ptr<int64_t>(),
        cos_sin.data_ptr<float>(), slot_mapping.data_ptr<int64_t>(), T,
        q.stride(0), k.stride(0), v.stride(0), q.size(1), k_cache.size(1),
        q.size(2), k_cache.size(2), current_stream()));
}

void attention_decode(torch::Tensor out, torch::Tensor q, torch::Tensor k_cache,
                      torch::Tensor v_cache, torch::Tensor block_tables,
                      torch::Tensor context_lens, double scale,
                      c10::optional<torch::Tensor> part_acc,
                      c10::optional<torch::Tensor> part_ml, int64_t nsplit,
                      c10::optional<torch::Tensor> k_scale,
                      c10::optional<torch::Tensor> v_scale, int64_t version) {
    check_cache(k_cache, "k_cache");
    TORCH_CHECK((k_cache.scalar_type() == torch::kUInt8) ==
                (k_scale.has_value() && k_scale->defined()),
                "fp8 cache needs scales; bf16 cache must not have them");
    check_bf16(q, "q");
    check_bf16(out, "out");
    TORCH_CHECK(out.is_contiguous());
    TORCH_CHECK(q.stride(2) == 1 && q.stride(1) == q.size(2));
    TORCH_CHECK(block_tables.scalar_type() == torch::kInt32 && block_tables.is_contiguous());
    TORCH_CHECK(context_lens.scalar_type() == torch::kInt32);
    const int B = q.size(0);
    const int Hq = q.size(1);
    const int D = q.size(2);
    const int Hkv = k_cache.size(1);
    const int block_size = k_cache.size(2);
    const int max_blocks = block_tables.size(1);
    float* pa = nullptr;
    float* pm = nullptr;
    if (nsplit > 1) {
        TORCH_CHECK(part_acc.has_value() && part_ml.has_value());
        TORCH_CHECK(part_acc->scalar_type() == torch::kFloat32 &&
                    part_acc->numel() >= (int64_t)B * Hq * nsplit * D);
        TORCH_CHECK(part_ml->scalar_type() == torch::kFloat32 &&
                    part_ml->numel() >= (int64_t)B * Hq * nsplit * 2);
        pa = part_acc->data_ptr<float>();
        pm = part_ml->data_ptr<float>();
    }
    CHECK_HIP(launch_attention_decode(
        out.data_ptr(), q.data_ptr(), k_cache.data_ptr(), v_cache.data_ptr(),
        scale_ptr(k_scale), scale_ptr(v_scale),
        block_tables.data_ptr<int>(), context_lens.data_ptr<int>(), (float)scale,
        B, Hq, Hkv, block_size, max_blocks, D, q.stride(0), pa, pm, (int)nsplit,
        (int)version, current_stream()));
}

void attention_prefill(torch::Tensor out, torch::Tensor q, torch::Tensor k,
                       torch::Tensor v, torch::Tensor cu_seqlens,
                       torch::Tensor tile_seq, torch::Tensor tile_off,
                       double scale,
                       c10::optional<torch::Tensor> k_cache,
                       c10::optional<torch::Tensor> v_cache,
                       c10::optional<torch::Tensor> block_tables,
                       c10::optional<torch::Tensor> cached_lens,
                       c10::optional<torch::Tensor> k_scale,
                       c10::optional<torch::Tensor> v_scale) {
    check_bf16(q, "q");
    check_bf16(out, "out");
    TORCH_CHECK(out.is_contiguous());
    TORCH_CHECK(q.stride(2) == 1 && q.stride(1) == q.size(2));
    TORCH_CHECK(k.stride(2) == 1 && k.stride(1) == k.size(2));
    TORCH_CHECK(v.stride(2) == 1 && v.stride(1) == v.size(2));
    TORCH_CHECK(cu_seqlens.scalar_type() == torch::kInt32);
    TORCH_CHECK(tile_seq.scalar_type() == torch::kInt32 &&
                tile_off.scalar_type() == torch::kInt32);
    const int ntiles = tile_seq.size(0);
    const void* kc = nullptr;
    const void* vc = nullptr;
    const int* bt = nullptr;
    const int* cl = nullptr;
    int block_size = 0, max_blocks = 0;
    if (cached_lens.has_value() && cached_lens->defined()) {
        TORCH_CHECK(k_cache.has_value() && block_tables.has_value());
        TORCH_CHECK(cached_lens->scalar_type() == torch::kInt32);
        TORCH_CHECK(block_tables->scalar_type() == torch::kInt32 &&
                    block_tables->is_contiguous());
        TORCH_CHECK(k_cache->is_contiguous() && v_cache->is_contiguous());
        kc = k_cache->data_ptr();
        vc = v_cache->data_ptr();
        bt = block_tables->data_ptr<int>();
        cl = cached_lens->data_ptr<int>();
        block_size = k_cache->size(2);
        max_blocks = block_tables->size(1);
    }
    CHECK_HIP(launch_attention_prefill(
        out.data_ptr(), q.data_ptr(), k.data_ptr(), v.data_ptr(),
        cu_seqlens.data_ptr<int>(), tile_seq.data_ptr<int>(),
        tile_off.data_ptr<int>(), ntiles, (float)scale, q.size(1), k.size(1),
        q.size(2), q.stride(0), k.stride(0), v.stride(0), kc, vc,
        kc ? scale_ptr(k_scale) : nullptr, vc ? scale_ptr(v_scale) : nullptr,
        bt, cl, block_size, max_blocks, current_stream()));
}

void sample(torch::Tensor out, torch::Tensor logits, torch::Tensor temperature,
            c10::optional<torch::Tensor> noise) {
    TORCH_CHECK(logits.is_cuda() && logits.scalar_type() == torch::kFloat32 &&
                logits.is_contiguous());
    TORCH_CHECK(out.scalar_type() == torch::kInt64);
    TORCH_CHECK(temperature.scalar_type() == torch::kFloat32);
    const float* noise_ptr = nullptr;
    if (noise.has_value() && noise->defined()) {
        TORCH_CHECK(noise->is_contiguous() && noise->scalar_type() == torch::kFloat32);
        noise_ptr = noise->data_ptr<float>();
    }
    const int B = logits.size(0);
    auto part_val = torch::empty(
        {B, 16}, torch::TensorOptions().dtype(torch::kFloat32).device(logits.device()));
    auto part_idx = torch::empty(
        {B, 16}, torch::TensorOptions().dtype(torch::kInt32).device(logits.device()));
    CHECK_HIP(launch_sample(out.data_ptr<int64_t>(), logits.data_ptr<float>(),
                            temperature.data_ptr<float>(), noise_ptr,
                            part_val.data_ptr<float>(), part_idx.data_ptr<int>(),
                            B, logits.size(1), current_stream()));
}

void topk_topp_filter(torch::Tensor logits, torch::Tensor topp, torch::Tensor topk) {
    TORCH_CHECK(logits.is_cuda() && logits.scalar_type() == torch::kFloat32 &&
                logits.is_contiguous() && logits.dim() == 2);
    TORCH_CHECK(topp.scalar_type() == torch::kFloat32 && topp.is_contiguous());
    TORCH_CHECK(topk.scalar_type() == torch::kInt32 && topk.is_contiguous());
    const int B = logits.size(0), V = logits.size(1);
    TORCH_CHECK(topp.numel() == B && topk.numel() == B);
    CHECK_HIP(launch_topk_topp_filter(logits.data_ptr<float>(),
                                      topp.data_ptr<float>(),
                                      topk.data_ptr<int>(), B, V,
                                      current_stream()));
}

void gemm_skinny(torch::Tensor y, torch::Tensor x, torch::Tensor w,
                 c10::optional<torch::Tensor> workspace, int64_t nsk,
                 bool swizzled) {
    check_bf16(x, "x");
    check_bf16(w, "w");
    check_bf16(y, "y");
    TORCH_CHECK(x.is_contiguous() && w.is_contiguous() && y.is_contiguous());
    TORCH_CHECK(x.dim() == 2 && y.dim() == 2);
    const int M = x.size(0), K = x.size(1);
    int N;
    if (swizzled) {  // w is [K/32, N, 32]
        TORCH_CHECK(w.dim() == 3 && w.size(0) == K / 32 && w.size(2) == 32);
        N = w.size(1);
    } else {
        TORCH_CHECK(w.dim() == 2 && w.size(1) == K);
        N = w.size(0);
    }
    TORCH_CHECK(y.size(0) == M && y.size(1) == N);
    float* ws = nullptr;
    if (workspace.has_value() && workspace->defined()) {
        TORCH_CHECK(workspace->scalar_type() == torch::kFloat32 &&
                    workspace->numel() >= nsk * (int64_t)M * N,
                    "workspace must be fp32 with >= nsk*M*N elements");
        ws = workspace->data_ptr<float>();
    }
    CHECK_HIP(launch_gemm_skinny(y.data_ptr(), ws, x.data_ptr(), w.data_ptr(),
                                 M, N, K, (int)nsk, swizzled ? 1 : 0,
                                 current_stream()));
}

void gemm_m256(torch::Tensor y, torch::Tensor x, torch::Tensor w_frag,
               c10::optional<torch::Tensor> workspace, int64_t nsk,
               int64_t nf, int64_t variant, int64_t pipe, int64_t swiglu) {
    check_bf16(x, "x");
    check_bf16(w_frag, "w_frag");
    check_bf16(y, "y");
    TORCH_CHECK(x.is_contiguous() && w_frag.is_contiguous() && y.is_contiguous());
    TORCH_CHECK(x.dim() == 2 && y.dim() == 2);
    const int M = x.size(0), K = x.size(1);
    // fragment-major twin: [K/32, N/16, 64, 8]
    TORCH_CHECK(w_frag.dim() == 4 && w_frag.size(0) == K / 32 &&
                w_frag.size(2) == 64 && w_frag.size(3) == 8,
                "w_frag must be the fragment-major [K/32, N/16, 64, 8] twin");
    const int N = w_frag.size(1) * 16;
    // fused swiglu epilogue: w_frag is the block-16 interleaved gate/up
    // twin and y holds silu(gate)*up, half the columns
    TORCH_CHECK(y.size(0) == M && y.size(1) == (swiglu ? N / 2 : N));
    float* ws = nullptr;
    if (workspace.has_value() && workspace->defined()) {
        TORCH_CHECK(workspace->scalar_type() == torch::kFloat32 &&
                    workspace->numel() >= nsk * (int64_t)M * N,
                    "workspace must be fp32 with >= nsk*M*N elements");
        ws = workspace->data_ptr<float>();
    }
    CHECK_HIP(launch_gemm_m256(y.data_ptr(), ws, x.data_ptr(),
                               w_frag.data_ptr(), M, N, K, (int)nsk, (int)nf,
                               (int)variant, (int)pipe, (int)swiglu,
                               current_stream()));
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
    m.def("rmsnorm", &rmsnorm, "RMSNorm (gfx950)");
    m.def("rmsnorm_residual", &rmsnorm_residual, "fused residual-add RMSNorm");
    m.def("rope_inplace", &rope_inplace, "rotate-half RoPE in place");
    m.def("swiglu", &swiglu, "fused SiLU-gate multiply");
    m.def("kv_cache_write", &kv_cache_write, "paged KV cache scatter");
    m.def("rope_kv_write", &rope_kv_write, "fused RoPE + paged KV scatter");
    m.def("attention_decode", &attention_decode, "paged decode attention");
    m.def("attention_prefill", &attention_prefill, "varlen causal prefill attention");
    m.def("sample", &sample, "greedy / gumbel-max sampling");
    m.def("topk_topp_filter", &topk_topp_filter,
          "in-place top-k/top-p logit filtering (histogram threshold)");
    m.def("gemm_skinny", &gemm_skinny,
          "skinny-M weight-streaming GEMM y = x @ w.T (decode projections)");
    m.def("gemm_m256", &gemm_m256,
          "macro-tile LDS-staged decode GEMM y = x @ w.T (M <= 256)");

    pybind11::class_<BlockAllocator>(m, "BlockAllocator")
        .def(pybind11::init<int64_t>())
        .def("num_free", &BlockAllocator::num_free)
        .def("allocate", &BlockAllocator::allocate)
        .def("free", &BlockAllocator::free_blocks)
        .def_property_readonly("num_blocks", &BlockAllocator::capacity);
}
