"""GPU numerics tests: every gfx950 HIP kernel vs the PyTorch fp32 reference
(ops/reference.py) on the same bf16 inputs. Transpose-detecting inputs
(asymmetric random) per cdna_hip_programming.md §5.4 rule 16."""

import math

import pytest
import torch

from llmapigateway_amd import ops
from llmapigateway_amd.ops import reference

pytestmark = pytest.mark.gpu

DEV = "cuda:0"


@pytest.fixture(autouse=True, scope="module")
def _require_native():
    assert torch.cuda.is_available(), "GPU required"
    assert ops.have_native(), (
        "HIP extension is not loaded on a GPU box — the native path MUST run"
    )
    torch.manual_seed(0)


def to_f32(x):
    return x.float().cpu()


def max_rel_err(got, ref, eps=1e-3):
    got, ref = to_f32(got), to_f32(ref)
    return ((got - ref).abs() / (ref.abs() + eps)).max().item()


# ---------- rmsnorm ----------

@pytest.mark.parametrize("shape", [(7, 256), (64, 4096), (3, 8192), (256, 1024)])
def test_rmsnorm(shape):
    x = torch.randn(shape, dtype=torch.bfloat16, device=DEV)
    w = torch.randn(shape[-1], dtype=torch.bfloat16, device=DEV)
    got = ops.rmsnorm(x, w, 1e-5)
    ref = reference.rmsnorm(x.cpu(), w.cpu(), 1e-5)
    assert max_rel_err(got, ref) < 0.05


def test_rmsnorm_residual():
    x = torch.randn(33, 4096, dtype=torch.bfloat16, device=DEV)
    res = torch.randn_like(x)
    w = torch.randn(4096, dtype=torch.bfloat16, device=DEV)
    res_cpu = res.cpu().clone()
    y, new_res = ops.rmsnorm_residual(x, res, w, 1e-5)
    y_ref, res_ref = reference.rmsnorm_residual(x.cpu(), res_cpu, w.cpu(), 1e-5)
    assert max_rel_err(new_res, res_ref) < 0.05
    assert max_rel_err(y, y_ref) < 0.05


# ---------- rope ----------

def test_rope_strided():
    T, Hq, Hkv, D = 19, 8, 2, 128
    # emulate the fused-qkv view (rows strided)
    qkv = torch.randn(T, (Hq + 2 * Hkv) * D, dtype=torch.bfloat16, device=DEV)
    q = qkv[:, : Hq * D].view(T, Hq, D)
    k = qkv[:, Hq * D : (Hq + Hkv) * D].view(T, Hkv, D)
    pos = torch.randint(0, 500, (T,), device=DEV)
    cs = ops.build_rope_cache(512, D, 500000.0, device=DEV)
    q_cpu, k_cpu = q.cpu().clone(), k.cpu().clone()
    ops.rope_inplace(q, k, pos, cs)
    reference.rope_inplace(q_cpu, k_cpu, pos.cpu(), cs.cpu())
    assert max_rel_err(q, q_cpu) < 0.05
    assert max_rel_err(k, k_cpu) < 0.05


# ---------- swiglu ----------

def test_swiglu():
    x = torch.randn(37, 2 * 14336, dtype=torch.bfloat16, device=DEV)
    got = ops.swiglu(x)
    ref = reference.swiglu(x.cpu())
    assert max_rel_err(got, ref) < 0.05


# ---------- kv cache write ----------

def test_kv_cache_write():
    T, Hkv, BS, D, NB = 50, 8, 16, 128, 32
    qkv = torch.randn(T, 3 * Hkv * D, dtype=torch.bfloat16, device=DEV)
    k = qkv[:, : Hkv * D].view(T, Hkv, D)
    v = qkv[:, Hkv * D : 2 * Hkv * D].view(T, Hkv, D)
    kc = torch.zeros(NB, Hkv, BS, D, dtype=torch.bfloat16, device=DEV)
    vc = torch.zeros_like(kc)
    slots = torch.randperm(NB * BS, device=DEV)[:T]
    kc_ref, vc_ref = kc.cpu().clone(), vc.cpu().clone()
    ops.kv_cache_write(k, v, kc, vc, slots)
    reference.kv_cache_write(k.cpu(), v.cpu(), kc_ref, vc_ref, slots.cpu())
    assert torch.equal(kc.cpu(), kc_ref)
    assert torch.equal(vc.cpu(), vc_ref)


# ---------- prefill attention ----------

@pytest.mark.parametrize(
    "lens,Hq,Hkv",
    [
        ([64], 4, 4),          # exact one tile, MHA
        ([128], 8, 2),         # GQA 4
        ([1, 63, 64, 200], 8, 2),  # ragged varlen
        ([300], 32, 8),        # llama-3-8b head config
    ],
)
def test_attention_prefill(lens, Hq, Hkv):
    D = 128
    T = sum(lens)
    q = torch.randn(T, Hq, D, dtype=torch.bfloat16, device=DEV)
    k = torch.randn(T, Hkv, D, dtype=torch.bfloat16, device=DEV)
    v = torch.randn(T, Hkv, D, dtype=torch.bfloat16, device=DEV)
    cu = [0]
    for L in lens:
        cu.append(cu[-1] + L)
    cu_t = torch.tensor(cu, dtype=torch.int32, device=DEV)
    got = ops.attention_prefill(q, k, v, cu_t, max(lens))
    ref = reference.attention_prefill(q.cpu(), k.cpu(), v.cpu(), cu_t.cpu())
    err = (to_f32(got) - to_f32(ref)).abs().max().item()
    assert err < 0.05, f"max abs err {err}"


def test_attention_prefill_strided_qkv():
    # q/k/v as views into one fused projection row (the model's real layout)
    D, Hq, Hkv, L = 128, 8, 2, 100
    qkv = torch.randn(L, (Hq + 2 * Hkv) * D, dtype=torch.bfloat16, device=DEV)
    q = qkv[:, : Hq * D].view(L, Hq, D)
    k = qkv[:, Hq * D : (Hq + Hkv) * D].view(L, Hkv, D)
    v = qkv[:, (Hq + Hkv) * D :].view(L, Hkv, D)
    cu = torch.tensor([0, L], dtype=torch.int32, device=DEV)
    got = ops.attention_prefill(q, k, v, cu, L)
    ref = reference.attention_prefill(q.cpu(), k.cpu(), v.cpu(), cu.cpu())
    assert (to_f32(got) - to_f32(ref)).abs().max().item() < 0.05


# ---------- decode attention ----------

@pytest.mark.parametrize(
    "B,Hq,Hkv,BS,lens",
    [
        (1, 4, 4, 16, [1]),
        (4, 8, 2, 16, [5, 16, 33, 200]),
        (8, 32, 8, 64, [7, 64, 65, 100, 128, 250, 300, 512]),
        (3, 4, 4, 16, [9, 70, 130]),    # MHA (G=1, llama-2 family)
        (3, 28, 4, 16, [9, 70, 130]),   # odd GQA group (G=7, qwen2 family)
    ],
)
def test_attention_decode(B, Hq, Hkv, BS, lens):
    D = 128
    max_blocks = (max(lens) + BS - 1) // BS
    NB = B * max_blocks + 4
    kc = torch.randn(NB, Hkv, BS, D, dtype=torch.bfloat16, device=DEV)
    vc = torch.randn_like(kc)
    # shuffled block tables
    perm = torch.randperm(NB)[: B * max_blocks].view(B, max_blocks).int().to(DEV)
    q = torch.randn(B, Hq, D, dtype=torch.bfloat16, device=DEV)
    ctx = torch.tensor(lens, dtype=torch.int32, device=DEV)
    got = ops.attention_decode(q, kc, vc, perm, ctx)
    ref = reference.attention_decode(q.cpu(), kc.cpu(), vc.cpu(), perm.cpu(), ctx.cpu())
    err = (to_f32(got) - to_f32(ref)).abs().max().item()
    assert err < 0.05, f"max abs err {err}"


def test_decode_matches_prefill_last_row():
    """Decode of the last position == prefill's last row (same K/V)."""
    D, Hq, Hkv, L, BS = 128, 8, 2, 90, 16
    q = torch.randn(L, Hq, D, dtype=torch.bfloat16, device=DEV)
    k = torch.randn(L, Hkv, D, dtype=torch.bfloat16, device=DEV)
    v = torch.randn(L, Hkv, D, dtype=torch.bfloat16, device=DEV)
    cu = torch.tensor([0, L], dtype=torch.int32, device=DEV)
    pre = ops.attention_prefill(q, k, v, cu, L)

    nb = (L + BS - 1) // BS
    kc = torch.zeros(nb + 2, Hkv, BS, D, dtype=torch.bfloat16, device=DEV)
    vc = torch.zeros_like(kc)
    table = torch.arange(1, nb + 1, dtype=torch.int32, device=DEV)  # skip block 0
    slots = (table[torch.arange(L, device=DEV) // BS] * BS + torch.arange(L, device=DEV) % BS).long()
    ops.kv_cache_write(k, v, kc, vc, slots)
    dec = ops.attention_decode(
        q[L - 1 : L], kc, vc, table.unsqueeze(0), torch.tensor([L], dtype=torch.int32, device=DEV)
    )
    err = (to_f32(dec[0]) - to_f32(pre[L - 1])).abs().max().item()
    assert err < 0.03, f"decode vs prefill last-row err {err}"


# ---------- sampling ----------

def test_sample_greedy():
    B, V = 16, 128256
    logits = torch.randn(B, V, device=DEV)
    temps = torch.zeros(B, device=DEV)
    got = ops.sample(logits, temps)
    ref = logits.argmax(dim=-1)
    assert torch.equal(got, ref)


def _assert_sample_equiv(got, logits, temps, noise):
    """GPU argmax must pick an entry whose gumbel-perturbed score ties the
    CPU reference winner (logf rounding can flip exact index on near-ties)."""
    ref = reference.sample(logits.cpu(), temps.cpu(), noise.cpu())
    lf = logits.float().cpu()
    u = noise.float().cpu().clamp_min(1e-20)
    g = -torch.log((-torch.log(u)).clamp_min(1e-20))
    for b in range(logits.shape[0]):
        if temps[b].item() <= 0:
            assert got[b].item() == ref[b].item()
            continue
        scored = lf[b] / temps[b].item() + g[b]
        assert scored[got[b].item()] >= scored[ref[b].item()] - 1e-3


def test_sample_temperature_matches_reference():
    B, V = 8, 50000
    logits = torch.randn(B, V, device=DEV)
    temps = torch.full((B,), 0.8, device=DEV)
    noise = torch.rand(B, V, device=DEV)
    got = ops.sample(logits, temps, noise).cpu()
    _assert_sample_equiv(got, logits, temps, noise)


def test_sample_mixed_greedy_and_temp():
    B, V = 4, 1000
    logits = torch.randn(B, V, device=DEV)
    temps = torch.tensor([0.0, 1.0, 0.0, 0.5], device=DEV)
    noise = torch.rand(B, V, device=DEV)
    got = ops.sample(logits, temps, noise).cpu()
    _assert_sample_equiv(got, logits, temps, noise)


# ---------- skinny-M streaming GEMM (gemm_skinny.hip) ----------

@pytest.mark.parametrize(
    "M,N,K",
    [
        (1, 6144, 4096),     # qkv, single seq (split-K 2)
        (7, 4096, 4096),     # o proj, odd M (split-K 2)
        (64, 4096, 14336),   # down proj (split-K 4)
        (200, 28672, 4096),  # gate_up, odd M (no split-K)
        (256, 4096, 14336),  # down proj, full M (split-K 4)
        (256, 128256, 4096), # lm_head (no split-K)
    ],
)
def test_gemm_skinny(M, N, K):
    torch.manual_seed(M * 31 + N)
    x = (torch.randn(M, K, device=DEV) * 0.5).bfloat16()
    w = (torch.randn(N, K, device=DEV) * 0.02).bfloat16()
    got = ops.linear(x, w)
    ref = (x.float() @ w.float().T)
    # bf16 output rounding on a fp32 accumulation: compare against the
    # fp32 reference with a tolerance scaled to the row norms
    err = (got.float() - ref).abs().max().item()
    scale = ref.abs().max().item() + 1e-3
    assert err / scale < 0.02, f"max_err={err} scale={scale}"


def test_gemm_skinny_graph_replay_rearms_workspace():
    # the split-K workspace must be re-zeroed by the kernel itself so
    # repeated launches (hipGraph replays) stay correct
    M, N, K = 32, 4096, 14336
    x = (torch.randn(M, K, device=DEV) * 0.5).bfloat16()
    w = (torch.randn(N, K, device=DEV) * 0.02).bfloat16()
    ref = x.float() @ w.float().T
    for _ in range(3):
        got = ops.linear(x, w)
        err = (got.float() - ref).abs().max().item()
        assert err / (ref.abs().max().item() + 1e-3) < 0.02


# ---------- macro-tile decode GEMM (gemm_m256.hip) ----------

@pytest.mark.parametrize(
    "M,N,K,nf",
    [
        (1, 6144, 4096, 4),      # MW=1, split-K
        (7, 4096, 4096, 4),      # odd M within one wave
        (33, 4096, 14336, 4),    # MW=2, K=14336 (224 k-tiles)
        (200, 28672, 4096, 4),   # MW=8 partial tail rows, no split-K
        (256, 6144, 4096, 4),    # qkv headline shape
        (256, 4096, 14336, 4),   # down proj, split-K
        (256, 28672, 4096, 8),   # gate_up with BN=128 tiles
        (100, 4096, 4096, 8),    # BN=128 + tail rows + split-K
    ],
)
def test_gemm_m256(M, N, K, nf):
    torch.manual_seed(M * 13 + N + nf)
    x = (torch.randn(M, K, device=DEV) * 0.5).bfloat16()
    w = (torch.randn(N, K, device=DEV) * 0.02).bfloat16()
    wf = ops.swizzle_weight_frag(w)
    ref = x.float() @ w.float().T
    for variant in (0, 1, 2):  # glds / register-staged / producer-consumer
        got = ops.gemm_m256(x, wf, nf=nf, variant=variant)
        err = (got.float() - ref).abs().max().item()
        scale = ref.abs().max().item() + 1e-3
        assert err / scale < 0.02, f"v{variant} max_err={err} scale={scale}"


def test_gemm_m256_via_linear_twin_dispatch():
    # linear() must route through the macro-tile kernel when the
    # fragment-major twin is passed, and match the library numerically
    M, N, K = 128, 6144, 4096
    torch.manual_seed(5)
    x = (torch.randn(M, K, device=DEV) * 0.5).bfloat16()
    w = (torch.randn(N, K, device=DEV) * 0.02).bfloat16()
    wf = ops.swizzle_weight_frag(w)
    got = ops.linear(x, w, wf)
    ref = x.float() @ w.float().T
    err = (got.float() - ref).abs().max().item() / (ref.abs().max().item() + 1e-3)
    assert err < 0.02


@pytest.mark.parametrize("N,K", [(8192, 8192), (8192, 28672)])
def test_gemm_m256_70b_dispatch(N, K):
    # 70B-class projections (hidden 8192) route to the custom kernel per
    # the measured table (profiles/r02_gemm_m256_sweep.md)
    from llmapigateway_amd.ops import _m256_config

    assert _m256_config(256, N, K) is not None
    M = 256
    torch.manual_seed(N + K)
    x = (torch.randn(M, K, device=DEV) * 0.5).bfloat16()
    w = (torch.randn(N, K, device=DEV) * 0.02).bfloat16()
    wf = ops.swizzle_weight_frag(w)
    got = ops.linear(x, w, wf)
    ref = x.float() @ w.float().T
    err = (got.float() - ref).abs().max().item() / (ref.abs().max().item() + 1e-3)
    assert err < 0.02


@pytest.mark.parametrize(
    "M,variant,nf",
    [(256, 0, 8), (256, 1, 8), (100, 1, 8), (100, 0, 8), (64, 0, 4)],
)
def test_gemm_m256_fused_swiglu(M, variant, nf):
    # fused gate_up+swiglu epilogue vs plain torch: silu(x@g.T) * (x@u.T)
    N, K = 28672, 4096
    torch.manual_seed(M + variant)
    x = (torch.randn(M, K, device=DEV) * 0.5).bfloat16()
    w = (torch.randn(N, K, device=DEV) * 0.02).bfloat16()
    wi = ops.swizzle_weight_frag(ops.interleave_gate_up(w))
    got = ops.gemm_m256_swiglu(x, wi, nf=nf, variant=variant)
    g, u = (x.float() @ w.float().T).chunk(2, dim=-1)
    ref = torch.nn.functional.silu(g) * u
    err = (got.float() - ref).abs().max().item() / (ref.abs().max().item() + 1e-3)
    assert got.shape == (M, N // 2)
    assert err < 0.02


def test_swiglu_linear_routes_fused():
    M, N, K = 128, 28672, 4096
    torch.manual_seed(11)
    x = (torch.randn(M, K, device=DEV) * 0.5).bfloat16()
    w = (torch.randn(N, K, device=DEV) * 0.02).bfloat16()
    wi = ops.swizzle_weight_frag(ops.interleave_gate_up(w))
    assert ops._m256_swiglu_config(M, N, K) is not None
    got = ops.swiglu_linear(x, w, None, wi)
    ref = ops.swiglu(torch.nn.functional.linear(x, w))
    err = (got.float() - ref.float()).abs().max().item() / (
        ref.float().abs().max().item() + 1e-3
    )
    assert err < 0.02


def test_gemm_m256_repeat_launches_stable():
    # split-K slab reuse across launches (hipGraph replay pattern)
    M, N, K = 64, 4096, 14336
    x = (torch.randn(M, K, device=DEV) * 0.5).bfloat16()
    w = (torch.randn(N, K, device=DEV) * 0.02).bfloat16()
    wf = ops.swizzle_weight_frag(w)
    ref = x.float() @ w.float().T
    for _ in range(3):
        got = ops.gemm_m256(x, wf)
        err = (got.float() - ref).abs().max().item()
        assert err / (ref.abs().max().item() + 1e-3) < 0.02


def test_rope_and_kv_write_matches_separate_ops():
    torch.manual_seed(7)
    T, Hq, Hkv, D, BS, NB = 9, 8, 2, 128, 16, 8
    qkv = torch.randn(T, (Hq + 2 * Hkv) * D, dtype=torch.bfloat16, device=DEV)
    q = qkv[:, : Hq * D].view(T, Hq, D)
    k = qkv[:, Hq * D : (Hq + Hkv) * D].view(T, Hkv, D)
    v = qkv[:, (Hq + Hkv) * D :].view(T, Hkv, D)
    q2, k2, v2 = q.clone().contiguous(), k.clone().contiguous(), v.clone().contiguous()
    kc = torch.zeros(NB, Hkv, BS, D, dtype=torch.bfloat16, device=DEV)
    vc = torch.zeros_like(kc)
    kc2, vc2 = kc.clone(), vc.clone()
    pos = torch.arange(T, dtype=torch.int64, device=DEV)
    slots = torch.tensor([3, 17, 40, -1, 9, 100, 55, 2, 77], dtype=torch.int64, device=DEV)
    cs = ops.build_rope_cache(64, D, 10000.0, device=DEV)

    ops.rope_and_kv_write(q, k, v, kc, vc, pos, cs, slots)
    ops.rope_inplace(q2, k2, pos, cs)
    ops.kv_cache_write(k2, v2, kc2, vc2, slots)
    torch.cuda.synchronize()
    assert torch.equal(q.contiguous(), q2) and torch.equal(k.contiguous(), k2)
    assert torch.equal(kc, kc2) and torch.equal(vc, vc2)


# ---------- prefill attention with cached context (prefix caching) ----------

@pytest.mark.parametrize("cached,fresh", [(64, 64), (48, 80), (128, 1), (0, 96)])
def test_attention_prefill_cached_context(cached, fresh):
    torch.manual_seed(cached * 7 + fresh)
    Hq, Hkv, D, BS, NB = 8, 2, 128, 16, 64
    L = cached + fresh
    # build full-sequence K/V, scatter the cached prefix into a paged cache
    kf = torch.randn(L, Hkv, D, dtype=torch.bfloat16, device=DEV)
    vf = torch.randn(L, Hkv, D, dtype=torch.bfloat16, device=DEV)
    qf = torch.randn(fresh, Hq, D, dtype=torch.bfloat16, device=DEV)
    kc = torch.zeros(NB, Hkv, BS, D, dtype=torch.bfloat16, device=DEV)
    vc = torch.zeros_like(kc)
    nblocks = (cached + BS - 1) // BS
    bt = torch.arange(3, 3 + max(nblocks, 1), dtype=torch.int32, device=DEV).unsqueeze(0)
    for pos in range(cached):
        blk = int(bt[0, pos // BS])
        kc[blk, :, pos % BS] = kf[pos]
        vc[blk, :, pos % BS] = vf[pos]
    cu = torch.tensor([0, fresh], dtype=torch.int32, device=DEV)
    got = ops.attention_prefill(
        qf, kf[cached:], vf[cached:], cu, fresh,
        k_cache=kc, v_cache=vc, block_tables=bt,
        cached_lens=torch.tensor([cached], dtype=torch.int32, device=DEV),
    )
    # reference: full causal attention over [cached | fresh], take fresh rows
    from llmapigateway_amd.ops import reference

    qfull = torch.zeros(L, Hq, D, dtype=torch.bfloat16)
    qfull[cached:] = qf.cpu()
    ref_full = reference.attention_prefill(
        qfull, kf.cpu(), vf.cpu(), torch.tensor([0, L], dtype=torch.int32)
    )
    ref = ref_full[cached:]
    err = (got.float().cpu() - ref.float()).abs().max().item()
    assert err < 0.03, f"cached={cached} fresh={fresh} err={err}"


def test_engine_prefix_hit_matches_uncached_gpu():
    from llmapigateway_amd.engine import LLMEngine, SamplingParams
    from llmapigateway_amd.models.configs import ModelConfig

    cfg = ModelConfig(
        name="gpu-tiny-pc", hidden_size=512, intermediate_size=1024,
        num_layers=2, num_heads=4, num_kv_heads=2, vocab_size=2048,
        head_dim=128, rope_theta=10000.0, max_positions=1024,
    )
    prompt = list(range(7, 87))  # 80 tokens -> 4 full blocks of 16
    def mk(prefix):
        return LLMEngine(
            model=cfg, device="cuda:0", dtype=torch.bfloat16,
            block_size=16, num_blocks=128, seed=5, prefix_caching=prefix,
        )
    clean = mk(False)
    ref = clean.generate(prompt, SamplingParams(max_tokens=8, ignore_eos=True))
    eng = mk(True)
    r1 = eng.generate(prompt, SamplingParams(max_tokens=8, ignore_eos=True))
    r2 = eng.generate(prompt, SamplingParams(max_tokens=8, ignore_eos=True))
    assert r1.num_cached == 0 and r2.num_cached == 64
    assert r1.out_ids == ref.out_ids == r2.out_ids


def test_attention_decode_split_context():
    """Small batch + long max context triggers the flash-decode split path
    (grid.z partials + combine kernel); results must match the reference."""
    torch.manual_seed(11)
    B, Hq, Hkv, BS, D = 2, 8, 2, 64, 128
    lens = [5000, 1800]  # crosses multiple 2048-token splits + one short
    max_blocks = (max(lens) + BS - 1) // BS
    NB = B * max_blocks + 2
    kc = torch.randn(NB, Hkv, BS, D, dtype=torch.bfloat16, device=DEV)
    vc = torch.randn_like(kc)
    perm = torch.randperm(NB)[: B * max_blocks].view(B, max_blocks).int().to(DEV)
    q = torch.randn(B, Hq, D, dtype=torch.bfloat16, device=DEV)
    ctx = torch.tensor(lens, dtype=torch.int32, device=DEV)
    # B*Hkv = 4 < 192 and max_ctx = 5056 > 2048 -> split path
    got = ops.attention_decode(q, kc, vc, perm, ctx)
    ref = reference.attention_decode(q.cpu(), kc.cpu(), vc.cpu(), perm.cpu(), ctx.cpu())
    err = (to_f32(got) - to_f32(ref)).abs().max().item()
    assert err < 0.05, f"split-path err {err}"


@pytest.mark.parametrize("M,N,K", [(7, 4096, 4096), (64, 4096, 14336), (256, 28672, 4096)])
def test_gemm_skinny_swizzled_matches_plain(M, N, K):
    torch.manual_seed(M + N)
    x = (torch.randn(M, K, device=DEV) * 0.5).bfloat16()
    w = (torch.randn(N, K, device=DEV) * 0.02).bfloat16()
    wz = ops.swizzle_weight(w)
    ref = x.float() @ w.float().T
    got = ops.linear(x, w, wz).float()
    err = (got - ref).abs().max().item() / (ref.abs().max().item() + 1e-3)
    assert err < 0.02, f"swizzled rel err {err}"


# ---------- top-k / top-p filter (sampling.hip histogram kernel) ----------

@pytest.mark.parametrize(
    "B,V,pk",
    [
        (8, 128256, [(0.9, 0), (1.0, 10), (0.5, 50), (1.0, 0),
                     (0.95, 100), (0.1, 0), (0.9, 5), (0.3, 1)]),
        (3, 4096, [(0.8, 0), (1.0, 1), (0.99, 2000)]),
    ],
)
def test_topk_topp_filter_matches_reference(B, V, pk):
    torch.manual_seed(B * V)
    logits = torch.randn(B, V, device=DEV) * 3.0
    topp = torch.tensor([p for p, _ in pk], dtype=torch.float32, device=DEV)
    topk = torch.tensor([k for _, k in pk], dtype=torch.int32, device=DEV)
    got = ops.topk_topp_filter(logits.clone(), topp, topk)
    ref = reference.topk_topp_filter(logits.cpu().clone(), topp.cpu(), topk.cpu())
    got_kept = (got > float("-inf")).cpu()
    ref_kept = ref > float("-inf")
    # the kernel's cut has 3.8e-6-logit granularity: identical kept sets
    # on continuous random logits (ties at that scale are measure-zero)
    assert torch.equal(got_kept, ref_kept), (
        f"kept mismatch: {got_kept.sum(1).tolist()} vs {ref_kept.sum(1).tolist()}"
    )
    # surviving entries keep their exact values
    assert torch.equal(got.cpu()[got_kept], logits.cpu()[got_kept])


def test_topk_topp_filter_untouched_rows():
    V = 8192
    logits = torch.randn(2, V, device=DEV)
    orig = logits.clone()
    topp = torch.tensor([1.0, 0.5], device=DEV)
    topk = torch.tensor([0, 4], dtype=torch.int32, device=DEV)
    got = ops.topk_topp_filter(logits, topp, topk)
    assert torch.equal(got[0], orig[0])  # disabled row untouched
    assert (got[1] > float("-inf")).sum().item() <= 4


def test_topk_topp_filter_argmax_survives():
    # extreme p: the argmax token must always remain sampleable
    logits = torch.randn(4, 128256, device=DEV)
    topp = torch.full((4,), 1e-6, device=DEV)
    topk = torch.zeros(4, dtype=torch.int32, device=DEV)
    am = logits.argmax(dim=-1)
    got = ops.topk_topp_filter(logits.clone(), topp, topk)
    for b in range(4):
        assert got[b, am[b]] > float("-inf")


# ---------- fp8 (e4m3) KV cache paths ----------

def _mk_fp8_cache(NB, Hkv, BS, D):
    kc = torch.zeros(NB, Hkv, BS, D, dtype=torch.uint8, device=DEV)
    vc = torch.zeros_like(kc)
    ks = torch.ones(NB, Hkv, BS, dtype=torch.float32, device=DEV)
    vs = torch.ones_like(ks)
    return kc, vc, ks, vs


def test_fp8_kv_cache_write_matches_reference():
    torch.manual_seed(11)
    T, Hkv, BS, D, NB = 40, 8, 16, 128, 32
    k = torch.randn(T, Hkv, D, dtype=torch.bfloat16, device=DEV)
    v = torch.randn_like(k)
    kc, vc, ks, vs = _mk_fp8_cache(NB, Hkv, BS, D)
    slots = torch.randperm(NB * BS, device=DEV)[:T]
    ops.kv_cache_write(k, v, kc, vc, slots, k_scale=ks, v_scale=vs)

    kc_r = torch.zeros(NB, Hkv, BS, D, dtype=torch.uint8)
    vc_r = torch.zeros_like(kc_r)
    ks_r = torch.ones(NB, Hkv, BS)
    vs_r = torch.ones_like(ks_r)
    reference.kv_cache_write(k.cpu(), v.cpu(), kc_r, vc_r, slots.cpu(),
                             k_scale=ks_r, v_scale=vs_r)
    # compare DEQUANTIZED values (GPU e4m3 rounding may differ by 1 ulp)
    got_k = reference.fp8_dequantize_rows(kc.cpu(), ks.cpu())
    ref_k = reference.fp8_dequantize_rows(kc_r, ks_r)
    err = (got_k - ref_k).abs().max().item()
    assert err < 0.05, err
    # scales agree
    assert (ks.cpu() - ks_r).abs().max().item() < 1e-3


def test_fp8_rope_kv_write_roundtrip():
    torch.manual_seed(3)
    T, Hq, Hkv, D, BS, NB = 17, 8, 2, 128, 16, 8
    qkv = torch.randn(T, (Hq + 2 * Hkv) * D, dtype=torch.bfloat16, device=DEV)
    q = qkv[:, : Hq * D].view(T, Hq, D)
    k = qkv[:, Hq * D : (Hq + Hkv) * D].view(T, Hkv, D)
    v = qkv[:, (Hq + Hkv) * D :].view(T, Hkv, D)
    kc, vc, ks, vs = _mk_fp8_cache(NB, Hkv, BS, D)
    pos = torch.arange(T, dtype=torch.int64, device=DEV)
    slots = torch.randperm(NB * BS, device=DEV)[:T].long()
    cs = ops.build_rope_cache(64, D, 10000.0, device=DEV)
    k_before = k.clone()
    ops.rope_and_kv_write(q, k, v, kc, vc, pos, cs, slots, k_scale=ks, v_scale=vs)
    # the roped k rows, dequantized from the cache, match the in-place k
    deq = reference.fp8_dequantize_rows(kc.cpu(), ks.cpu())
    blocks = (slots // BS).cpu()
    offs = (slots % BS).cpu()
    got = deq[blocks, :, offs, :]
    rel = (got - k.float().cpu()).abs().max() / (k.float().abs().max() + 1e-6)
    assert rel < 0.05, rel
    assert not torch.equal(k, k_before)  # rope really ran


@pytest.mark.parametrize("B,lens", [(4, [5, 16, 33, 200]), (2, [7, 300])])
def test_fp8_attention_decode_matches_dequant_reference(B, lens):
    torch.manual_seed(B)
    Hq, Hkv, BS, D = 8, 2, 16, 128
    max_blocks = (max(lens) + BS - 1) // BS
    NB = B * max_blocks + 2
    bits = torch.randint(0, 255, (NB, Hkv, BS, D), dtype=torch.uint8, device=DEV)
    bits[bits == 127] = 0  # avoid e4m3 NaN encodings (0x7f/0xff)
    bits[bits == 255] = 0
    kc, vc = bits, bits.flip(0).contiguous()
    ks = (torch.rand(NB, Hkv, BS, device=DEV) * 0.01 + 0.001)
    vs = (torch.rand(NB, Hkv, BS, device=DEV) * 0.01 + 0.001)
    perm = torch.randperm(NB)[: B * max_blocks].view(B, max_blocks).int().to(DEV)
    q = torch.randn(B, Hq, D, dtype=torch.bfloat16, device=DEV)
    ctx = torch.tensor(lens, dtype=torch.int32, device=DEV)
    got = ops.attention_decode(q, kc, vc, perm, ctx, k_scale=ks, v_scale=vs)
    ref = reference.attention_decode(
        q.cpu(), kc.cpu(), vc.cpu(), perm.cpu(), ctx.cpu(),
        k_scale=ks.cpu(), v_scale=vs.cpu(),
    )
    err = (to_f32(got) - to_f32(ref)).abs().max().item()
    assert err < 0.05, f"max abs err {err}"


def test_fp8_prefill_cached_context_matches_reference():
    torch.manual_seed(9)
    Hq, Hkv, D, BS, NB = 8, 2, 128, 16, 64
    cached, fresh = 48, 80
    kf = torch.randn(cached + fresh, Hkv, D, dtype=torch.bfloat16, device=DEV)
    vf = torch.randn_like(kf)
    qf = torch.randn(fresh, Hq, D, dtype=torch.bfloat16, device=DEV)
    kc, vc, ks, vs = _mk_fp8_cache(NB, Hkv, BS, D)
    nblocks = (cached + BS - 1) // BS
    bt = torch.arange(3, 3 + nblocks, dtype=torch.int32, device=DEV).unsqueeze(0)
    slots = (bt[0, torch.arange(cached, device=DEV) // BS] * BS
             + torch.arange(cached, device=DEV) % BS).long()
    ops.kv_cache_write(kf[:cached], vf[:cached], kc, vc, slots, k_scale=ks, v_scale=vs)
    cu = torch.tensor([0, fresh], dtype=torch.int32, device=DEV)
    got = ops.attention_prefill(
        qf, kf[cached:], vf[cached:], cu, fresh,
        k_cache=kc, v_cache=vc, block_tables=bt,
        cached_lens=torch.tensor([cached], dtype=torch.int32, device=DEV),
        k_scale=ks, v_scale=vs,
    )
    ref = reference.attention_prefill(
        qf.cpu(), kf[cached:].cpu(), vf[cached:].cpu(), cu.cpu(),
        k_cache=kc.cpu(), v_cache=vc.cpu(), block_tables=bt.cpu(),
        cached_lens=torch.tensor([cached], dtype=torch.int32),
        k_scale=ks.cpu(), v_scale=vs.cpu(),
    )
    err = (to_f32(got) - to_f32(ref)).abs().max().item()
    assert err < 0.06, err


def test_fp8_engine_decode_close_to_bf16_gpu():
    """End-to-end: an fp8-KV llama-3-8b engine's greedy output stays close
    to the bf16-KV engine (same seed/weights): first tokens agree."""
    import llmapigateway_amd.engine as E

    a = E.LLMEngine(model="llama-3-8b", device=DEV, dtype=torch.bfloat16,
                    max_batch_size=4, max_model_len=128, seed=3,
                    num_blocks=64)
    b = E.LLMEngine(model="llama-3-8b", device=DEV, dtype=torch.bfloat16,
                    max_batch_size=4, max_model_len=128, seed=3,
                    num_blocks=64, kv_dtype="fp8")
    prompt = list(range(5, 37))
    ra = a.generate(prompt, E.SamplingParams(max_tokens=8, ignore_eos=True))
    rb = b.generate(prompt, E.SamplingParams(max_tokens=8, ignore_eos=True))
    # the FIRST token's attention reads only fresh bf16 K/V (prefill), so
    # it must match exactly; later tokens read the quantized cache and
    # random-init logits are clustered enough that argmax may flip —
    # require determinism of the fp8 path instead
    assert ra.out_ids[0] == rb.out_ids[0], (ra.out_ids, rb.out_ids)
    rb2 = b.generate(prompt, E.SamplingParams(max_tokens=8, ignore_eos=True))
    assert rb2.out_ids == rb.out_ids
