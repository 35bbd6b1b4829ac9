"""The measured GEMM dispatch table (ops._m256_config) is data the
serving hot path depends on: these CPU tests pin which shapes route to
the custom macro-tile kernel vs the tuned library so a table edit can't
silently drop a measured entry (profiles/r02_gemm_m256_sweep.md)."""

import pytest

from llmapigateway_amd.ops import _m256_config


# llama-3-8b tp=1 decode projections (hidden 4096, inter 14336)
def test_8b_routing():
    assert _m256_config(256, 6144, 4096) is None        # qkv -> library
    assert _m256_config(256, 4096, 4096) is not None    # o -> custom
    assert _m256_config(256, 28672, 4096) is None       # gate_up -> library
    assert _m256_config(256, 4096, 14336) is not None   # down -> custom


# llama-3-70b tp=1 decode projections (hidden 8192, inter 28672)
def test_70b_routing():
    assert _m256_config(256, 10240, 8192) is None       # qkv -> library
    cfg = _m256_config(256, 8192, 8192)                 # o -> custom
    assert cfg == {"nf": 8, "nsk": 4, "variant": 0, "pipe": 0}
    assert _m256_config(256, 57344, 8192) is None       # gate_up -> library
    cfg = _m256_config(256, 8192, 28672)                # down -> custom
    assert cfg == {"nf": 8, "nsk": 4, "variant": 1, "pipe": 0}


# llama-3-8b tp=4 per-rank shard shapes
def test_8b_tp4_routing():
    assert _m256_config(256, 1536, 4096) is None        # qkv shard -> library
    cfg = _m256_config(256, 4096, 1024)                 # o shard -> custom
    assert cfg == {"nf": 4, "nsk": 1, "variant": 1, "pipe": 0}
    cfg = _m256_config(256, 7168, 4096)                 # gate_up shard -> custom
    assert cfg == {"nf": 4, "nsk": 3, "variant": 0, "pipe": 4}
    assert _m256_config(256, 4096, 3584) is not None    # down shard -> custom


def test_prefill_m_goes_to_library():
    # chunked-prefill token counts exceed the macro-tile M ceiling
    assert _m256_config(4096, 4096, 4096) is None


@pytest.mark.parametrize("N,K", [(4096, 4096), (4096, 14336), (8192, 8192), (8192, 28672)])
def test_custom_configs_are_launchable_shapes(N, K):
    cfg = _m256_config(256, N, K)
    assert cfg is not None
    assert N % (16 * cfg["nf"]) == 0          # whole column tiles
    assert (K // 64) % cfg["nsk"] == 0 or cfg["nsk"] <= (K // 64)  # split-K fits


def test_swiglu_config_routing():
    from llmapigateway_amd.ops import _m256_swiglu_config

    assert _m256_swiglu_config(256, 28672, 4096) == {
        "nf": 8, "variant": 1, "pipe": 0}
    assert _m256_swiglu_config(100, 28672, 4096) == {
        "nf": 8, "variant": 0, "pipe": 0}          # M<=128 -> glds variant
    assert _m256_swiglu_config(256, 7168, 4096) is not None   # tp4 shard
    assert _m256_swiglu_config(4, 28672, 4096) is None        # skinny regime
    assert _m256_swiglu_config(256, 57344, 8192) is None      # 70B: library


def test_interleave_gate_up_mapping():
    # block-16 interleave must place gate block b at rows 32b..32b+16 and
    # the matching up block right after it — the fused epilogue contract
    import torch

    from llmapigateway_amd import ops

    I, K = 64, 8
    w = torch.arange(2 * I * K, dtype=torch.float32).view(2 * I, K)
    wi = ops.interleave_gate_up(w)
    for b in range(I // 16):
        assert torch.equal(wi[32 * b : 32 * b + 16], w[16 * b : 16 * b + 16])
        assert torch.equal(
            wi[32 * b + 16 : 32 * b + 32], w[I + 16 * b : I + 16 * b + 16]
        )


def test_interleaved_fused_matches_reference_math():
    # silu(gate-block) * up-block over the interleaved product equals
    # swiglu of the plain product (pure torch; the GPU kernel epilogue
    # implements exactly this block mapping)
    import torch
    import torch.nn.functional as F

    from llmapigateway_amd import ops

    torch.manual_seed(3)
    M, N, K = 8, 128, 64
    x = torch.randn(M, K)
    w = torch.randn(N, K)
    y = x @ ops.interleave_gate_up(w).T
    fused = torch.cat(
        [
            F.silu(y[:, 32 * b : 32 * b + 16]) * y[:, 32 * b + 16 : 32 * b + 32]
            for b in range(N // 32)
        ],
        dim=-1,
    )
    g, u = (x @ w.T).chunk(2, dim=-1)
    assert torch.allclose(fused, F.silu(g) * u, atol=1e-5)
