"""Op dispatch: gfx950 HIP kernels on GPU, PyTorch reference on CPU.

The HIP extension (llmapigateway_amd/ops/_C*.so, built in-tree by
``python setup.py build_ext --inplace`` / __graft_entry__.build) is the ONLY
compute path on GPU tensors — if a CUDA tensor reaches an op and the
extension is missing, we raise instead of silently falling back to eager
PyTorch. CPU tensors use ops.reference (fp32 semantics contract).
"""

from __future__ import annotations

import os
from typing import Optional, Tuple

# Enable the pre-tuned hipBLASLt/rocBLAS GEMM selections for gfx950 (3.2x on
# the decode lm_head GEMM). torch appends the device ordinal before ".csv",
# so the shipped table is tunableop_gfx950<N>.csv and FILENAME names the
# base. Must be set before the first GEMM executes; honor user overrides.
_TUNE_BASE = os.path.join(os.path.dirname(os.path.abspath(__file__)), "tunableop_gfx950.csv")
if os.path.exists(_TUNE_BASE.replace(".csv", "0.csv")):
    os.environ.setdefault("PYTORCH_TUNABLEOP_ENABLED", "1")
    os.environ.setdefault("PYTORCH_TUNABLEOP_TUNING", "0")
    os.environ.setdefault("PYTORCH_TUNABLEOP_FILENAME", _TUNE_BASE)

import torch

# Belt-and-braces: activate TunableOp through the python API too (the env
# vars alone were not picked up when torch was imported first elsewhere).
if torch.version.hip and os.path.exists(_TUNE_BASE.replace(".csv", "0.csv")):
    try:
        import torch.cuda.tunable as _tunable

        _tunable.enable(True)
        # honor an explicit tuning request (tools/tune_gemms.py sets "1");
        # otherwise lock to lookup-only so serving never pays tuning cost
        if os.environ.get("PYTORCH_TUNABLEOP_TUNING") != "1":
            _tunable.tuning_enable(False)
            _tunable.set_filename(_TUNE_BASE, insert_device_ordinal=True)
            _tunable.read_file(_TUNE_BASE.replace(".csv", "0.csv"))
    except Exception:  # pragma: no cover - best effort
        pass

from . import reference
from .reference import build_rope_cache  # re-export (host-side table builder)

_C = None
_C_ERR: Optional[str] = None
try:
    import importlib

    _C = importlib.import_module("._C", __name__)
except ImportError as e:  # extension not built (CPU-only envs are fine)
    _C_ERR = str(e)


def have_native() -> bool:
    return _C is not None


def _native():
    if _C is None:
        raise RuntimeError(
            "llmapigateway_amd.ops._C (gfx950 HIP extension) is not built but a GPU tensor "
            "reached the ops layer. Build it with `python setup.py build_ext --inplace` "
            f"(import error: {_C_ERR})"
        )
    return _C


def rmsnorm(x: torch.Tensor, weight: torch.Tensor, eps: float = 1e-5) -> torch.Tensor:
    if x.is_cuda:
        out = torch.empty_like(x)
        _native().rmsnorm(out, x, weight, eps)
        return out
    return reference.rmsnorm(x, weight, eps)


def rmsnorm_residual(
    x: torch.Tensor, residual: torch.Tensor, weight: torch.Tensor, eps: float = 1e-5
) -> Tuple[torch.Tensor, torch.Tensor]:
    if x.is_cuda:
        # fused: residual += x; x_out = rmsnorm(residual) — one HBM pass
        out = torch.empty_like(x)
        _native().rmsnorm_residual(out, x, residual, weight, eps)
        return out, residual
    return reference.rmsnorm_residual(x, residual, weight, eps)


def rope_inplace(
    q: torch.Tensor, k: torch.Tensor, positions: torch.Tensor, cos_sin: torch.Tensor
) -> None:
    if q.is_cuda:
        _native().rope_inplace(q, k, positions, cos_sin)
        return
    reference.rope_inplace(q, k, positions, cos_sin)


def swiglu(x: torch.Tensor) -> torch.Tensor:
    if x.is_cuda:
        inter = x.shape[-1] // 2
        out = torch.empty(
            (*x.shape[:-1], inter), dtype=x.dtype, device=x.device
        )
        _native().swiglu(out, x)
        return out
    return reference.swiglu(x)


def rope_and_kv_write(
    q: torch.Tensor,
    k: torch.Tensor,
    v: torch.Tensor,
    k_cache: torch.Tensor,
    v_cache: torch.Tensor,
    positions: torch.Tensor,
    cos_sin: torch.Tensor,
    slot_mapping: torch.Tensor,
    k_scale: Optional[torch.Tensor] = None,
    v_scale: Optional[torch.Tensor] = None,
) -> None:
    """Fused rope_inplace(q, k) + kv_cache_write(k, v): one launch on GPU.
    With k_scale/v_scale the cache is fp8 e4m3 (per-row quantization)."""
    if q.is_cuda:
        _native().rope_kv_write(
            q, k, v, k_cache, v_cache, positions, cos_sin, slot_mapping,
            k_scale, v_scale,
        )
        return
    reference.rope_inplace(q, k, positions, cos_sin)
    reference.kv_cache_write(k, v, k_cache, v_cache, slot_mapping, k_scale, v_scale)


def kv_cache_write(
    k: torch.Tensor,
    v: torch.Tensor,
    k_cache: torch.Tensor,
    v_cache: torch.Tensor,
    slot_mapping: torch.Tensor,
    k_scale: Optional[torch.Tensor] = None,
    v_scale: Optional[torch.Tensor] = None,
) -> None:
    if k.is_cuda:
        _native().kv_cache_write(k, v, k_cache, v_cache, slot_mapping, k_scale, v_scale)
        return
    reference.kv_cache_write(k, v, k_cache, v_cache, slot_mapping, k_scale, v_scale)


QTILE = 64  # q rows per prefill workgroup (must match attention_prefill.hip)


def build_prefill_tiles(seqlens, device) -> Tuple[torch.Tensor, torch.Tensor]:
    """Host-side tile map for the prefill kernel: one entry per 64-row q tile."""
    tile_seq, tile_off = [], []
    for i, L in enumerate(seqlens):
        for off in range(0, int(L), QTILE):
            tile_seq.append(i)
            tile_off.append(off)
    return (
        torch.tensor(tile_seq, dtype=torch.int32, device=device),
        torch.tensor(tile_off, dtype=torch.int32, device=device),
    )


def attention_prefill(
    q: torch.Tensor,
    k: torch.Tensor,
    v: torch.Tensor,
    cu_seqlens: torch.Tensor,
    max_seqlen: int,
    scale: Optional[float] = None,
    causal: bool = True,
    tile_seq: Optional[torch.Tensor] = None,
    tile_off: Optional[torch.Tensor] = None,
    k_cache: Optional[torch.Tensor] = None,
    v_cache: Optional[torch.Tensor] = None,
    block_tables: Optional[torch.Tensor] = None,
    cached_lens: Optional[torch.Tensor] = None,
    k_scale: Optional[torch.Tensor] = None,
    v_scale: Optional[torch.Tensor] = None,
) -> torch.Tensor:
    """Varlen causal prefill. With cached_lens set, each sequence also
    attends (unmasked) to its first cached_lens[i] positions read from the
    paged KV cache via block_tables — the prefix-caching / chunked-prefill
    context phase."""
    if scale is None:
        scale = float(q.shape[-1]) ** -0.5
    if q.is_cuda:
        if not causal:
            raise NotImplementedError("GPU prefill kernel is causal-only")
        if tile_seq is None or tile_off is None:
            cu = cu_seqlens.cpu().tolist()
            seqlens = [cu[i + 1] - cu[i] for i in range(len(cu) - 1)]
            tile_seq, tile_off = build_prefill_tiles(seqlens, q.device)
        out = torch.empty_like(q)
        _native().attention_prefill(
            out, q, k, v, cu_seqlens.int(), tile_seq, tile_off, float(scale),
            k_cache, v_cache, block_tables, cached_lens, k_scale, v_scale,
        )
        return out
    return reference.attention_prefill(
        q, k, v, cu_seqlens, scale, causal,
        k_cache=k_cache, v_cache=v_cache,
        block_tables=block_tables, cached_lens=cached_lens,
        k_scale=k_scale, v_scale=v_scale,
    )


_SPLIT_SPAN = 2048  # keep in sync with SPLIT_SPAN in attention_decode.hip
# decode-attention geometry: 2 = shared-LDS-chunk / wave-per-head form
# (coalesced staging, no cross-wave combine); 1 = the round-1 form
_DECODE_VER = int(os.environ.get("LLMAPI_DECODE_VER", "2"))
_decode_ws: dict = {}
_decode_ws_retired: list = []


def _decode_scratch(device, n_acc: int, n_ml: int):
    ws = _decode_ws.get(device.index)
    if ws is None or ws[0].numel() < n_acc or ws[1].numel() < n_ml:
        if ws is not None:
            _decode_ws_retired.append(ws)  # captured graphs may still use it
        ws = (
            torch.empty(n_acc, dtype=torch.float32, device=device),
            torch.empty(n_ml, dtype=torch.float32, device=device),
        )
        _decode_ws[device.index] = ws
    return ws


def attention_decode(
    q: torch.Tensor,
    k_cache: torch.Tensor,
    v_cache: torch.Tensor,
    block_tables: torch.Tensor,
    context_lens: torch.Tensor,
    scale: Optional[float] = None,
    k_scale: Optional[torch.Tensor] = None,
    v_scale: Optional[torch.Tensor] = None,
) -> torch.Tensor:
    if scale is None:
        scale = float(q.shape[-1]) ** -0.5
    if q.is_cuda:
        out = torch.empty_like(q)
        # flash-decode context splits when (B x Hkv) underfills the 256 CUs
        # but the block table allows long contexts; derived only from
        # capture-stable shapes so hipGraph replays stay valid
        B, Hq = q.shape[0], q.shape[1]
        max_ctx = block_tables.shape[1] * k_cache.shape[2]
        nsplit = 1
        if B * k_cache.shape[1] < 192 and max_ctx > _SPLIT_SPAN:
            nsplit = min(8, -(-max_ctx // _SPLIT_SPAN))
        if nsplit > 1:
            pa, pm = _decode_scratch(
                q.device, B * Hq * nsplit * q.shape[2], B * Hq * nsplit * 2
            )
            # measured: the v1 geometry wins once the flash-decode split
            # path engages (small-batch long-context), v2 everywhere else
            _native().attention_decode(
                out, q, k_cache, v_cache, block_tables, context_lens,
                float(scale), pa, pm, nsplit, k_scale, v_scale, 1,
            )
        else:
            _native().attention_decode(
                out, q, k_cache, v_cache, block_tables, context_lens,
                float(scale), None, None, 1, k_scale, v_scale, _DECODE_VER,
            )
        return out
    return reference.attention_decode(
        q, k_cache, v_cache, block_tables, context_lens, scale,
        k_scale=k_scale, v_scale=v_scale,
    )


# The 8-wave MF=2 depth-4 pipeline ties the tuned library when W is
# L3-resident (tools/gemm_probe.py rotates too little data for >100 MB
# weights, so its M>=64 rows are L3-warm and misled an M<=256 dispatch:
# in-engine, where the 16 GB weight cycle is always L3-cold, batch-256
# decode dropped 272 -> 129 req/s). In the true cold regime the library
# keeps winning at M >= 64; the custom kernel stays dispatched for the
# latency regime only.
_SKINNY_MAX_M = int(os.environ.get("SKINNY_GEMM_MAX_M", "16"))
_SKINNY_MAX_N = 28672
# per-device split-K fp32 slab scratch — each workgroup fully overwrites
# its slab stripe, so no zeroing is needed and the address is stable
# across hipGraph replays.
_skinny_ws: dict = {}


def _skinny_nsk(N: int, K: int) -> int:
    """Split-K factor: target exactly 256 workgroups (one 8-wave WG fills
    a CU; extra WGs only queue) — more splits just multiply the fp32 slab
    traffic, which at the old 512-WG target added ~60% to the weight
    stream and halved in-engine decode throughput."""
    tiles = N // 64
    nsk = max(1, -(-256 // tiles))
    nsk = min(nsk, max(1, (K // 32) // 8), 8)
    return nsk


_skinny_ws_retired: list = []  # keep old slabs alive for captured graphs


def _skinny_scratch(device, numel: int):
    ws = _skinny_ws.get(device.index)
    if ws is None or ws.numel() < numel:
        if ws is not None:
            _skinny_ws_retired.append(ws)  # a hipGraph may still replay into it
        ws = torch.empty(numel, dtype=torch.float32, device=device)
        _skinny_ws[device.index] = ws
    return ws


def swizzle_weight(w: torch.Tensor) -> torch.Tensor:
    """[N, K] -> k-major [K/32, N, 32] for gemm_skinny's contiguous
    B-tile streams (one 4 KB block per 64-row n-stripe per k-step)."""
    N, K = w.shape
    assert K % 32 == 0 and N % 64 == 0
    return w.view(N, K // 32, 32).permute(1, 0, 2).contiguous()


def swizzle_weight_frag(w: torch.Tensor) -> torch.Tensor:
    """[N, K] -> fragment-major [K/32, N/16, 64, 8] for gemm_m256.

    Element [k32][n16][lane][e] = W[n16*16 + (lane&15)][k32*32 + (lane>>4)*8
    + e]: each 1 KiB row is exactly one wave's mfma_f32_16x16x32_bf16
    B-fragment in lane order, so the kernel's global_load_lds staging is a
    straight contiguous copy and its ds_read_b128 is lane-linear
    (conflict-free per the gfx950 b128 lane groups)."""
    N, K = w.shape
    assert K % 64 == 0 and N % 64 == 0
    return (
        w.view(N // 16, 16, K // 32, 4, 8)
        .permute(2, 0, 3, 1, 4)
        .reshape(K // 32, N // 16, 64, 8)
        .contiguous()
    )


def _m256_config(M: int, N: int, K: int) -> Optional[dict]:
    """Measured dispatch table for the decode projections (cold-L3 sweep,
    profiles/r02_gemm_m256_sweep.md). Returns gemm_m256 kwargs, or None
    when the tuned library wins the shape (gate_up N=28672, lm_head)."""
    if M > 256:
        return None
    if N == 8192 and K % 64 == 0 and K >= 8192:
        # 70B-class projections at tp=1 (hidden 8192): o-proj K=8192
        # (51.2 vs library 58.2 us) and down-proj K=28672 (147.5 vs
        # 192.4 us) — register-staged variant wins the deep-K shape
        if K > 8192:
            return {"nf": 8, "nsk": 4, "variant": 1, "pipe": 0}  # 70B down 1.30x
        return {"nf": 8, "nsk": 4, "variant": 0, "pipe": 0}      # 70B o 1.14x
    if N == 7168 and K <= 8192:
        # tp=4 gate_up shard (unfused path): 37.3 vs library 43.4 us
        return {"nf": 4, "nsk": 3, "variant": 0, "pipe": 4}
    if N > 4096:
        # library wins the very wide shapes: gate_up N=28672/57344
        # (custom best 73.3 vs 70.3 us / 287.9 vs 258.0), qkv N=6144
        # (36.4 vs ~28 us — hipBLASLt's MT112x256 kernel is strong
        # exactly there) and N=10240 (77.7 vs 66.7), lm_head
        return None
    if K > 8192 and N % 128 == 0:
        return {"nf": 8, "nsk": 8, "variant": 0, "pipe": 0}  # down 1.34x
    if N < 2048:
        # narrow column shards (tp=4 qkv N=1536): too few tiles to fill
        # the chip — library 1.8x faster (21.7 vs 39.7 us)
        return None
    if K < 2048:
        # shallow-K row shards (tp=4 o-proj K=1024): single-pass
        # register-staged wins 1.67x (12.7 vs 21.2 us); split-K slab
        # traffic would dominate this little work
        return {"nf": 4, "nsk": 1, "variant": 1, "pipe": 0}
    # o-proj class (N<=4096, K<=8192): BK64/NBUF2 2-blocks/CU + split-K
    tiles = N // 64
    nsk = max(1, min(-(-256 // tiles), (K // 64) // 2, 8))
    return {"nf": 4, "nsk": nsk, "variant": 0, "pipe": 4}  # o 1.30x


def _m256_nsk(N: int, K: int, nf: int) -> int:
    """Split-K factor for gemm_m256: fill the 256 CUs (one 8-wave block
    per CU at the 120 KB LDS ring) when the column-tile count alone
    cannot. The fp32 slab traffic (2*nsk*M*N*4 B) is the price, so tiles
    >= 256 never split; smaller-N shapes are load-path/latency-bound and
    the slab round trip is cheap relative to the CU-fill win
    (profiles/r02_gemm_m256_probe.md)."""
    tiles = N // (16 * nf)
    if tiles >= 256:
        return 1
    nsk = -(-256 // tiles)
    return max(1, min(nsk, (K // 64) // 2, 8))


# swizzled weights raise the profitable dispatch ceiling (contiguous
# streams); plain-layout dispatch stays in the latency regime
_SKINNY_SWZ_MAX_M = int(os.environ.get("SKINNY_GEMM_SWZ_MAX_M", "256"))


# 0 = glds-staged, 1 = register-staged T14 (see gemm_m256.hip header)
_M256_VARIANT = int(os.environ.get("LLMAPI_M256_VARIANT", "0"))


def gemm_m256(
    x: torch.Tensor, w_frag: torch.Tensor, nf: Optional[int] = None,
    nsk: Optional[int] = None, variant: Optional[int] = None,
    pipe: Optional[int] = None,
) -> torch.Tensor:
    """y = x @ w.T with w pre-swizzled fragment-major (swizzle_weight_frag).
    The macro-tile LDS-staged decode GEMM (csrc/gemm_m256.hip); M <= 256.
    pipe selects the DMA ring geometry for the glds variant:
    0=(BK64,N3) 1=(BK64,N4,nf4) 2=(BK32,N4,nf8) 3=(BK32,N6,nf8)."""
    M, K = x.shape
    N = w_frag.shape[1] * 16
    if nf is None:
        if N % 128 == 0 and M > 64:
            nf = 8  # deep BK32 ring needs BN=128
        else:
            nf = 8 if N // 64 >= 448 else 4
    if N % (16 * nf) != 0:
        nf = 4
    if nsk is None:
        nsk = _m256_nsk(N, K, nf)
    if variant is None:
        variant = _M256_VARIANT
    if pipe is None:
        pipe = 3 if (variant == 0 and nf == 8 and K % 32 == 0) else 0
    if variant == 0:
        if pipe in (2, 3) and nf != 8:
            pipe = 0
        if pipe in (1, 4, 5) and nf != 4:
            pipe = 0
    y = torch.empty((M, N), dtype=torch.bfloat16, device=x.device)
    ws = _skinny_scratch(x.device, nsk * M * N) if nsk > 1 else None
    _native().gemm_m256(y, x, w_frag, ws, nsk, nf, variant, pipe, 0)
    return y


def interleave_gate_up(w: torch.Tensor) -> torch.Tensor:
    """Row-permute a [gate; up] stacked weight [2I, K] into block-16
    interleaved order [g0..15, u0..15, g16..31, ...]: after
    swizzle_weight_frag, the gate/up accumulators for the same output
    column land in the SAME lane of adjacent 16-col fragments, which is
    what gemm_m256's fused swiglu epilogue consumes."""
    two_i, K = w.shape
    assert two_i % 32 == 0
    return w.view(2, two_i // 32, 16, K).transpose(0, 1).reshape(two_i, K)


def gemm_m256_swiglu(
    x: torch.Tensor, w_frag: torch.Tensor, nf: int = 8, variant: int = 1,
    pipe: int = 0,
) -> torch.Tensor:
    """silu(x @ gate.T) * (x @ up.T) in ONE kernel: w_frag is
    swizzle_weight_frag(interleave_gate_up(gate_up)) and the swiglu runs
    in the GEMM epilogue on the fp32 accumulators — the [M, 2I]
    intermediate round trip and the separate swiglu kernel disappear
    (~72 MB of HBM traffic per llama-3-8b layer at M=256)."""
    M, K = x.shape
    N = w_frag.shape[1] * 16
    y = torch.empty((M, N // 2), dtype=torch.bfloat16, device=x.device)
    _native().gemm_m256(y, x, w_frag, None, 1, nf, variant, pipe, 1)
    return y


def _m256_swiglu_config(M: int, N: int, K: int) -> Optional[dict]:
    """Measured dispatch for the FUSED gate_up+swiglu decode MLP front
    (profiles/r02_gemm_m256_sweep.md). N is the stacked 2*intermediate.
    Only the 8B shape measured ahead of library+swiglu (77.2 vs 79.4 us
    cold-L3); the 70B shape loses (304.8 vs 226.6) and stays unfused."""
    if os.environ.get("LLMAPI_NO_FUSED_SWIGLU"):
        return None
    if not (8 < M <= 256):
        return None
    if N == 28672 and K == 4096:  # llama-3-8b tp=1
        if M <= 128:  # 55.0 vs 61.8 us at M=128 (glds variant)
            return {"nf": 8, "variant": 0, "pipe": 0}
        return {"nf": 8, "variant": 1, "pipe": 0}
    if N == 7168 and K == 4096:  # tp=4 gate_up shard: 45.9 vs 46.8 us
        return {"nf": 4, "variant": 0, "pipe": 4}
    return None


def swiglu_linear(
    x: torch.Tensor, w: torch.Tensor,
    w_swz: Optional[torch.Tensor] = None,
    w_int: Optional[torch.Tensor] = None,
) -> torch.Tensor:
    """MLP front: swiglu(x @ w.T) with w = [gate; up] stacked. Routes to
    the fused gemm_m256 epilogue when the interleaved twin exists and the
    measured table says it wins; otherwise library GEMM + swiglu kernel."""
    if w_int is not None and x.dim() == 2:
        cfg = _m256_swiglu_config(x.shape[0], w_int.shape[1] * 16, x.shape[1])
        if cfg is not None:
            return gemm_m256_swiglu(x, w_int, **cfg)
    return swiglu(linear(x, w, w_swz))


def linear(
    x: torch.Tensor, w: torch.Tensor, w_swz: Optional[torch.Tensor] = None
) -> torch.Tensor:
    """y = x @ w.T. Decode-shaped bf16 GEMMs (M <= 256) go through the
    macro-tile gfx950 kernel (csrc/gemm_m256.hip) when the fragment-major
    weight twin exists (models/llama.py builds them), through the
    streaming skinny kernel (csrc/gemm_skinny.hip) in the latency regime
    otherwise; everything else through the TunableOp-tuned library GEMMs."""
    M = x.shape[0]
    if (
        x.is_cuda
        and x.dtype == torch.bfloat16
        and w.dtype == torch.bfloat16
        and 0 < M
        and w.shape[0] % 64 == 0
        and w.shape[0] <= _SKINNY_MAX_N
        and w.shape[1] % 32 == 0
        and x.is_contiguous()
        and w.is_contiguous()
    ):
        N, K = w.shape
        if (
            w_swz is not None
            and w_swz.dim() == 4
            and 8 < M <= 256
            and K % 64 == 0
        ):
            cfg = _m256_config(M, N, K)
            if cfg is not None:
                return gemm_m256(x, w_swz, **cfg)
            # fall through to library (measured faster for this shape)
            return torch.nn.functional.linear(x, w)
        if w_swz is not None and w_swz.dim() == 3 and M <= _SKINNY_SWZ_MAX_M:
            # legacy k-major twin -> streaming skinny kernel
            y = torch.empty((M, N), dtype=torch.bfloat16, device=x.device)
            nsk = _skinny_nsk(N, K)
            ws = _skinny_scratch(x.device, nsk * M * N) if nsk > 1 else None
            _native().gemm_skinny(y, x, w_swz, ws, nsk, True)
            return y
        if M <= _SKINNY_MAX_M:
            y = torch.empty((M, N), dtype=torch.bfloat16, device=x.device)
            nsk = _skinny_nsk(N, K)
            ws = _skinny_scratch(x.device, nsk * M * N) if nsk > 1 else None
            _native().gemm_skinny(y, x, w, ws, nsk, False)
            return y
    return torch.nn.functional.linear(x, w)


def topk_topp_filter(
    logits: torch.Tensor, topp: torch.Tensor, topk: torch.Tensor
) -> torch.Tensor:
    """Mask logits outside the per-row top-k / top-p set to -inf, in
    place, and return logits. GPU: histogram-threshold kernel
    (csrc/sampling.hip, no sort, no host loop); CPU: batched torch sort
    (reference semantics)."""
    if logits.is_cuda:
        _native().topk_topp_filter(logits, topp, topk)
        return logits
    return reference.topk_topp_filter(logits, topp, topk)


def sample(
    logits: torch.Tensor,
    temperature: torch.Tensor,
    noise: Optional[torch.Tensor] = None,
) -> torch.Tensor:
    if logits.is_cuda:
        out = torch.empty(logits.shape[0], dtype=torch.long, device=logits.device)
        _native().sample(out, logits, temperature, noise)
        return out
    return reference.sample(logits, temperature, noise)


__all__ = [
    "have_native",
    "build_rope_cache",
    "rmsnorm",
    "rmsnorm_residual",
    "rope_inplace",
    "rope_and_kv_write",
    "swiglu",
    "kv_cache_write",
    "linear",
    "gemm_m256",
    "swizzle_weight",
    "swizzle_weight_frag",
    "attention_prefill",
    "attention_decode",
    "sample",
    "topk_topp_filter",
    "reference",
]
