"""Pure-PyTorch fp32 reference implementations of every engine op.

These are the *semantic contract* for the HIP kernels in csrc/: GPU numerics
tests compare each gfx950 kernel against these (run in fp32) per-shape, and
the CPU engine path runs on them directly so the whole gateway + engine
stack is testable without a GPU.

Shapes (T = total tokens in a varlen batch, B = sequences, D = head dim):
- rmsnorm:            x [T, H]          -> y [T, H]
- rmsnorm_residual:   x, residual       -> (y, x+residual)
- rope_inplace:       q [T, Hq, D], k [T, Hkv, D], positions [T]
- swiglu:             x [T, 2I]         -> silu(x[:, :I]) * x[:, I:]
- kv_cache_write:     k/v [T, Hkv, D] -> cache [nblocks, Hkv, block, D]
- attention_prefill:  varlen causal flash (q [T,Hq,D], k/v [T,Hkv,D], cu_seqlens)
- attention_decode:   q [B, Hq, D] vs paged cache via block_tables
- sample:             logits [B, V] (+ per-seq temperature, optional noise)
"""

from __future__ import annotations

import math
from typing import Optional, Tuple

import torch


def rmsnorm(x: torch.Tensor, weight: torch.Tensor, eps: float = 1e-5) -> torch.Tensor:
    dtype = x.dtype
    xf = x.float()
    var = xf.pow(2).mean(dim=-1, keepdim=True)
    y = xf * torch.rsqrt(var + eps) * weight.float()
    return y.to(dtype)


def rmsnorm_residual(
    x: torch.Tensor, residual: torch.Tensor, weight: torch.Tensor, eps: float = 1e-5
) -> Tuple[torch.Tensor, torch.Tensor]:
    r = (x.float() + residual.float()).to(x.dtype)
    return rmsnorm(r, weight, eps), r


def build_rope_cache(
    max_positions: int, head_dim: int, theta: float = 500000.0, device="cpu"
) -> torch.Tensor:
    """Return [max_positions, head_dim] fp32 cache: [cos(half) | sin(half)]."""
    half = head_dim // 2
    inv_freq = 1.0 / (theta ** (torch.arange(half, dtype=torch.float32, device=device) / half))
    pos = torch.arange(max_positions, dtype=torch.float32, device=device)
    ang = torch.outer(pos, inv_freq)  # [P, half]
    return torch.cat([ang.cos(), ang.sin()], dim=-1).contiguous()


def rope_inplace(
    q: torch.Tensor, k: torch.Tensor, positions: torch.Tensor, cos_sin: torch.Tensor
) -> None:
    """Llama-style rotate-half RoPE applied in place to q and k."""
    half = q.shape[-1] // 2
    cs = cos_sin[positions]  # [T, D]
    cos = cs[:, :half].unsqueeze(1)  # [T, 1, half]
    sin = cs[:, half:].unsqueeze(1)
    for t in (q, k):
        tf = t.float()
        x1, x2 = tf[..., :half], tf[..., half:]
        t[..., :half] = (x1 * cos - x2 * sin).to(t.dtype)
        t[..., half:] = (x2 * cos + x1 * sin).to(t.dtype)


def swiglu(x: torch.Tensor) -> torch.Tensor:
    inter = x.shape[-1] // 2
    g = x[..., :inter].float()
    u = x[..., inter:].float()
    return (g * torch.sigmoid(g) * u).to(x.dtype)


FP8_MAX = 448.0  # e4m3fn


def fp8_quantize_rows(x: torch.Tensor):
    """Per-row (last-dim) e4m3 quantization: returns (uint8 bits, scales).
    x: [..., D] -> bits [..., D] uint8, scales [...] fp32."""
    xf = x.float()
    amax = xf.abs().amax(dim=-1)
    scale = torch.where(amax > 0, amax / FP8_MAX, torch.ones_like(amax))
    q = (xf / scale.unsqueeze(-1)).to(torch.float8_e4m3fn)
    return q.view(torch.uint8), scale


def fp8_dequantize_rows(bits: torch.Tensor, scale: torch.Tensor) -> torch.Tensor:
    return bits.view(torch.float8_e4m3fn).float() * scale.unsqueeze(-1).float()


def kv_cache_write(
    k: torch.Tensor,
    v: torch.Tensor,
    k_cache: torch.Tensor,
    v_cache: torch.Tensor,
    slot_mapping: torch.Tensor,
    k_scale: "Optional[torch.Tensor]" = None,
    v_scale: "Optional[torch.Tensor]" = None,
) -> None:
    """Scatter new K/V rows into the paged cache.

    cache layout: [num_blocks, num_kv_heads, block_size, head_dim];
    slot_mapping[t] = block_id * block_size + block_offset.
    """
    block_size = k_cache.shape[2]
    blocks = torch.div(slot_mapping, block_size, rounding_mode="floor")
    offs = slot_mapping % block_size
    if k_scale is not None:  # fp8 e4m3 cache with per-row scales
        kq, ks = fp8_quantize_rows(k)
        vq, vs = fp8_quantize_rows(v)
        k_cache[blocks, :, offs, :] = kq
        v_cache[blocks, :, offs, :] = vq
        k_scale[blocks, :, offs] = ks
        v_scale[blocks, :, offs] = vs
        return
    k_cache[blocks, :, offs, :] = k.to(k_cache.dtype)
    v_cache[blocks, :, offs, :] = v.to(v_cache.dtype)


def attention_prefill(
    q: torch.Tensor,
    k: torch.Tensor,
    v: torch.Tensor,
    cu_seqlens: torch.Tensor,
    scale: Optional[float] = None,
    causal: bool = True,
    k_cache: Optional[torch.Tensor] = None,
    v_cache: Optional[torch.Tensor] = None,
    block_tables: Optional[torch.Tensor] = None,
    cached_lens: Optional[torch.Tensor] = None,
    k_scale: Optional[torch.Tensor] = None,
    v_scale: Optional[torch.Tensor] = None,
) -> torch.Tensor:
    """Varlen causal attention with GQA over fresh K/V; with cached_lens,
    each sequence also attends (unmasked) to its cached prefix gathered
    from the paged cache (prefix caching / chunked prefill)."""
    T, Hq, D = q.shape
    Hkv = k.shape[1]
    group = Hq // Hkv
    if scale is None:
        scale = 1.0 / math.sqrt(D)
    out = torch.empty_like(q)
    cu = cu_seqlens.tolist()
    for i in range(len(cu) - 1):
        s, e = cu[i], cu[i + 1]
        L = e - s
        qi = q[s:e].float()  # [L, Hq, D]
        ki = k[s:e].float().repeat_interleave(group, dim=1)
        vi = v[s:e].float().repeat_interleave(group, dim=1)
        nc = int(cached_lens[i]) if cached_lens is not None else 0
        if nc > 0:
            bs = k_cache.shape[2]
            rows_k, rows_v = [], []
            for pos in range(nc):
                blk = int(block_tables[i, pos // bs])
                kr = k_cache[blk, :, pos % bs, :]
                vr = v_cache[blk, :, pos % bs, :]
                if k_scale is not None:
                    kr = fp8_dequantize_rows(kr, k_scale[blk, :, pos % bs])
                    vr = fp8_dequantize_rows(vr, v_scale[blk, :, pos % bs])
                rows_k.append(kr)
                rows_v.append(vr)
            kc = torch.stack(rows_k).float().repeat_interleave(group, dim=1)
            vc = torch.stack(rows_v).float().repeat_interleave(group, dim=1)
            ki = torch.cat([kc, ki], dim=0)
            vi = torch.cat([vc, vi], dim=0)
        scores = torch.einsum("qhd,khd->hqk", qi, ki) * scale
        if causal:
            mask = torch.triu(
                torch.ones(L, L, dtype=torch.bool, device=q.device), diagonal=1
            )
            if nc > 0:  # cached keys precede every fresh row: never masked
                mask = torch.cat(
                    [torch.zeros(L, nc, dtype=torch.bool, device=q.device), mask],
                    dim=1,
                )
            scores.masked_fill_(mask, float("-inf"))
        p = torch.softmax(scores, dim=-1)
        out[s:e] = torch.einsum("hqk,khd->qhd", p, vi).to(q.dtype)
    return out


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
    """Single-token decode attention over the paged KV cache."""
    B, Hq, D = q.shape
    Hkv = k_cache.shape[1]
    block_size = k_cache.shape[2]
    group = Hq // Hkv
    if scale is None:
        scale = 1.0 / math.sqrt(D)
    out = torch.empty_like(q)
    for b in range(B):
        L = int(context_lens[b])
        nblocks = (L + block_size - 1) // block_size
        blocks = block_tables[b, :nblocks].long()
        # gather [L, Hkv, D]
        if k_scale is not None:
            kd = fp8_dequantize_rows(k_cache[blocks], k_scale[blocks])
            vd = fp8_dequantize_rows(v_cache[blocks], v_scale[blocks])
            kk = kd.permute(0, 2, 1, 3).reshape(-1, Hkv, D)[:L].float()
            vv = vd.permute(0, 2, 1, 3).reshape(-1, Hkv, D)[:L].float()
        else:
            kk = k_cache[blocks].permute(0, 2, 1, 3).reshape(-1, Hkv, D)[:L].float()
            vv = v_cache[blocks].permute(0, 2, 1, 3).reshape(-1, Hkv, D)[:L].float()
        kk = kk.repeat_interleave(group, dim=1)  # [L, Hq, D]
        vv = vv.repeat_interleave(group, dim=1)
        qb = q[b].float()  # [Hq, D]
        scores = torch.einsum("hd,khd->hk", qb, kk) * scale
        p = torch.softmax(scores, dim=-1)
        out[b] = torch.einsum("hk,khd->hd", p, vv).to(q.dtype)
    return out


def sample(
    logits: torch.Tensor,
    temperature: torch.Tensor,
    noise: Optional[torch.Tensor] = None,
) -> torch.Tensor:
    """Greedy when temperature<=0, else Gumbel-max sampling.

    temperature: [B] fp32; noise: [B, V] uniform(0,1) — supplied by the
    caller so GPU and CPU paths can be compared with identical randomness.
    """
    B, V = logits.shape
    lf = logits.float()
    temps = temperature.view(B, 1).float()
    greedy = temps <= 0
    if noise is None or bool(greedy.all()):
        return lf.argmax(dim=-1)
    u = noise.float().clamp_min(1e-20)
    inner = (-torch.log(u)).clamp_min(1e-20)
    gumbel = -torch.log(inner)
    scored = torch.where(greedy, lf, lf / temps.clamp_min(1e-6) + gumbel)
    return scored.argmax(dim=-1)


def topk_topp_filter(
    logits: torch.Tensor, topp: torch.Tensor, topk: torch.Tensor
) -> torch.Tensor:
    """Batched top-k / top-p filtering: mask entries outside the kept set
    to -inf, in place, and return logits. Semantics: top-p keeps the
    smallest prefix of the descending-sorted row whose softmax mass
    reaches p (the crossing token included); top-k keeps rank < k; both
    given -> intersection. Rows with topk==0 and topp>=1 are untouched."""
    B, V = logits.shape
    active = (topk > 0) | (topp < 1.0)
    if not bool(active.any()):
        return logits
    lf = logits.float()
    sorted_logits, sorted_idx = torch.sort(lf, descending=True, dim=-1)
    pos = torch.arange(V, device=logits.device).unsqueeze(0)  # [1, V]
    k = torch.where(topk > 0, topk, torch.full_like(topk, V)).unsqueeze(1)
    mask_sorted = pos >= k  # beyond top-k rank
    probs = torch.softmax(sorted_logits, dim=-1)
    cum = probs.cumsum(dim=-1)
    # keep through the first position where cumulative >= p
    crossed = (cum - probs) >= topp.unsqueeze(1)
    mask_sorted |= crossed
    mask_sorted &= active.unsqueeze(1)
    mask = torch.zeros_like(mask_sorted).scatter_(1, sorted_idx, mask_sorted)
    logits.masked_fill_(mask, float("-inf"))
    return logits
