"""Llama-family decoder (Llama-3, Mistral) built directly on the MI355X ops.

Inference-only, bf16 weights, random init (no network for checkpoints —
BASELINE.json: synthetic data / random-init weights). The forward pass is a
flat varlen batch (ForwardBatch) in one of two modes:

- prefill: fresh prompts; ops.attention_prefill over the batch's own K/V
  (and K/V written to the paged cache for later decode);
- decode: one token per running sequence; ops.attention_decode over the
  paged cache via block tables.

Projections run through ops.linear: decode-shaped batches (M<=256) hit the
hand-written weight-streaming gfx950 GEMM (ops/csrc/gemm_skinny.hip),
prefill batches the TunableOp-tuned hipBLASLt/rocBLAS library GEMMs; the
fused hot ops (rmsnorm+residual, rope, swiglu, attention, sampling) are the
hand-written CDNA4 kernels behind llmapigateway_amd.ops.

Tensor parallelism: construct with ``tp_group`` + a config pre-sharded via
ModelConfig.scaled_for_tp; qkv/gate_up are column-sharded, o/down
row-sharded with an RCCL all-reduce over xGMI after each (2 per layer).
"""

from __future__ import annotations

import math
from dataclasses import dataclass
from typing import Dict, List, Optional

import torch
import torch.distributed as dist
import torch.nn.functional as F

from .. import ops
from .configs import ModelConfig


@dataclass
class ForwardBatch:
    kind: str  # "prefill" | "decode" | "mixed"
    token_ids: torch.Tensor          # [T] int64
    positions: torch.Tensor          # [T] int64
    slot_mapping: torch.Tensor       # [T] int64 (global KV slots; -1 = don't write)
    # prefill:
    cu_seqlens: Optional[torch.Tensor] = None   # [B+1] int32
    max_seqlen: int = 0
    tile_seq: Optional[torch.Tensor] = None     # [ntiles] int32 (GPU kernel tile map)
    tile_off: Optional[torch.Tensor] = None
    # decode:
    block_tables: Optional[torch.Tensor] = None  # [B, max_blocks] int32
    context_lens: Optional[torch.Tensor] = None  # [B] int32
    # prefill with cached prefix (prefix caching): per-seq number of
    # positions already in the paged cache (block_tables set too)
    cached_lens: Optional[torch.Tensor] = None   # [B] int32
    # rows of the flat batch at which logits are needed (last token per seq)
    logits_indices: Optional[torch.Tensor] = None  # [B] int64
    # mixed steps: rows [0, n_prefill_tokens) are prefill (cu_seqlens etc.
    # apply to them), rows [n_prefill_tokens, T) are decode rows with their
    # own tables/lens (decode never shares block_tables with the cached-
    # prefill phase in a mixed step)
    n_prefill_tokens: int = 0
    dec_block_tables: Optional[torch.Tensor] = None  # [Bd, max_blocks] int32
    dec_context_lens: Optional[torch.Tensor] = None  # [Bd] int32


class LlamaModel:
    def __init__(
        self,
        config: ModelConfig,
        device: torch.device | str = "cpu",
        dtype: torch.dtype = torch.bfloat16,
        seed: int = 0,
        tp_group: Optional[object] = None,
        tp_rank: int = 0,
        tp_size: int = 1,
        full_config: Optional[ModelConfig] = None,
    ):
        self.config = config
        self.full_config = full_config or config
        self.device = torch.device(device)
        self.dtype = dtype
        self.tp_group = tp_group
        self.tp_rank = tp_rank
        self.tp_size = tp_size
        self.scale = config.head_dim ** -0.5
        self.layers: List[Dict[str, torch.Tensor]] = []
        self._init_weights(seed)
        self.cos_sin = ops.build_rope_cache(
            config.max_positions, config.head_dim, config.rope_theta, device=self.device
        )

    # ---- weights ----
    def _init_weights(self, seed: int) -> None:
        c = self.config
        hidden = c.hidden_size
        # tiny models init on CPU (deterministic across devices, for tests);
        # big models init directly on the GPU (CPU RNG would take minutes at 8B+)
        approx_params = (
            2 * self.full_config.vocab_size * hidden
            + c.num_layers * hidden * (c.q_size + 2 * c.kv_size + c.q_size + 3 * c.intermediate_size) // 1
        )
        init_device = torch.device("cpu") if approx_params < 10**8 else self.device
        gen = torch.Generator(device=init_device).manual_seed(seed)

        def mk(rows: int, cols: int, std: float) -> torch.Tensor:
            w = torch.empty(rows, cols, dtype=torch.float32, device=init_device)
            w.normal_(0.0, std, generator=gen)
            return w.to(self.dtype).to(self.device)

        # TP: every rank draws the same full-shape weights (same seed) and
        # keeps its Megatron-style slice (parallel/tp.py), so TP=N is a pure
        # sharding of the TP=1 model (gloo CPU test asserts logits parity).
        from ..parallel.tp import shard_column, shard_gate_up, shard_qkv, shard_row

        fc = self.full_config
        tpr, tps = self.tp_rank, self.tp_size

        std = 0.02
        out_std = 0.02 / math.sqrt(2 * c.num_layers)
        self.embed = mk(fc.vocab_size, hidden, std)
        for _ in range(c.num_layers):
            qkv_full = mk(fc.q_size + 2 * fc.kv_size, hidden, std)
            o_full = mk(hidden, fc.q_size, out_std)
            gu_full = mk(2 * fc.intermediate_size, hidden, std)
            down_full = mk(hidden, fc.intermediate_size, out_std)
            self.layers.append(
                {
                    "input_norm": torch.ones(hidden, dtype=self.dtype, device=self.device),
                    "qkv": shard_qkv(qkv_full, tpr, tps, fc.q_size, fc.kv_size)
                    if tps > 1
                    else qkv_full,
                    "o": shard_row(o_full, tpr, tps) if tps > 1 else o_full,
                    "post_norm": torch.ones(hidden, dtype=self.dtype, device=self.device),
                    "gate_up": shard_gate_up(gu_full, tpr, tps, fc.intermediate_size)
                    if tps > 1
                    else gu_full,
                    "down": shard_row(down_full, tpr, tps) if tps > 1 else down_full,
                }
            )
        self.final_norm = torch.ones(hidden, dtype=self.dtype, device=self.device)
        self.lm_head = self.embed if c.tie_embeddings else mk(self.full_config.vocab_size, hidden, std)

        # Fragment-major twins of the decode projections for the
        # macro-tile LDS-staged GEMM (ops/csrc/gemm_m256.hip) — ON by
        # default: the twin copy costs ~weights-again for the projections
        # (~14 GB for llama-3-8b, cheap against 288 GB HBM) and buys the
        # decode step its dominant GEMM time back from the library
        # (profiles/r02_gemm_m256_probe.md). LLMAPI_NO_FRAG_WEIGHTS=1
        # opts out; LLMAPI_SWZ_WEIGHTS=1 selects the legacy k-major twins
        # (gemm_skinny streaming form) for comparison runs.
        import os as _os

        if self.device.type == "cuda" and not _os.environ.get("LLMAPI_NO_FRAG_WEIGHTS"):
            legacy = bool(_os.environ.get("LLMAPI_SWZ_WEIGHTS"))
            for layer in self.layers:
                for name in ("qkv", "o", "gate_up", "down"):
                    w = layer[name]
                    if legacy:
                        if w.shape[0] % 64 == 0 and w.shape[1] % 32 == 0:
                            layer[name + "_swz"] = ops.swizzle_weight(w)
                    elif (
                        w.shape[0] % 64 == 0
                        and w.shape[1] % 64 == 0
                        # only shapes the measured dispatch actually routes to
                        # the custom kernel get a twin — qkv/gate_up twins
                        # would be dead HBM (the library wins those shapes),
                        # ~9 GB at 8B and ~89 GB at 70B
                        and ops._m256_config(256, w.shape[0], w.shape[1]) is not None
                        # gate_up with a fused-swiglu twin never uses the
                        # plain twin — don't hold both in HBM
                        and not (
                            name == "gate_up"
                            and ops._m256_swiglu_config(
                                256, w.shape[0], w.shape[1]
                            )
                            is not None
                        )
                    ):
                        layer[name + "_swz"] = ops.swizzle_weight_frag(w)
                gu = layer["gate_up"]
                if (
                    not legacy
                    and ops._m256_swiglu_config(256, gu.shape[0], gu.shape[1])
                    is not None
                ):
                    # block-16 interleaved twin for the FUSED gate_up+swiglu
                    # decode path (ops.swiglu_linear)
                    layer["gate_up_int"] = ops.swizzle_weight_frag(
                        ops.interleave_gate_up(gu)
                    )

    def param_bytes(self) -> int:
        total = self.embed.numel() + self.final_norm.numel()
        if self.lm_head is not self.embed:
            total += self.lm_head.numel()
        for layer in self.layers:
            total += sum(t.numel() for t in layer.values())
        return total * self.embed.element_size()

    # ---- forward ----
    def _maybe_all_reduce(self, x: torch.Tensor) -> torch.Tensor:
        if self.tp_group is not None:
            dist.all_reduce(x, group=self.tp_group)
        return x

    def _row_parallel(self, x: torch.Tensor, w: torch.Tensor,
                      w_swz=None) -> torch.Tensor:
        """Row-parallel linear with comm/compute overlap: split the token
        batch in two, launch the first half's all-reduce asynchronously
        while the second half's GEMM runs (overlap the 2
        all-reduces/layer with compute). Falls back to the plain form for
        tiny batches or TP=1."""
        if self.tp_group is None:
            return ops.linear(x, w, w_swz)
        T = x.shape[0]
        if T < 32:  # split overhead exceeds the overlap win
            y = ops.linear(x, w, w_swz)
            dist.all_reduce(y, group=self.tp_group)
            return y
        half = T // 2
        y1 = ops.linear(x[:half], w, w_swz)
        h1 = dist.all_reduce(y1, group=self.tp_group, async_op=True)
        y2 = ops.linear(x[half:], w, w_swz)
        h2 = dist.all_reduce(y2, group=self.tp_group, async_op=True)
        h1.wait()
        h2.wait()
        return torch.cat([y1, y2], dim=0)

    @torch.inference_mode()
    def forward(
        self,
        batch: ForwardBatch,
        k_caches: List[torch.Tensor],
        v_caches: List[torch.Tensor],
        k_scales: "Optional[List[torch.Tensor]]" = None,
        v_scales: "Optional[List[torch.Tensor]]" = None,
    ) -> torch.Tensor:
        """Returns logits [B, vocab] at batch.logits_indices. With
        k_scales/v_scales the caches are fp8 e4m3 (per-row scales)."""
        c = self.config
        ksc = k_scales if k_scales is not None else [None] * len(self.layers)
        vsc = v_scales if v_scales is not None else [None] * len(self.layers)
        h = F.embedding(batch.token_ids, self.embed)
        residual = h  # placeholder; layer 0 sets the real residual stream
        T = h.shape[0]

        for i, layer in enumerate(self.layers):
            if i == 0:
                x, residual = ops.rmsnorm(h, layer["input_norm"], c.rms_eps), h
            else:
                x, residual = ops.rmsnorm_residual(h, residual, layer["input_norm"], c.rms_eps)

            qkv = ops.linear(x, layer["qkv"], layer.get("qkv_swz"))
            q, k, v = qkv.split([c.q_size, c.kv_size, c.kv_size], dim=-1)
            q = q.view(T, c.num_heads, c.head_dim)
            k = k.view(T, c.num_kv_heads, c.head_dim)
            v = v.view(T, c.num_kv_heads, c.head_dim)
            ops.rope_and_kv_write(
                q, k, v, k_caches[i], v_caches[i],
                batch.positions, self.cos_sin, batch.slot_mapping,
                k_scale=ksc[i], v_scale=vsc[i],
            )

            if batch.kind == "prefill":
                attn = ops.attention_prefill(
                    q, k, v, batch.cu_seqlens, batch.max_seqlen, self.scale,
                    tile_seq=batch.tile_seq, tile_off=batch.tile_off,
                    k_cache=k_caches[i] if batch.cached_lens is not None else None,
                    v_cache=v_caches[i] if batch.cached_lens is not None else None,
                    block_tables=batch.block_tables,
                    cached_lens=batch.cached_lens,
                    k_scale=ksc[i] if batch.cached_lens is not None else None,
                    v_scale=vsc[i] if batch.cached_lens is not None else None,
                )
            elif batch.kind == "mixed":
                Tp = batch.n_prefill_tokens
                attn_p = ops.attention_prefill(
                    q[:Tp], k[:Tp], v[:Tp], batch.cu_seqlens, batch.max_seqlen,
                    self.scale, tile_seq=batch.tile_seq, tile_off=batch.tile_off,
                    k_cache=k_caches[i] if batch.cached_lens is not None else None,
                    v_cache=v_caches[i] if batch.cached_lens is not None else None,
                    block_tables=batch.block_tables,
                    cached_lens=batch.cached_lens,
                    k_scale=ksc[i] if batch.cached_lens is not None else None,
                    v_scale=vsc[i] if batch.cached_lens is not None else None,
                )
                attn_d = ops.attention_decode(
                    q[Tp:], k_caches[i], v_caches[i],
                    batch.dec_block_tables, batch.dec_context_lens, self.scale,
                    k_scale=ksc[i], v_scale=vsc[i],
                )
                attn = torch.cat([attn_p, attn_d], dim=0)
            else:
                attn = ops.attention_decode(
                    q, k_caches[i], v_caches[i], batch.block_tables,
                    batch.context_lens, self.scale,
                    k_scale=ksc[i], v_scale=vsc[i],
                )

            h = self._row_parallel(
                attn.reshape(T, c.q_size), layer["o"], layer.get("o_swz")
            )

            x, residual = ops.rmsnorm_residual(h, residual, layer["post_norm"], c.rms_eps)
            h = self._row_parallel(
                ops.swiglu_linear(
                    x, layer["gate_up"], layer.get("gate_up_swz"),
                    layer.get("gate_up_int"),
                ),
                layer["down"], layer.get("down_swz"),
            )

        x, _ = ops.rmsnorm_residual(h, residual, self.final_norm, c.rms_eps)
        if batch.logits_indices is not None:
            x = x[batch.logits_indices]
        return ops.linear(x, self.lm_head).float()
