"""hipGraph-captured decode step (torch.cuda.CUDAGraph == hipGraph on ROCm).

A decode step at batch 64 on Llama-3-8B issues ~300 kernel launches (32
layers x ~9 kernels + head); at ~10 us launch overhead that is several ms
of pure CPU launch cost per step — comparable to the HBM time of the step
itself. Capturing the whole decode forward per batch-size bucket replays it
as ONE hipGraph launch.

Mechanics: persistent device buffers (token ids, positions, slot mapping,
block tables, context lens) sized at max batch; per-bucket graphs captured
lazily on first use (standard side-stream warmup then capture); each step
fills pinned staging, async-copies into the static buffers and replays.
Padding rows use slot=-1 (kv write kernel skips) and context_len=1.
"""

from __future__ import annotations

import logging
from typing import Dict, List, Optional

import torch

from ..models.llama import ForwardBatch

logger = logging.getLogger(__name__)


class DecodeGraphRunner:
    def __init__(self, model, k_caches, v_caches, max_batch: int, max_blocks: int,
                 k_scales=None, v_scales=None):
        self.model = model
        self.k_caches = k_caches
        self.v_caches = v_caches
        self.k_scales = k_scales
        self.v_scales = v_scales
        self.device = model.device
        self.max_batch = max_batch
        self.max_blocks = max_blocks

        self.buckets = [b for b in (1, 2, 4, 8, 16, 24, 32, 48, 64, 96, 128, 192, 256) if b < max_batch]
        self.buckets.append(max_batch)

        dev = self.device
        B, NB = max_batch, max_blocks
        self.token_ids = torch.zeros(B, dtype=torch.long, device=dev)
        self.positions = torch.zeros(B, dtype=torch.long, device=dev)
        self.slot_mapping = torch.full((B,), -1, dtype=torch.long, device=dev)
        self.block_tables = torch.zeros(B, NB, dtype=torch.int32, device=dev)
        self.context_lens = torch.ones(B, dtype=torch.int32, device=dev)

        pin = dev.type == "cuda"
        self.h_token_ids = torch.zeros(B, dtype=torch.long, pin_memory=pin)
        self.h_positions = torch.zeros(B, dtype=torch.long, pin_memory=pin)
        self.h_slot_mapping = torch.full((B,), -1, dtype=torch.long, pin_memory=pin)
        self.h_block_tables = torch.zeros(B, NB, dtype=torch.int32, pin_memory=pin)
        self.h_context_lens = torch.ones(B, dtype=torch.int32, pin_memory=pin)

        self.graphs: Dict[int, torch.cuda.CUDAGraph] = {}
        self.graph_logits: Dict[int, torch.Tensor] = {}
        self.pool = None

    def _batch_view(self, b: int) -> ForwardBatch:
        return ForwardBatch(
            kind="decode",
            token_ids=self.token_ids[:b],
            positions=self.positions[:b],
            slot_mapping=self.slot_mapping[:b],
            block_tables=self.block_tables[:b],
            context_lens=self.context_lens[:b],
            logits_indices=None,
        )

    def _capture(self, bucket: int) -> None:
        logger.info("capturing decode hipGraph for batch bucket %d", bucket)
        batch = self._batch_view(bucket)
        # warmup on a side stream (cuBLAS/hipBLASLt workspace allocs etc.)
        s = torch.cuda.Stream(device=self.device)
        s.wait_stream(torch.cuda.current_stream(self.device))
        with torch.cuda.stream(s):
            for _ in range(2):
                self.model.forward(batch, self.k_caches, self.v_caches,
                                   k_scales=self.k_scales, v_scales=self.v_scales)
        torch.cuda.current_stream(self.device).wait_stream(s)
        torch.cuda.synchronize(self.device)

        graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(graph, pool=self.pool):
            logits = self.model.forward(batch, self.k_caches, self.v_caches,
                                        k_scales=self.k_scales, v_scales=self.v_scales)
        if self.pool is None:
            self.pool = graph.pool()  # share the memory pool across buckets
        self.graphs[bucket] = graph
        self.graph_logits[bucket] = logits

    def capture_all(self) -> None:
        """Capture every bucket up front (serving engines: avoids a
        multi-second capture stall on the first request of each batch
        size). The static buffers hold benign defaults (slot=-1 skips KV
        writes, ctx=1 reads block 0, token 0 embeds)."""
        for b in self.buckets:
            if b not in self.graphs:
                self._capture(b)
        torch.cuda.synchronize(self.device)

    def bucket_for(self, n: int) -> int:
        for b in self.buckets:
            if b >= n:
                return b
        return self.max_batch

    def run(
        self,
        tokens,          # np.int64 [n] | int64 CUDA tensor (deferred sampling)
        positions,       # np.int64 [n]
        slots,           # np.int64 [n]
        block_tables,    # np.int32 [n, <=max_blocks]
        context_lens,    # np.int32 [n]
    ) -> torch.Tensor:
        n = len(tokens)
        bucket = self.bucket_for(n)

        # fill pinned staging via numpy views (no per-request tensors)
        if isinstance(tokens, torch.Tensor):
            self.token_ids[:n].copy_(tokens[:n])  # device-to-device
        else:
            self.h_token_ids.numpy()[:n] = tokens
        self.h_positions.numpy()[:n] = positions
        sm = self.h_slot_mapping.numpy()
        sm[:n] = slots
        sm[n:bucket] = -1
        cl = self.h_context_lens.numpy()
        cl[:n] = context_lens
        cl[n:bucket] = 1  # padding rows read 1 stale key; output discarded
        self.h_block_tables.numpy()[:n, : block_tables.shape[1]] = block_tables

        if not isinstance(tokens, torch.Tensor):
            self.token_ids[:bucket].copy_(self.h_token_ids[:bucket], non_blocking=True)
        self.positions[:bucket].copy_(self.h_positions[:bucket], non_blocking=True)
        self.slot_mapping[:bucket].copy_(self.h_slot_mapping[:bucket], non_blocking=True)
        self.context_lens[:bucket].copy_(self.h_context_lens[:bucket], non_blocking=True)
        self.block_tables[:bucket].copy_(self.h_block_tables[:bucket], non_blocking=True)

        if bucket not in self.graphs:
            self._capture(bucket)
        self.graphs[bucket].replay()
        return self.graph_logits[bucket][:n]
