"""Paged KV cache: HBM-resident cache tensors + block allocator.

Replaces the reference's httpx connection bookkeeping as the per-request
resource being managed. Layout per layer:
``[num_blocks, num_kv_heads, block_size, head_dim]`` — for a fixed kv head,
a key/value row of head_dim is contiguous and rows within a block are
contiguous, which is what the decode-attention kernel streams.

The block allocator has two interchangeable implementations with identical
semantics: the C++ one in ops/csrc/kv_manager.cpp (used when the extension
is built — the native runtime path) and a pure-Python fallback. Sizing: on
GPU the default number of blocks is computed from *free HBM after model
weights* × settings.engine_hbm_fraction (288 GB per MI355X makes this the
dominant tensor).
"""

from __future__ import annotations

import logging
from typing import List, Optional

import torch

from ..models.configs import ModelConfig

logger = logging.getLogger(__name__)


class _PyBlockAllocator:
    """LIFO free-list block allocator (Python twin of kv_manager.cpp)."""

    def __init__(self, num_blocks: int):
        self._free = list(range(num_blocks - 1, -1, -1))
        self.num_blocks = num_blocks

    def num_free(self) -> int:
        return len(self._free)

    def allocate(self, n: int) -> List[int]:
        if n > len(self._free):
            raise RuntimeError(f"KV cache out of blocks: need {n}, have {len(self._free)}")
        out = [self._free.pop() for _ in range(n)]
        return out

    def free(self, blocks: List[int]) -> None:
        self._free.extend(reversed(blocks))


def make_block_allocator(num_blocks: int):
    try:
        from ..ops import _C  # type: ignore

        return _C.BlockAllocator(num_blocks)
    except Exception:
        return _PyBlockAllocator(num_blocks)


class BlockManager:
    """Per-sequence block-table bookkeeping on top of the allocator.

    With ``prefix_caching=True``, FULL prompt blocks are content-addressed
    by a (parent_block, token_bytes) chain key: a new request whose prompt
    shares a cached prefix reuses those blocks (refcounted) and only
    prefills the suffix; refcount-0 cached blocks stay resident (data
    retained) in an LRU and are evicted only under allocation pressure.
    Shared blocks are never written again: only FULL prompt blocks are
    registered, and generation always appends into a fresh block.
    """

    def __init__(self, num_blocks: int, block_size: int, prefix_caching: bool = False):
        self.block_size = block_size
        self.allocator = make_block_allocator(num_blocks)
        self.prefix_caching = prefix_caching
        # cached-prefix state (block ids are allocator-owned while cached)
        self._table: dict = {}      # (parent_block, chunk_bytes) -> block_id
        self._block_key: dict = {}  # block_id -> its table key
        self._refs: dict = {}       # block_id -> request refcount
        from collections import OrderedDict

        self._evictable: "OrderedDict[int, None]" = OrderedDict()
        self.stats_prefix_hits = 0
        self.stats_prefix_tokens = 0

    @property
    def num_free_blocks(self) -> int:
        return self.allocator.num_free()

    def blocks_needed(self, num_tokens: int) -> int:
        return (num_tokens + self.block_size - 1) // self.block_size

    def can_allocate(self, num_tokens: int) -> bool:
        avail = self.allocator.num_free() + len(self._evictable)
        return self.blocks_needed(num_tokens) <= avail

    def _alloc_raw(self, n: int) -> List[int]:
        while self.allocator.num_free() < n and self._evictable:
            blk, _ = self._evictable.popitem(last=False)  # oldest first
            key = self._block_key.pop(blk)
            del self._table[key]
            self._refs.pop(blk, None)
            self.allocator.free([blk])
        return list(self.allocator.allocate(n))

    def allocate(self, num_tokens: int) -> List[int]:
        return self._alloc_raw(self.blocks_needed(num_tokens))

    # ---- prefix caching ----
    def _chunk_key(self, parent: int, prompt_ids, i: int):
        bs = self.block_size
        import numpy as _np

        chunk = _np.asarray(prompt_ids[i * bs : (i + 1) * bs], dtype=_np.int64)
        return (parent, chunk.tobytes())

    def _ref(self, blk: int) -> None:
        self._refs[blk] = self._refs.get(blk, 0) + 1
        self._evictable.pop(blk, None)

    def _unref(self, blk: int) -> None:
        self._refs[blk] -= 1
        if self._refs[blk] == 0:
            self._evictable[blk] = None  # LRU tail (data stays resident)

    def allocate_with_prefix(self, prompt_ids) -> tuple:
        """Returns (block_table, num_cached_tokens)."""
        bs = self.block_size
        cached: List[int] = []
        parent = -1
        max_full = (len(prompt_ids) - 1) // bs  # leave >= 1 token to prefill
        for i in range(max_full):
            blk = self._table.get(self._chunk_key(parent, prompt_ids, i))
            if blk is None:
                break
            cached.append(blk)
            parent = blk
        total = self.blocks_needed(len(prompt_ids))
        # Pin the matched cached blocks FIRST: _alloc_raw evicts refcount-0
        # cached blocks under pressure, and without the ref it could hand a
        # matched prefix block back as this request's own suffix block
        # (read-as-prefix + write-as-suffix aliasing -> silent corruption).
        for blk in cached:
            self._ref(blk)
        try:
            private = self._alloc_raw(total - len(cached))
        except RuntimeError:
            for blk in cached:
                self._unref(blk)
            raise
        if cached:
            self.stats_prefix_hits += 1
            self.stats_prefix_tokens += len(cached) * bs
        return cached + private, len(cached) * bs

    def register_prefix(self, prompt_ids, block_table: List[int]) -> None:
        """Register a request's FULL prompt blocks (KV now written) so later
        prompts can reuse them. Call after the prefill forward is issued."""
        if not self.prefix_caching:
            return
        bs = self.block_size
        parent = -1
        for i in range(len(prompt_ids) // bs):
            key = self._chunk_key(parent, prompt_ids, i)
            existing = self._table.get(key)
            if existing is not None:
                parent = existing
                continue
            blk = block_table[i]
            if blk in self._block_key:
                # this block is already registered under a different chain
                # (it IS a cached block we reused) — just walk on
                parent = blk
                continue
            self._table[key] = blk
            self._block_key[blk] = key
            self._refs[blk] = self._refs.get(blk, 0) + 1  # owner's reference
            parent = blk

    def extend(self, block_table: List[int], old_tokens: int, new_tokens: int) -> None:
        need = self.blocks_needed(new_tokens) - len(block_table)
        if need > 0:
            block_table.extend(self._alloc_raw(need))

    def free(self, block_table: List[int]) -> None:
        plain = []
        for blk in block_table:
            if blk in self._block_key:
                self._unref(blk)
            else:
                plain.append(blk)
        if plain:
            self.allocator.free(plain)


class PagedKVCache:
    def __init__(
        self,
        config: ModelConfig,
        num_blocks: int,
        block_size: int,
        device: torch.device | str,
        dtype: torch.dtype = torch.bfloat16,
        prefix_caching: bool = False,
        kv_dtype: str = "auto",
    ):
        """kv_dtype: "auto" (= compute dtype) or "fp8" — e4m3 bytes with
        per-(token, head) fp32 row scales: half the decode KV traffic and
        ~2x the resident-token capacity on the same HBM budget."""
        self.config = config
        self.block_size = block_size
        self.num_blocks = num_blocks
        self.fp8 = kv_dtype == "fp8"
        shape = (num_blocks, config.num_kv_heads, block_size, config.head_dim)
        store = torch.uint8 if self.fp8 else dtype
        self.k_caches = [
            torch.zeros(shape, dtype=store, device=device) for _ in range(config.num_layers)
        ]
        self.v_caches = [
            torch.zeros(shape, dtype=store, device=device) for _ in range(config.num_layers)
        ]
        if self.fp8:
            sshape = (num_blocks, config.num_kv_heads, block_size)
            self.k_scales = [
                torch.ones(sshape, dtype=torch.float32, device=device)
                for _ in range(config.num_layers)
            ]
            self.v_scales = [
                torch.ones(sshape, dtype=torch.float32, device=device)
                for _ in range(config.num_layers)
            ]
        else:
            self.k_scales = [None] * config.num_layers
            self.v_scales = [None] * config.num_layers
        self.manager = BlockManager(num_blocks, block_size, prefix_caching=prefix_caching)

    @staticmethod
    def block_bytes(
        config: ModelConfig, block_size: int, dtype: torch.dtype,
        kv_dtype: str = "auto",
    ) -> int:
        if kv_dtype == "fp8":
            # 1 byte per element + a 4-byte scale per (token, head) row
            per_row = config.head_dim + 4
            return 2 * config.num_layers * config.num_kv_heads * block_size * per_row
        elem = torch.empty(0, dtype=dtype).element_size()
        return 2 * config.num_layers * config.num_kv_heads * block_size * config.head_dim * elem

    @classmethod
    def fit_num_blocks(
        cls,
        config: ModelConfig,
        block_size: int,
        device: torch.device,
        dtype: torch.dtype,
        hbm_fraction: float = 0.90,
        max_blocks: Optional[int] = None,
        kv_dtype: str = "auto",
    ) -> int:
        per_block = cls.block_bytes(config, block_size, dtype, kv_dtype)
        if device.type == "cuda":
            free, _total = torch.cuda.mem_get_info(device)
            budget = int(free * hbm_fraction)
        else:
            budget = 256 * 1024 * 1024  # CPU tests: small cache
        n = max(budget // per_block, 16)
        if max_blocks is not None:
            n = min(n, max_blocks)
        logger.info(
            "KV cache: %d blocks x %d tokens (%.2f GiB)",
            n,
            block_size,
            n * per_block / (1 << 30),
        )
        return int(n)
