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
    """Per-sequence block-table bookkeeping on top of the allocator."""

    def __init__(self, num_blocks: int, block_size: int):
        self.block_size = block_size
        self.allocator = make_block_allocator(num_blocks)

    @property
    def num_free_blocks(self) -> int:
        return self.allocator.num_free()

    def blocks_needed(self, num_tokens: int) -> int:
        return (num_tokens + self.block_size - 1) // self.block_size

    def can_allocate(self, num_tokens: int) -> bool:
        return self.blocks_needed(num_tokens) <= self.allocator.num_free()

    def allocate(self, num_tokens: int) -> List[int]:
        return list(self.allocator.allocate(self.blocks_needed(num_tokens)))

    def extend(self, block_table: List[int], old_tokens: int, new_tokens: int) -> None:
        need = self.blocks_needed(new_tokens) - len(block_table)
        if need > 0:
            block_table.extend(self.allocator.allocate(need))

    def free(self, block_table: List[int]) -> None:
        if block_table:
            self.allocator.free(list(block_table))


class PagedKVCache:
    def __init__(
        self,
        config: ModelConfig,
        num_blocks: int,
        block_size: int,
        device: torch.device | str,
        dtype: torch.dtype = torch.bfloat16,
    ):
        self.config = config
        self.block_size = block_size
        self.num_blocks = num_blocks
        shape = (num_blocks, config.num_kv_heads, block_size, config.head_dim)
        self.k_caches = [
            torch.zeros(shape, dtype=dtype, device=device) for _ in range(config.num_layers)
        ]
        self.v_caches = [
            torch.zeros(shape, dtype=dtype, device=device) for _ in range(config.num_layers)
        ]
        self.manager = BlockManager(num_blocks, block_size)

    @staticmethod
    def block_bytes(config: ModelConfig, block_size: int, dtype: torch.dtype) -> int:
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
    ) -> int:
        per_block = cls.block_bytes(config, block_size, dtype)
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
