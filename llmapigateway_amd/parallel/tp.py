"""Tensor-parallel weight sharding (Megatron-style row/column split).

Every rank draws the SAME full-shape random weights (same seed) and keeps
its slice — so TP=N is numerically a sharding of the TP=1 model and the
gloo CPU test can assert logits parity. Per layer, the two RCCL all-reduces
over xGMI happen after the row-parallel o-projection and down-projection
(LlamaModel._maybe_all_reduce).
"""

from __future__ import annotations

import torch


def shard_column(full: torch.Tensor, rank: int, world: int) -> torch.Tensor:
    """Column-parallel linear stored as [out, in]: split the OUT dim."""
    out = full.shape[0]
    assert out % world == 0
    step = out // world
    return full[rank * step : (rank + 1) * step].contiguous()


def shard_row(full: torch.Tensor, rank: int, world: int) -> torch.Tensor:
    """Row-parallel linear stored as [out, in]: split the IN dim."""
    in_ = full.shape[1]
    assert in_ % world == 0
    step = in_ // world
    return full[:, rank * step : (rank + 1) * step].contiguous()


def shard_qkv(
    full: torch.Tensor, rank: int, world: int, q_size: int, kv_size: int
) -> torch.Tensor:
    """Fused qkv [q + 2kv, in]: shard q, k and v blocks independently."""
    q = shard_column(full[:q_size], rank, world)
    k = shard_column(full[q_size : q_size + kv_size], rank, world)
    v = shard_column(full[q_size + kv_size :], rank, world)
    return torch.cat([q, k, v], dim=0).contiguous()


def shard_gate_up(full: torch.Tensor, rank: int, world: int, inter: int) -> torch.Tensor:
    """Fused gate|up [2I, in]: shard each half independently."""
    g = shard_column(full[:inter], rank, world)
    u = shard_column(full[inter:], rank, world)
    return torch.cat([g, u], dim=0).contiguous()
