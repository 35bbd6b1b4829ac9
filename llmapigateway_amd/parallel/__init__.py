from .dist import init_from_env, get_tp_info
from .tp import shard_qkv, shard_column, shard_row

__all__ = ["init_from_env", "get_tp_info", "shard_qkv", "shard_column", "shard_row"]
