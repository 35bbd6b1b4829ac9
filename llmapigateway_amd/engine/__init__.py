from .engine import LLMEngine, EngineRequest, SamplingParams
from .kvcache import BlockManager, PagedKVCache
from .tokenizer import ByteTokenizer

__all__ = [
    "LLMEngine",
    "EngineRequest",
    "SamplingParams",
    "BlockManager",
    "PagedKVCache",
    "ByteTokenizer",
]
