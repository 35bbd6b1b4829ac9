"""Deterministic byte-level tokenizer (no network for tokenizer files).

Token ids: 0=PAD, 1=BOS, 2=EOS, byte b -> 3+b. Any model vocab >= 259 can
decode these ids; generation over random-init weights produces arbitrary
ids, which detokenize via modulo into the byte range — output text is
synthetic either way (BASELINE.json: synthetic data / random-init weights).
"""

from __future__ import annotations

from typing import List


class ByteTokenizer:
    PAD, BOS, EOS = 0, 1, 2
    OFFSET = 3

    def __init__(self, vocab_size: int = 128256):
        self.vocab_size = vocab_size
        self.eos_ids = frozenset({self.EOS})

    def encode(self, text: str, add_bos: bool = True) -> List[int]:
        ids = [self.BOS] if add_bos else []
        ids.extend(self.OFFSET + b for b in text.encode("utf-8"))
        return ids

    def decode(self, ids: List[int]) -> str:
        data = bytearray()
        for i in ids:
            if i in (self.PAD, self.BOS, self.EOS):
                continue
            # fold out-of-range ids (random-weight sampling) into byte space,
            # keeping to printable ASCII so SSE payloads stay clean
            b = (int(i) - self.OFFSET) % 256
            if 32 <= b < 127 or b in (9, 10):
                data.append(b)
            else:
                data.append(32 + (b % 95))
        return data.decode("utf-8", errors="replace")

    def render_chat(self, messages: List[dict]) -> str:
        """Minimal chat template: 'role: content' lines + assistant cue."""
        parts = []
        for m in messages or []:
            role = m.get("role", "user")
            content = m.get("content", "")
            if isinstance(content, list):  # OpenAI content-parts form
                content = " ".join(
                    p.get("text", "") for p in content if isinstance(p, dict)
                )
            parts.append(f"{role}: {content}")
        parts.append("assistant:")
        return "\n".join(parts)
