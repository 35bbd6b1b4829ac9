"""Tokenizers for the engine.

BPETokenizer — the production tokenizer: a real byte-level BPE (32k
merges trained offline from in-image text, vocab checked in at
assets/bpe32k.json — tools/train_tokenizer.py; no network) with
llama-3-style special tokens and chat template. Prompt token counts,
stop-string semantics and usage accounting behave like a real
deployment's; ids the random-init models sample beyond the trained vocab
decode through a printable fallback so SSE output stays text.

ByteTokenizer — the round-1 deterministic byte-level stand-in, kept for
tiny-model tests (vocab 512 presets cannot hold BPE ids).

Reference anchor: usage accounting consumes these counts exactly where
the reference parses provider-reported usage
(/root/reference/llm_gateway_core/middleware/chat_logging.py:233-263).
"""

from __future__ import annotations

import os
import threading
from typing import List, Optional


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


class BPETokenizer:
    """Byte-level BPE with llama-3-style specials and chat template."""

    _shared = None
    _shared_lock = threading.Lock()

    def __init__(self, vocab_size: int = 128256, path: Optional[str] = None):
        from tokenizers import Tokenizer as _HFTokenizer

        if path is None:
            path = os.path.join(
                os.path.dirname(os.path.abspath(__file__)), "assets", "bpe32k.json"
            )
        self._tok = _HFTokenizer.from_file(path)
        self.vocab_size = vocab_size  # model head size (>= trained vocab)
        self.trained_vocab = self._tok.get_vocab_size()
        self.BOS = self._tok.token_to_id("<|begin_of_text|>")
        self.EOS = self._tok.token_to_id("<|end_of_text|>")
        self.EOT = self._tok.token_to_id("<|eot_id|>")
        self._hdr_s = self._tok.token_to_id("<|start_header_id|>")
        self._hdr_e = self._tok.token_to_id("<|end_header_id|>")
        self.eos_ids = frozenset({self.EOS, self.EOT})
        self._specials = {
            self.BOS, self.EOS, self.EOT, self._hdr_s, self._hdr_e,
            self._tok.token_to_id("<|pad|>"),
        }

    @classmethod
    def shared(cls) -> "BPETokenizer":
        """Process-wide instance (the vocab file parse is ~100 ms)."""
        with cls._shared_lock:
            if cls._shared is None:
                cls._shared = cls()
            return cls._shared

    def encode(self, text: str, add_bos: bool = True) -> List[int]:
        ids = self._tok.encode(text, add_special_tokens=False).ids
        return ([self.BOS] + ids) if add_bos else ids

    def decode(self, ids: List[int]) -> str:
        known: List[int] = []
        out: List[str] = []
        for i in ids:
            i = int(i)
            if 0 <= i < self.trained_vocab and i not in self._specials:
                known.append(i)
                continue
            if known:
                out.append(self._tok.decode(known, skip_special_tokens=True))
                known = []
            if i in self._specials:
                continue
            # random-weight sampling can emit any id < vocab_size: fold
            # out-of-vocab ids into printable ASCII so SSE stays text
            out.append(chr(32 + (i % 95)))
        if known:
            out.append(self._tok.decode(known, skip_special_tokens=True))
        return "".join(out)

    def render_chat(self, messages: List[dict]) -> str:
        """llama-3 chat template (header tokens spelled out as text; they
        encode to their single special ids)."""
        parts = []
        for m in messages or []:
            role = m.get("role", "user")
            content = m.get("content", "")
            if isinstance(content, list):  # OpenAI content-parts form
                content = " ".join(
                    p.get("text", "") for p in content if isinstance(p, dict)
                )
            parts.append(
                f"<|start_header_id|>{role}<|end_header_id|>\n\n{content}<|eot_id|>"
            )
        parts.append("<|start_header_id|>assistant<|end_header_id|>\n\n")
        return "".join(parts)


def get_tokenizer(name: str, vocab_size: int):
    """Tokenizer for a model preset: real BPE whenever the vocab can hold
    its ids, the byte-level stand-in for tiny test presets."""
    if name == "byte" or vocab_size < 32000:
        return ByteTokenizer(vocab_size)
    return BPETokenizer.shared()
