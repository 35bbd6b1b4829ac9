"""Train the in-tree byte-level BPE tokenizer (no network: the corpus is
text already present in the image — Python stdlib sources and system docs,
i.e. code + English prose, which matches gateway traffic well enough for
synthetic serving).

Produces llmapigateway_amd/engine/assets/bpe32k.json (checked in). The
vocabulary will not match Meta's Llama tokenizer (that file cannot be
fetched offline); what matters for the gateway is REAL BPE semantics:
multi-byte merges, whitespace-prefixed word pieces, lossless byte-level
round-trip, llama-3-style special tokens and chat template — so prompt
token counts, stop-string handling and usage accounting behave like a
real deployment.

Reproduce: python tools/train_tokenizer.py
"""

import glob
import gzip
import io
import os
import sys

from tokenizers import Tokenizer, models, pre_tokenizers, decoders, trainers

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
OUT = os.path.join(ROOT, "llmapigateway_amd", "engine", "assets", "bpe32k.json")

SPECIALS = [
    "<|begin_of_text|>",
    "<|end_of_text|>",
    "<|start_header_id|>",
    "<|end_header_id|>",
    "<|eot_id|>",
    "<|pad|>",
]


def corpus():
    # NOTE: globbing /usr/share/doc recursively hangs on special files —
    # the stdlib sources alone are ~11 MB of code+prose and train in ~6 s.
    paths = sorted(glob.glob("/usr/lib/python3.10/**/*.py", recursive=True))
    paths += sorted(glob.glob(os.path.join(ROOT, "docs", "*.md")))
    paths += [os.path.join(ROOT, "README.md")]
    n = 0
    for p in paths:
        try:
            if p.endswith(".gz"):
                with gzip.open(p, "rt", errors="ignore") as f:
                    text = f.read()
            else:
                with io.open(p, "r", errors="ignore") as f:
                    text = f.read()
        except OSError:
            continue
        n += len(text)
        yield text
    print(f"corpus: {len(paths)} files, {n/1e6:.1f} MB", file=sys.stderr)


def main():
    tok = Tokenizer(models.BPE(unk_token=None))
    tok.pre_tokenizer = pre_tokenizers.ByteLevel(add_prefix_space=False)
    tok.decoder = decoders.ByteLevel()
    trainer = trainers.BpeTrainer(
        vocab_size=32000,
        min_frequency=2,
        special_tokens=SPECIALS,
        initial_alphabet=pre_tokenizers.ByteLevel.alphabet(),
        show_progress=False,
    )
    tok.train_from_iterator(corpus(), trainer)
    os.makedirs(os.path.dirname(OUT), exist_ok=True)
    tok.save(OUT)
    print(f"saved {OUT} ({os.path.getsize(OUT)/1e6:.2f} MB, "
          f"vocab {tok.get_vocab_size()})")


if __name__ == "__main__":
    main()
