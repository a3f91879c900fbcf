"""Deterministic CLIP-shaped tokenizer.

There is no network in this environment (no downloadable BPE vocab), and all
benchmark configs run random-init weights on synthetic prompts, so token
IDENTITY does not matter — only the CLIP token-stream SHAPE does (bos/eos
ids, 77-token padding, eos = max id so pooled-EOT lookup works). This
tokenizer hashes whitespace/punctuation-split words to stable ids in the
CLIP vocab range. If a real CLIP BPE vocab/merges pair is available on disk
it can be dropped in behind the same interface later.
"""

from __future__ import annotations

import hashlib
import re

import torch

_WORD_RE = re.compile(r"[a-z0-9]+|[^\sa-z0-9]")


class SimpleTokenizer:
    def __init__(self, vocab_size: int = 49408, max_length: int = 77):
        self.vocab_size = vocab_size
        self.max_length = max_length
        self.bos_token_id = vocab_size - 2  # 49406, CLIP <|startoftext|>
        self.eos_token_id = vocab_size - 1  # 49407, CLIP <|endoftext|>

    def _word_id(self, word: str) -> int:
        h = int.from_bytes(hashlib.sha1(word.encode()).digest()[:4], "little")
        return h % (self.vocab_size - 2)  # keep below bos/eos

    def __call__(self, text: str | list[str], device=None) -> torch.Tensor:
        if isinstance(text, str):
            text = [text]
        rows = []
        for t in text:
            words = _WORD_RE.findall(t.lower())[: self.max_length - 2]
            ids = [self.bos_token_id] + [self._word_id(w) for w in words] + [self.eos_token_id]
            # CLIP pads with eos up to 77 — but pooled-EOT uses argmax, which
            # then finds the FIRST eos since all eos ids are equal; pad with 0
            # after the first eos to keep argmax on the real EOT (matches HF
            # behavior where argmax finds the first occurrence of the max id).
            ids = ids + [0] * (self.max_length - len(ids))
            rows.append(ids)
        return torch.tensor(rows, dtype=torch.long, device=device)
