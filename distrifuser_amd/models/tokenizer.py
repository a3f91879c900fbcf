"""CLIP tokenizers.

Two implementations behind one ``__call__(text, device) -> LongTensor[B, 77]``
interface:

* :class:`CLIPBPETokenizer` — a real CLIP byte-pair-encoding tokenizer that
  loads ``vocab.json`` + ``merges.txt`` from a local checkpoint directory
  (the files HF ``CLIPTokenizer`` ships; reference obtains this through
  ``transformers``, /root/reference/distrifuser/pipelines.py:39-41 via the
  diffusers pipeline). Matches HF semantics: lowercased, whitespace-cleaned,
  GPT-2 byte encoder, ``</w>`` end-of-word merges, bos/eos wrapping, and
  eos-padding (CLIP-L) or 0-padding (open-CLIP big-G, SDXL's tokenizer_2).
* :class:`SimpleTokenizer` — deterministic hash fallback for the offline
  random-init benchmark configs where token IDENTITY cannot matter (no vocab
  on disk, weights are random) — only the CLIP token-stream SHAPE does.
  ``pipelines.from_pretrained`` refuses to pair this with a real checkpoint
  unless the checkpoint was written by our own ``save_pretrained`` with a
  SimpleTokenizer (marker file) or the caller opts in explicitly.
"""

from __future__ import annotations

import functools
import hashlib
import json
import os
import re

import torch

_WORD_RE = re.compile(r"[a-z0-9]+|[^\sa-z0-9]")

# CLIP's token split pattern (HF CLIPTokenizer); \p{L}/\p{N} rewritten for
# the stdlib re module.
_BPE_PAT = re.compile(
    r"<\|startoftext\|>|<\|endoftext\|>|'s|'t|'re|'ve|'m|'ll|'d"
    r"|[^\W\d_]+|\d|[^\s\w]+",
    re.IGNORECASE | re.UNICODE,
)


@functools.lru_cache()
def _bytes_to_unicode() -> dict[int, str]:
    """GPT-2 reversible byte<->unicode map (printable chars for all 256 bytes)."""
    bs = list(range(ord("!"), ord("~") + 1)) + list(range(ord("¡"), ord("¬") + 1)) + \
        list(range(ord("®"), ord("ÿ") + 1))
    cs = bs[:]
    n = 0
    for b in range(256):
        if b not in bs:
            bs.append(b)
            cs.append(256 + n)
            n += 1
    return dict(zip(bs, [chr(c) for c in cs]))


def _clean_text(text: str) -> str:
    return re.sub(r"\s+", " ", text).strip().lower()


class CLIPBPETokenizer:
    """Real CLIP BPE from local ``vocab.json`` + ``merges.txt``."""

    def __init__(self, vocab_path: str, merges_path: str, max_length: int = 77,
                 pad_with_zero: bool = False):
        self.vocab_path = vocab_path
        self.merges_path = merges_path
        with open(vocab_path, encoding="utf-8") as f:
            self.encoder: dict[str, int] = json.load(f)
        merges: list[tuple[str, str]] = []
        with open(merges_path, encoding="utf-8") as f:
            for line in f:
                line = line.strip()
                if not line or line.startswith("#"):
                    continue
                a, _, b = line.partition(" ")
                if b:
                    merges.append((a, b))
        self.bpe_ranks = {pair: i for i, pair in enumerate(merges)}
        self.byte_encoder = _bytes_to_unicode()
        self.max_length = max_length
        self.vocab_size = max(self.encoder.values()) + 1
        self.bos_token_id = self.encoder.get("<|startoftext|>", self.vocab_size - 2)
        self.eos_token_id = self.encoder.get("<|endoftext|>", self.vocab_size - 1)
        # CLIP-L pads with eos; SDXL's tokenizer_2 (open-CLIP) pads with "!"=0.
        self.pad_token_id = 0 if pad_with_zero else self.eos_token_id
        self._cache: dict[str, list[int]] = {}

    # -- BPE core ------------------------------------------------------------

    def _bpe(self, token: str) -> list[str]:
        word: tuple[str, ...] = tuple(token[:-1]) + (token[-1] + "</w>",)
        if len(word) == 1:
            return [word[0]]
        while True:
            pairs = {(word[i], word[i + 1]) for i in range(len(word) - 1)}
            best = min(pairs, key=lambda p: self.bpe_ranks.get(p, 1 << 30))
            if best not in self.bpe_ranks:
                break
            a, b = best
            merged: list[str] = []
            i = 0
            while i < len(word):
                if i < len(word) - 1 and word[i] == a and word[i + 1] == b:
                    merged.append(a + b)
                    i += 2
                else:
                    merged.append(word[i])
                    i += 1
            word = tuple(merged)
            if len(word) == 1:
                break
        return list(word)

    def encode_words(self, text: str) -> list[int]:
        ids: list[int] = []
        for tok in _BPE_PAT.findall(_clean_text(text)):
            if tok in self._cache:
                ids.extend(self._cache[tok])
                continue
            btok = "".join(self.byte_encoder[b] for b in tok.encode("utf-8"))
            sub = [self.encoder.get(piece, self.eos_token_id) for piece in self._bpe(btok)]
            self._cache[tok] = sub
            ids.extend(sub)
        return ids

    def __call__(self, text: str | list[str], device=None) -> torch.Tensor:
        if isinstance(text, str):
            text = [text]
        rows = []
        for t in text:
            ids = self.encode_words(t)[: self.max_length - 2]
            ids = [self.bos_token_id] + ids + [self.eos_token_id]
            # pooled-EOT lookup is argmax(ids) (models/clip.py); eos is the
            # max id and argmax returns the FIRST occurrence, so eos-padding
            # keeps the pooled position on the real EOT (HF behavior).
            ids = ids + [self.pad_token_id] * (self.max_length - len(ids))
            rows.append(ids)
        return torch.tensor(rows, dtype=torch.long, device=device)


def load_clip_tokenizer(root: str, subfolder: str = "tokenizer",
                        pad_with_zero: bool = False) -> "CLIPBPETokenizer | None":
    """Load vocab.json+merges.txt from ``<root>/<subfolder>/`` if present."""
    d = os.path.join(root, subfolder)
    vocab, merges = os.path.join(d, "vocab.json"), os.path.join(d, "merges.txt")
    if os.path.isfile(vocab) and os.path.isfile(merges):
        return CLIPBPETokenizer(vocab, merges, pad_with_zero=pad_with_zero)
    return None


class SimpleTokenizer:
    """Hash-based CLIP-shaped fallback (offline random-init runs only)."""

    def __init__(self, vocab_size: int = 49408, max_length: int = 77):
        self.vocab_size = vocab_size
        self.max_length = max_length
        self.bos_token_id = vocab_size - 2  # 49406, CLIP <|startoftext|>
        self.eos_token_id = vocab_size - 1  # 49407, CLIP <|endoftext|>
        self.pad_token_id = 0

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
            # pad with 0 ("!" in CLIP vocab — SDXL tokenizer_2's convention);
            # argmax still lands on the single eos.
            ids = ids + [self.pad_token_id] * (self.max_length - len(ids))
            rows.append(ids)
        return torch.tensor(rows, dtype=torch.long, device=device)
