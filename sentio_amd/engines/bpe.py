"""Byte-level BPE tokenizer (offline-trained, deterministic, no network).

The r1 engines tokenized at the byte level (1 token/char), which skews
prompt-length statistics ~4x versus any real deployment.
This tokenizer keeps the byte tokenizer's exactness properties — fully
reversible on arbitrary UTF-8, same special ids — while emitting multi-byte
tokens from a merge table trained by scripts/train_tokenizer.py on a
deterministic synthetic corpus (committed to engines/assets/bpe_merges.json;
there is no network for a real vocab).

Id layout (superset of ByteTokenizer's): 0=PAD, 1=BOS, 2=EOS, 3..258 raw
bytes, 259.. merge tokens.  Compute costs per token stay honest: the model
embedding tables keep their real vocab sizes (e.g. 128256 for the
Llama-3-class lm_head); the tokenizer simply emits low ids.

Pre-tokenization splits on whitespace GPT-style (a space joins the WORD
THAT FOLLOWS it), and merges never cross pre-token boundaries — so any
boundary between pre-tokens is a split-exact point, which prefix_split
relies on (prefix-KV caching needs encode(prefix)+encode(suffix) ==
encode(full)).
"""

from __future__ import annotations

import json
import os
import re
from functools import lru_cache

from sentio_amd.engines.tokenizer import (
    BOS_ID,
    BYTE_OFFSET,
    EOS_ID,
    PAD_ID,
    ByteTokenizer,
)

MERGE_OFFSET = BYTE_OFFSET + 256           # 259: first merge id
_ASSET = os.path.join(os.path.dirname(__file__), "assets", "bpe_merges.json")

# a pre-token = optional single leading space + run of non-space, or a run
# of whitespace (newlines etc. group separately)
_PRETOKEN = re.compile(rb" ?[^\s]+|\s+")


class BPETokenizer:
    """Greedy pair-merge BPE over raw bytes; merge ranks define priority."""

    def __init__(self, merges: list[list[int]] | None = None):
        if merges is None:
            with open(_ASSET) as f:
                merges = json.load(f)["merges"]
        # token id -> byte string
        self.token_bytes: list[bytes] = [b""] * MERGE_OFFSET
        for b in range(256):
            self.token_bytes[BYTE_OFFSET + b] = bytes([b])
        self.ranks: dict[tuple[int, int], int] = {}
        for rank, (a, b) in enumerate(merges):
            tid = MERGE_OFFSET + rank
            self.token_bytes.append(self.token_bytes[a] + self.token_bytes[b])
            self.ranks[(a, b)] = rank
        self.vocab_size = len(self.token_bytes)
        self._word_cache = lru_cache(maxsize=65536)(self._encode_word)

    # ----- encoding -----
    def _encode_word(self, word: bytes) -> tuple[int, ...]:
        ids = [BYTE_OFFSET + b for b in word]
        while len(ids) > 1:
            best_rank = None
            best_i = -1
            for i in range(len(ids) - 1):
                r = self.ranks.get((ids[i], ids[i + 1]))
                if r is not None and (best_rank is None or r < best_rank):
                    best_rank = r
                    best_i = i
            if best_rank is None:
                break
            ids[best_i: best_i + 2] = [MERGE_OFFSET + best_rank]
        return tuple(ids)

    def encode(self, text: str, max_len: int | None = None,
               add_bos: bool = True, add_eos: bool = False) -> list[int]:
        ids = [BOS_ID] if add_bos else []
        for m in _PRETOKEN.finditer(text.encode("utf-8")):
            ids.extend(self._word_cache(m.group(0)))
            if max_len is not None and len(ids) >= max_len + 1:
                break
        if add_eos:
            ids.append(EOS_ID)
        if max_len is not None:
            ids = ids[:max_len]
        return ids

    # ----- decoding -----
    _PLACEHOLDER = ByteTokenizer._PLACEHOLDER

    def decode(self, ids: list[int]) -> str:
        parts: list[str] = []
        run = bytearray()
        for i in ids:
            if BYTE_OFFSET <= i < self.vocab_size:
                run += self.token_bytes[i]
            elif i >= self.vocab_size:
                # model vocab exceeds tokenizer vocab (random-init logits
                # sample the whole 128k table): deterministic placeholders
                if run:
                    parts.append(run.decode("utf-8", errors="replace"))
                    run = bytearray()
                parts.append(" " + self._PLACEHOLDER[i % len(self._PLACEHOLDER)])
        if run:
            parts.append(run.decode("utf-8", errors="replace"))
        return "".join(parts)

    def encode_batch(self, texts: list[str], max_len: int,
                     add_bos: bool = True) -> tuple[list[list[int]], list[int]]:
        seqs = [self.encode(t, max_len, add_bos=add_bos) for t in texts]
        lens = [len(s) for s in seqs]
        width = max(lens) if lens else 1
        return [s + [PAD_ID] * (width - len(s)) for s in seqs], lens

    def count_tokens(self, text: str) -> int:
        return len(self.encode(text, None))

    def prefix_split(self, text: str, max_tokens: int) -> tuple[int, list[int]]:
        """Largest char count n with encode(text[:n]) <= max_tokens ids AND
        encode(text[:n]) + encode(text[n:], add_bos=False) == encode(text).
        Splits only at pre-token boundaries (merges never cross them)."""
        budget = max_tokens - 1  # BOS
        if budget <= 0:
            return 0, []
        data = text.encode("utf-8")
        matches = list(_PRETOKEN.finditer(data))
        used = 0
        end_byte = 0
        # the FINAL pre-token of the prefix text is never a safe boundary:
        # in the full string it may continue (a trailing space joins the
        # suffix's first word; a partial word merges with the suffix)
        for m in matches[:-1]:
            n_ids = len(self._word_cache(m.group(0)))
            if used + n_ids > budget:
                break
            used += n_ids
            end_byte = m.end()
        if end_byte == 0:
            return 0, []
        n_chars = len(data[:end_byte].decode("utf-8"))  # boundary is exact:
        # pre-tokens never split inside a UTF-8 char (non-space runs keep
        # multi-byte chars whole)
        return n_chars, self.encode(text[:n_chars], None)


_default: BPETokenizer | None = None


def get_tokenizer():
    """The engine tokenizer: trained BPE when the committed vocab exists
    (SENTIO_TOKENIZER=byte forces the byte fallback)."""
    global _default
    if os.environ.get("SENTIO_TOKENIZER", "bpe") == "byte":
        return ByteTokenizer()
    if _default is None:
        if not os.path.exists(_ASSET):
            return ByteTokenizer()
        _default = BPETokenizer()
    return _default
