"""Deterministic offline byte-level tokenizer.

There is no network to fetch a trained vocab, so the engine tokenizes at the
byte level: ids 0..2 are specials, 3..258 are raw bytes.  Model embedding
tables may be much larger (e.g. the Llama-3-8B-class generator keeps its
real 128256-entry vocab so the lm_head GEMM has the true cost); the tokenizer
simply only emits low ids.  Fully reversible, no data files.
"""

from __future__ import annotations

PAD_ID = 0
BOS_ID = 1
EOS_ID = 2
BYTE_OFFSET = 3


class ByteTokenizer:
    vocab_size = 259

    def encode(self, text: str, max_len: int | None = None, add_bos: bool = True,
               add_eos: bool = False) -> list[int]:
        ids = [BOS_ID] if add_bos else []
        ids += [b + BYTE_OFFSET for b in text.encode("utf-8")]
        if add_eos:
            ids.append(EOS_ID)
        if max_len is not None:
            ids = ids[:max_len]
        return ids

    # ids beyond the byte range appear when a model's vocab is larger than
    # the tokenizer's (random-init logits sample the whole 128k vocab) —
    # render them as deterministic placeholder words so generations are
    # non-empty, readable text.
    _PLACEHOLDER = (
        "alpha beta gamma delta epsilon zeta eta theta iota kappa lambda mu "
        "nu xi omicron pi rho sigma tau upsilon phi chi psi omega node edge "
        "graph wave tile lane block grid cache token shard index fuse rank"
    ).split()

    def decode(self, ids: list[int]) -> str:
        parts: list[str] = []
        run: list[int] = []
        for i in ids:
            if BYTE_OFFSET <= i < BYTE_OFFSET + 256:
                run.append(i - BYTE_OFFSET)
            elif i >= BYTE_OFFSET + 256:
                if run:
                    parts.append(bytes(run).decode("utf-8", errors="replace"))
                    run = []
                parts.append(" " + self._PLACEHOLDER[i % len(self._PLACEHOLDER)])
        if run:
            parts.append(bytes(run).decode("utf-8", errors="replace"))
        return "".join(parts)

    def encode_batch(
        self, texts: list[str], max_len: int, add_bos: bool = True
    ) -> tuple[list[list[int]], list[int]]:
        """Returns (padded id lists, true lengths)."""
        seqs = [self.encode(t, max_len, add_bos=add_bos) for t in texts]
        lens = [len(s) for s in seqs]
        width = max(lens) if lens else 1
        padded = [s + [PAD_ID] * (width - len(s)) for s in seqs]
        return padded, lens

    def count_tokens(self, text: str) -> int:
        return 1 + len(text.encode("utf-8"))

    def prefix_split(self, text: str, max_tokens: int) -> tuple[int, list[int]]:
        """Largest char count n such that encode(text[:n]) fits in max_tokens
        ids AND encode(text[:n]) + encode(text[n:], add_bos=False) ==
        encode(text) (split-exactness).  The budget is counted in TOKENS —
        multi-byte UTF-8 chars produce several ids each, so a char-based cap
        can overshoot a token budget.  Returns (n_chars, encode(text[:n]))."""
        budget = max_tokens - 1  # BOS
        if budget <= 0:
            return 0, []
        n = used = 0
        for ch in text:
            b = len(ch.encode("utf-8"))
            if used + b > budget:
                break
            used += b
            n += 1
        return n, self.encode(text[:n], None)
